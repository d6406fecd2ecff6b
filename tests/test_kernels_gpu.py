"""Numerics tests: every gfx950 HIP kernel vs the fp32 PyTorch reference.

Run on an MI355X via:  gpurun -- 'python -m pytest tests -m gpu -x -q'
"""
import math

import pytest
import torch

import kubeai_amd.ops as ops
from kubeai_amd.ops import ref

pytestmark = pytest.mark.gpu

DEV = "cuda"


def assert_close_bf16(out, ref_f32, atol=2e-2, rtol=2e-2):
    ref_bf16 = ref_f32.to(torch.bfloat16).float()
    torch.testing.assert_close(out.float(), ref_bf16, atol=atol, rtol=rtol)


@pytest.fixture(autouse=True)
def _seed():
    torch.manual_seed(1234)


def test_hip_ext_loaded():
    # GPU boxes must run the native path — no silent eager fallback
    assert ops.have_hip_ext()


@pytest.mark.parametrize("shape", [(1, 4096), (17, 4096), (256, 8192), (33, 256)])
def test_rmsnorm(shape):
    x = torch.randn(shape, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(shape[-1], dtype=torch.bfloat16, device=DEV)
    out = ops.rmsnorm(x, w, 1e-5)
    expected = ref.rmsnorm(x.float(), w.float(), 1e-5)
    assert_close_bf16(out, expected)


@pytest.mark.parametrize("shape", [(9, 4096), (128, 8192)])
def test_fused_add_rmsnorm(shape):
    x = torch.randn(shape, dtype=torch.bfloat16, device=DEV)
    res = torch.randn(shape, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(shape[-1], dtype=torch.bfloat16, device=DEV)
    y_ref, res_ref = ref.fused_add_rmsnorm(x.float(), res.float(), w.float(), 1e-5)
    y, res_out = ops.fused_add_rmsnorm(x.clone(), res.clone(), w, 1e-5)
    # residual is the exact bf16 sum; tolerance for the bf16 round trip
    assert_close_bf16(res_out, res_ref, atol=3e-2, rtol=3e-2)
    assert_close_bf16(y, y_ref, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("nq,nkv", [(32, 8), (16, 16), (8, 1)])
def test_rope(nq, nkv):
    T, hd = 37, 128
    q = torch.randn(T, nq, hd, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(T, nkv, hd, dtype=torch.bfloat16, device=DEV)
    pos = torch.randint(0, 4000, (T,), dtype=torch.int32, device=DEV)
    cs = ref.make_cos_sin_cache(hd, 4096, 500000.0).to(DEV)
    q_ref, k_ref = ref.rope(q.float(), k.float(), pos, cs)
    q_out, k_out = ops.rope(q.clone(), k.clone(), pos, cs)
    assert_close_bf16(q_out, q_ref)
    assert_close_bf16(k_out, k_ref)


def test_reshape_and_cache():
    T, nkv, hd, nb, bs = 50, 8, 128, 16, 16
    k = torch.randn(T, nkv, hd, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(T, nkv, hd, dtype=torch.bfloat16, device=DEV)
    kc = torch.zeros(nb, nkv, bs, hd, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    slots = torch.randperm(nb * bs, device=DEV)[:T].to(torch.int64)
    slots[3] = -1  # dropped token
    kc_ref, vc_ref = kc.clone(), vc.clone()
    ref.reshape_and_cache(k, v, kc_ref, vc_ref, slots)
    ops.reshape_and_cache(k, v, kc, vc, slots)
    torch.testing.assert_close(kc, kc_ref)
    torch.testing.assert_close(vc, vc_ref)


def _rand_cache(nb, nkv, bs, hd):
    kc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=DEV)
    return kc, vc


@pytest.mark.parametrize("nq,nkv", [(32, 8), (8, 8), (16, 2), (64, 8)])
@pytest.mark.parametrize("lens", [[1], [16], [30, 7], [257, 64, 19, 100]])
def test_paged_decode(nq, nkv, lens):
    hd, bs = 128, 16
    B = len(lens)
    max_blocks = max((L + bs - 1) // bs for L in lens)
    nb = B * max_blocks + 1
    kc, vc = _rand_cache(nb, nkv, bs, hd)
    bt = torch.arange(1, nb, dtype=torch.int32, device=DEV).reshape(B, max_blocks)
    seq_lens = torch.tensor(lens, dtype=torch.int32, device=DEV)
    q = torch.randn(B, nq, hd, dtype=torch.bfloat16, device=DEV)
    scale = 1.0 / math.sqrt(hd)
    expected = ref.paged_attention_decode(
        q.float(), kc.float(), vc.float(), bt, seq_lens, scale
    )
    out = ops.paged_attention_decode(q, kc, vc, bt, seq_lens, scale)
    assert_close_bf16(out, expected, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("nq,nkv", [(32, 8), (8, 8), (64, 8)])
@pytest.mark.parametrize(
    "q_lens,ctx_lens",
    [
        ([16], [0]),          # single fresh tile
        ([100], [0]),         # fresh prefill, ragged tail
        ([40], [60]),         # chunked / prefix-hit continuation
        ([1], [37]),          # decode-shaped via prefill path
        ([33, 64, 5], [0, 16, 91]),  # mixed batch
    ],
)
def test_paged_prefill(nq, nkv, q_lens, ctx_lens):
    hd, bs = 128, 16
    B = len(q_lens)
    seq_lens_l = [q + c for q, c in zip(q_lens, ctx_lens)]
    max_blocks = max((L + bs - 1) // bs for L in seq_lens_l)
    nb = B * max_blocks + 1
    kc, vc = _rand_cache(nb, nkv, bs, hd)
    bt = torch.arange(1, nb, dtype=torch.int32, device=DEV).reshape(B, max_blocks)
    seq_lens = torch.tensor(seq_lens_l, dtype=torch.int32, device=DEV)
    qsl = torch.tensor(
        [0] + list(torch.tensor(q_lens).cumsum(0)), dtype=torch.int32, device=DEV
    )
    Tq = sum(q_lens)
    q = torch.randn(Tq, nq, hd, dtype=torch.bfloat16, device=DEV)
    scale = 1.0 / math.sqrt(hd)
    expected = ref.paged_attention_prefill(
        q.float(), kc.float(), vc.float(), bt, qsl, seq_lens, scale
    )
    out = ops.paged_attention_prefill(q, kc, vc, bt, qsl, seq_lens, scale)
    assert_close_bf16(out, expected, atol=3e-2, rtol=3e-2)


def test_strided_qkv_views():
    """rope/cache/decode must accept row-strided views into the fused qkv
    buffer (the no-copy path the model uses)."""
    T, nq, nkv, hd, bs = 24, 32, 8, 128, 16
    W = (nq + 2 * nkv) * hd
    qkv = torch.randn(T, W, dtype=torch.bfloat16, device=DEV)
    qkv_ref = qkv.clone()
    pos = torch.randint(0, 1000, (T,), dtype=torch.int32, device=DEV)
    cs = ref.make_cos_sin_cache(hd, 2048, 500000.0).to(DEV)

    q = qkv[:, : nq * hd].unflatten(-1, (nq, hd))
    k = qkv[:, nq * hd : (nq + nkv) * hd].unflatten(-1, (nkv, hd))
    v = qkv[:, (nq + nkv) * hd :].unflatten(-1, (nkv, hd))
    ops.rope(q, k, pos, cs)

    qc = qkv_ref[:, : nq * hd].reshape(T, nq, hd).contiguous()
    kc_ = qkv_ref[:, nq * hd : (nq + nkv) * hd].reshape(T, nkv, hd).contiguous()
    q_ref, k_ref = ref.rope(qc.float(), kc_.float(), pos, cs)
    assert_close_bf16(q, q_ref)
    assert_close_bf16(k, k_ref)

    # cache write from strided views
    nb = T + 1
    kcache = torch.zeros(nb, nkv, bs, hd, dtype=torch.bfloat16, device=DEV)
    vcache = torch.zeros_like(kcache)
    slots = torch.arange(T, dtype=torch.int64, device=DEV)
    ops.reshape_and_cache(k, v, kcache, vcache, slots)
    kcache_ref = torch.zeros_like(kcache)
    vcache_ref = torch.zeros_like(vcache)
    ref.reshape_and_cache(
        k.contiguous(), v.contiguous(), kcache_ref, vcache_ref, slots
    )
    torch.testing.assert_close(kcache, kcache_ref)
    torch.testing.assert_close(vcache, vcache_ref)

    # decode from a strided q view
    B = T
    bt = torch.arange(0, 1, dtype=torch.int32, device=DEV).repeat(B, 1)
    seq_lens = torch.full((B,), bs, dtype=torch.int32, device=DEV)
    kcache2 = torch.randn(2, nkv, bs, hd, dtype=torch.bfloat16, device=DEV)
    vcache2 = torch.randn_like(kcache2)
    out = ops.paged_attention_decode(q, kcache2, vcache2, bt, seq_lens, 0.1)
    expected = ref.paged_attention_decode(
        q.contiguous().float(), kcache2.float(), vcache2.float(), bt, seq_lens, 0.1
    )
    assert_close_bf16(out, expected, atol=3e-2, rtol=3e-2)


def test_silu_and_mul():
    x = torch.randn(77, 2 * 14336, dtype=torch.bfloat16, device=DEV)
    out = ops.silu_and_mul(x)
    expected = ref.silu_and_mul(x.float())
    assert_close_bf16(out, expected)


@pytest.mark.parametrize("M", [1, 7, 16, 33, 40, 64])
@pytest.mark.parametrize(
    "N,K", [(6144, 4096), (4096, 4096), (4096, 14336), (128256, 4096)]
)
def test_skinny_gemm(M, N, K):
    if N == 128256 and M > 16:
        pytest.skip("one big-vocab case is enough")
    from kubeai_amd import _C

    x = torch.randn(M, K, dtype=torch.bfloat16, device=DEV) / 8
    w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV) / 8
    out = torch.empty(M, N, dtype=torch.bfloat16, device=DEV)
    _C.skinny_gemm(out, x, w)
    expected = (x.float() @ w.float().t())
    # bf16 inputs, fp32 accumulation on both sides
    torch.testing.assert_close(out.float(), expected, atol=0.3, rtol=3e-2)


@pytest.mark.parametrize("R", [8, 16, 64])
def test_sgmv(R):
    from kubeai_amd import _C

    T, H, out = 33, 4096, 6144
    x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
    y = torch.randn(T, out, dtype=torch.bfloat16, device=DEV)
    A = torch.randn(R, H, dtype=torch.bfloat16, device=DEV) * 0.05
    B = torch.randn(out, R, dtype=torch.bfloat16, device=DEV) * 0.5
    idx = torch.tensor([0, 3, 7, 20, 32], dtype=torch.int64, device=DEV)
    scale = 0.125
    expected = y.float().clone()
    expected[idx] += (x[idx].float() @ A.float().T @ B.float().T) * scale
    y2 = y.clone()
    _C.sgmv(y2, x, A, B, idx, scale)
    untouched = torch.ones(T, dtype=torch.bool)
    untouched[idx.cpu()] = False
    torch.testing.assert_close(y2[untouched], y[untouched])
    assert_close_bf16(y2[idx], expected[idx], atol=5e-2, rtol=5e-2)


def _dequant(q8, scale):
    return q8.float() * scale.view(-1, 1)


@pytest.mark.parametrize("T,H", [(7, 4096), (64, 8192)])
def test_rmsnorm_fp8(T, H):
    x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(H, dtype=torch.bfloat16, device=DEV)
    q8, scale = ops.rmsnorm_fp8(x, w, 1e-5)
    expected = ref.rmsnorm(x.float(), w.float(), 1e-5)
    # scale = amax/448 per row
    torch.testing.assert_close(
        scale, expected.abs().amax(dim=-1) / 448.0, atol=1e-2, rtol=1e-2
    )
    torch.testing.assert_close(
        _dequant(q8, scale), expected, atol=0.07, rtol=0.07
    )


def test_fused_add_rmsnorm_fp8():
    T, H = 9, 4096
    x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
    res = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(H, dtype=torch.bfloat16, device=DEV)
    y_ref, res_ref = ref.fused_add_rmsnorm(x.float(), res.float(), w.float(), 1e-5)
    q8, scale = ops.fused_add_rmsnorm_fp8(x, res, w, 1e-5)
    assert_close_bf16(res, res_ref, atol=3e-2, rtol=3e-2)  # updated in place
    torch.testing.assert_close(_dequant(q8, scale), y_ref, atol=0.07, rtol=0.07)


def test_silu_and_mul_fp8():
    x = torch.randn(13, 2 * 14336, dtype=torch.bfloat16, device=DEV)
    q8, scale = ops.silu_and_mul_fp8(x)
    expected = ref.silu_and_mul(x.float())
    torch.testing.assert_close(_dequant(q8, scale), expected, atol=0.07, rtol=0.07)


def test_quant_fp8_roundtrip():
    x = torch.randn(21, 4096, dtype=torch.bfloat16, device=DEV) * 3
    q8, scale = ops.quant_fp8(x)
    torch.testing.assert_close(_dequant(q8, scale), x.float(), atol=0.1, rtol=0.07)
    # matches torch's cast given the same scale
    expected_q = (x.float() / scale.view(-1, 1)).clamp(-448, 448).to(
        torch.float8_e4m3fn
    )
    assert (q8.view(torch.uint8) == expected_q.view(torch.uint8)).float().mean() > 0.99


def test_greedy_sample():
    logits = torch.randn(64, 128256, dtype=torch.float32, device=DEV)
    out = ops.greedy_sample(logits)
    assert torch.equal(out, logits.argmax(dim=-1))


def test_gumbel_sample_matches_reference_rng():
    B, V = 32, 4096
    logits = torch.randn(B, V, dtype=torch.float32, device=DEV) * 4
    temps = torch.full((B,), 0.8, dtype=torch.float32, device=DEV)
    temps[0] = 0.0  # greedy row
    seeds = torch.arange(100, 100 + B, dtype=torch.int64, device=DEV)
    out = ops.gumbel_sample(logits, temps, seeds, step=7)
    expected = ref.gumbel_sample(logits.cpu(), temps.cpu(), seeds.cpu(), step=7)
    # identical RNG spec; allow <=2 mismatches from fast-math exp/log ties
    mismatches = (out.cpu() != expected).sum().item()
    assert mismatches <= 2, f"{mismatches} mismatches"
    # determinism
    out2 = ops.gumbel_sample(logits, temps, seeds, step=7)
    assert torch.equal(out, out2)
    out3 = ops.gumbel_sample(logits, temps, seeds, step=8)
    assert not torch.equal(out, out3)


def test_nucleus_kernels_match_reference():
    """Fused nucleus_stats/nucleus_accept vs the plain torch formula."""
    torch.manual_seed(0)
    S, V = 9, 5000
    logits = (torch.randn(S, V, device=DEV) * 3).float().contiguous()
    temps = torch.tensor([0.8] * 4 + [0.0] + [1.3] * 4, device=DEV)
    top_ps = torch.tensor([0.9, 0.5, 0.99, 1.0, 0.9, 0.7, 0.9, 0.9, 0.2],
                          device=DEV)
    top_ks = torch.tensor([0, 0, 50, 0, 0, 0, 3, 100000, 0],
                          dtype=torch.int32, device=DEV)
    m, z = ops.nucleus_stats(logits, temps)
    # reference stats
    t = temps.clamp_min(1e-6).unsqueeze(1)
    m_ref = logits.max(dim=-1).values
    z_ref = torch.exp((logits - m_ref.unsqueeze(1)) / t).sum(-1)
    assert torch.allclose(m, m_ref)
    assert torch.allclose(z, z_ref, rtol=1e-3)
    for trial in range(5):
        cand = torch.randint(0, V, (S,), dtype=torch.int64, device=DEV)
        ok = ops.nucleus_accept(logits, cand, m, z, temps, top_ps, top_ks)
        lt = logits.gather(1, cand.view(-1, 1))
        above = logits > lt
        mass = (torch.exp((logits - m_ref.unsqueeze(1)) / t) * above).sum(-1) / z_ref
        cnt = above.sum(-1)
        want = ((mass < top_ps) & ((top_ks <= 0) | (cnt < top_ks))
                | (temps <= 0)).to(torch.uint8)
        assert torch.equal(ok.cpu(), want.cpu()), (trial, ok, want)


def test_paged_prefill_long_context():
    """Chunked continuation against a LONG cached context (16k)."""
    torch.manual_seed(3)
    nq, nkv, hd, bs = 32, 8, 128, 16
    Tq, ctx = 1024, 15360
    L = Tq + ctx
    nb = L // bs + 1
    kc, vc = _rand_cache(nb, nkv, bs, hd)
    bt = torch.arange(1, nb, dtype=torch.int32, device=DEV).reshape(1, -1)
    sl = torch.tensor([L], dtype=torch.int32, device=DEV)
    qsl = torch.tensor([0, Tq], dtype=torch.int32, device=DEV)
    q = torch.randn(Tq, nq, hd, dtype=torch.bfloat16, device=DEV)
    scale = 1.0 / math.sqrt(hd)
    out = ops.paged_attention_prefill(q, kc, vc, bt, qsl, sl, scale)
    want = ref.paged_attention_prefill(
        q.float(), kc.float(), vc.float(), bt, qsl, sl, scale
    )
    assert_close_bf16(out, want, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("lens,window", [([100, 37], 48), ([2048], 512),
                                         ([17], 64)])
@pytest.mark.gpu
def test_paged_decode_sliding_window(lens, window):
    nq, nkv, hd, bs = 32, 8, 128, 16
    B = len(lens)
    max_blocks = max((L + bs - 1) // bs for L in lens)
    nb = B * max_blocks + 1
    kc, vc = _rand_cache(nb, nkv, bs, hd)
    bt = torch.arange(1, nb, dtype=torch.int32, device=DEV).reshape(B, max_blocks)
    seq_lens = torch.tensor(lens, dtype=torch.int32, device=DEV)
    q = torch.randn(B, nq, hd, dtype=torch.bfloat16, device=DEV)
    scale = 1.0 / math.sqrt(hd)
    expected = ref.paged_attention_decode(
        q.float(), kc.float(), vc.float(), bt, seq_lens, scale, window=window
    )
    out = ops.paged_attention_decode(q, kc, vc, bt, seq_lens, scale,
                                     window=window)
    assert_close_bf16(out, expected, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("q_lens,ctx_lens,window",
                         [([120], [0], 32), ([40], [60], 24),
                          ([33, 64], [0, 16], 48)])
@pytest.mark.gpu
def test_paged_prefill_sliding_window(q_lens, ctx_lens, window):
    nq, nkv, hd, bs = 32, 8, 128, 16
    B = len(q_lens)
    lens = [q + c for q, c in zip(q_lens, ctx_lens)]
    max_blocks = max((L + bs - 1) // bs for L in lens)
    nb = B * max_blocks + 1
    kc, vc = _rand_cache(nb, nkv, bs, hd)
    bt = torch.arange(1, nb, dtype=torch.int32, device=DEV).reshape(B, max_blocks)
    seq_lens = torch.tensor(lens, dtype=torch.int32, device=DEV)
    qsl = torch.tensor([0] + list(torch.tensor(q_lens).cumsum(0)),
                       dtype=torch.int32, device=DEV)
    Tq = sum(q_lens)
    q = torch.randn(Tq, nq, hd, dtype=torch.bfloat16, device=DEV)
    scale = 1.0 / math.sqrt(hd)
    expected = ref.paged_attention_prefill(
        q.float(), kc.float(), vc.float(), bt, qsl, seq_lens, scale,
        window=window,
    )
    out = ops.paged_attention_prefill(q, kc, vc, bt, qsl, seq_lens, scale,
                                      window=window)
    assert_close_bf16(out, expected, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("nq,nkv,lens,softcap,window",
                         [(8, 4, [100, 37], 0.0, 0),
                          (8, 4, [257], 50.0, 0),
                          (2, 1, [2048], 50.0, 512),
                          (8, 4, [16], 0.0, 0)])
@pytest.mark.gpu
def test_paged_decode_hd256(nq, nkv, lens, softcap, window):
    """gemma2-shaped decode: head_dim 256, optional softcap + window."""
    hd, bs = 256, 16
    B = len(lens)
    max_blocks = max((L + bs - 1) // bs for L in lens)
    nb = B * max_blocks + 1
    kc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=DEV)
    bt = torch.arange(1, nb, dtype=torch.int32, device=DEV).reshape(B, max_blocks)
    seq_lens = torch.tensor(lens, dtype=torch.int32, device=DEV)
    q = torch.randn(B, nq, hd, dtype=torch.bfloat16, device=DEV)
    scale = 1.0 / math.sqrt(hd)
    expected = ref.paged_attention_decode(
        q.float(), kc.float(), vc.float(), bt, seq_lens, scale,
        window=window, softcap=softcap,
    )
    out = ops.paged_attention_decode(q, kc, vc, bt, seq_lens, scale,
                                     window=window, softcap=softcap)
    assert_close_bf16(out, expected, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("q_lens,ctx_lens,softcap,window",
                         [([100], [0], 0.0, 0),
                          ([64], [30], 50.0, 0),
                          ([40, 17], [0, 50], 50.0, 48)])
@pytest.mark.gpu
def test_paged_prefill_hd256(q_lens, ctx_lens, softcap, window):
    """gemma2-shaped prefill: head_dim 256, optional softcap + window."""
    nq, nkv, hd, bs = 8, 4, 256, 16
    B = len(q_lens)
    lens = [a + c for a, c in zip(q_lens, ctx_lens)]
    max_blocks = max((L + bs - 1) // bs for L in lens)
    nb = B * max_blocks + 1
    kc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=DEV)
    bt = torch.arange(1, nb, dtype=torch.int32, device=DEV).reshape(B, max_blocks)
    seq_lens = torch.tensor(lens, dtype=torch.int32, device=DEV)
    qsl = torch.tensor([0] + list(torch.tensor(q_lens).cumsum(0)),
                       dtype=torch.int32, device=DEV)
    q = torch.randn(sum(q_lens), nq, hd, dtype=torch.bfloat16, device=DEV)
    scale = 1.0 / math.sqrt(hd)
    expected = ref.paged_attention_prefill(
        q.float(), kc.float(), vc.float(), bt, qsl, seq_lens, scale,
        window=window, softcap=softcap,
    )
    out = ops.paged_attention_prefill(q, kc, vc, bt, qsl, seq_lens, scale,
                                      window=window, softcap=softcap)
    assert_close_bf16(out, expected, atol=3e-2, rtol=3e-2)


@pytest.mark.gpu
def test_gelu_and_mul_gpu():
    x = torch.randn(33, 512, dtype=torch.bfloat16, device=DEV)
    out = ops.gelu_and_mul(x)
    want = ref.gelu_and_mul(x.float())
    assert_close_bf16(out, want, atol=2e-2, rtol=2e-2)


@pytest.mark.gpu
def test_gemma2_engine_e2e_gpu():
    from kubeai_amd.engine import EngineConfig, LLMEngine, SamplingParams

    eng = LLMEngine(EngineConfig(model="gemma2-tiny", device="cuda",
                                 num_gpu_blocks=256, max_model_len=512))
    prompt = [2] + list(range(100, 200))
    eng.add_request(prompt, SamplingParams(max_tokens=6, ignore_eos=True),
                    request_id="gg")
    out = None
    for _ in range(100):
        if not eng.has_work():
            break
        for o in eng.step():
            if o.finished:
                out = o
    assert out is not None and len(out.output_token_ids) == 6


@pytest.mark.gpu
def test_paged_attention_hd256_fp8_cache():
    """gemma2 shapes with the fp8 e5m2 KV cache (decode + prefill)."""
    nq, nkv, hd, bs = 8, 4, 256, 16
    L = 200
    nb = (L + bs - 1) // bs + 1
    kc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=DEV)
    kc8 = kc.to(torch.float8_e5m2)
    vc8 = vc.to(torch.float8_e5m2)
    bt = torch.arange(1, nb, dtype=torch.int32, device=DEV).reshape(1, -1)
    sl = torch.tensor([L], dtype=torch.int32, device=DEV)
    scale = 1.0 / math.sqrt(hd)
    qd = torch.randn(1, nq, hd, dtype=torch.bfloat16, device=DEV)
    want = ref.paged_attention_decode(
        qd.float(), kc8.float(), vc8.float(), bt, sl, scale)
    out = ops.paged_attention_decode(qd, kc8, vc8, bt, sl, scale)
    assert_close_bf16(out, want, atol=3e-2, rtol=3e-2)
    Tq = 64
    qp = torch.randn(Tq, nq, hd, dtype=torch.bfloat16, device=DEV)
    qsl = torch.tensor([0, Tq], dtype=torch.int32, device=DEV)
    wantp = ref.paged_attention_prefill(
        qp.float(), kc8.float(), vc8.float(), bt, qsl, sl, scale)
    outp = ops.paged_attention_prefill(qp, kc8, vc8, bt, qsl, sl, scale)
    assert_close_bf16(outp, wantp, atol=3e-2, rtol=3e-2)
