"""Sliding-window attention (Mistral family; half of Gemma2).

Reference parity: the reference serves Mistral via vLLM engine images;
here the in-house kernels implement the window natively
(csrc/attention_decode.hip w0 clamp + mask, csrc/attention_prefill.hip
kt0 skip + mask; v2 ladder routes windowed models to v1).
"""
import math

import pytest
import torch

from kubeai_amd.engine import EngineConfig, LLMEngine, SamplingParams
from kubeai_amd.models.config import PRESETS
from kubeai_amd.ops import ref


def _naive(q, k, v, scale, window, causal_pos=None):
    # q [h, d], k/v [L, h, d]; window over the last positions
    L = k.shape[0]
    s = torch.einsum("hd,lhd->hl", q.float(), k.float()) * scale
    if window > 0 and L > window:
        s[:, : L - window] = float("-inf")
    p = torch.softmax(s, dim=-1)
    return torch.einsum("hl,lhd->hd", p, v.float())


def test_ref_decode_window_matches_naive():
    torch.manual_seed(0)
    B, nq, nkv, hd, bs = 2, 4, 2, 128, 16
    lens = [100, 37]
    maxb = (max(lens) + bs - 1) // bs
    nb = B * maxb + 1
    kc = torch.randn(nb, nkv, bs, hd)
    vc = torch.randn(nb, nkv, bs, hd)
    bt = torch.arange(1, nb, dtype=torch.int32).reshape(B, maxb)
    sl = torch.tensor(lens, dtype=torch.int32)
    q = torch.randn(B, nq, hd)
    scale = 1.0 / math.sqrt(hd)
    W = 48
    out = ref.paged_attention_decode(q, kc, vc, bt, sl, scale, window=W)
    for b in range(B):
        L = lens[b]
        ks, vs = [], []
        for t in range(L):
            blk = int(bt[b, t // bs])
            ks.append(kc[blk, :, t % bs].repeat_interleave(nq // nkv, 0))
            vs.append(vc[blk, :, t % bs].repeat_interleave(nq // nkv, 0))
        k = torch.stack(ks)
        v = torch.stack(vs)
        want = _naive(q[b], k, v, scale, W)
        torch.testing.assert_close(out[b].float(), want, atol=1e-4, rtol=1e-4)


def test_ref_prefill_window_restricts_context():
    torch.manual_seed(1)
    nq, nkv, hd, bs = 4, 2, 128, 16
    L = 120
    W = 32
    nb = (L + bs - 1) // bs + 1
    kc = torch.randn(nb, nkv, bs, hd)
    vc = torch.randn(nb, nkv, bs, hd)
    bt = torch.arange(1, nb, dtype=torch.int32).reshape(1, -1)
    qsl = torch.tensor([0, L], dtype=torch.int32)
    sl = torch.tensor([L], dtype=torch.int32)
    q = torch.randn(L, nq, hd)
    scale = 1.0 / math.sqrt(hd)
    full = ref.paged_attention_prefill(q, kc, vc, bt, qsl, sl, scale)
    win = ref.paged_attention_prefill(q, kc, vc, bt, qsl, sl, scale, window=W)
    # early rows (pos < W) see identical context; late rows differ
    torch.testing.assert_close(win[: W - 1], full[: W - 1])
    assert not torch.allclose(win[-1], full[-1])
    # the last row must equal attention over only its window
    ks, vs = [], []
    for t in range(L - W, L):
        blk = int(bt[0, t // bs])
        ks.append(kc[blk, :, t % bs].repeat_interleave(nq // nkv, 0))
        vs.append(vc[blk, :, t % bs].repeat_interleave(nq // nkv, 0))
    want = _naive(q[-1], torch.stack(ks), torch.stack(vs), scale, 0)
    torch.testing.assert_close(win[-1].float(), want, atol=1e-4, rtol=1e-4)


def test_mistral_engine_e2e_cpu():
    eng = LLMEngine(EngineConfig(model="mistral-tiny", device="cpu",
                                 num_gpu_blocks=128, enable_graphs=False,
                                 max_model_len=512))
    assert eng.arch.sliding_window == 64
    prompt = [1] + list(range(100, 220))  # 121 tokens > window 64
    eng.add_request(prompt, SamplingParams(max_tokens=5, ignore_eos=True),
                    request_id="w1")
    out = None
    for _ in range(100):
        if not eng.has_work():
            break
        for o in eng.step():
            if o.finished:
                out = o
    assert out is not None and len(out.output_token_ids) == 5


def test_mistral_preset_shapes():
    m = PRESETS["mistral-7b"]
    assert m.sliding_window == 4096 and m.num_key_value_heads == 8
    assert m.vocab_size == 32000 and m.rope_theta == 10000.0
