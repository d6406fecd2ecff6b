"""Reference (pure PyTorch) implementations of the engine ops.

These are the *semantic contract* for the hand-written gfx950 HIP kernels in
``kubeai_amd/csrc``: every HIP kernel has a numerics test comparing it against
the fp32 version of the op here (tests/test_kernels_gpu.py).

They are also the CPU execution path, so the whole engine (scheduler, paged KV
cache, prefix reuse, sampling) is testable without a GPU.

Shapes / layout conventions (MI355X-first):
  - token-packed activations:   [T, H] (no batch dim; ragged batches are
    described by query_start_loc, vLLM-v1 style)
  - KV cache:                   k/v_cache [num_blocks, n_kv_heads, block_size,
    head_dim] — one (block, kv-head) tile is block_size*head_dim contiguous
    elements = 4 KiB at bf16/128hd/16bs, a coalesced unit for decode loads.
  - block_tables:               [B, max_blocks_per_seq] int32
  - slot_mapping:               [T] int64, slot = block_id * block_size + off
"""
from __future__ import annotations

import math

import torch


# ---------------------------------------------------------------------------
# normalization
# ---------------------------------------------------------------------------

def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    """y = x / rms(x) * weight, computed in fp32, cast back to x.dtype."""
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps) * weight.float()
    return y.to(x.dtype)


def fused_add_rmsnorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
) -> tuple[torch.Tensor, torch.Tensor]:
    """residual' = x + residual ; y = rmsnorm(residual').

    Returns (y, residual'). The HIP kernel does this in one pass in-place,
    saving one full HBM round-trip of the hidden states per layer.
    """
    res = (x.float() + residual.float())
    y = rmsnorm(res, weight, eps)
    return y.to(x.dtype), res.to(x.dtype)


# ---------------------------------------------------------------------------
# rotary embedding
# ---------------------------------------------------------------------------

def make_cos_sin_cache(
    head_dim: int,
    max_positions: int,
    base: float = 10000.0,
    scaling: float = 1.0,
    dtype: torch.dtype = torch.float32,
    rope_scaling: dict | None = None,
) -> torch.Tensor:
    """[max_positions, head_dim] — first half cos, second half sin (NeoX).

    rope_scaling follows the HF config.json `rope_scaling` dict:
      {"rope_type": "llama3", "factor", "low_freq_factor",
       "high_freq_factor", "original_max_position_embeddings"}  (Llama 3.1)
      {"rope_type": "linear", "factor"}                         (PI)
    """
    inv_freq = 1.0 / (
        base ** (torch.arange(0, head_dim, 2, dtype=torch.float64) / head_dim)
    )
    if rope_scaling:
        rtype = rope_scaling.get("rope_type") or rope_scaling.get("type")
        factor = float(rope_scaling.get("factor", 1.0))
        if rtype == "llama3":
            lo_f = float(rope_scaling.get("low_freq_factor", 1.0))
            hi_f = float(rope_scaling.get("high_freq_factor", 4.0))
            orig = float(
                rope_scaling.get("original_max_position_embeddings", 8192)
            )
            wavelen = 2 * math.pi / inv_freq
            lo_wl = orig / lo_f  # long wavelengths: fully rescale
            hi_wl = orig / hi_f  # short wavelengths: keep
            smooth = ((orig / wavelen - lo_f) / (hi_f - lo_f)).clamp(0.0, 1.0)
            scaled = (1 - smooth) * inv_freq / factor + smooth * inv_freq
            inv_freq = torch.where(wavelen > lo_wl, inv_freq / factor, inv_freq)
            mid = (wavelen <= lo_wl) & (wavelen >= hi_wl)
            inv_freq = torch.where(mid, scaled, inv_freq)
        elif rtype == "linear":
            scaling = scaling * factor
        elif rtype in (None, "default"):
            pass
        else:
            # longrope/yarn/dynamic would silently produce wrong
            # positions — refuse loudly instead
            raise ValueError(f"unsupported rope_scaling type {rtype!r}")
    t = torch.arange(max_positions, dtype=torch.float64) / scaling
    freqs = torch.outer(t, inv_freq)
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).to(dtype)


def rope(
    q: torch.Tensor,  # [T, n_heads, head_dim]
    k: torch.Tensor,  # [T, n_kv_heads, head_dim]
    positions: torch.Tensor,  # [T] int
    cos_sin: torch.Tensor,  # [max_pos, head_dim]
) -> tuple[torch.Tensor, torch.Tensor]:
    """NeoX-style rotary embedding (rotate halves), out-of-place reference."""

    def _apply(x: torch.Tensor) -> torch.Tensor:
        hd = x.shape[-1]
        cs = cos_sin[positions.long()].to(torch.float32)  # [T, hd]
        cos = cs[:, : hd // 2].unsqueeze(1)  # [T, 1, hd/2]
        sin = cs[:, hd // 2 :].unsqueeze(1)
        xf = x.float()
        x1, x2 = xf[..., : hd // 2], xf[..., hd // 2 :]
        o1 = x1 * cos - x2 * sin
        o2 = x2 * cos + x1 * sin
        return torch.cat([o1, o2], dim=-1).to(x.dtype)

    return _apply(q), _apply(k)


# ---------------------------------------------------------------------------
# paged KV cache
# ---------------------------------------------------------------------------

def reshape_and_cache(
    k: torch.Tensor,  # [T, n_kv_heads, head_dim]
    v: torch.Tensor,  # [T, n_kv_heads, head_dim]
    k_cache: torch.Tensor,  # [num_blocks, n_kv_heads, block_size, head_dim]
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,  # [T] int64; -1 = skip (dropped token)
) -> None:
    block_size = k_cache.shape[2]
    valid = slot_mapping >= 0
    slots = slot_mapping[valid]
    blk = torch.div(slots, block_size, rounding_mode="floor").long()
    off = (slots % block_size).long()
    k_cache[blk, :, off] = k[valid].to(k_cache.dtype)
    v_cache[blk, :, off] = v[valid].to(v_cache.dtype)


def _gather_kv(
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_table: torch.Tensor,  # [max_blocks] int32
    seq_len: int,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Gather a sequence's KV from the paged cache → [seq_len, n_kv, hd]."""
    block_size = k_cache.shape[2]
    n_blocks = (seq_len + block_size - 1) // block_size
    blocks = block_table[:n_blocks].long()
    # [n_blocks, n_kv, bs, hd] -> [n_blocks, bs, n_kv, hd] -> [n_blocks*bs, ...]
    k = k_cache[blocks].transpose(1, 2).reshape(-1, k_cache.shape[1], k_cache.shape[3])
    v = v_cache[blocks].transpose(1, 2).reshape(-1, v_cache.shape[1], v_cache.shape[3])
    return k[:seq_len], v[:seq_len]


def paged_attention_decode(
    q: torch.Tensor,  # [B, n_heads, head_dim] (one new token per seq)
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [B, max_blocks]
    seq_lens: torch.Tensor,  # [B] int32 (total length incl. the new token)
    scale: float,
    window: int = 0,  # >0: sliding window, keys in (L-1-window, L-1]
    softcap: float = 0.0,  # >0: gemma2 logit cap s = c*tanh(s/c)
) -> torch.Tensor:
    B, n_heads, head_dim = q.shape
    n_kv = k_cache.shape[1]
    g = n_heads // n_kv
    out = torch.empty_like(q, dtype=torch.float32)
    for b in range(B):
        L = int(seq_lens[b])
        k, v = _gather_kv(k_cache, v_cache, block_tables[b], L)  # [L, n_kv, hd]
        kf = k.float().repeat_interleave(g, dim=1)  # [L, n_heads, hd]
        vf = v.float().repeat_interleave(g, dim=1)
        s = torch.einsum("hd,lhd->hl", q[b].float(), kf) * scale
        if softcap > 0:
            s = softcap * torch.tanh(s / softcap)
        if window > 0 and L > window:
            s[:, : L - window] = float("-inf")
        p = torch.softmax(s, dim=-1)
        out[b] = torch.einsum("hl,lhd->hd", p, vf)
    return out.to(q.dtype)


def paged_attention_prefill(
    q: torch.Tensor,  # [Tq, n_heads, head_dim] packed query tokens
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [B, max_blocks]
    query_start_loc: torch.Tensor,  # [B+1] int32 (prefix sums of q lens)
    seq_lens: torch.Tensor,  # [B] total kv length incl. this chunk
    scale: float,
    window: int = 0,  # >0: query p attends keys in (p-window, p]
    softcap: float = 0.0,  # >0: gemma2 logit cap s = c*tanh(s/c)
) -> torch.Tensor:
    """Causal attention of new (chunk) tokens against the full paged KV.

    Query token i of seq b sits at absolute position seq_len - q_len + i and
    attends to kv positions [0, abs_pos].  Covers fresh prefill, prefix-cache
    hits and chunked prefill uniformly (KV must already be in the cache).
    """
    out = torch.empty_like(q, dtype=torch.float32)
    B = seq_lens.shape[0]
    for b in range(B):
        s0, s1 = int(query_start_loc[b]), int(query_start_loc[b + 1])
        q_len = s1 - s0
        L = int(seq_lens[b])
        k, v = _gather_kv(k_cache, v_cache, block_tables[b], L)
        n_kv = k.shape[1]
        g = q.shape[1] // n_kv
        kf = k.float().repeat_interleave(g, dim=1)
        vf = v.float().repeat_interleave(g, dim=1)
        qf = q[s0:s1].float()  # [q_len, n_heads, hd]
        s = torch.einsum("qhd,lhd->hql", qf, kf) * scale
        if softcap > 0:
            s = softcap * torch.tanh(s / softcap)
        # causal mask: query i (abs pos L - q_len + i) sees kv j <= abs pos
        qpos = torch.arange(L - q_len, L, device=q.device).unsqueeze(1)
        kpos = torch.arange(L, device=q.device).unsqueeze(0)
        s.masked_fill_((kpos > qpos).unsqueeze(0), float("-inf"))
        if window > 0:
            s.masked_fill_((kpos <= qpos - window).unsqueeze(0), float("-inf"))
        p = torch.softmax(s, dim=-1)
        out[s0:s1] = torch.einsum("hql,lhd->qhd", p, vf)
    return out.to(q.dtype)


# ---------------------------------------------------------------------------
# activations
# ---------------------------------------------------------------------------

def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    """x = [T, 2*I] (gate | up) -> silu(gate) * up, fp32 internally."""
    gate, up = x.float().chunk(2, dim=-1)
    return (torch.nn.functional.silu(gate) * up).to(x.dtype)


def gelu_and_mul(x: torch.Tensor) -> torch.Tensor:
    """x = [T, 2*I] (gate | up) -> gelu_tanh(gate) * up (gemma GeGLU)."""
    gate, up = x.float().chunk(2, dim=-1)
    act = torch.nn.functional.gelu(gate, approximate="tanh")
    return (act * up).to(x.dtype)


# ---------------------------------------------------------------------------
# sampling
# ---------------------------------------------------------------------------

def greedy_sample(logits: torch.Tensor) -> torch.Tensor:
    return logits.argmax(dim=-1).to(torch.int64)


def _lshr(z: torch.Tensor, n: int) -> torch.Tensor:
    """Logical right shift on int64 treated as uint64."""
    return (z >> n) & ((1 << (64 - n)) - 1)


_U64 = (1 << 64) - 1


def _splitmix64(z: torch.Tensor) -> torch.Tensor:
    """splitmix64 over int64 tensors (two's-complement wraparound ≡ u64)."""
    z = z + (-0x61C8864680B583EB)  # 0x9e3779b97f4a7c15 as i64
    z = (z ^ _lshr(z, 30)) * (-0x40A7B892E31B1A47)  # 0xbf58476d1ce4e5b9
    z = (z ^ _lshr(z, 27)) * (-0x6B2FB644ECCEEE15)  # 0x94d049bb133111eb
    return z ^ _lshr(z, 31)


def hash_uniform(key: int, idx: torch.Tensor) -> torch.Tensor:
    """Counter-based uniform in (0,1], spec-identical to csrc/common.h."""
    mult = -0x2ECBDABC217D106B  # 0xd1342543de82ef95 as i64
    k = key & _U64
    if k >= 1 << 63:
        k -= 1 << 64
    key_t = torch.tensor(k, dtype=torch.int64)
    h = _splitmix64(key_t ^ (idx.to(torch.int64) * mult))
    m = _lshr(h, 40).to(torch.float32)
    return (m + 1.0) * (1.0 / 16777216.0)


def gumbel_sample(
    logits: torch.Tensor,  # [B, V] fp32/bf16
    temperature: torch.Tensor,  # [B] fp32; 0 => greedy
    seeds: torch.Tensor,  # [B] int64 per-request seed
    step: int,
) -> torch.Tensor:
    """Temperature sampling via the Gumbel-argmax trick.

    argmax(logits/T + G) with G ~ Gumbel(0,1) samples exactly from
    softmax(logits/T) without materialising the softmax or sorting — a
    wave-reduction-shaped op that maps to one kernel on the GPU.  The RNG is
    the counter-based splitmix64 hash (csrc/common.h), so the HIP kernel and
    this reference draw the same noise for a given (seed, step).
    """
    lf = logits.float()
    B, V = lf.shape
    out = torch.empty(B, dtype=torch.int64, device=logits.device)
    idx = torch.arange(V)
    for b in range(B):
        t = float(temperature[b])
        if t <= 0.0:
            out[b] = int(lf[b].argmax())
            continue
        key = (int(seeds[b]) * 1000003 + step) & _U64
        u = hash_uniform(key, idx).to(lf.device)
        g = -torch.log(-torch.log(u))
        out[b] = int((lf[b] / t + g).argmax())
    return out


def topk_topp_sample(
    logits: torch.Tensor,  # [B, V]
    temperature: torch.Tensor,  # [B]
    top_p: torch.Tensor,  # [B] in (0, 1]
    top_k: torch.Tensor,  # [B] int; 0 = disabled
    seeds: torch.Tensor,
    step: int,
) -> torch.Tensor:
    """Sort-based top-k/top-p reference."""
    lf = logits.float()
    B, V = lf.shape
    out = torch.empty(B, dtype=torch.int64, device=logits.device)
    for b in range(B):
        t = float(temperature[b])
        if t <= 0.0:
            out[b] = int(lf[b].argmax())
            continue
        row = lf[b] / t
        probs = torch.softmax(row, dim=-1)
        sp, si = probs.sort(descending=True)
        k = int(top_k[b])
        if k > 0:
            sp, si = sp[:k], si[:k]
        csum = sp.cumsum(0)
        keep = (csum - sp) < float(top_p[b])
        sp, si = sp[keep], si[keep]
        gen = torch.Generator(device="cpu")
        gen.manual_seed(int(seeds[b]) * 1000003 + step)
        r = float(torch.rand((), generator=gen)) * float(sp.sum())
        idx = int(torch.searchsorted(sp.cumsum(0), torch.tensor(r)).clamp(max=sp.shape[0] - 1))
        out[b] = int(si[idx])
    return out


def quant_fp8(x: torch.Tensor):
    """Per-row dynamic OCP-e4m3 quantization (CPU reference of
    csrc/quant_fp8.hip::quant_fp8): scale = amax/448, symmetric."""
    xf = x.float()
    amax = xf.abs().amax(dim=-1).clamp_min(1e-6)
    scale = amax / 448.0
    q = (xf / scale[:, None]).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
    return q, scale
