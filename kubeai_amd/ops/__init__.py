"""Op dispatch: hand-written gfx950 HIP kernels on GPU, torch reference on CPU.

Policy (deliberate, see repo instructions): on a ROCm GPU the HIP extension is
MANDATORY — if ``kubeai_amd._C`` failed to import, GPU-tensor calls raise
instead of silently falling back to eager PyTorch.  CPU tensors always use the
reference implementations (ref.py) so the engine logic is testable anywhere.
"""
from __future__ import annotations

import torch

from . import ref

_C = None
_C_IMPORT_ERROR: Exception | None = None
try:  # built in-tree by `python setup.py build_ext --inplace` / __graft_entry__.build()
    from kubeai_amd import _C  # type: ignore[attr-defined,no-redef]
except Exception as e:  # pragma: no cover - exercised only when ext missing
    _C_IMPORT_ERROR = e


def have_hip_ext() -> bool:
    return _C is not None


def _require_ext() -> None:
    if _C is None:
        raise RuntimeError(
            "kubeai_amd HIP extension (kubeai_amd._C) is not built but a GPU "
            "tensor reached the ops layer. Build it with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Original import error: {_C_IMPORT_ERROR!r}"
        )


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if x.is_cuda:
        _require_ext()
        out = torch.empty_like(x)
        _C.rmsnorm(out, x, weight, eps)
        return out
    return ref.rmsnorm(x, weight, eps)


def fused_add_rmsnorm(x, residual, weight, eps: float):
    """In-place on GPU: x <- rmsnorm(x + residual), residual <- x + residual."""
    if x.is_cuda:
        _require_ext()
        _C.fused_add_rmsnorm(x, residual, weight, eps)
        return x, residual
    return ref.fused_add_rmsnorm(x, residual, weight, eps)


def rope(q, k, positions, cos_sin):
    """In-place on GPU; returns (q, k)."""
    if q.is_cuda:
        _require_ext()
        _C.rope(q, k, positions, cos_sin)
        return q, k
    return ref.rope(q, k, positions, cos_sin)


def reshape_and_cache(k, v, k_cache, v_cache, slot_mapping) -> None:
    if k.is_cuda:
        _require_ext()
        if k_cache.dtype == torch.float8_e5m2:
            _C.reshape_and_cache_fp8(k, v, k_cache, v_cache, slot_mapping)
        else:
            _C.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)
        return
    ref.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


def paged_attention_decode(
    q, k_cache, v_cache, block_tables, seq_lens, scale: float, out=None,
    window: int = 0, softcap: float = 0.0,
):
    """q may be a row-strided view (fused qkv output); out must be
    contiguous (allocated here if not supplied). window > 0 enables
    sliding-window attention (Mistral/Gemma2 style: keys in
    (pos-window, pos])."""
    if q.is_cuda:
        _require_ext()
        if out is None:
            out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        _C.paged_attention_decode(
            out, q, k_cache, v_cache, block_tables, seq_lens, scale, window,
            softcap
        )
        return out
    res = ref.paged_attention_decode(
        q, k_cache, v_cache, block_tables, seq_lens, scale, window=window,
        softcap=softcap,
    )
    if out is not None:
        out.copy_(res)
        return out
    return res


def paged_attention_prefill(
    q, k_cache, v_cache, block_tables, query_start_loc, seq_lens, scale: float,
    out=None, window: int = 0, softcap: float = 0.0,
):
    if q.is_cuda:
        _require_ext()
        if out is None:
            out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        _C.paged_attention_prefill(
            out, q, k_cache, v_cache, block_tables, query_start_loc, seq_lens,
            scale, window, softcap
        )
        return out
    res = ref.paged_attention_prefill(
        q, k_cache, v_cache, block_tables, query_start_loc, seq_lens, scale,
        window=window, softcap=softcap,
    )
    if out is not None:
        out.copy_(res)
        return out
    return res


def silu_and_mul(x):
    if x.is_cuda:
        _require_ext()
        T, two_i = x.shape
        out = torch.empty((T, two_i // 2), dtype=x.dtype, device=x.device)
        _C.silu_and_mul(out, x)
        return out
    return ref.silu_and_mul(x)


def gelu_and_mul(x):
    """GeGLU (gemma family): gelu_tanh(gate) * up over packed [T, 2I]."""
    if x.is_cuda:
        _require_ext()
        T, two_i = x.shape
        out = torch.empty((T, two_i // 2), dtype=x.dtype, device=x.device)
        _C.gelu_and_mul(out, x)
        return out
    return ref.gelu_and_mul(x)


import os

_USE_SKINNY = os.environ.get("KUBEAI_SKINNY_GEMM", "0") == "1"


def linear(x, weight):
    """GEMM dispatch. The hand-written weight-streaming kernel (skinny_gemm)
    currently loses to hipBLASLt on MI355X (direct 16 B row gathers do not
    coalesce; measured 0.3-0.65x — profiles/r01_results.md), so it is
    opt-in via KUBEAI_SKINNY_GEMM=1 until the LDS-staged variant lands."""
    if (
        _USE_SKINNY
        and x.is_cuda
        and x.dtype == torch.bfloat16
        and x.dim() == 2
        and 1 <= x.shape[0] <= 64
        and weight.shape[1] % 64 == 0
        and weight.shape[0] % 64 == 0
        and x.is_contiguous()
        and weight.is_contiguous()
    ):
        _require_ext()
        y = torch.empty(
            (x.shape[0], weight.shape[0]), dtype=x.dtype, device=x.device
        )
        _C.skinny_gemm(y, x, weight)
        return y
    return torch.nn.functional.linear(x, weight)


def rmsnorm_fp8(x, weight, eps: float):
    """GPU-only: rmsnorm with fused per-row fp8 quantization."""
    _require_ext()
    out = torch.empty(x.shape, dtype=torch.float8_e4m3fn, device=x.device)
    scale = torch.empty(x.shape[0], dtype=torch.float32, device=x.device)
    _C.rmsnorm_fp8(out, scale, x, weight, eps)
    return out, scale


def fused_add_rmsnorm_fp8(x, residual, weight, eps: float):
    """GPU-only, in-place residual update; returns (fp8 out, row scales)."""
    _require_ext()
    out = torch.empty(x.shape, dtype=torch.float8_e4m3fn, device=x.device)
    scale = torch.empty(x.shape[0], dtype=torch.float32, device=x.device)
    _C.fused_add_rmsnorm_fp8(out, scale, x, residual, weight, eps)
    return out, scale


def silu_and_mul_fp8(x):
    if not x.is_cuda:
        return ref.quant_fp8(ref.silu_and_mul(x))
    _require_ext()
    T, two_i = x.shape
    out = torch.empty((T, two_i // 2), dtype=torch.float8_e4m3fn, device=x.device)
    scale = torch.empty(T, dtype=torch.float32, device=x.device)
    _C.silu_and_mul_fp8(out, scale, x)
    return out, scale


def quant_fp8(x):
    if not x.is_cuda:
        return ref.quant_fp8(x)
    _require_ext()
    out = torch.empty(x.shape, dtype=torch.float8_e4m3fn, device=x.device)
    scale = torch.empty(x.shape[0], dtype=torch.float32, device=x.device)
    _C.quant_fp8(out, scale, x)
    return out, scale


def greedy_sample(logits):
    if logits.is_cuda:
        _require_ext()
        out = torch.empty(logits.shape[0], dtype=torch.int64, device=logits.device)
        _C.greedy_sample(out, logits)
        return out
    return ref.greedy_sample(logits)


def gumbel_sample(logits, temperature, seeds, step: int):
    if logits.is_cuda:
        _require_ext()
        out = torch.empty(logits.shape[0], dtype=torch.int64, device=logits.device)
        _C.gumbel_sample(out, logits, temperature, seeds, step)
        return out
    return ref.gumbel_sample(logits, temperature, seeds, step)


def nucleus_stats(logits, temps):
    """Per-row (max, sum exp((x-max)/t)) for rejection nucleus sampling."""
    if logits.is_cuda:
        _require_ext()
        S = logits.shape[0]
        m = torch.empty(S, dtype=torch.float32, device=logits.device)
        z = torch.empty_like(m)
        _C.nucleus_stats(m, z, logits, temps)
        return m, z
    t = temps.clamp_min(1e-6).unsqueeze(1)
    m = logits.max(dim=-1).values
    z = torch.exp((logits - m.unsqueeze(1)) / t).sum(-1)
    return m, z


def nucleus_accept(logits, cand, m, z, temps, top_ps, top_ks):
    """uint8 mask: sampled token lies in the top-p/top-k nucleus
    (probability mass strictly above it < top_p, rank < top_k);
    greedy rows always accept."""
    if logits.is_cuda:
        _require_ext()
        ok = torch.empty(logits.shape[0], dtype=torch.uint8,
                         device=logits.device)
        _C.nucleus_accept(ok, logits, cand, m, z, temps, top_ps, top_ks)
        return ok
    t = temps.clamp_min(1e-6).unsqueeze(1)
    lt = logits.gather(1, cand.view(-1, 1))
    above = logits > lt
    mass = (torch.exp((logits - m.unsqueeze(1)) / t) * above).sum(-1) / z
    cnt = above.sum(-1)
    ok = (mass < top_ps) & ((top_ks <= 0) | (cnt < top_ks)) | (temps <= 0)
    return ok.to(torch.uint8)
