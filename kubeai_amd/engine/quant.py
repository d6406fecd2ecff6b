"""FP8 (OCP e4m3fn) weight + dynamic per-token activation quantization.

MI355X-first: gfx950's fp8 MFMA peak is ~2x bf16 and — what actually
matters for skinny decode GEMMs — fp8 weights halve the HBM bytes per
step. The BASELINE headline config is Llama-3.1-8B-Instruct-FP8, so this
is the config-faithful serving mode. GEMMs go through torch._scaled_mm
(hipBLASLt fp8 path); gfx950 uses OCP e4m3fn, not the MI300X fnuz variant.

Scheme: W8A8 dynamic — per-output-channel weight scales, per-token
activation scales (amax/448), bf16 output.
"""
from __future__ import annotations

import torch
import torch.nn as nn

FP8_DTYPE = torch.float8_e4m3fn
FP8_MAX = 448.0


class Fp8Linear(nn.Module):
    """Drop-in replacement for bias-free nn.Linear, fp8 weights."""

    def __init__(self, weight_bf16: torch.Tensor, bias: torch.Tensor | None = None):
        super().__init__()
        if bias is not None:
            self.register_buffer("bias", bias.detach().to(torch.bfloat16))
        else:
            self.bias = None
        w = weight_bf16.detach().float()
        w_amax = w.abs().amax(dim=1, keepdim=True).clamp_min(1e-6)  # [out,1]
        w_scale = w_amax / FP8_MAX
        wq = (w / w_scale).clamp(-FP8_MAX, FP8_MAX).to(FP8_DTYPE)
        # scaled_mm wants mat2 column-major: keep [out, in] row-major and
        # pass .t() (a column-major view) at call time
        self.register_buffer("weight_fp8", wq.contiguous())
        self.register_buffer("weight_scale", w_scale.reshape(1, -1).contiguous())
        self.out_features, self.in_features = weight_bf16.shape

    @property
    def weight(self) -> torch.Tensor:
        """bf16 view for code paths that read .weight (LoRA base, tests)."""
        return (
            self.weight_fp8.float() * self.weight_scale.reshape(-1, 1)
        ).to(torch.bfloat16)

    def forward_quantized(self, xq: torch.Tensor, x_scale: torch.Tensor) -> torch.Tensor:
        """Pre-quantized input from a fused producer kernel (quant is free)."""
        if not xq.is_cuda:
            # CPU semantics fallback (tests): dequantize and matmul
            x = xq.float() * x_scale.view(-1, 1)
            w = self.weight_fp8.float() * self.weight_scale.reshape(-1, 1)
            y = (x @ w.t()).to(torch.bfloat16)
            return y + self.bias if self.bias is not None else y
        return torch._scaled_mm(
            xq,
            self.weight_fp8.t(),
            scale_a=x_scale.view(-1, 1),
            scale_b=self.weight_scale,
            bias=self.bias,
            out_dtype=torch.bfloat16,
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        a_amax = x.float().abs().amax(dim=-1, keepdim=True).clamp_min(1e-6)
        a_scale = a_amax / FP8_MAX
        xq = (x.float() / a_scale).clamp(-FP8_MAX, FP8_MAX).to(FP8_DTYPE)
        if not x.is_cuda:
            return self.forward_quantized(xq, a_scale.squeeze(-1))
        return torch._scaled_mm(
            xq,
            self.weight_fp8.t(),
            scale_a=a_scale,
            scale_b=self.weight_scale,
            bias=self.bias,
            out_dtype=torch.bfloat16,
        )


_TARGETS = ("qkv_proj", "o_proj", "gate_up_proj", "down_proj", "lm_head")


def convert_to_fp8(model: nn.Module) -> int:
    """Swap target nn.Linear modules for Fp8Linear; returns count.

    Dense decoder layers additionally get the FUSED activation-quant path
    (llama.DecoderLayer checks _fp8_fused): producer kernels emit fp8 +
    per-row scales, so per-linear dynamic quantization disappears.
    """
    n = 0
    for mod in model.modules():
        for name in _TARGETS:
            child = getattr(mod, name, None)
            if isinstance(child, nn.Linear):
                setattr(mod, name, Fp8Linear(child.weight, child.bias))
                n += 1
    for layer in getattr(model, "layers", []):
        attn = getattr(layer, "self_attn", None)
        mlp = getattr(layer, "mlp", None)
        if attn is None or not isinstance(getattr(attn, "qkv_proj", None), Fp8Linear):
            continue
        if isinstance(getattr(mlp, "gate_up_proj", None), Fp8Linear):
            layer._fp8_fused = True  # dense: norms/activations emit fp8
        elif hasattr(mlp, "experts"):
            # MoE: attention runs the fused-fp8 path; the MLP input stays
            # bf16 (router needs it) and MoEMLP quantizes once itself
            layer._fp8_fused = True
    return n
