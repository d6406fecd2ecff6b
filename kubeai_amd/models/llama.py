"""Llama-family decoder for the kubeai_amd engine.

MI355X-first execution model (one process per GPU):
  - token-packed forward ([T, H], no padding) over the ForwardBatch;
  - plain GEMMs (qkv/o/gate_up/down/lm_head) go through torch.nn.functional
    .linear, i.e. hipBLASLt/rocBLAS on ROCm — library GEMMs per the design
    rules; every fused hot op (rmsnorm, rope, cache write, paged attention,
    SwiGLU) is a hand-written gfx950 kernel via kubeai_amd.ops;
  - KV cache is paged [num_blocks, n_kv, block_size, head_dim] bf16, owned by
    the runner and passed in.

Reference parity: this is the in-house engine the reference delegates to
vLLM pods for; no reference code exists for it.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from kubeai_amd import ops
from kubeai_amd.engine.batch import ForwardBatch
from .config import ModelArchConfig


class EngineLinear(nn.Linear):
    """nn.Linear routed through ops.linear: the hand-written weight-
    streaming GEMM takes skinny decode batches (M<=64); hipBLASLt keeps
    the rest. Bias (Qwen2 qkv) is added outside the skinny path."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = ops.linear(x, self.weight)
        if self.bias is not None:
            y = y + self.bias
        return y


def _rms_head(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    """Per-head RMSNorm over the last dim (gemma3 qk-norm).

    On GPU this routes through the rmsnorm HIP kernel on a [T*heads, hd]
    view (one copy + one kernel instead of ~6 eager elementwise ops —
    these run inside the decode hipGraph every layer)."""
    if x.is_cuda and x.dtype == torch.bfloat16:
        flat = x.contiguous().view(-1, x.shape[-1])
        return ops.rmsnorm(flat, w, eps).view(x.shape)
    xf = x.float()
    xf = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * w.float()).to(x.dtype)


class Attention(nn.Module):
    def __init__(self, cfg: ModelArchConfig, layer_idx: int):
        super().__init__()
        self.layer_idx = layer_idx
        self.n_q = cfg.num_attention_heads
        self.n_kv = cfg.num_key_value_heads
        self.hd = cfg.head_dim
        # per-layer window: gemma2 interleaves windowed ("even" layers)
        # and full-causal attention; gemma3 makes every Nth layer GLOBAL
        # (others windowed, local rope base); mistral windows every layer
        if cfg.global_layer_interval > 0:
            self.is_global = (layer_idx + 1) % cfg.global_layer_interval == 0
            use_win = not self.is_global
        else:
            self.is_global = True
            use_win = (
                cfg.sliding_window_pattern != "even" or layer_idx % 2 == 0
            )
        self.window = (cfg.sliding_window or 0) if use_win else 0
        self.qk_norm = cfg.qk_norm
        if cfg.qk_norm:
            self.q_norm = nn.Parameter(torch.ones(cfg.head_dim))
            self.k_norm = nn.Parameter(torch.ones(cfg.head_dim))
        self.norm_eps = cfg.rms_norm_eps
        self.softcap = cfg.attn_logit_softcap or 0.0
        self.scale = (
            cfg.attention_multiplier
            or 1.0 / math.sqrt(cfg.query_pre_attn_scalar or self.hd)
        )
        q_size = self.n_q * self.hd
        kv_size = self.n_kv * self.hd
        self.qkv_proj = EngineLinear(
            cfg.hidden_size, q_size + 2 * kv_size, bias=cfg.attention_bias
        )
        self.o_proj = EngineLinear(q_size, cfg.hidden_size, bias=False)

    def forward(
        self,
        x: torch.Tensor,  # [T, H]
        fb: ForwardBatch,
        kv_cache: tuple[torch.Tensor, torch.Tensor],
        cos_sin: torch.Tensor,
    ) -> torch.Tensor:
        fp8_in = isinstance(x, tuple)  # (fp8 tensor, row scales) from a
        # fused producer kernel (DecoderLayer fp8 path)
        if fp8_in:
            xq, xs = x
            T = xq.shape[0]
            qkv = self.qkv_proj.forward_quantized(xq, xs)
            lm = None
        else:
            T = x.shape[0]
            qkv = self.qkv_proj(x)
            lm = getattr(self, "_lora_manager", None)
            if lm is not None and lm.active and fb.lora_ids is not None:
                lm.apply(self.layer_idx, "qkv", x, qkv, fb.lora_ids)
        q, k, v = qkv.split(
            [self.n_q * self.hd, self.n_kv * self.hd, self.n_kv * self.hd], dim=-1
        )
        # strided views into the qkv buffer — the HIP kernels take a row
        # stride, so no .contiguous() copies (saves 3 HBM round trips/layer)
        q = q.unflatten(-1, (self.n_q, self.hd))
        k = k.unflatten(-1, (self.n_kv, self.hd))
        v = v.unflatten(-1, (self.n_kv, self.hd))
        if self.qk_norm:
            # gemma3: per-head RMSNorm ((1+w) stored as w) before rope;
            # fp32 math, contiguous bf16 out. v follows to contiguous —
            # the cache-write kernel wants k/v with one shared row stride
            q = _rms_head(q, self.q_norm, self.norm_eps)
            k = _rms_head(k, self.k_norm, self.norm_eps)
            v = v.contiguous()
        if isinstance(cos_sin, tuple):
            # (global, local) caches — gemma3 dual rope bases
            cos_sin = cos_sin[0] if self.is_global else cos_sin[1]
        q, k = ops.rope(q, k, fb.positions, cos_sin)
        k_cache, v_cache = kv_cache
        ops.reshape_and_cache(k, v, k_cache, v_cache, fb.slot_mapping)

        out = torch.empty(
            (T, self.n_q, self.hd), dtype=qkv.dtype, device=qkv.device
        )
        nd = fb.n_decode
        if nd > 0:
            ops.paged_attention_decode(
                q[:nd],
                k_cache,
                v_cache,
                fb.decode_block_tables,
                fb.decode_seq_lens,
                self.scale,
                out=out[:nd],
                window=self.window,
                softcap=self.softcap,
            )
        if fb.n_prefill > 0:
            ops.paged_attention_prefill(
                q[nd:],
                k_cache,
                v_cache,
                fb.prefill_block_tables,
                fb.prefill_query_start_loc,
                fb.prefill_seq_lens,
                self.scale,
                out=out[nd:],
                window=self.window,
                softcap=self.softcap,
            )
        attn_flat = out.view(T, -1)
        if fp8_in:
            a8, ascale = ops.quant_fp8(attn_flat)
            return self.o_proj.forward_quantized(a8, ascale)
        result = self.o_proj(attn_flat)
        if lm is not None and lm.active and fb.lora_ids is not None:
            lm.apply(self.layer_idx, "o", attn_flat, result, fb.lora_ids)
        return result


class MLP(nn.Module):
    def __init__(self, cfg: ModelArchConfig, layer_idx: int | None = None):
        super().__init__()
        # layer_idx set only for the DENSE per-layer MLP: LoRA targets it
        # (MoE expert MLPs pass None — adapters stay attention-only there)
        self.layer_idx = layer_idx
        self.gate_up_proj = EngineLinear(
            cfg.hidden_size, 2 * cfg.intermediate_size, bias=False
        )
        self.down_proj = EngineLinear(cfg.intermediate_size, cfg.hidden_size, bias=False)
        self.gelu = cfg.hidden_act == "gelu_pytorch_tanh"

    def forward(self, x, fb: ForwardBatch | None = None) -> torch.Tensor:
        if isinstance(x, tuple):  # fused fp8 path (never active with LoRA)
            h = self.gate_up_proj.forward_quantized(*x)
            a8, ascale = ops.silu_and_mul_fp8(h)
            return self.down_proj.forward_quantized(a8, ascale)
        h = self.gate_up_proj(x)
        lm = getattr(self, "_lora_manager", None)
        lora_live = (
            lm is not None and lm.active and fb is not None
            and fb.lora_ids is not None and self.layer_idx is not None
        )
        if lora_live:
            lm.apply(self.layer_idx, "gate_up", x, h, fb.lora_ids)
        a = ops.gelu_and_mul(h) if self.gelu else ops.silu_and_mul(h)
        y = self.down_proj(a)
        if lora_live:
            lm.apply(self.layer_idx, "down", a, y, fb.lora_ids)
        return y


class MoEMLP(nn.Module):
    """Mixtral-style sparse MoE block (token-level top-k routing).

    Expert GEMMs are grouped per expert (sort tokens by expert) so each
    expert runs one hipBLASLt GEMM over its token group.
    """

    def __init__(self, cfg: ModelArchConfig):
        super().__init__()
        self.n_experts = cfg.num_local_experts
        self.top_k = cfg.num_experts_per_tok
        self.gate = nn.Linear(cfg.hidden_size, self.n_experts, bias=False)
        self.experts = nn.ModuleList([MLP(cfg) for _ in range(self.n_experts)])

    def forward(self, x: torch.Tensor, fb: ForwardBatch | None = None) -> torch.Tensor:
        T, H = x.shape
        router = F.linear(x.float(), self.gate.weight.float())  # [T, E]
        weights, selected = torch.topk(router, self.top_k, dim=-1)
        weights = torch.softmax(weights, dim=-1).to(x.dtype)  # [T, k]
        if T <= 256:
            # decode-sized batches: run EVERY expert densely. At these M
            # the expert GEMMs are weight-bound, and a batch this size
            # activates nearly all experts anyway — so the HBM cost is
            # the same as the sparse gather path, but there are no
            # dynamic shapes (topk/nonzero gathers), which is what lets
            # the decode step be hipGraph-captured. (The sparse path's
            # nonzero() aborts capture -> Mixtral decoded EAGER before
            # this: 47 ms/step of launch overhead.)
            wfull = torch.zeros(T, self.n_experts, dtype=x.dtype,
                                device=x.device)
            wfull.scatter_(1, selected, weights)
            fp8 = hasattr(self.experts[0].gate_up_proj, "forward_quantized")
            if fp8:
                xq, xs = ops.quant_fp8(x)
            out = None
            for e in range(self.n_experts):
                contrib = (
                    self.experts[e]((xq, xs)) if fp8 else self.experts[e](x)
                )
                contrib = contrib * wfull[:, e : e + 1]
                out = contrib if out is None else out + contrib
            return out
        # fp8 experts: quantize the hidden ONCE per layer and hand each
        # expert a row gather of the shared (fp8, scale) pair — per-expert
        # dynamic quantization costs more in launches than fp8 wins back
        # (measured: naive fp8 Mixtral 517 vs 769 bf16 tok/s, NOTES.md)
        fp8 = hasattr(self.experts[0].gate_up_proj, "forward_quantized")
        if fp8:
            xq, xs = ops.quant_fp8(x)
            xq_bytes = xq.view(torch.uint8)  # float8 lacks index_select
        out = torch.zeros_like(x)
        for e in range(self.n_experts):
            mask = selected == e  # [T, k]
            tok_idx, k_idx = mask.nonzero(as_tuple=True)
            if tok_idx.numel() == 0:
                continue
            if fp8:
                sub = (xq_bytes[tok_idx].view(xq.dtype), xs[tok_idx])
                contrib = self.experts[e](sub)
            else:
                contrib = self.experts[e](x[tok_idx])
            out.index_add_(0, tok_idx, contrib * weights[tok_idx, k_idx, None])
        return out


class DecoderLayer(nn.Module):
    def __init__(self, cfg: ModelArchConfig, layer_idx: int):
        super().__init__()
        self.input_layernorm = nn.Parameter(torch.ones(cfg.hidden_size))
        self.post_attention_layernorm = nn.Parameter(torch.ones(cfg.hidden_size))
        self.self_attn = Attention(cfg, layer_idx)
        self.mlp = (
            MoEMLP(cfg) if cfg.num_local_experts > 0 else MLP(cfg, layer_idx)
        )
        self.eps = cfg.rms_norm_eps
        # gemma2: branch outputs are normed BEFORE the residual add, with
        # separate pre/post feed-forward norms (4 norms per layer)
        self.post_norms = cfg.post_norms
        # granite: branch outputs scale by residual_multiplier before the
        # residual add
        self.res_mult = cfg.residual_multiplier or 1.0
        if cfg.post_norms:
            self.pre_feedforward_layernorm = nn.Parameter(
                torch.ones(cfg.hidden_size)
            )
            self.post_feedforward_layernorm = nn.Parameter(
                torch.ones(cfg.hidden_size)
            )

    def forward(self, x, residual, fb, kv_cache, cos_sin):
        if self.post_norms:
            # incoming invariant: hidden = x + residual (residual None at
            # layer 0); outgoing (m_normed, hidden_after_attn) keeps it,
            # so the model-level final fused_add_rmsnorm still applies
            hidden = x if residual is None else x + residual
            a = self.self_attn(
                ops.rmsnorm(hidden, self.input_layernorm, self.eps),
                fb, kv_cache, cos_sin,
            )
            a = ops.rmsnorm(a, self.post_attention_layernorm, self.eps)
            h2 = hidden + a
            mm = self.mlp(
                ops.rmsnorm(h2, self.pre_feedforward_layernorm, self.eps), fb
            )
            mm = ops.rmsnorm(mm, self.post_feedforward_layernorm, self.eps)
            return mm, h2
        if (
            getattr(self, "_fp8_fused", False)
            and x.is_cuda
            and fb.lora_ids is None
        ):
            # fp8 serving: norms/activations emit fp8 + per-row scales
            # directly (quant_fp8.hip), so activation quantization costs
            # nothing extra; residual stays bf16
            if residual is None:
                residual = x
                xq, xs = ops.rmsnorm_fp8(x, self.input_layernorm, self.eps)
            else:
                xq, xs = ops.fused_add_rmsnorm_fp8(
                    x, residual, self.input_layernorm, self.eps
                )
            x = self.self_attn((xq, xs), fb, kv_cache, cos_sin)
            if isinstance(self.mlp, MoEMLP):
                # router consumes bf16; MoEMLP quantizes once internally
                x, residual = ops.fused_add_rmsnorm(
                    x, residual, self.post_attention_layernorm, self.eps
                )
                x = self.mlp(x)
            else:
                xq, xs = ops.fused_add_rmsnorm_fp8(
                    x, residual, self.post_attention_layernorm, self.eps
                )
                x = self.mlp((xq, xs))
            return x, residual
        if residual is None:
            residual = x
            x = ops.rmsnorm(x, self.input_layernorm, self.eps)
        else:
            x, residual = ops.fused_add_rmsnorm(
                x, residual, self.input_layernorm, self.eps
            )
        x = self.self_attn(x, fb, kv_cache, cos_sin)
        if self.res_mult != 1.0:
            x = x * self.res_mult
        x, residual = ops.fused_add_rmsnorm(
            x, residual, self.post_attention_layernorm, self.eps
        )
        x = self.mlp(x, fb)
        if self.res_mult != 1.0:
            x = x * self.res_mult
        return x, residual


class LlamaForCausalLM(nn.Module):
    """Covers dense Llama and (with num_local_experts>0) Mixtral MoE."""

    def __init__(self, cfg: ModelArchConfig, device=None, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        # build weights directly in the target dtype/device: a 70B model
        # constructed fp32-first would transiently need ~280 GB
        prev_dtype = torch.get_default_dtype()
        torch.set_default_dtype(dtype)
        try:
            with torch.device(device if device is not None else "cpu"):
                self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
                self.layers = nn.ModuleList(
                    [DecoderLayer(cfg, i) for i in range(cfg.num_hidden_layers)]
                )
                self.norm = nn.Parameter(torch.ones(cfg.hidden_size))
                if cfg.tie_word_embeddings:
                    self.lm_head = None
                else:
                    self.lm_head = EngineLinear(
                        cfg.hidden_size, cfg.vocab_size, bias=False
                    )
        finally:
            torch.set_default_dtype(prev_dtype)
        from kubeai_amd.ops import ref as ops_ref

        self.register_buffer(
            "cos_sin",
            ops_ref.make_cos_sin_cache(
                cfg.head_dim, cfg.max_position_embeddings, cfg.rope_theta,
                rope_scaling=cfg.rope_scaling,
            ).to(device=device),
            persistent=False,
        )
        if cfg.rope_local_base_freq:
            # gemma3: windowed layers rope with the LOCAL base, unscaled
            self.register_buffer(
                "cos_sin_local",
                ops_ref.make_cos_sin_cache(
                    cfg.head_dim, cfg.max_position_embeddings,
                    cfg.rope_local_base_freq,
                ).to(device=device),
                persistent=False,
            )
        else:
            self.cos_sin_local = None

    @torch.inference_mode()
    def forward(self, fb: ForwardBatch) -> torch.Tensor:
        cs = (
            (self.cos_sin, self.cos_sin_local)
            if self.cos_sin_local is not None
            else self.cos_sin
        )
        x = self.embed_tokens(fb.input_ids.long())
        if self.cfg.scale_embeddings:
            # gemma: embeddings scaled by sqrt(hidden), cast to model dtype
            # first (HF normalizer semantics)
            x = x * torch.tensor(
                self.cfg.hidden_size ** 0.5, dtype=x.dtype
            )
        elif self.cfg.embedding_multiplier:
            x = x * self.cfg.embedding_multiplier  # granite
        if fb.mm_embeds is not None:
            # image-placeholder positions take the vision-tower embeddings
            x = x.index_copy(0, fb.mm_idx, fb.mm_embeds.to(x.dtype))
        residual = None
        for i, layer in enumerate(self.layers):
            x, residual = layer(x, residual, fb, self.kv_caches[i], cs)
        x, _ = ops.fused_add_rmsnorm(x, residual, self.norm, self.cfg.rms_norm_eps)
        return x

    @torch.inference_mode()
    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        if self.lm_head is None:
            logits = F.linear(hidden, self.embed_tokens.weight).float()
        else:
            logits = self.lm_head(hidden).float()
        if self.cfg.logits_scaling:
            logits = logits / self.cfg.logits_scaling  # granite
        cap = self.cfg.final_logit_softcap
        if cap > 0:
            logits = cap * torch.tanh(logits / cap)
        return logits

    def bind_kv_caches(self, kv_caches: list[tuple[torch.Tensor, torch.Tensor]]):
        self.kv_caches = kv_caches
