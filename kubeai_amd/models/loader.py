"""Checkpoint loading: HF safetensors -> engine model layout.

Maps HF Llama/Mixtral parameter names onto the engine's fused layout
(q/k/v -> qkv_proj, gate/up -> gate_up_proj); supports sharded
model-*.safetensors. TP loading slices each full tensor per rank with the
same sharding as parallel/tp.py.
"""
from __future__ import annotations

import glob
import json
import os
from typing import Iterator

import torch


# quantization artifact suffixes: consumed alongside their weight (fp8)
# or a loud error (formats we cannot dequantize) — ADVICE r1: never copy
# fp8-coded bytes as if they were bf16 weights.
_SCALE_SUFFIXES = (".weight_scale", ".weight_scale_inv", ".input_scale")
_FP8_DTYPES = tuple(
    getattr(torch, n)
    for n in ("float8_e4m3fn", "float8_e5m2", "float8_e4m3fnuz", "float8_e5m2fnuz")
    if hasattr(torch, n)
)
_UNSUPPORTED_QUANT_KEYS = ("g_idx", "scales_zp")  # GPTQ act-order etc.

# AWQ "gemm" nibble interleave (AutoAWQ unpack_awq + reverse_awq_order):
# nibble i of each int32 holds logical column REVERSE_AWQ_ORDER.index(i);
# applying this permutation to the shift-ordered unpack restores logical
# column order. Self-consistency is covered by tests (synthetic packing);
# no real AWQ checkpoint exists in this offline environment to cross-check.
_REVERSE_AWQ_ORDER = (0, 4, 1, 5, 2, 6, 3, 7)


def _awq_unpack(q: torch.Tensor) -> torch.Tensor:
    """int32 [r, c/8] -> int [r, c] nibbles in logical column order."""
    shifts = torch.arange(0, 32, 4, dtype=torch.int32)
    nibbles = (q.unsqueeze(-1) >> shifts) & 0xF  # [r, c/8, 8] shift order
    logical = nibbles[..., list(_REVERSE_AWQ_ORDER)]
    return logical.reshape(q.shape[0], -1)


def _awq_dequant(qweight, qzeros, scales) -> torch.Tensor:
    """AWQ gemm-format group -> bf16 [out, in] dense weight.

    qweight int32 [in, out/8]; qzeros int32 [in/g, out/8];
    scales fp16 [in/g, out]. w = (nibble - zero) * scale.
    """
    iw = _awq_unpack(qweight).float()           # [in, out]
    iz = _awq_unpack(qzeros).float()            # [in/g, out]
    g = qweight.shape[0] // qzeros.shape[0]
    z = iz.repeat_interleave(g, dim=0)
    s = scales.float().repeat_interleave(g, dim=0)
    return ((iw - z) * s).t().contiguous().to(torch.bfloat16)


def iter_safetensors(model_dir: str) -> Iterator[tuple[str, torch.Tensor]]:
    """Yield (name, tensor) over all shards; pre-quantized fp8 checkpoints
    (e.g. Llama-3.1-*-FP8: fp8 weights + per-tensor `weight_scale`) are
    dequantized on the fly so callers always see plain float weights.
    Unsupported quant formats (AWQ/GPTQ packed ints) raise instead of
    silently producing garbage."""
    from safetensors import safe_open

    files = sorted(glob.glob(os.path.join(model_dir, "*.safetensors")))
    if not files:
        raise FileNotFoundError(f"no *.safetensors under {model_dir}")
    # AWQ (4-bit, "gemm" packing) dequantizes on the fly to bf16 —
    # reference catalog entry: hugging-quants/...-70B-Instruct-AWQ-INT4
    awq = False
    cfg_path = os.path.join(model_dir, "config.json")
    if os.path.exists(cfg_path):
        try:
            with open(cfg_path) as cf:
                qc = (json.load(cf).get("quantization_config") or {})
            awq = (
                qc.get("quant_method") == "awq"
                and int(qc.get("bits", 4)) == 4
                and qc.get("version", "gemm").lower() == "gemm"
            )
        except Exception:  # noqa: BLE001
            awq = False
    for f in files:
        with safe_open(f, framework="pt", device="cpu") as sf:
            keys = set(sf.keys())
            for key in sf.keys():
                if key.endswith(_SCALE_SUFFIXES):
                    continue  # consumed with its weight below
                if key.endswith((".qzeros", ".scales")) and awq:
                    continue  # consumed with their .qweight below
                if key.endswith(".qweight"):
                    pre = key[: -len(".qweight")]
                    if (
                        not awq
                        or pre + ".qzeros" not in keys
                        or pre + ".scales" not in keys
                    ):
                        raise ValueError(
                            f"{model_dir}: packed-quantized tensor {key!r} "
                            "is not a recognized AWQ-gemm group — provide "
                            "bf16/fp16, fp8(+weight_scale), or AWQ-INT4 "
                            "gemm checkpoints"
                        )
                    yield pre + ".weight", _awq_dequant(
                        sf.get_tensor(key),
                        sf.get_tensor(pre + ".qzeros"),
                        sf.get_tensor(pre + ".scales"),
                    )
                    continue
                if any(key.endswith("." + s) for s in _UNSUPPORTED_QUANT_KEYS):
                    raise ValueError(
                        f"{model_dir}: packed-quantized checkpoint tensor "
                        f"{key!r} is not supported — provide a bf16/fp16, "
                        "fp8(+weight_scale), or AWQ-INT4 gemm checkpoint"
                    )
                w = sf.get_tensor(key)
                if w.dtype in _FP8_DTYPES:
                    scale_key = None
                    for cand in (key + "_scale", key + "_scale_inv"):
                        if cand in keys:
                            scale_key = cand
                            break
                    if scale_key is None:
                        raise ValueError(
                            f"{model_dir}: fp8 tensor {key!r} has no "
                            "weight_scale companion — cannot dequantize"
                        )
                    scale = sf.get_tensor(scale_key).float()
                    if scale_key.endswith("_scale_inv"):
                        scale = 1.0 / scale
                    if scale.numel() not in (1, w.shape[0]):
                        raise ValueError(
                            f"{model_dir}: {scale_key!r} shape "
                            f"{tuple(scale.shape)} unsupported (per-tensor "
                            f"or per-row expected for {tuple(w.shape)})"
                        )
                    scale = scale.reshape(-1, *([1] * (w.dim() - 1)))
                    w = w.float() * scale
                yield key, w


def _strip(name: str) -> str:
    for prefix in ("model.", "language_model.model.", "language_model."):
        if name.startswith(prefix):
            return name[len(prefix) :]
    return name


def load_weights_tp(model, model_dir: str, vision=None) -> int:
    """Load HF weights into TPLlamaForCausalLM: every rank reads the full
    tensors and keeps its shard (same slicing as parallel/tp.py init).
    The vision tower (llava) is replicated on every rank, not sharded."""
    cfg = model.cfg
    tp = model.tp
    hd = cfg.head_dim
    nq, nkv = cfg.num_attention_heads, cfg.num_key_value_heads
    nq_l, nkv_l = nq // tp.world, nkv // tp.world
    i_l = cfg.intermediate_size // tp.world
    params = dict(model.named_parameters())
    filled: set[str] = set()

    def put(target: str, tensor: torch.Tensor) -> None:
        p = params[target]
        if p.shape != tensor.shape:
            raise ValueError(
                f"{target}: shape {tuple(tensor.shape)} != {tuple(p.shape)}"
            )
        with torch.no_grad():
            p.copy_(tensor.to(p.dtype))
        filled.add(target)

    def shard_rows_heads(w, n_heads, n_local):
        return w.view(n_heads, hd, -1)[
            tp.rank * n_local : (tp.rank + 1) * n_local
        ].reshape(n_local * hd, -1)

    pending: dict[str, dict[str, torch.Tensor]] = {}

    def fuse_qkv(layer: int) -> None:
        ps = pending.get(f"qkv.{layer}", {})
        if len(ps) == 3:
            q = shard_rows_heads(ps["q"], nq, nq_l)
            k = shard_rows_heads(ps["k"], nkv, nkv_l)
            v = shard_rows_heads(ps["v"], nkv, nkv_l)
            put(f"layers.{layer}.self_attn.qkv_proj.weight",
                torch.cat([q, k, v], dim=0))

    def fuse_gate_up(layer: int) -> None:
        ps = pending.get(f"gu.{layer}", {})
        if len(ps) == 2:
            g = ps["gate"][tp.rank * i_l : (tp.rank + 1) * i_l]
            u = ps["up"][tp.rank * i_l : (tp.rank + 1) * i_l]
            put(f"layers.{layer}.mlp.gate_up_proj.weight", torch.cat([g, u], dim=0))

    for name, w in iter_safetensors(model_dir):
        if name.startswith(("vision_tower.", "multi_modal_projector.")):
            if vision is not None and not vision.load_hf_tensor(name, w):
                raise ValueError(f"unrecognized vision tensor {name}")
            continue
        if name == "image_newline":  # llava-1.6 packing artifact, unused
            continue
        n = _strip(name)
        if n == "embed_tokens.weight":
            put("embed_tokens.weight", w)
        elif n == "lm_head.weight":
            if model.lm_head is not None:
                put("lm_head.weight", w)
        elif n == "norm.weight":
            put("norm", w)
        elif n.startswith("layers."):
            parts = n.split(".")
            layer = int(parts[1])
            rest = ".".join(parts[2:])
            if rest == "input_layernorm.weight":
                put(f"layers.{layer}.input_layernorm", w)
            elif rest == "post_attention_layernorm.weight":
                put(f"layers.{layer}.post_attention_layernorm", w)
            elif rest == "self_attn.qkv_proj.weight":
                # phi-3/phi-4: pre-fused [q;k;v] rows; split and feed the
                # same per-tensor sharding paths
                qr, kr, vr = torch.split(
                    w,
                    [cfg.num_attention_heads * hd,
                     cfg.num_key_value_heads * hd,
                     cfg.num_key_value_heads * hd],
                    dim=0,
                )
                ps = pending.setdefault(f"qkv.{layer}", {})
                ps["q"], ps["k"], ps["v"] = qr, kr, vr
                fuse_qkv(layer)
            elif rest == "mlp.gate_up_proj.weight":
                gr, ur = w.chunk(2, dim=0)
                ps = pending.setdefault(f"gu.{layer}", {})
                ps["gate"], ps["up"] = gr, ur
                fuse_gate_up(layer)
            elif rest == "self_attn.q_proj.weight":
                pending.setdefault(f"qkv.{layer}", {})["q"] = w
                fuse_qkv(layer)
            elif rest == "self_attn.k_proj.weight":
                pending.setdefault(f"qkv.{layer}", {})["k"] = w
                fuse_qkv(layer)
            elif rest == "self_attn.v_proj.weight":
                pending.setdefault(f"qkv.{layer}", {})["v"] = w
                fuse_qkv(layer)
            elif rest == "self_attn.o_proj.weight":
                put(
                    f"layers.{layer}.self_attn.o_proj.weight",
                    w[:, tp.rank * nq_l * hd : (tp.rank + 1) * nq_l * hd],
                )
            elif rest == "mlp.gate_proj.weight":
                pending.setdefault(f"gu.{layer}", {})["gate"] = w
                fuse_gate_up(layer)
            elif rest == "mlp.up_proj.weight":
                pending.setdefault(f"gu.{layer}", {})["up"] = w
                fuse_gate_up(layer)
            elif rest == "mlp.down_proj.weight":
                put(
                    f"layers.{layer}.mlp.down_proj.weight",
                    w[:, tp.rank * i_l : (tp.rank + 1) * i_l],
                )
    if model.lm_head is not None and "lm_head.weight" not in filled:
        with torch.no_grad():
            model.lm_head.weight.copy_(model.embed_tokens.weight)
        filled.add("lm_head.weight")
    missing = set(params) - filled
    if missing:
        raise ValueError(f"unfilled parameters after TP load: {sorted(missing)[:8]}")
    return len(filled)


def load_weights(model, model_dir: str, vision=None) -> int:
    """Load HF weights into LlamaForCausalLM (dense or MoE). Returns the
    number of engine parameters filled; raises if any stays unset.

    vision: optional models/vision.py VisionTower — llava checkpoints
    route their vision_tower.* / multi_modal_projector.* tensors there
    (HF llava nests the text model under language_model.*)."""
    cfg = model.cfg
    hd, nq, nkv = cfg.head_dim, cfg.num_attention_heads, cfg.num_key_value_heads
    params = dict(model.named_parameters())
    filled: set[str] = set()

    def put(target: str, tensor: torch.Tensor) -> None:
        p = params[target]
        if p.shape != tensor.shape:
            raise ValueError(f"{target}: shape {tuple(tensor.shape)} != {tuple(p.shape)}")
        with torch.no_grad():
            p.copy_(tensor.to(p.dtype))
        filled.add(target)

    # staging for fused tensors
    pending: dict[str, dict[str, torch.Tensor]] = {}

    def fuse_qkv(layer: int) -> None:
        ps = pending.get(f"qkv.{layer}", {})
        if len(ps) == 3:
            put(
                f"layers.{layer}.self_attn.qkv_proj.weight",
                torch.cat([ps["q"], ps["k"], ps["v"]], dim=0),
            )

    def fuse_qkv_bias(layer: int) -> None:
        ps = pending.get(f"qkvb.{layer}", {})
        if len(ps) == 3:
            put(
                f"layers.{layer}.self_attn.qkv_proj.bias",
                torch.cat([ps["q"], ps["k"], ps["v"]], dim=0),
            )

    def fuse_gate_up(layer: int, expert: int | None) -> None:
        key = f"gu.{layer}" + ("" if expert is None else f".{expert}")
        ps = pending.get(key, {})
        if len(ps) == 2:
            tgt = (
                f"layers.{layer}.mlp.gate_up_proj.weight"
                if expert is None
                else f"layers.{layer}.mlp.experts.{expert}.gate_up_proj.weight"
            )
            put(tgt, torch.cat([ps["gate"], ps["up"]], dim=0))

    for name, w in iter_safetensors(model_dir):
        if name.startswith(("vision_tower.", "multi_modal_projector.")):
            if vision is not None and not vision.load_hf_tensor(name, w):
                raise ValueError(f"unrecognized vision tensor {name}")
            continue
        if name == "image_newline":  # llava-1.6 packing artifact, unused
            continue
        n = _strip(name)
        # gemma RMSNorm convention: y = x * (1 + w). The engine's norm
        # kernel multiplies by w, so gemma checkpoints load as w+1 (init
        # zeros -> ones matches).
        norm_w = (lambda t: t.float() + 1.0) if cfg.norm_plus_one else (lambda t: t)
        if n == "embed_tokens.weight":
            put("embed_tokens.weight", w)
        elif n in ("lm_head.weight",):
            if model.lm_head is not None:
                put("lm_head.weight", w)
        elif n == "norm.weight":
            put("norm", norm_w(w))
        elif ".layers." in n or n.startswith("layers."):
            parts = n.split(".")
            layer = int(parts[parts.index("layers") + 1])
            rest = ".".join(parts[parts.index("layers") + 2 :])
            if rest == "input_layernorm.weight":
                put(f"layers.{layer}.input_layernorm", norm_w(w))
            elif rest == "post_attention_layernorm.weight":
                put(f"layers.{layer}.post_attention_layernorm", norm_w(w))
            elif rest == "pre_feedforward_layernorm.weight":
                put(f"layers.{layer}.pre_feedforward_layernorm", norm_w(w))
            elif rest == "post_feedforward_layernorm.weight":
                put(f"layers.{layer}.post_feedforward_layernorm", norm_w(w))
            elif rest == "self_attn.q_norm.weight":
                put(f"layers.{layer}.self_attn.q_norm", norm_w(w))
            elif rest == "self_attn.k_norm.weight":
                put(f"layers.{layer}.self_attn.k_norm", norm_w(w))
            elif rest == "self_attn.qkv_proj.weight":
                # phi-3/phi-4 checkpoints pre-fuse qkv ([q;k;v] rows —
                # same layout as the engine's fused projection)
                qr, kr, vr = torch.split(
                    w, [nq * hd, nkv * hd, nkv * hd], dim=0
                )
                ps = pending.setdefault(f"qkv.{layer}", {})
                ps["q"], ps["k"], ps["v"] = qr, kr, vr
                fuse_qkv(layer)
            elif rest == "mlp.gate_up_proj.weight":
                # phi family pre-fuses gate|up as well
                gr, ur = w.chunk(2, dim=0)
                ps = pending.setdefault(f"gu.{layer}", {})
                ps["gate"], ps["up"] = gr, ur
                fuse_gate_up(layer, None)
            elif rest == "self_attn.q_proj.weight":
                pending.setdefault(f"qkv.{layer}", {})["q"] = w
                fuse_qkv(layer)
            elif rest == "self_attn.k_proj.weight":
                pending.setdefault(f"qkv.{layer}", {})["k"] = w
                fuse_qkv(layer)
            elif rest == "self_attn.v_proj.weight":
                pending.setdefault(f"qkv.{layer}", {})["v"] = w
                fuse_qkv(layer)
            elif rest == "self_attn.q_proj.bias":
                pending.setdefault(f"qkvb.{layer}", {})["q"] = w
                fuse_qkv_bias(layer)
            elif rest == "self_attn.k_proj.bias":
                pending.setdefault(f"qkvb.{layer}", {})["k"] = w
                fuse_qkv_bias(layer)
            elif rest == "self_attn.v_proj.bias":
                pending.setdefault(f"qkvb.{layer}", {})["v"] = w
                fuse_qkv_bias(layer)
            elif rest == "self_attn.o_proj.weight":
                put(f"layers.{layer}.self_attn.o_proj.weight", w)
            elif rest == "mlp.gate_proj.weight":
                pending.setdefault(f"gu.{layer}", {})["gate"] = w
                fuse_gate_up(layer, None)
            elif rest == "mlp.up_proj.weight":
                pending.setdefault(f"gu.{layer}", {})["up"] = w
                fuse_gate_up(layer, None)
            elif rest == "mlp.down_proj.weight":
                put(f"layers.{layer}.mlp.down_proj.weight", w)
            elif rest == "block_sparse_moe.gate.weight":
                put(f"layers.{layer}.mlp.gate.weight", w)
            elif ".block_sparse_moe.experts." in "." + rest:
                ep = rest.split(".")
                e = int(ep[ep.index("experts") + 1])
                which = ep[-2]  # w1/w2/w3
                if which == "w1":  # gate
                    pending.setdefault(f"gu.{layer}.{e}", {})["gate"] = w
                    fuse_gate_up(layer, e)
                elif which == "w3":  # up
                    pending.setdefault(f"gu.{layer}.{e}", {})["up"] = w
                    fuse_gate_up(layer, e)
                elif which == "w2":  # down
                    put(f"layers.{layer}.mlp.experts.{e}.down_proj.weight", w)

    if model.lm_head is not None and "lm_head.weight" not in filled:
        # tied embeddings checkpoints
        with torch.no_grad():
            model.lm_head.weight.copy_(model.embed_tokens.weight)
        filled.add("lm_head.weight")

    missing = set(params) - filled
    if missing:
        raise ValueError(f"unfilled parameters after load: {sorted(missing)[:8]}...")
    return len(filled)


def save_hf_checkpoint(model, out_dir: str) -> None:
    """Write the model back out in HF layout (tests + weight-prep tooling)."""
    from safetensors.torch import save_file

    os.makedirs(out_dir, exist_ok=True)
    cfg = model.cfg
    state: dict[str, torch.Tensor] = {
        "model.embed_tokens.weight": model.embed_tokens.weight.detach().cpu(),
    }
    if model.lm_head is not None:
        state["lm_head.weight"] = model.lm_head.weight.detach().cpu()
    nq, nkv, hd = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
    # gemma convention stores norms as w (engine holds 1+w); reverse it
    norm_out = (
        (lambda t: t.detach().cpu().float() - 1.0)
        if cfg.norm_plus_one
        else (lambda t: t.detach().cpu())
    )
    state["model.norm.weight"] = norm_out(model.norm)
    for i, layer in enumerate(model.layers):
        pre = f"model.layers.{i}."
        state[pre + "input_layernorm.weight"] = norm_out(layer.input_layernorm)
        state[pre + "post_attention_layernorm.weight"] = norm_out(
            layer.post_attention_layernorm
        )
        if getattr(layer, "post_norms", False):
            state[pre + "pre_feedforward_layernorm.weight"] = norm_out(
                layer.pre_feedforward_layernorm
            )
            state[pre + "post_feedforward_layernorm.weight"] = norm_out(
                layer.post_feedforward_layernorm
            )
        if getattr(layer.self_attn, "qk_norm", False):
            state[pre + "self_attn.q_norm.weight"] = norm_out(
                layer.self_attn.q_norm
            )
            state[pre + "self_attn.k_norm.weight"] = norm_out(
                layer.self_attn.k_norm
            )
        qkv = layer.self_attn.qkv_proj.weight.detach().cpu()
        state[pre + "self_attn.q_proj.weight"] = qkv[: nq * hd].clone()
        state[pre + "self_attn.k_proj.weight"] = qkv[nq * hd : (nq + nkv) * hd].clone()
        state[pre + "self_attn.v_proj.weight"] = qkv[(nq + nkv) * hd :].clone()
        qkv_b = layer.self_attn.qkv_proj.bias
        if qkv_b is not None:
            b = qkv_b.detach().cpu()
            state[pre + "self_attn.q_proj.bias"] = b[: nq * hd].clone()
            state[pre + "self_attn.k_proj.bias"] = b[nq * hd : (nq + nkv) * hd].clone()
            state[pre + "self_attn.v_proj.bias"] = b[(nq + nkv) * hd :].clone()
        state[pre + "self_attn.o_proj.weight"] = (
            layer.self_attn.o_proj.weight.detach().cpu()
        )
        mlp = layer.mlp
        if hasattr(mlp, "experts"):
            state[pre + "block_sparse_moe.gate.weight"] = mlp.gate.weight.detach().cpu()
            for e, ex in enumerate(mlp.experts):
                gu = ex.gate_up_proj.weight.detach().cpu()
                I = cfg.intermediate_size
                epre = pre + f"block_sparse_moe.experts.{e}."
                state[epre + "w1.weight"] = gu[:I].clone()
                state[epre + "w3.weight"] = gu[I:].clone()
                state[epre + "w2.weight"] = ex.down_proj.weight.detach().cpu()
        else:
            gu = mlp.gate_up_proj.weight.detach().cpu()
            I = cfg.intermediate_size
            state[pre + "mlp.gate_proj.weight"] = gu[:I].clone()
            state[pre + "mlp.up_proj.weight"] = gu[I:].clone()
            state[pre + "mlp.down_proj.weight"] = mlp.down_proj.weight.detach().cpu()
    save_file(state, os.path.join(out_dir, "model.safetensors"))
    with open(os.path.join(out_dir, "config.json"), "w") as f:
        json.dump(
            {
                "architectures": [
                    "MixtralForCausalLM" if cfg.num_local_experts
                    else "Gemma3ForCausalLM" if cfg.qk_norm
                    else "Gemma2ForCausalLM" if cfg.post_norms
                    else "GemmaForCausalLM" if cfg.norm_plus_one
                    else "GraniteForCausalLM" if cfg.residual_multiplier
                    else "LlamaForCausalLM"
                ],
                "vocab_size": cfg.vocab_size,
                "hidden_size": cfg.hidden_size,
                "intermediate_size": cfg.intermediate_size,
                "num_hidden_layers": cfg.num_hidden_layers,
                "num_attention_heads": cfg.num_attention_heads,
                "num_key_value_heads": cfg.num_key_value_heads,
                "head_dim": cfg.head_dim,
                "max_position_embeddings": cfg.max_position_embeddings,
                "rms_norm_eps": cfg.rms_norm_eps,
                "rope_theta": cfg.rope_theta,
                "tie_word_embeddings": cfg.tie_word_embeddings,
                "bos_token_id": cfg.bos_token_id,
                "eos_token_id": cfg.eos_token_id,
                "num_local_experts": cfg.num_local_experts,
                "num_experts_per_tok": cfg.num_experts_per_tok,
                "attention_bias": cfg.attention_bias,
                "rope_scaling": cfg.rope_scaling,
                "sliding_window": cfg.sliding_window or None,
                "attn_logit_softcapping": cfg.attn_logit_softcap or None,
                "final_logit_softcapping": cfg.final_logit_softcap or None,
                "query_pre_attn_scalar": cfg.query_pre_attn_scalar or None,
                "hidden_act": cfg.hidden_act,
                "sliding_window_pattern": cfg.global_layer_interval or None,
                "rope_local_base_freq": cfg.rope_local_base_freq or None,
                "embedding_multiplier": cfg.embedding_multiplier or None,
                "residual_multiplier": cfg.residual_multiplier or None,
                "logits_scaling": cfg.logits_scaling or None,
                "attention_multiplier": cfg.attention_multiplier or None,
                "torch_dtype": "bfloat16",
            },
            f,
        )
