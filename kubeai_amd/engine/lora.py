"""LoRA adapter manager (multi-adapter serving, vLLM-style semantics).

Adapters target the attention projections (q/k/v/o). A mixed batch carries
per-token adapter ids (ForwardBatch.lora_ids); apply() computes

    y[tok] += B_a (A_a x[tok]) * (alpha / r)      for tok with adapter a

grouped per adapter — the segmented-gather form of SGMV. Weight format:
HF PEFT safetensors when `path` exists; otherwise deterministic random-init
(rank 16) for synthetic/benchmark use (no network for real checkpoints).

KV-cache isolation: the adapter id doubles as the BlockManager salt, so
prefix blocks are never shared across adapters (kvcache.py).
"""
from __future__ import annotations

import os
from typing import Optional

import torch


class LoRAManager:
    DEFAULT_RANK = 16
    TARGETS = ("qkv", "o", "gate_up", "down")

    def __init__(self, model, device, dtype):
        self.model = model
        self.device = device
        self.dtype = dtype
        # adapters[lora_id][target] = list over layers of (A [r,in], B [out,r], scale)
        self.adapters: dict[int, dict[str, list[tuple[torch.Tensor, torch.Tensor, float]]]] = {}
        model.lora_manager = self
        for m in model.modules():
            if hasattr(m, "qkv_proj"):  # Attention modules
                object.__setattr__(m, "_lora_manager", self)
            # dense per-layer MLPs (layer_idx set); MoE expert MLPs excluded
            if hasattr(m, "gate_up_proj") and getattr(m, "layer_idx", None) is not None:
                object.__setattr__(m, "_lora_manager", self)

    # ------------------------------------------------------------------
    def load(self, lora_id: int, path: Optional[str]) -> None:
        if lora_id in self.adapters:
            return
        if path and os.path.isdir(path) and _peft_weights_present(path):
            self.adapters[lora_id] = self._load_peft(path)
        else:
            self.adapters[lora_id] = self._random_adapter(lora_id)

    def unload(self, lora_id: int) -> None:
        self.adapters.pop(lora_id, None)

    @property
    def active(self) -> bool:
        return bool(self.adapters)

    # ------------------------------------------------------------------
    def _random_adapter(self, lora_id: int):
        g = torch.Generator(device="cpu").manual_seed(lora_id)
        r = self.DEFAULT_RANK
        cfg = self.model.cfg
        H = cfg.hidden_size
        qkv_out = (cfg.num_attention_heads + 2 * cfg.num_key_value_heads) * cfg.head_dim
        o_in = cfg.num_attention_heads * cfg.head_dim
        I = cfg.intermediate_size
        moe = cfg.num_local_experts > 0  # MoE: adapters stay attention-only
        out = {t: [] for t in self.TARGETS}
        for _ in range(cfg.num_hidden_layers):
            shapes = [("qkv", (H, qkv_out)), ("o", (o_in, H))]
            if not moe:
                shapes += [("gate_up", (H, 2 * I)), ("down", (I, H))]
            for tgt, (din, dout) in shapes:
                A = (torch.randn(r, din, generator=g) * 0.05).to(self.device, self.dtype)
                B = (torch.randn(dout, r, generator=g) * 0.5).to(self.device, self.dtype)
                out[tgt].append((A, B, 2.0 / r))
        return out

    def _load_peft(self, path: str):
        from safetensors.torch import load_file
        import json

        alpha, r = 16.0, self.DEFAULT_RANK
        cfg_path = os.path.join(path, "adapter_config.json")
        if os.path.exists(cfg_path):
            with open(cfg_path) as f:
                acfg = json.load(f)
            alpha = float(acfg.get("lora_alpha", 16))
            r = int(acfg.get("r", 16))
        weights = load_file(_peft_weights_present(path))
        cfg = self.model.cfg
        out = {t: [] for t in self.TARGETS}
        for layer in range(cfg.num_hidden_layers):
            # fuse q/k/v adapters into the qkv slot (block-diagonal A, stacked B)
            parts = []
            for proj in ("q_proj", "k_proj", "v_proj"):
                key = _find_key(weights, layer, proj)
                parts.append(key)
            As, Bs = [], []
            for key in parts:
                if key is None:
                    As.append(None)
                    Bs.append(None)
                else:
                    As.append(weights[key + ".lora_A.weight"])
                    Bs.append(weights[key + ".lora_B.weight"])
            H = cfg.hidden_size
            q_out = cfg.num_attention_heads * cfg.head_dim
            kv_out = cfg.num_key_value_heads * cfg.head_dim
            outs = (q_out, kv_out, kv_out)
            n = sum(1 for a in As if a is not None)
            if n == 0:
                A = torch.zeros(1, H)
                B = torch.zeros(q_out + 2 * kv_out, 1)
            else:
                A = torch.cat([a for a in As if a is not None], dim=0)  # [n*r, H]
                B = torch.zeros(q_out + 2 * kv_out, A.shape[0])
                row, col = 0, 0
                for a, b, dout in zip(As, Bs, outs):
                    if a is not None:
                        B[row : row + dout, col : col + a.shape[0]] = b
                        col += a.shape[0]
                    row += dout
            out["qkv"].append(
                (A.to(self.device, self.dtype), B.to(self.device, self.dtype), alpha / r)
            )
            okey = _find_key(weights, layer, "o_proj")
            if okey is None:
                out["o"].append(
                    (
                        torch.zeros(1, q_out, device=self.device, dtype=self.dtype),
                        torch.zeros(H, 1, device=self.device, dtype=self.dtype),
                        0.0,
                    )
                )
            else:
                out["o"].append(
                    (
                        weights[okey + ".lora_A.weight"].to(self.device, self.dtype),
                        weights[okey + ".lora_B.weight"].to(self.device, self.dtype),
                        alpha / r,
                    )
                )
            # MLP targets: gate+up fused block-diagonally, down direct
            I = cfg.intermediate_size
            gkey = _find_key(weights, layer, "gate_proj", "mlp")
            ukey = _find_key(weights, layer, "up_proj", "mlp")
            if gkey is None and ukey is None:
                out["gate_up"].append(
                    (torch.zeros(1, H, device=self.device, dtype=self.dtype),
                     torch.zeros(2 * I, 1, device=self.device, dtype=self.dtype),
                     0.0)
                )
            else:
                As2, Bs2 = [], []
                for key, row in ((gkey, 0), (ukey, I)):
                    if key is not None:
                        As2.append((weights[key + ".lora_A.weight"], row))
                        Bs2.append(weights[key + ".lora_B.weight"])
                A2 = torch.cat([a for a, _ in As2], dim=0)  # [n*r, H]
                B2 = torch.zeros(2 * I, A2.shape[0])
                col = 0
                for (a, row), bmat in zip(As2, Bs2):
                    B2[row : row + I, col : col + a.shape[0]] = bmat
                    col += a.shape[0]
                out["gate_up"].append(
                    (A2.to(self.device, self.dtype),
                     B2.to(self.device, self.dtype), alpha / r)
                )
            dkey = _find_key(weights, layer, "down_proj", "mlp")
            if dkey is None:
                out["down"].append(
                    (torch.zeros(1, I, device=self.device, dtype=self.dtype),
                     torch.zeros(H, 1, device=self.device, dtype=self.dtype),
                     0.0)
                )
            else:
                out["down"].append(
                    (weights[dkey + ".lora_A.weight"].to(self.device, self.dtype),
                     weights[dkey + ".lora_B.weight"].to(self.device, self.dtype),
                     alpha / r)
                )
        return out

    # ------------------------------------------------------------------
    def apply(
        self,
        layer_idx: int,
        target: str,
        x: torch.Tensor,  # [T, in]
        y: torch.Tensor,  # [T, out] += delta (in-place)
        lora_ids: torch.Tensor,  # [T] int32
    ) -> None:
        """Segmented LoRA matmul: per-adapter token groups; one hand-written
        SGMV kernel launch per adapter on GPU (csrc/sgmv.hip)."""
        present = torch.unique(lora_ids)
        for lid_t in present:
            lid = int(lid_t)
            if lid == 0 or lid not in self.adapters:
                continue
            A, B, s = self.adapters[lid][target][layer_idx]
            idx = (lora_ids == lid_t).nonzero(as_tuple=True)[0]
            if x.is_cuda:
                from kubeai_amd import _C

                _C.sgmv(y, x, A, B, idx, s)
            else:
                xs = x[idx]
                y[idx] += (xs @ A.T @ B.T) * s


def _peft_weights_present(path: str) -> Optional[str]:
    for f in ("adapter_model.safetensors", "model.safetensors"):
        p = os.path.join(path, f)
        if os.path.exists(p):
            return p
    return None


def _find_key(
    weights: dict, layer: int, proj: str, sub: str = "self_attn"
) -> Optional[str]:
    for prefix in (
        f"base_model.model.model.layers.{layer}.{sub}.{proj}",
        f"model.layers.{layer}.{sub}.{proj}",
    ):
        if prefix + ".lora_A.weight" in weights:
            return prefix
    return None
