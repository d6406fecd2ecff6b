"""CLIP-ViT vision tower + LLaVA-style multimodal projector.

Reference parity: the reference serves vision models by passing content
parts through to vLLM (`/root/reference/api/openai/v1/chat_completions.go`
preserves image_url parts; the 11B-vision entry in BASELINE.md implies
multimodal serving). The in-house engine implements the LLaVA
architecture natively: CLIP vision encoder -> 2-layer GELU projector ->
image embeddings spliced into the language model's token-embedding
stream (models/llama.py forward; runner builds the splice indices).

The tower runs once per image at prefill admission — a few GEMMs via
hipBLASLt — so plain torch modules are the right tool here; the hot
serving path (attention/decode) stays on the hand-written HIP kernels.

HF weight names follow llava-hf checkpoints:
  vision_tower.vision_model.embeddings.{class_embedding,patch_embedding,
    position_embedding}, .encoder.layers.N.{self_attn.{q,k,v,out}_proj,
    layer_norm1, layer_norm2, mlp.fc1, mlp.fc2}, .{pre_layrnorm,post_layernorm}
  multi_modal_projector.linear_1 / linear_2
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from .config import ModelArchConfig


class _ViTBlock(nn.Module):
    def __init__(self, hidden: int, heads: int, mlp: int):
        super().__init__()
        self.layer_norm1 = nn.LayerNorm(hidden)
        self.layer_norm2 = nn.LayerNorm(hidden)
        self.q_proj = nn.Linear(hidden, hidden)
        self.k_proj = nn.Linear(hidden, hidden)
        self.v_proj = nn.Linear(hidden, hidden)
        self.out_proj = nn.Linear(hidden, hidden)
        self.fc1 = nn.Linear(hidden, mlp)
        self.fc2 = nn.Linear(mlp, hidden)
        self.heads = heads
        self.head_dim = hidden // heads

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        b, t, h = x.shape
        y = self.layer_norm1(x)
        q = self.q_proj(y).view(b, t, self.heads, self.head_dim).transpose(1, 2)
        k = self.k_proj(y).view(b, t, self.heads, self.head_dim).transpose(1, 2)
        v = self.v_proj(y).view(b, t, self.heads, self.head_dim).transpose(1, 2)
        a = F.scaled_dot_product_attention(q, k, v)
        a = a.transpose(1, 2).reshape(b, t, h)
        x = x + self.out_proj(a)
        y = self.layer_norm2(x)
        x = x + self.fc2(F.gelu(self.fc1(y), approximate="tanh"))
        return x


class VisionTower(nn.Module):
    """CLIP encoder + projector; encode() -> per-image text-space embeds."""

    def __init__(self, cfg: ModelArchConfig, device=None,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        v = cfg.vision or {}
        self.image_size = int(v.get("image_size", 336))
        self.patch_size = int(v.get("patch_size", 14))
        hidden = int(v.get("hidden_size", 1024))
        layers = int(v.get("num_hidden_layers", 24))
        heads = int(v.get("num_attention_heads", 16))
        mlp = int(v.get("intermediate_size", hidden * 4))
        # llava default: features from the penultimate encoder layer
        self.feature_layer = int(v.get("vision_feature_layer", -2))
        self.n_patches = (self.image_size // self.patch_size) ** 2
        self.dtype = dtype
        prev = torch.get_default_dtype()
        torch.set_default_dtype(torch.float32)
        try:
            with torch.device(device if device is not None else "cpu"):
                self.patch_embedding = nn.Conv2d(
                    3, hidden, kernel_size=self.patch_size,
                    stride=self.patch_size, bias=False,
                )
                self.class_embedding = nn.Parameter(torch.randn(hidden) * 0.02)
                self.position_embedding = nn.Embedding(
                    self.n_patches + 1, hidden
                )
                self.pre_layrnorm = nn.LayerNorm(hidden)  # (sic, CLIP name)
                self.blocks = nn.ModuleList(
                    [_ViTBlock(hidden, heads, mlp) for _ in range(layers)]
                )
                self.linear_1 = nn.Linear(hidden, cfg.hidden_size)
                self.linear_2 = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        finally:
            torch.set_default_dtype(prev)

    @torch.inference_mode()
    def encode(self, pixel_values: torch.Tensor) -> torch.Tensor:
        """[B, 3, S, S] float -> [B * n_patches, text_hidden] (self.dtype)."""
        dev = self.patch_embedding.weight.device
        x = pixel_values.to(device=dev, dtype=torch.float32)
        p = self.patch_embedding(x)  # [B, H, g, g]
        b, h = p.shape[0], p.shape[1]
        p = p.flatten(2).transpose(1, 2)  # [B, n_patches, H]
        cls = self.class_embedding.expand(b, 1, h)
        x = torch.cat([cls, p], dim=1)
        x = x + self.position_embedding.weight.unsqueeze(0)
        x = self.pre_layrnorm(x)
        n_keep = len(self.blocks) + 1 + self.feature_layer  # -2 -> skip last
        for blk in self.blocks[: max(n_keep, 0)]:
            x = blk(x)
        feats = x[:, 1:]  # drop CLS (llava vision_feature_select "default")
        y = self.linear_2(F.gelu(self.linear_1(feats), approximate="tanh"))
        return y.reshape(-1, y.shape[-1]).to(self.dtype)

    # checkpoint key mapping (HF llava names -> this module)
    def load_hf_tensor(self, name: str, tensor: torch.Tensor) -> bool:
        t = tensor.float()
        pfx = "vision_tower.vision_model."
        if name == "multi_modal_projector.linear_1.weight":
            self.linear_1.weight.data.copy_(t)
        elif name == "multi_modal_projector.linear_1.bias":
            self.linear_1.bias.data.copy_(t)
        elif name == "multi_modal_projector.linear_2.weight":
            self.linear_2.weight.data.copy_(t)
        elif name == "multi_modal_projector.linear_2.bias":
            self.linear_2.bias.data.copy_(t)
        elif name == pfx + "embeddings.class_embedding":
            self.class_embedding.data.copy_(t.reshape(-1))
        elif name == pfx + "embeddings.patch_embedding.weight":
            self.patch_embedding.weight.data.copy_(t)
        elif name == pfx + "embeddings.position_embedding.weight":
            self.position_embedding.weight.data.copy_(t)
        elif name.startswith(pfx + "pre_layrnorm."):
            p = self.pre_layrnorm
            (p.weight if name.endswith(".weight") else p.bias).data.copy_(t)
        elif name.startswith(pfx + "post_layernorm."):
            pass  # unused: llava takes features from the penultimate layer
        elif name.startswith(pfx + "encoder.layers."):
            rest = name[len(pfx + "encoder.layers."):]
            idx_s, _, tail = rest.partition(".")
            blk = self.blocks[int(idx_s)]
            attr_map = {
                "self_attn.q_proj": blk.q_proj,
                "self_attn.k_proj": blk.k_proj,
                "self_attn.v_proj": blk.v_proj,
                "self_attn.out_proj": blk.out_proj,
                "layer_norm1": blk.layer_norm1,
                "layer_norm2": blk.layer_norm2,
                "mlp.fc1": blk.fc1,
                "mlp.fc2": blk.fc2,
            }
            mod_name, _, leaf = tail.rpartition(".")
            mod = attr_map.get(mod_name)
            if mod is None:
                return False
            getattr(mod, leaf).data.copy_(t)
        else:
            return False
        return True
