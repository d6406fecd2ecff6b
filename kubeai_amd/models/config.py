"""Model architecture configs.

Mirrors the fields of HF `config.json` that the engine needs; presets cover
the BASELINE.json configs (Llama-3 8B / 70B, plus tiny variants for tests).
"""
from __future__ import annotations

import dataclasses
import json
import os


@dataclasses.dataclass
class ModelArchConfig:
    architecture: str = "llama"
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    head_dim: int = 128
    max_position_embeddings: int = 8192
    rms_norm_eps: float = 1e-5
    rope_theta: float = 500000.0
    tie_word_embeddings: bool = False
    bos_token_id: int = 128000
    eos_token_id: int = 128001
    # MoE (Mixtral-style); n_experts == 0 means dense MLP
    num_local_experts: int = 0
    num_experts_per_tok: int = 2
    # Qwen2-style: q/k/v projections carry a bias
    attention_bias: bool = False
    # HF rope_scaling dict (llama3 / linear), None = unscaled
    rope_scaling: dict | None = None
    # sliding-window attention (Mistral family; Gemma2 uses it on
    # alternate layers): 0 = full causal
    sliding_window: int = 0
    # "all": every layer windowed (Mistral); "even": layers 0,2,4,..
    # windowed (Gemma2 layer_types pattern)
    sliding_window_pattern: str = "all"
    # gemma3: every Nth layer is GLOBAL attention (no window, global
    # rope base); the rest are windowed with rope_local_base_freq.
    # 0 disables the pattern.
    global_layer_interval: int = 0
    rope_local_base_freq: float = 0.0
    # gemma3: per-head RMSNorm on q and k before rope
    qk_norm: bool = False
    # granite-3 scalar multipliers (0 = disabled/default)
    embedding_multiplier: float = 0.0
    residual_multiplier: float = 0.0
    logits_scaling: float = 0.0
    attention_multiplier: float = 0.0
    # gemma2 family knobs
    hidden_act: str = "silu"  # or "gelu_pytorch_tanh" (GeGLU)
    norm_plus_one: bool = False  # RMSNorm multiplies by (1 + w)
    scale_embeddings: bool = False  # x *= sqrt(hidden) after embedding
    post_norms: bool = False  # extra post-attn/pre+post-FFN norms
    attn_logit_softcap: float = 0.0  # s = c*tanh(s/c) on attention logits
    final_logit_softcap: float = 0.0  # same cap on lm_head logits
    query_pre_attn_scalar: float = 0.0  # 0 -> head_dim (scale = x^-0.5)
    # multimodal (LLaVA-style): vision-tower config dict + the token id
    # that marks an image slot in the prompt (expanded to n_patches
    # placeholder positions at admission; models/vision.py)
    vision: dict | None = None
    image_token_id: int = -1

    @property
    def n_kv_heads(self) -> int:
        return self.num_key_value_heads

    @staticmethod
    def from_hf_config(path: str) -> "ModelArchConfig":
        with open(os.path.join(path, "config.json")) as f:
            cfg = json.load(f)
        hidden = cfg["hidden_size"]
        n_heads = cfg["num_attention_heads"]
        arch = (cfg.get("architectures") or ["LlamaForCausalLM"])[0]
        gemma2 = arch == "Gemma2ForCausalLM"
        gemma1 = arch == "GemmaForCausalLM"
        gemma3 = arch in ("Gemma3ForCausalLM", "Gemma3ForConditionalGeneration")
        granite = arch == "GraniteForCausalLM"
        vision = None
        image_token_id = -1
        if arch == "Gemma3ForConditionalGeneration":
            # text-only serving of the text tower (the gemma3 SigLIP
            # vision tower is not implemented; LLaVA covers multimodal)
            cfg = {**cfg, **(cfg.get("text_config") or {})}
            hidden = cfg["hidden_size"]
            n_heads = cfg["num_attention_heads"]
        if arch == "LlavaForConditionalGeneration":
            # llava config nests text_config + vision_config
            vision = dict(cfg.get("vision_config") or {})
            vision.setdefault(
                "vision_feature_layer", cfg.get("vision_feature_layer", -2)
            )
            image_token_id = cfg.get("image_token_index", 32000)
            cfg = {**cfg, **(cfg.get("text_config") or {})}
            hidden = cfg["hidden_size"]
            n_heads = cfg["num_attention_heads"]
        eos = cfg.get("eos_token_id", 2)
        if isinstance(eos, list):
            eos = eos[0]
        return ModelArchConfig(
            architecture="mixtral" if "Mixtral" in arch else "llama",
            # Qwen2 always uses qkv bias; Llama-family configs may say so
            attention_bias=bool(
                cfg.get("attention_bias", arch.startswith("Qwen2"))
            ),
            vocab_size=cfg["vocab_size"],
            hidden_size=hidden,
            intermediate_size=cfg["intermediate_size"],
            num_hidden_layers=cfg["num_hidden_layers"],
            num_attention_heads=n_heads,
            num_key_value_heads=cfg.get("num_key_value_heads", n_heads),
            head_dim=cfg.get("head_dim", hidden // n_heads),
            max_position_embeddings=cfg.get("max_position_embeddings", 8192),
            rms_norm_eps=cfg.get("rms_norm_eps", 1e-5),
            rope_theta=cfg.get("rope_theta", 10000.0),
            tie_word_embeddings=cfg.get(
                "tie_word_embeddings", gemma2 or gemma1 or gemma3
            ),
            bos_token_id=cfg.get("bos_token_id", 1),
            eos_token_id=eos,
            num_local_experts=cfg.get("num_local_experts", 0),
            num_experts_per_tok=cfg.get("num_experts_per_tok", 2),
            rope_scaling=cfg.get("rope_scaling"),
            sliding_window=int(cfg.get("sliding_window") or 0),
            sliding_window_pattern="even" if gemma2 else "all",
            hidden_act=(
                "gelu_pytorch_tanh"
                if gemma2 or gemma1 or gemma3
                or "gelu" in str(cfg.get("hidden_act") or "")
                else "silu"
            ),
            norm_plus_one=gemma2 or gemma1 or gemma3,
            scale_embeddings=gemma2 or gemma1 or gemma3,
            post_norms=gemma2 or gemma3,
            global_layer_interval=(
                int(cfg.get("sliding_window_pattern") or 6) if gemma3 else 0
            ),
            rope_local_base_freq=(
                float(cfg.get("rope_local_base_freq") or 10000.0)
                if gemma3 else 0.0
            ),
            qk_norm=gemma3,
            embedding_multiplier=float(
                cfg.get("embedding_multiplier") or 0.0
            ) if granite else 0.0,
            residual_multiplier=float(
                cfg.get("residual_multiplier") or 0.0
            ) if granite else 0.0,
            logits_scaling=float(
                cfg.get("logits_scaling") or 0.0
            ) if granite else 0.0,
            attention_multiplier=float(
                cfg.get("attention_multiplier") or 0.0
            ) if granite else 0.0,
            attn_logit_softcap=float(
                cfg.get("attn_logit_softcapping") or 0.0
            ) if gemma2 else 0.0,
            final_logit_softcap=float(
                cfg.get("final_logit_softcapping") or 0.0
            ) if gemma2 else 0.0,
            query_pre_attn_scalar=float(
                cfg.get("query_pre_attn_scalar") or 0.0
            ) if (gemma2 or gemma3) else 0.0,
            vision=vision,
            image_token_id=image_token_id,
        )


# Presets (BASELINE.json configs + test-size models)
PRESETS: dict[str, ModelArchConfig] = {
    "llama-3-8b": ModelArchConfig(
        # Llama-3.1-8B-Instruct shape (the BASELINE model): 128k context
        # via llama3 rope scaling
        max_position_embeddings=131072,
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192},
    ),
    "llama-3-70b": ModelArchConfig(
        hidden_size=8192,
        intermediate_size=28672,
        num_hidden_layers=80,
        num_attention_heads=64,
        num_key_value_heads=8,
        max_position_embeddings=131072,
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192},
    ),
    # tiny: CPU tests / smoke; same head_dim=128 so HIP kernel paths match
    # Qwen2 family (also the DeepSeek-R1-Distill-Qwen arch): qkv bias,
    # tied embeddings on the small models
    "qwen2-tiny": ModelArchConfig(
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=2,
        num_attention_heads=2,
        num_key_value_heads=1,
        head_dim=128,
        max_position_embeddings=2048,
        bos_token_id=1,
        eos_token_id=2,
        attention_bias=True,
        tie_word_embeddings=True,
        rope_theta=1000000.0,
    ),
    "qwen2-7b": ModelArchConfig(
        vocab_size=152064,
        hidden_size=3584,
        intermediate_size=18944,
        num_hidden_layers=28,
        num_attention_heads=28,
        num_key_value_heads=4,
        head_dim=128,
        max_position_embeddings=32768,
        rms_norm_eps=1e-6,
        rope_theta=1000000.0,
        bos_token_id=151643,
        eos_token_id=151645,
        attention_bias=True,
    ),
    "llama-tiny": ModelArchConfig(
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=2,
        num_attention_heads=2,
        num_key_value_heads=1,
        head_dim=128,
        max_position_embeddings=2048,
        bos_token_id=1,
        eos_token_id=2,
    ),
    # Gemma-2b (v1) shape: MQA (1 kv head, G=8), GeGLU, (1+w) norms,
    # scaled embeddings, head_dim 256 — no post-norms/softcaps/window
    "gemma-2b": ModelArchConfig(
        vocab_size=256000,
        hidden_size=2048,
        intermediate_size=16384,
        num_hidden_layers=18,
        num_attention_heads=8,
        num_key_value_heads=1,
        head_dim=256,
        max_position_embeddings=8192,
        rope_theta=10000.0,
        rms_norm_eps=1e-6,
        tie_word_embeddings=True,
        bos_token_id=2,
        eos_token_id=1,
        hidden_act="gelu_pytorch_tanh",
        norm_plus_one=True,
        scale_embeddings=True,
    ),
    "gemma-tiny": ModelArchConfig(
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=2,
        num_attention_heads=2,
        num_key_value_heads=1,
        head_dim=256,
        max_position_embeddings=2048,
        rope_theta=10000.0,
        rms_norm_eps=1e-6,
        tie_word_embeddings=True,
        bos_token_id=2,
        eos_token_id=1,
        hidden_act="gelu_pytorch_tanh",
        norm_plus_one=True,
        scale_embeddings=True,
    ),
    # Phi-4 shape (Phi3ForCausalLM): llama structure with fused
    # qkv_proj / gate_up_proj checkpoint tensors, untied head
    "phi-4": ModelArchConfig(
        vocab_size=100352,
        hidden_size=5120,
        intermediate_size=17920,
        num_hidden_layers=40,
        num_attention_heads=40,
        num_key_value_heads=10,
        head_dim=128,
        max_position_embeddings=16384,
        rope_theta=250000.0,
        rms_norm_eps=1e-5,
        bos_token_id=100257,
        eos_token_id=100257,
    ),
    # Gemma2-2b shape: GeGLU, (1+w) norms, post-norms, logit softcaps,
    # head_dim 256, alternate-layer sliding window
    "gemma2-2b": ModelArchConfig(
        vocab_size=256000,
        hidden_size=2304,
        intermediate_size=9216,
        num_hidden_layers=26,
        num_attention_heads=8,
        num_key_value_heads=4,
        head_dim=256,
        max_position_embeddings=8192,
        rope_theta=10000.0,
        rms_norm_eps=1e-6,
        tie_word_embeddings=True,
        bos_token_id=2,
        eos_token_id=1,
        sliding_window=4096,
        sliding_window_pattern="even",
        hidden_act="gelu_pytorch_tanh",
        norm_plus_one=True,
        scale_embeddings=True,
        post_norms=True,
        attn_logit_softcap=50.0,
        final_logit_softcap=30.0,
        query_pre_attn_scalar=256.0,
    ),
    # tiny gemma2 for CPU tests: every architectural wrinkle, small dims
    "gemma2-tiny": ModelArchConfig(
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=2,
        num_attention_heads=2,
        num_key_value_heads=1,
        head_dim=256,
        max_position_embeddings=2048,
        rope_theta=10000.0,
        rms_norm_eps=1e-6,
        tie_word_embeddings=True,
        bos_token_id=2,
        eos_token_id=1,
        sliding_window=64,
        sliding_window_pattern="even",
        hidden_act="gelu_pytorch_tanh",
        norm_plus_one=True,
        scale_embeddings=True,
        post_norms=True,
        attn_logit_softcap=50.0,
        final_logit_softcap=30.0,
        query_pre_attn_scalar=256.0,
    ),
    # Granite-3.1-8b dense shape: llama structure + scalar multipliers
    "granite-3-8b": ModelArchConfig(
        vocab_size=49155,
        hidden_size=4096,
        intermediate_size=12800,
        num_hidden_layers=40,
        num_attention_heads=32,
        num_key_value_heads=8,
        head_dim=128,
        max_position_embeddings=131072,
        rope_theta=10000000.0,
        tie_word_embeddings=True,
        bos_token_id=0,
        eos_token_id=0,
        embedding_multiplier=12.0,
        residual_multiplier=0.22,
        logits_scaling=16.0,
        attention_multiplier=0.0078125,
    ),
    "granite-tiny": ModelArchConfig(
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=2,
        num_attention_heads=2,
        num_key_value_heads=1,
        head_dim=128,
        max_position_embeddings=2048,
        tie_word_embeddings=True,
        bos_token_id=1,
        eos_token_id=2,
        embedding_multiplier=12.0,
        residual_multiplier=0.22,
        logits_scaling=16.0,
        attention_multiplier=0.0078125,
    ),
    # Gemma3-4b text shape: qk-norm, 5 local(1024-window, theta 10k) :
    # 1 global(theta 1M, linear x8) layer pattern, post-norms, GeGLU
    "gemma3-4b": ModelArchConfig(
        vocab_size=262208,
        hidden_size=2560,
        intermediate_size=10240,
        num_hidden_layers=34,
        num_attention_heads=8,
        num_key_value_heads=4,
        head_dim=256,
        max_position_embeddings=131072,
        rope_theta=1000000.0,
        rope_scaling={"rope_type": "linear", "factor": 8.0},
        rope_local_base_freq=10000.0,
        rms_norm_eps=1e-6,
        tie_word_embeddings=True,
        bos_token_id=2,
        eos_token_id=1,
        sliding_window=1024,
        global_layer_interval=6,
        hidden_act="gelu_pytorch_tanh",
        norm_plus_one=True,
        scale_embeddings=True,
        post_norms=True,
        qk_norm=True,
        query_pre_attn_scalar=256.0,
    ),
    "gemma3-tiny": ModelArchConfig(
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=3,
        num_attention_heads=2,
        num_key_value_heads=1,
        head_dim=256,
        max_position_embeddings=2048,
        rope_theta=1000000.0,
        rope_local_base_freq=10000.0,
        rms_norm_eps=1e-6,
        tie_word_embeddings=True,
        bos_token_id=2,
        eos_token_id=1,
        sliding_window=32,
        global_layer_interval=3,
        hidden_act="gelu_pytorch_tanh",
        norm_plus_one=True,
        scale_embeddings=True,
        post_norms=True,
        qk_norm=True,
        query_pre_attn_scalar=256.0,
    ),
    # Mistral-7B v0.1/v0.2 shape: llama arch + 4096-token sliding window
    "mistral-7b": ModelArchConfig(
        vocab_size=32000,
        hidden_size=4096,
        intermediate_size=14336,
        num_hidden_layers=32,
        num_attention_heads=32,
        num_key_value_heads=8,
        head_dim=128,
        max_position_embeddings=32768,
        rope_theta=10000.0,
        sliding_window=4096,
        bos_token_id=1,
        eos_token_id=2,
    ),
    # tiny windowed model for CPU tests (window smaller than the
    # prompts the tests use, so masking actually bites)
    "mistral-tiny": ModelArchConfig(
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=2,
        num_attention_heads=2,
        num_key_value_heads=1,
        head_dim=128,
        max_position_embeddings=2048,
        rope_theta=10000.0,
        sliding_window=64,
        bos_token_id=1,
        eos_token_id=2,
    ),
    # multimodal tiny: llama-tiny text + 2-layer CLIP tower; 16 patches
    # per image (64px / 16px patches). Image slot token = 99 (synthetic
    # tokenizer's char region stops at 97).
    "llava-tiny": ModelArchConfig(
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=2,
        num_attention_heads=2,
        num_key_value_heads=1,
        head_dim=128,
        max_position_embeddings=2048,
        bos_token_id=1,
        eos_token_id=2,
        image_token_id=99,
        vision={
            "image_size": 64,
            "patch_size": 16,
            "hidden_size": 128,
            "num_hidden_layers": 2,
            "num_attention_heads": 2,
            "intermediate_size": 256,
        },
    ),
    # LLaVA-1.5-7B shape: Vicuna-7B text + CLIP ViT-L/14-336 (576
    # patches/image, features from the penultimate layer)
    "llava-1.5-7b": ModelArchConfig(
        vocab_size=32064,
        hidden_size=4096,
        intermediate_size=11008,
        num_hidden_layers=32,
        num_attention_heads=32,
        num_key_value_heads=32,
        head_dim=128,
        max_position_embeddings=4096,
        rope_theta=10000.0,
        bos_token_id=1,
        eos_token_id=2,
        image_token_id=32000,
        vision={
            "image_size": 336,
            "patch_size": 14,
            "hidden_size": 1024,
            "num_hidden_layers": 24,
            "num_attention_heads": 16,
            "intermediate_size": 4096,
        },
    ),
    # long-context tiny model: exercises chunked prefill + paged decode
    # at tens of thousands of tokens without big weights (the engine
    # sizes KV from free HBM; presets elsewhere are 8k-faithful)
    "llama-tiny-32k": ModelArchConfig(
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=2,
        num_attention_heads=2,
        num_key_value_heads=1,
        head_dim=128,
        max_position_embeddings=32768,
        bos_token_id=1,
        eos_token_id=2,
    ),
    # small: 1-GPU quick bench model (head_dim=128 to stay on the HIP path)
    "llama-1b": ModelArchConfig(
        vocab_size=128256,
        hidden_size=2048,
        intermediate_size=8192,
        num_hidden_layers=16,
        num_attention_heads=16,
        num_key_value_heads=4,
        head_dim=128,
    ),
    # tiny with 2 KV heads so TP=2 shards cleanly (tests/test_tp.py)
    "llama-tiny-tp": ModelArchConfig(
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=128,
        max_position_embeddings=2048,
        bos_token_id=1,
        eos_token_id=2,
    ),
    # TP-shardable multimodal tiny (2 kv heads so TP=2 splits evenly)
    "llava-tiny-tp": ModelArchConfig(
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=128,
        max_position_embeddings=2048,
        bos_token_id=1,
        eos_token_id=2,
        image_token_id=99,
        vision={
            "image_size": 64,
            "patch_size": 16,
            "hidden_size": 128,
            "num_hidden_layers": 2,
            "num_attention_heads": 2,
            "intermediate_size": 256,
        },
    ),
    "mixtral-tiny": ModelArchConfig(
        architecture="mixtral",
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_hidden_layers=2,
        num_attention_heads=2,
        num_key_value_heads=1,
        head_dim=128,
        max_position_embeddings=2048,
        bos_token_id=1,
        eos_token_id=2,
        num_local_experts=4,
        num_experts_per_tok=2,
    ),
    "mixtral-8x7b": ModelArchConfig(
        architecture="mixtral",
        vocab_size=32000,
        hidden_size=4096,
        intermediate_size=14336,
        num_hidden_layers=32,
        num_attention_heads=32,
        num_key_value_heads=8,
        head_dim=128,
        rope_theta=1000000.0,
        bos_token_id=1,
        eos_token_id=2,
        num_local_experts=8,
        num_experts_per_tok=2,
    ),
}
