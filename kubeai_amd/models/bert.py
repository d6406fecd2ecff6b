"""BERT-family encoder for TextEmbedding / Reranking.

The reference serves these features through the Infinity engine
(internal/modelcontroller/engine_infinity.go) with BGE/e5-style models;
here it is an in-house bidirectional encoder behind the same
`/v1/embeddings` and `/v1/rerank` surface. Decoder-based embedding models
(e5-mistral) are already covered by LLMEngine.embed(); this module covers
the BERT-architecture ones (bge-base, e5-large, bge-reranker).

Execution: one-shot batched encode (no KV cache, bidirectional attention)
— plain PyTorch modules; GEMM-bound work lands on hipBLASLt via rocm
torch. `load_weights_bert` maps HF safetensors names when a checkpoint
directory is provided; random-init under the synthetic tokenizer
otherwise (no offline checkpoints in this environment).
"""
from __future__ import annotations

import dataclasses
import math

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclasses.dataclass
class BertConfig:
    vocab_size: int = 2048
    hidden_size: int = 128
    num_hidden_layers: int = 2
    num_attention_heads: int = 4
    intermediate_size: int = 256
    max_position_embeddings: int = 512
    type_vocab_size: int = 2
    layer_norm_eps: float = 1e-12
    pad_token_id: int = 0
    # cross-encoder rerank head (bge-reranker / ms-marco cross-encoders):
    # score = head(hidden[CLS]) over "[CLS] query [SEP] doc [SEP]"
    cross_encoder: bool = False
    cls_token_id: int = 101
    sep_token_id: int = 102


PRESETS = {
    "bert-tiny": BertConfig(),
    "reranker-tiny": BertConfig(cross_encoder=True),
    # bge-base-en-v1.5 / e5-base dimensions
    "bge-base": BertConfig(
        vocab_size=30522, hidden_size=768, num_hidden_layers=12,
        num_attention_heads=12, intermediate_size=3072,
    ),
    # e5-large / bge-large dimensions
    "e5-large": BertConfig(
        vocab_size=30522, hidden_size=1024, num_hidden_layers=24,
        num_attention_heads=16, intermediate_size=4096,
    ),
}


class _BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        H = cfg.hidden_size
        self.n_head = cfg.num_attention_heads
        self.query = nn.Linear(H, H)
        self.key = nn.Linear(H, H)
        self.value = nn.Linear(H, H)
        self.attn_out = nn.Linear(H, H)
        self.attn_ln = nn.LayerNorm(H, eps=cfg.layer_norm_eps)
        self.ffn_in = nn.Linear(H, cfg.intermediate_size)
        self.ffn_out = nn.Linear(cfg.intermediate_size, H)
        self.ffn_ln = nn.LayerNorm(H, eps=cfg.layer_norm_eps)

    def forward(self, x: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
        B, T, H = x.shape
        h = self.n_head

        def heads(t):
            return t.view(B, T, h, H // h).transpose(1, 2)

        # bidirectional attention; padding masked via additive bias
        bias = (~mask)[:, None, None, :].float() * torch.finfo(torch.float32).min
        o = F.scaled_dot_product_attention(
            heads(self.query(x)), heads(self.key(x)), heads(self.value(x)),
            attn_mask=bias.to(x.dtype),
        )
        o = o.transpose(1, 2).reshape(B, T, H)
        x = self.attn_ln(x + self.attn_out(o))
        return self.ffn_ln(x + self.ffn_out(F.gelu(self.ffn_in(x))))


class _ClsHead(nn.Module):
    """Sequence-classification head on [CLS]: plain linear (BERT
    cross-encoders) or Roberta-style dense+tanh+out_proj (bge-reranker's
    XLMRobertaForSequenceClassification)."""

    def __init__(self, hidden: int, roberta: bool = False):
        super().__init__()
        self.dense = nn.Linear(hidden, hidden) if roberta else None
        self.out_proj = nn.Linear(hidden, 1)

    def forward(self, cls_hidden: torch.Tensor) -> torch.Tensor:
        x = cls_hidden
        if self.dense is not None:
            x = torch.tanh(self.dense(x))
        return self.out_proj(x).squeeze(-1)


class BertEncoder(nn.Module):
    """Encoder + mean pooling; `score_pairs` reranks — cross-encoder
    scoring when the model has a classification head (reference analog:
    Infinity serving bge-reranker), else bi-encoder cosine."""

    def __init__(self, cfg: BertConfig, device: str = "cpu", seed: int = 0):
        super().__init__()
        self.cfg = cfg
        torch.manual_seed(seed)
        H = cfg.hidden_size
        self.word_embeddings = nn.Embedding(cfg.vocab_size, H)
        self.position_embeddings = nn.Embedding(cfg.max_position_embeddings, H)
        self.token_type_embeddings = nn.Embedding(cfg.type_vocab_size, H)
        self.emb_ln = nn.LayerNorm(H, eps=cfg.layer_norm_eps)
        self.layers = nn.ModuleList(
            _BertLayer(cfg) for _ in range(cfg.num_hidden_layers)
        )
        self.cls_head = _ClsHead(H) if cfg.cross_encoder else None
        self.to(device)
        self.device_ = device
        self.eval()

    @torch.inference_mode()
    def encode(self, token_lists: list[list[int]]) -> torch.Tensor:
        """Batched encode -> L2-normalised mean-pooled embeddings [N, H]."""
        if not token_lists:
            return torch.empty(0, self.cfg.hidden_size)
        T = min(
            max(len(t) for t in token_lists), self.cfg.max_position_embeddings
        )
        pad = self.cfg.pad_token_id
        ids = torch.full((len(token_lists), T), pad, dtype=torch.int64)
        mask = torch.zeros((len(token_lists), T), dtype=torch.bool)
        for i, toks in enumerate(token_lists):
            toks = toks[:T]
            ids[i, : len(toks)] = torch.tensor(toks, dtype=torch.int64)
            mask[i, : len(toks)] = True
        ids, mask = ids.to(self.device_), mask.to(self.device_)
        x = self._hidden(ids, mask, torch.zeros_like(ids))
        m = mask[:, :, None].to(x.dtype)
        pooled = (x * m).sum(dim=1) / m.sum(dim=1).clamp_min(1e-9)
        return F.normalize(pooled.float(), dim=-1)

    def _hidden(self, ids, mask, type_ids) -> torch.Tensor:
        pos = torch.arange(ids.shape[1], device=ids.device)[None, :].expand_as(ids)
        x = (
            self.word_embeddings(ids)
            + self.position_embeddings(pos)
            + self.token_type_embeddings(type_ids)
        )
        x = self.emb_ln(x)
        for layer in self.layers:
            x = layer(x, mask)
        return x

    @torch.inference_mode()
    def score_pairs(self, query: list[int], docs: list[list[int]]) -> list[float]:
        if self.cls_head is not None:
            return self._score_cross(query, docs)
        vecs = self.encode([query] + docs)
        q, d = vecs[0], vecs[1:]
        return (d @ q).cpu().tolist()

    def _score_cross(self, query: list[int], docs: list[list[int]]) -> list[float]:
        """Cross-encoder: one joint forward per (query, doc) pair,
        score from the classification head on [CLS]."""
        cfg = self.cfg
        Tmax = cfg.max_position_embeddings
        pairs, types = [], []
        for d in docs:
            toks = [cfg.cls_token_id] + list(query) + [cfg.sep_token_id]
            tt = [0] * len(toks)
            toks += list(d) + [cfg.sep_token_id]
            tt += [1] * (len(toks) - len(tt))
            pairs.append(toks[:Tmax])
            types.append(tt[:Tmax])
        T = max(len(p) for p in pairs)
        pad = cfg.pad_token_id
        ids = torch.full((len(pairs), T), pad, dtype=torch.int64)
        mask = torch.zeros((len(pairs), T), dtype=torch.bool)
        tids = torch.zeros((len(pairs), T), dtype=torch.int64)
        for i, (p, tt) in enumerate(zip(pairs, types)):
            ids[i, : len(p)] = torch.tensor(p, dtype=torch.int64)
            mask[i, : len(p)] = True
            tids[i, : len(tt)] = torch.tensor(tt, dtype=torch.int64)
        ids, mask, tids = (ids.to(self.device_), mask.to(self.device_),
                           tids.to(self.device_))
        x = self._hidden(ids, mask, tids)
        return self.cls_head(x[:, 0].float()).cpu().tolist()


# ----------------------------------------------------------- checkpoint IO
_HF_MAP = [
    ("word_embeddings.", "embeddings.word_embeddings."),
    ("position_embeddings.", "embeddings.position_embeddings."),
    ("token_type_embeddings.", "embeddings.token_type_embeddings."),
    ("emb_ln.", "embeddings.LayerNorm."),
    (".query.", ".attention.self.query."),
    (".key.", ".attention.self.key."),
    (".value.", ".attention.self.value."),
    (".attn_out.", ".attention.output.dense."),
    (".attn_ln.", ".attention.output.LayerNorm."),
    (".ffn_in.", ".intermediate.dense."),
    (".ffn_out.", ".output.dense."),
    (".ffn_ln.", ".output.LayerNorm."),
    ("layers.", "encoder.layer."),
]


def _to_hf(name: str) -> str:
    for ours, hf in _HF_MAP:
        name = name.replace(ours, hf)
    return name


def save_bert_checkpoint(model: BertEncoder, out_dir: str) -> None:
    import json
    import os

    from safetensors.torch import save_file

    os.makedirs(out_dir, exist_ok=True)
    state = {
        _to_hf(n): p.detach().cpu().contiguous()
        for n, p in model.state_dict().items()
    }
    save_file(state, os.path.join(out_dir, "model.safetensors"))
    cfg = model.cfg
    with open(os.path.join(out_dir, "config.json"), "w") as f:
        json.dump({"architectures": ["BertModel"],
                   "vocab_size": cfg.vocab_size,
                   "hidden_size": cfg.hidden_size,
                   "num_hidden_layers": cfg.num_hidden_layers,
                   "num_attention_heads": cfg.num_attention_heads,
                   "intermediate_size": cfg.intermediate_size,
                   "max_position_embeddings": cfg.max_position_embeddings,
                   "type_vocab_size": cfg.type_vocab_size,
                   "layer_norm_eps": cfg.layer_norm_eps,
                   "pad_token_id": cfg.pad_token_id}, f)


def config_from_hf(path: str) -> BertConfig:
    import json
    import os

    with open(os.path.join(path, "config.json")) as f:
        hc = json.load(f)
    return BertConfig(
        vocab_size=hc.get("vocab_size", 30522),
        hidden_size=hc.get("hidden_size", 768),
        num_hidden_layers=hc.get("num_hidden_layers", 12),
        num_attention_heads=hc.get("num_attention_heads", 12),
        intermediate_size=hc.get("intermediate_size", 3072),
        max_position_embeddings=hc.get("max_position_embeddings", 512),
        type_vocab_size=hc.get("type_vocab_size", 2),
        layer_norm_eps=hc.get("layer_norm_eps", 1e-12),
        pad_token_id=hc.get("pad_token_id", 0),
    )


def load_weights_bert(model: BertEncoder, model_dir: str) -> int:
    import glob
    import os

    from safetensors import safe_open

    inverse = {_to_hf(n): n for n in model.state_dict()}
    sd = model.state_dict()
    filled = set()
    files = sorted(glob.glob(os.path.join(model_dir, "*.safetensors")))
    if not files:
        raise FileNotFoundError(f"no *.safetensors under {model_dir}")
    cls_keys = {}
    for fpath in files:
        with safe_open(fpath, framework="pt", device="cpu") as sf:
            for key in sf.keys():
                if key.startswith("classifier."):
                    cls_keys[key] = sf.get_tensor(key)
                    continue
                k = key[len("bert."):] if key.startswith("bert.") else key
                k = k[len("roberta."):] if k.startswith("roberta.") else k
                tgt = inverse.get(k)
                if tgt is None:
                    continue  # pooler etc.
                with torch.no_grad():
                    sd[tgt].copy_(sf.get_tensor(key).to(sd[tgt].dtype))
                filled.add(tgt)
    if cls_keys:
        roberta = "classifier.dense.weight" in cls_keys
        if model.cls_head is None:
            model.cls_head = _ClsHead(model.cfg.hidden_size, roberta=roberta)
            model.cls_head.to(model.device_)
        with torch.no_grad():
            if roberta:
                model.cls_head.dense.weight.copy_(cls_keys["classifier.dense.weight"])
                model.cls_head.dense.bias.copy_(cls_keys["classifier.dense.bias"])
                model.cls_head.out_proj.weight.copy_(
                    cls_keys["classifier.out_proj.weight"])
                model.cls_head.out_proj.bias.copy_(
                    cls_keys["classifier.out_proj.bias"])
            else:
                model.cls_head.out_proj.weight.copy_(cls_keys["classifier.weight"])
                model.cls_head.out_proj.bias.copy_(cls_keys["classifier.bias"])
    missing = set(sd) - filled
    if missing:
        raise ValueError(f"unfilled bert parameters: {sorted(missing)[:8]}")
    return len(filled)
