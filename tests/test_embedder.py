"""BERT-architecture embedding/reranking engine (TextEmbedding/Reranking
features; reference analog: the Infinity engine serving bge/e5 models —
decoder-based embedding models are covered by LLMEngine.embed())."""
import numpy as np
import pytest
import torch
from fastapi.testclient import TestClient

from kubeai_amd.models.bert import (
    PRESETS,
    BertEncoder,
    load_weights_bert,
    save_bert_checkpoint,
)


def test_encode_shapes_and_mask():
    enc = BertEncoder(PRESETS["bert-tiny"], seed=0)
    vecs = enc.encode([[5, 6, 7], [8, 9, 10, 11, 12, 13]])
    assert vecs.shape == (2, 128)
    assert torch.allclose(vecs.norm(dim=-1), torch.ones(2), atol=1e-4)
    # padding must not change a sequence's embedding (mask correctness):
    # same batch, one short + one long vs short alone
    alone = enc.encode([[5, 6, 7]])
    assert torch.allclose(vecs[0], alone[0], atol=1e-5)


def test_encode_deterministic_and_content_sensitive():
    enc = BertEncoder(PRESETS["bert-tiny"], seed=0)
    a = enc.encode([[5, 6, 7, 8]])
    b = enc.encode([[5, 6, 7, 8]])
    c = enc.encode([[5, 6, 7, 9]])
    assert torch.equal(a, b)
    assert not torch.allclose(a, c)


def test_checkpoint_roundtrip(tmp_path):
    src = BertEncoder(PRESETS["bert-tiny"], seed=0)
    ckpt = str(tmp_path / "bert")
    save_bert_checkpoint(src, ckpt)
    dst = BertEncoder(PRESETS["bert-tiny"], seed=777)
    toks = [[4, 5, 6, 7, 8]]
    assert not torch.allclose(dst.encode(toks), src.encode(toks))
    n = load_weights_bert(dst, ckpt)
    assert n > 10
    assert torch.allclose(dst.encode(toks), src.encode(toks))


@pytest.fixture(scope="module")
def embed_client():
    from kubeai_amd.engine.engine import EngineConfig
    from kubeai_amd.engine.server import EngineServer, build_app

    cfg = EngineConfig(model="bert-tiny", device="cpu", num_gpu_blocks=16)
    server = EngineServer(cfg, "embed-model", task="embed")
    server.start()
    server._ready.wait(timeout=60)
    with TestClient(build_app(server)) as c:
        yield c
    server.stop()


def test_embeddings_endpoint(embed_client):
    assert embed_client.get("/health").status_code == 200
    r = embed_client.post(
        "/v1/embeddings",
        json={"model": "embed-model", "input": ["hello world", "other words"]},
    )
    assert r.status_code == 200
    data = r.json()["data"]
    assert len(data) == 2
    v = data[0]["embedding"]
    assert len(v) == 128  # bert-tiny hidden
    assert abs(sum(x * x for x in v) ** 0.5 - 1.0) < 1e-3


def test_rerank_endpoint(embed_client):
    r = embed_client.post(
        "/v1/rerank",
        json={"query": "alpha beta", "top_n": 2,
              "documents": ["alpha beta gamma", "unrelated words here",
                            "alpha beta"]},
    )
    assert r.status_code == 200
    results = r.json()["results"]
    assert len(results) == 2
    assert results[0]["relevance_score"] >= results[1]["relevance_score"]
    # the token-overlapping docs must outrank the unrelated one
    assert 1 not in {res["index"] for res in results}


def test_text_endpoints_rejected(embed_client):
    r = embed_client.post("/v1/completions",
                          json={"prompt": "x", "max_tokens": 1})
    assert r.status_code == 400 and "task=embed" in r.text


@pytest.mark.gpu
def test_bert_gpu_matches_cpu():
    enc_g = BertEncoder(PRESETS["bert-tiny"], device="cuda", seed=0)
    enc_c = BertEncoder(PRESETS["bert-tiny"], device="cpu", seed=0)
    toks = [[5, 6, 7, 8, 9]]
    vg, vc = enc_g.encode(toks).cpu(), enc_c.encode(toks)
    cos = torch.nn.functional.cosine_similarity(
        vg.flatten(), vc.flatten(), dim=0
    )
    assert cos > 0.999


def test_engine_embed_batched_equals_sequential():
    """Packed multi-sequence embed must match one-at-a-time embed."""
    import torch

    from kubeai_amd.engine import EngineConfig, LLMEngine

    eng = LLMEngine(
        EngineConfig(model="llama-tiny", device="cpu", num_gpu_blocks=256,
                     max_model_len=512, seed=5)
    )
    seqs = [list(range(10, 10 + n)) for n in (7, 33, 18, 64, 5)]
    batched = eng.embed(seqs)
    single = [eng.embed([s])[0] for s in seqs]
    for b, s in zip(batched, single):
        cos = torch.nn.functional.cosine_similarity(
            torch.tensor(b), torch.tensor(s), dim=0
        )
        assert cos > 0.9999, cos
    # no blocks leaked
    assert eng.block_manager.usage() == 0.0


def test_tp_lora_rejected():
    from kubeai_amd.engine import EngineConfig
    from kubeai_amd.engine.server import EngineServer

    srv = EngineServer(
        EngineConfig(model="llama-tiny", device="cpu"), "m", tp_size=2
    )
    import pytest as _pytest

    with _pytest.raises(ValueError, match="tensor-parallel"):
        srv.load_lora("a", None)


def test_cross_encoder_rerank_head():
    """Cross-encoder scoring: joint pair forward + classification head
    (bge-reranker analog); differs from bi-encoder cosine and responds
    to the head weights."""
    import torch

    from kubeai_amd.models import bert as bert_mod

    enc = bert_mod.BertEncoder(
        bert_mod.PRESETS["reranker-tiny"], device="cpu", seed=3
    )
    assert enc.cls_head is not None
    q = list(range(200, 220))
    docs = [list(range(300, 330)), list(range(400, 420)), q]
    with torch.no_grad():
        enc.cls_head.out_proj.bias.zero_()
    scores = enc.score_pairs(q, docs)
    assert len(scores) == 3
    # scores change when the head changes (really flows through the head)
    with torch.no_grad():
        enc.cls_head.out_proj.weight.mul_(-1.0)
    flipped = enc.score_pairs(q, docs)
    for a, b in zip(scores, flipped):
        assert abs(a + b) < 1e-3, (a, b)
    # bi-encoder model scores the same inputs differently (cosine path)
    bi = bert_mod.BertEncoder(bert_mod.PRESETS["bert-tiny"], device="cpu",
                              seed=3)
    bi_scores = bi.score_pairs(q, docs)
    assert bi_scores != scores


def test_cross_encoder_checkpoint_roundtrip(tmp_path):
    """classifier.* keys in an HF checkpoint create and fill the head
    (XLMRobertaForSequenceClassification / bge-reranker layout)."""
    import json
    import os

    import torch
    from safetensors.torch import save_file

    from kubeai_amd.models import bert as bert_mod

    src = bert_mod.BertEncoder(bert_mod.PRESETS["bert-tiny"], device="cpu",
                               seed=7)
    ckpt = str(tmp_path / "rr")
    bert_mod.save_bert_checkpoint(src, ckpt)
    # append a roberta-style classification head to the checkpoint
    from safetensors.torch import load_file

    tensors = load_file(os.path.join(ckpt, "model.safetensors"))
    H = src.cfg.hidden_size
    torch.manual_seed(9)
    tensors["classifier.dense.weight"] = torch.randn(H, H) * 0.05
    tensors["classifier.dense.bias"] = torch.zeros(H)
    tensors["classifier.out_proj.weight"] = torch.randn(1, H) * 0.05
    tensors["classifier.out_proj.bias"] = torch.zeros(1)
    save_file(tensors, os.path.join(ckpt, "model.safetensors"))

    dst = bert_mod.BertEncoder(bert_mod.config_from_hf(ckpt), device="cpu")
    bert_mod.load_weights_bert(dst, ckpt)
    assert dst.cls_head is not None and dst.cls_head.dense is not None
    q = list(range(50, 70))
    docs = [list(range(80, 100)), list(range(120, 160))]
    s = dst.score_pairs(q, docs)
    assert len(s) == 2 and s[0] != s[1]
