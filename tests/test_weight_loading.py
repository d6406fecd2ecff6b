"""Checkpoint round trip: save engine weights in HF safetensors layout,
reload through the model-dir path, and reproduce greedy generation."""
import pytest

from kubeai_amd.engine import EngineConfig, LLMEngine, SamplingParams
from kubeai_amd.models.loader import save_hf_checkpoint


def gen(eng, prompt, rid):
    eng.add_request(prompt, SamplingParams(max_tokens=6), request_id=rid)
    for _ in range(200):
        if not eng.has_work():
            break
        for o in eng.step():
            if o.finished:
                return o.output_token_ids
    raise AssertionError("did not finish")


@pytest.mark.parametrize("preset", ["llama-tiny", "mixtral-tiny"])
def test_checkpoint_roundtrip(tmp_path, preset):
    src = LLMEngine(
        EngineConfig(model=preset, device="cpu", num_gpu_blocks=128,
                     max_model_len=512, seed=3)
    )
    ckpt = str(tmp_path / "ckpt")
    save_hf_checkpoint(src.runner.model, ckpt)
    dst = LLMEngine(
        EngineConfig(model=ckpt, device="cpu", num_gpu_blocks=128,
                     max_model_len=512, seed=999)  # different init seed
    )
    prompt = list(range(10, 90))
    assert gen(src, prompt, "a") == gen(dst, prompt, "b")


def test_missing_weights_raise(tmp_path):
    import json
    import os

    from safetensors.torch import save_file
    import torch

    ckpt = str(tmp_path / "bad")
    os.makedirs(ckpt)
    # config for llama-tiny but only an embedding tensor
    from kubeai_amd.models.loader import save_hf_checkpoint  # noqa: F401

    cfgd = {
        "architectures": ["LlamaForCausalLM"],
        "vocab_size": 2048, "hidden_size": 256, "intermediate_size": 512,
        "num_hidden_layers": 2, "num_attention_heads": 2,
        "num_key_value_heads": 1, "head_dim": 128,
        "max_position_embeddings": 2048, "rms_norm_eps": 1e-5,
        "rope_theta": 10000.0, "bos_token_id": 1, "eos_token_id": 2,
    }
    with open(os.path.join(ckpt, "config.json"), "w") as f:
        json.dump(cfgd, f)
    save_file(
        {"model.embed_tokens.weight": torch.zeros(2048, 256)},
        os.path.join(ckpt, "model.safetensors"),
    )
    with pytest.raises(ValueError, match="unfilled"):
        LLMEngine(
            EngineConfig(model=ckpt, device="cpu", num_gpu_blocks=64,
                         max_model_len=256)
        )
