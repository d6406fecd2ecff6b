"""Engine integration on the GPU (HIP kernel path end-to-end).

Validates cross-kernel consistency (prefill vs decode vs prefix-cached
paths must yield identical greedy generations) and CPU-reference logits
agreement with shared weights.
"""
import pytest
import torch

from kubeai_amd.engine import EngineConfig, LLMEngine, SamplingParams

pytestmark = pytest.mark.gpu


def make_engine(device, **kw):
    base = dict(
        model="llama-tiny",
        device=device,
        max_model_len=512,
        num_gpu_blocks=256,
        seed=0,
    )
    base.update(kw)
    return LLMEngine(EngineConfig(**base))


def drain(eng, max_steps=300):
    outs = {}
    for _ in range(max_steps):
        if not eng.has_work():
            break
        for o in eng.step():
            if o.finished:
                outs[o.request_id] = o
    assert not eng.has_work()
    return outs


def test_greedy_deterministic_gpu():
    prompt = list(range(10, 140))
    a = []
    for _ in range(2):
        eng = make_engine("cuda")
        eng.add_request(prompt, SamplingParams(max_tokens=12), request_id="r")
        a.append(drain(eng)["r"].output_token_ids)
    assert a[0] == a[1]


def test_prefix_cache_equivalence_gpu():
    prompt = list(range(5, 120))
    eng = make_engine("cuda")
    eng.add_request(prompt, SamplingParams(max_tokens=8), request_id="cold")
    cold = drain(eng)["cold"]
    eng.add_request(prompt, SamplingParams(max_tokens=8), request_id="warm")
    warm = drain(eng)["warm"]
    assert warm.num_cached_tokens >= 96
    assert warm.output_token_ids == cold.output_token_ids


def test_chunked_prefill_equivalence_gpu():
    prompt = list(range(5, 260))
    eng1 = make_engine("cuda", max_num_batched_tokens=4096)
    eng1.add_request(prompt, SamplingParams(max_tokens=6), request_id="a")
    full = drain(eng1)["a"].output_token_ids
    eng2 = make_engine("cuda", max_num_batched_tokens=96)
    eng2.add_request(prompt, SamplingParams(max_tokens=6), request_id="a")
    chunked = drain(eng2)["a"].output_token_ids
    assert full == chunked


def test_gpu_logits_match_cpu_reference():
    # identical weights on both devices; compare last-token logits of a
    # prefill — full-stack numerics check of the HIP path vs torch ref
    gpu = make_engine("cuda")
    cpu = make_engine("cpu")
    cpu.runner.model.load_state_dict(
        {k: v.cpu() for k, v in gpu.runner.model.state_dict().items()}
    )
    prompt = list(range(20, 150))
    for eng, rid in ((gpu, "g"), (cpu, "c")):
        eng.add_request(prompt, SamplingParams(max_tokens=4), request_id=rid)
    og = drain(gpu)["g"].output_token_ids
    oc = drain(cpu)["c"].output_token_ids
    # bf16 GEMM backends differ (hipBLASLt vs CPU); require the greedy
    # argmax to agree on at least the first generated token
    assert og[0] == oc[0], f"gpu={og} cpu={oc}"


def test_lora_adapter_gpu():
    """Adapter serving on the HIP path: the SGMV kernel must shift the
    adapter stream's logits while leaving the base stream bit-stable."""
    import torch

    from kubeai_amd.engine.scheduler import (
        Request,
        SamplingParams as SP,
        ScheduledSeq,
        SchedulerOutput,
    )

    eng = make_engine("cuda")
    eng.runner.load_lora(7, None)  # synthetic rank-16 adapter
    prompt = list(range(10, 100))

    def last_logits(lora_id, salt):
        toks = list(prompt)
        table, _ = eng.block_manager.allocate(toks, salt=salt, max_cached=0)
        req = Request(toks, SP(max_tokens=1), lora_id=lora_id)
        req.block_table = table
        so = SchedulerOutput(
            decode=[], prefill=[ScheduledSeq(req, 0, len(toks))], preempted=[]
        )
        fb = eng.runner.build_batch(so)
        h = eng.runner.model(fb)
        logits = eng.runner.model.compute_logits(h[len(toks) - 1 : len(toks)])[0]
        eng.block_manager.free(table)
        return logits

    base1 = last_logits(0, -10)
    ad = last_logits(7, -11)
    base2 = last_logits(0, -12)
    assert torch.equal(base1, base2)  # base stream untouched by the adapter
    diff = (ad - base1).abs().max().item()
    assert diff > 0.05, f"adapter did not move logits (max diff {diff})"
    ad2 = last_logits(7, -13)
    torch.testing.assert_close(ad, ad2)  # deterministic adapter stream


def test_mixtral_moe_gpu():
    """Mixtral-style MoE: runs through the HIP kernel path, deterministic,
    prefix-cache-consistent."""
    eng = LLMEngine(
        EngineConfig(
            model="mixtral-tiny",
            device="cuda",
            max_model_len=512,
            num_gpu_blocks=256,
            seed=0,
        )
    )
    prompt = list(range(10, 100))
    eng.add_request(prompt, SamplingParams(max_tokens=8), request_id="a")
    a = drain(eng)["a"]
    assert len(a.output_token_ids) == 8
    eng.add_request(prompt, SamplingParams(max_tokens=8), request_id="b")
    b = drain(eng)["b"]
    assert b.num_cached_tokens >= 80
    assert b.output_token_ids == a.output_token_ids


def test_fp8_quantized_engine():
    """fp8 W8A8 serving: logits must track the bf16 model closely and
    generation must run NaN-free end to end."""
    import torch

    bf16 = make_engine("cuda")
    fp8 = LLMEngine(
        EngineConfig(
            model="llama-tiny",
            device="cuda",
            max_model_len=512,
            num_gpu_blocks=256,
            seed=0,
            quantization="fp8",
        )
    )
    prompt = list(range(10, 120))
    for eng, rid in ((bf16, "a"), (fp8, "b")):
        eng.add_request(prompt, SamplingParams(max_tokens=8), request_id=rid)
    oa = drain(bf16)["a"]
    ob = drain(fp8)["b"]
    assert len(ob.output_token_ids) == 8
    # compare prefill logits of identical weights under both precisions
    from kubeai_amd.engine.scheduler import (
        Request,
        SamplingParams as SP,
        ScheduledSeq,
        SchedulerOutput,
    )

    def last_logits(eng):
        toks = list(range(10, 120))
        table, _ = eng.block_manager.allocate(toks, salt=-2, max_cached=0)
        req = Request(toks, SP(max_tokens=1))
        req.block_table = table
        so = SchedulerOutput(
            decode=[], prefill=[ScheduledSeq(req, 0, len(toks))], preempted=[]
        )
        fb = eng.runner.build_batch(so)
        h = eng.runner.model(fb)
        logits = eng.runner.model.compute_logits(h[len(toks) - 1 : len(toks)])
        eng.block_manager.free(table)
        return logits[0]

    la, lb = last_logits(bf16), last_logits(fp8)
    assert torch.isfinite(lb).all()
    cos = torch.nn.functional.cosine_similarity(la, lb, dim=0)
    assert cos > 0.98, f"fp8 logits diverged: cos={cos}"


def test_batch_throughput_many_seqs_gpu():
    eng = make_engine("cuda", max_num_seqs=64)
    for i in range(32):
        eng.add_request(
            list(range(i + 3, i + 90)),
            SamplingParams(max_tokens=8),
            request_id=f"r{i}",
        )
    outs = drain(eng, max_steps=600)
    assert len(outs) == 32
    for o in outs.values():
        assert len(o.output_token_ids) == 8


def test_fp8_moe_engine():
    """Fused MoE fp8 (one activation quant per layer, fp8 expert gathers):
    generation NaN-free; logits track the bf16 MoE model."""
    import torch

    bf16 = make_engine("cuda", model="mixtral-tiny", seed=4)
    fp8 = make_engine("cuda", model="mixtral-tiny", seed=4, quantization="fp8")
    mlp = fp8.runner.model.layers[0].mlp
    assert hasattr(mlp.experts[0].gate_up_proj, "forward_quantized")
    prompt = list(range(10, 120))
    for eng, rid in ((bf16, "a"), (fp8, "b")):
        eng.add_request(prompt, SamplingParams(max_tokens=8), request_id=rid)
    oa = drain(bf16)["a"]
    ob = drain(fp8)["b"]
    assert len(ob.output_token_ids) == 8

    def logits_of(eng):
        from kubeai_amd.engine.kvcache import BlockManager
        from kubeai_amd.engine.scheduler import Request, Scheduler
        from kubeai_amd.engine.scheduler import SamplingParams as SP

        s = Scheduler(BlockManager(64, 16), max_num_batched_tokens=512,
                      max_model_len=256)
        s.add_request(Request(list(range(10, 58)), SP(max_tokens=1),
                              request_id="x"))
        out = s.schedule()
        fb = eng.runner.build_batch(out)
        h = eng.runner.model(fb)
        return eng.runner.model.compute_logits(h[fb.logits_indices]).float()

    a, b = logits_of(bf16), logits_of(fp8)
    cos = torch.nn.functional.cosine_similarity(a.flatten(), b.flatten(), dim=0)
    assert cos > 0.95, f"MoE fp8 cos {cos}"
    assert torch.isfinite(b).all()


def test_qwen2_engine_gpu():
    """Qwen2 arch on the HIP path: qkv bias + tied embeddings generate
    deterministically and match a fresh engine's stream."""
    a = make_engine("cuda", model="qwen2-tiny", seed=11)
    b = make_engine("cuda", model="qwen2-tiny", seed=11)
    prompt = list(range(10, 100))
    for eng, rid in ((a, "x"), (b, "y")):
        eng.add_request(prompt, SamplingParams(max_tokens=8, ignore_eos=True),
                        request_id=rid)
    oa = drain(a)["x"]
    ob = drain(b)["y"]
    assert len(oa.output_token_ids) == 8
    assert oa.output_token_ids == ob.output_token_ids
    assert a.runner.model.layers[0].self_attn.qkv_proj.bias is not None


def test_hf_checkpoint_dir_e2e_gpu(tmp_path):
    """Real-weights e2e on device: an HF-format safetensors dir loads
    through the engine (cache-dir layout) and reproduces the greedy
    generation of the in-memory model it was saved from."""
    import torch

    from kubeai_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from kubeai_amd.models.loader import save_hf_checkpoint

    def gen(eng, rid):
        eng.add_request(list(range(10, 200)),
                        SamplingParams(max_tokens=12), request_id=rid)
        for _ in range(200):
            if not eng.has_work():
                break
            for o in eng.step():
                if o.finished:
                    return o.output_token_ids
        raise AssertionError("did not finish")

    src = LLMEngine(EngineConfig(model="llama-tiny", device="cuda",
                                 num_gpu_blocks=256, max_model_len=512,
                                 seed=11))
    ckpt = str(tmp_path / "hf-model")
    save_hf_checkpoint(src.runner.model, ckpt)
    want = gen(src, "src")
    del src
    torch.cuda.empty_cache()
    dst = LLMEngine(EngineConfig(model=ckpt, device="cuda",
                                 num_gpu_blocks=256, max_model_len=512,
                                 seed=999))
    got = gen(dst, "dst")
    assert got == want


def test_long_context_generation_gpu():
    """Long-context serving path: a 20k-token prompt prefills in chunks
    through the paged cache and decodes correctly (presets
    were 8k-capped with no long run)."""
    eng = LLMEngine(
        EngineConfig(model="llama-tiny-32k", device="cuda",
                     max_model_len=24576, max_num_batched_tokens=8192,
                     gpu_memory_utilization=0.5)
    )
    prompt = [(7 + 13 * i) % 1900 + 100 for i in range(20000)]
    eng.add_request(prompt, SamplingParams(max_tokens=8, ignore_eos=True),
                    request_id="long")
    out = None
    for _ in range(200):
        if not eng.has_work():
            break
        for o in eng.step():
            if o.finished:
                out = o
    assert out is not None and len(out.output_token_ids) == 8
    # prefix-cache equivalence at long length: resubmitting the same
    # prompt hits the cache and reproduces the same greedy tokens
    eng.add_request(prompt, SamplingParams(max_tokens=8, ignore_eos=True),
                    request_id="long2")
    out2 = None
    for _ in range(200):
        if not eng.has_work():
            break
        for o in eng.step():
            if o.finished:
                out2 = o
    assert out2.num_cached_tokens > 15000
    assert out2.output_token_ids == out.output_token_ids


@pytest.mark.gpu
def test_multimodal_llava_gpu():
    """LLaVA path on device: image embeddings spliced into prefill, greedy
    reproducibility with the prefix cache salted by image content."""
    import base64

    from kubeai_amd.utils import imaging

    g = torch.Generator().manual_seed(11)
    img = torch.randint(0, 256, (40, 56, 3), generator=g, dtype=torch.uint8)
    px = imaging.preprocess(img, 64)
    eng = LLMEngine(EngineConfig(model="llava-tiny", device="cuda",
                                 num_gpu_blocks=256, max_model_len=512))
    img_id = eng.arch.image_token_id
    prompt = [1, 4, img_id, 200, 300, 400]

    def run(rid, pixels):
        eng.add_request(prompt, SamplingParams(max_tokens=6, ignore_eos=True),
                        request_id=rid, images=[pixels])
        out = None
        for _ in range(100):
            if not eng.has_work():
                break
            for o in eng.step():
                if o.finished and o.request_id == rid:
                    out = o
        return out

    o1 = run("mm1", px)
    o2 = run("mm2", px)  # same image: cache hit + identical greedy tokens
    assert o1 is not None and len(o1.output_token_ids) == 6
    assert o2.output_token_ids == o1.output_token_ids
    assert o2.num_cached_tokens > 0
    # different image, same tokens: salt isolates the prefix cache
    g2 = torch.Generator().manual_seed(99)
    img2 = torch.randint(0, 256, (40, 56, 3), generator=g2, dtype=torch.uint8)
    o3 = run("mm3", imaging.preprocess(img2, 64))
    assert o3 is not None and o3.num_cached_tokens == 0


@pytest.mark.gpu
def test_gemma3_engine_e2e_gpu():
    """gemma3 on device: qk-norm + dual rope + local:global layers."""
    eng = LLMEngine(EngineConfig(model="gemma3-tiny", device="cuda",
                                 num_gpu_blocks=256, max_model_len=512))
    prompt = [2] + list(range(100, 170))
    eng.add_request(prompt, SamplingParams(max_tokens=6, ignore_eos=True),
                    request_id="g3")
    out = None
    for _ in range(100):
        if not eng.has_work():
            break
        for o in eng.step():
            if o.finished:
                out = o
    assert out is not None and len(out.output_token_ids) == 6


@pytest.mark.gpu
def test_granite_engine_e2e_gpu():
    eng = LLMEngine(EngineConfig(model="granite-tiny", device="cuda",
                                 num_gpu_blocks=256, max_model_len=512))
    eng.add_request([1] + list(range(100, 140)),
                    SamplingParams(max_tokens=6, ignore_eos=True),
                    request_id="gr")
    out = None
    for _ in range(100):
        if not eng.has_work():
            break
        for o in eng.step():
            if o.finished:
                out = o
    assert out is not None and len(out.output_token_ids) == 6


@pytest.mark.gpu
def test_mistral_sliding_window_e2e_gpu():
    """Windowed model end to end on device: prompt longer than the
    window, greedy reproduction through the prefix cache."""
    eng = LLMEngine(EngineConfig(model="mistral-tiny", device="cuda",
                                 num_gpu_blocks=256, max_model_len=512))
    assert eng.arch.sliding_window == 64
    prompt = [1] + list(range(100, 220))  # 121 tokens > window

    def run(rid):
        eng.add_request(prompt, SamplingParams(max_tokens=6, ignore_eos=True),
                        request_id=rid)
        out = None
        for _ in range(100):
            if not eng.has_work():
                break
            for o in eng.step():
                if o.finished:
                    out = o
        return out

    a = run("w1")
    b = run("w2")
    assert a is not None and len(a.output_token_ids) == 6
    assert b.output_token_ids == a.output_token_ids
    assert b.num_cached_tokens > 0
