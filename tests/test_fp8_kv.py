"""fp8(e5m2) KV cache: CPU semantics + conversion-spec tests.

The GPU kernels (reshape_and_cache_fp8 + fp8-templated attention) are
compiled for gfx950 and dispatched when the cache dtype is float8_e5m2;
device validation is scheduled for round 2 (NOTES.md) — the feature stays
off by default (kv_cache_dtype="auto")."""
import pytest
import torch

from kubeai_amd.engine import EngineConfig, LLMEngine, SamplingParams


def _e5m2_to_f32_spec(b: int) -> float:
    """Python mirror of csrc/common.h::e5m2_to_f32 (the kernel's
    conversion); must match torch's float8_e5m2 semantics exactly."""
    import math
    import struct

    s = (b & 0x80) << 24
    e = (b >> 2) & 0x1F
    m = b & 0x3
    if e == 0:
        if m == 0:
            out = s
        else:
            shift = 0 if (m & 2) else 1
            e32 = 127 - 15 + 1 - 1 - shift
            m32 = (m << (shift + 1)) & 0x3
            out = s | (e32 << 23) | (m32 << 21)
    elif e == 0x1F:
        out = s | 0x7F800000 | (m << 21)
    else:
        out = s | ((e - 15 + 127) << 23) | (m << 21)
    return struct.unpack("f", struct.pack("I", out & 0xFFFFFFFF))[0]


def test_e5m2_conversion_spec_matches_torch():
    """All 256 e5m2 byte patterns: the kernel's bit-conversion spec equals
    torch.float8_e5m2 -> float32 (incl. subnormals, inf, nan)."""
    import math

    raw = torch.arange(256, dtype=torch.uint8)
    want = raw.view(torch.float8_e5m2).float()
    for b in range(256):
        got = _e5m2_to_f32_spec(b)
        w = float(want[b])
        if math.isnan(w):
            assert math.isnan(got), f"byte {b:#x}: want nan got {got}"
        else:
            assert got == w, f"byte {b:#x}: want {w} got {got}"


def _gen(kv_dtype, seed=3, max_tokens=8):
    eng = LLMEngine(
        EngineConfig(model="llama-tiny", device="cpu", num_gpu_blocks=128,
                     max_model_len=512, seed=seed, kv_cache_dtype=kv_dtype)
    )
    eng.add_request(list(range(10, 100)),
                    SamplingParams(max_tokens=max_tokens, ignore_eos=True),
                    request_id="r")
    for _ in range(200):
        if not eng.has_work():
            break
        for o in eng.step():
            if o.finished:
                return eng, o.output_token_ids
    raise AssertionError("did not finish")


def test_fp8_kv_cache_cpu_semantics():
    """Engine with an e5m2 KV cache: caches really are fp8, generation
    completes, and logits track the bf16-cache model closely."""
    eng, toks = _gen("fp8_e5m2")
    assert eng.runner.kv_caches[0][0].dtype == torch.float8_e5m2
    assert len(toks) == 8
    # logits comparison on a fresh prefill
    from kubeai_amd.engine.kvcache import BlockManager
    from kubeai_amd.engine.scheduler import Request, Scheduler

    def logits_of(kv_dtype):
        e = LLMEngine(
            EngineConfig(model="llama-tiny", device="cpu", num_gpu_blocks=64,
                         max_model_len=512, seed=3, kv_cache_dtype=kv_dtype)
        )
        s = Scheduler(BlockManager(64, 16), max_num_batched_tokens=512,
                      max_model_len=256)
        s.add_request(Request(list(range(10, 58)), SamplingParams(max_tokens=1),
                              request_id="x"))
        out = s.schedule()
        fb = e.runner.build_batch(out)
        h = e.runner.model(fb)
        return e.runner.model.compute_logits(h[fb.logits_indices]).float()

    a, b = logits_of("auto"), logits_of("fp8_e5m2")
    cos = torch.nn.functional.cosine_similarity(a.flatten(), b.flatten(), dim=0)
    assert cos > 0.98, f"fp8 kv cos {cos}"


def test_fp8_kv_prefix_cache_equivalence():
    """Prefix-cache hits must reproduce the uncached stream under fp8 KV
    (both paths read the same quantized blocks)."""
    eng, first = _gen("fp8_e5m2")
    eng.add_request(list(range(10, 100)),
                    SamplingParams(max_tokens=8, ignore_eos=True),
                    request_id="again")
    for _ in range(200):
        if not eng.has_work():
            break
        for o in eng.step():
            if o.finished:
                assert o.num_cached_tokens >= 80
                assert o.output_token_ids == first
                return
    raise AssertionError("did not finish")


def test_fp8_kv_doubles_capacity():
    from kubeai_amd.engine.runner import ModelRunner
    from kubeai_amd.models.config import PRESETS

    bf16 = ModelRunner(PRESETS["llama-tiny"], device="cpu", num_gpu_blocks=32)
    fp8 = ModelRunner(PRESETS["llama-tiny"], device="cpu", num_gpu_blocks=32,
                      kv_cache_dtype="fp8_e5m2")
    per_bf16 = bf16.kv_caches[0][0].element_size()
    per_fp8 = fp8.kv_caches[0][0].element_size()
    assert per_bf16 == 2 and per_fp8 == 1
