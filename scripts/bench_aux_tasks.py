"""Auxiliary-task throughput on one MI355X: STT, embeddings, rerank.

Random-init weights, synthetic inputs — structural throughput numbers
for the non-generate features (reference analogs: FasterWhisper,
Infinity engines).
"""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bench_whisper():
    import numpy as np

    from kubeai_amd.models import whisper as wh

    cfg = wh.PRESETS["whisper-large"]
    model = wh.SpeechToText(cfg, device="cuda")
    sr = 16000
    audio = (np.random.RandomState(0).randn(30 * sr) * 0.1).astype("float32")
    model.transcribe_tokens(audio, sr, max_tokens=32)  # warm
    torch.cuda.synchronize()
    t0 = time.monotonic()
    n = 3
    for _ in range(n):
        model.transcribe_tokens(audio, sr, max_tokens=32)
    torch.cuda.synchronize()
    dt = (time.monotonic() - t0) / n
    print(f"whisper-large 30s clip, 32 tokens greedy: {dt*1e3:.0f} ms "
          f"({30/dt:.1f}x realtime)")


def bench_embed_bert():
    from kubeai_amd.models import bert as bm

    cfg = bm.PRESETS["bge-base"]
    enc = bm.BertEncoder(cfg, device="cuda")
    toks = [[1] + list(range(10, 10 + 126)) + [2] for _ in range(256)]
    enc.encode(toks[:8])
    torch.cuda.synchronize()
    t0 = time.monotonic()
    enc.encode(toks)
    torch.cuda.synchronize()
    dt = time.monotonic() - t0
    total_tok = sum(len(t) for t in toks)
    print(f"bert embed bge-base ({cfg.num_hidden_layers}L/"
          f"{cfg.hidden_size}h): 256 x 128-token docs in {dt*1e3:.0f} ms "
          f"({total_tok/dt:.0f} tok/s)")


def bench_embed_causal():
    from kubeai_amd.engine import EngineConfig, LLMEngine

    eng = LLMEngine(EngineConfig(model="llama-3-8b", device="cuda",
                                 quantization="fp8"))
    toks = [[1] + list(range(100, 100 + 255)) for _ in range(128)]
    eng.embed(toks[:8])
    torch.cuda.synchronize()
    t0 = time.monotonic()
    eng.embed(toks)
    torch.cuda.synchronize()
    dt = time.monotonic() - t0
    total = sum(len(t) for t in toks)
    print(f"causal embed (llama-3-8b fp8): 128 x 256-token docs in "
          f"{dt*1e3:.0f} ms ({total/dt:.0f} tok/s)")


def main():
    for fn in (bench_embed_bert, bench_embed_causal, bench_whisper):
        try:
            fn()
        except Exception as e:  # noqa: BLE001
            print(f"{fn.__name__}: SKIP ({type(e).__name__}: {e})")


if __name__ == "__main__":
    main()
