"""Prefill kernel A/B: run once with KUBEAI_PREFILL_V2=0 (v1) and once
with =1 (v2, default) — the dispatch flag is read once per process.

    KUBEAI_PREFILL_V2=0 python scripts/bench_prefill_v2.py
    KUBEAI_PREFILL_V2=1 python scripts/bench_prefill_v2.py

Prints achieved attention TFLOP/s per shape + numerics vs the fp32
reference (so a fast-but-wrong kernel can never look good).
"""
import math
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import kubeai_amd.ops as ops
from kubeai_amd.ops import ref

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from bench_ops import timeit  # noqa: E402

dev = "cuda"
torch.manual_seed(0)
print(f"KUBEAI_PREFILL_V2={os.environ.get('KUBEAI_PREFILL_V2', '<default 1>')}")

# (nq, nkv, Tq, ctx) — 8B GQA shape, 70B (G=8), chunked continuation
SHAPES = [
    (32, 8, 4096, 0),
    (32, 8, 2048, 0),
    (32, 8, 1024, 0),
    (32, 8, 512, 3584),   # chunked prefill continuation
    (64, 8, 2048, 0),     # G=8 (70B)
    (32, 8, 8192, 0),
]

for nq, nkv, Tq, ctx in SHAPES:
    hd, bs = 128, 16
    L = Tq + ctx
    nb = (L + bs - 1) // bs + 1
    kc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=dev)
    vc = torch.randn_like(kc)
    bt = torch.arange(1, nb, dtype=torch.int32, device=dev).reshape(1, -1)
    sl = torch.tensor([L], dtype=torch.int32, device=dev)
    qsl = torch.tensor([0, Tq], dtype=torch.int32, device=dev)
    q = torch.randn(Tq, nq, hd, dtype=torch.bfloat16, device=dev)
    scale = 1.0 / math.sqrt(hd)

    out = ops.paged_attention_prefill(q, kc, vc, bt, qsl, sl, scale)
    want = ref.paged_attention_prefill(
        q.float(), kc.float(), vc.float(), bt, qsl, sl, scale
    )
    cos = torch.nn.functional.cosine_similarity(
        out.float().flatten(), want.flatten(), dim=0
    ).item()
    err = (out.float() - want).abs().max().item()

    us = timeit(lambda: ops.paged_attention_prefill(q, kc, vc, bt, qsl, sl, scale))
    # causal flops: per q row i attends ctx+i+1 keys; QK + PV => 4*nq*hd
    toks = sum(ctx + i + 1 for i in range(Tq))
    flops = 4.0 * nq * hd * toks
    tf = flops / (us * 1e-6) / 1e12
    status = "OK" if cos > 0.999 and err < 0.1 else "FAIL"
    print(f"nq={nq:3d} Tq={Tq:5d} ctx={ctx:5d}: {us:9.1f} us {tf:7.1f} TF"
          f"  cos={cos:.5f} err={err:.4f} {status}", flush=True)
