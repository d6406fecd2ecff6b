"""Sweep KUBEAI_DECODE_SPLIT_TARGET for the flash-decoding kernel.

The split target is read once per process (static), so run this script
once per value:

    for t in 512 1024 2048 4096 8192; do
        KUBEAI_DECODE_SPLIT_TARGET=$t python scripts/sweep_decode_splits.py
    done

Shapes: the bench steady state (B=40, L~1.5k), a long-context case and a
large-batch case. Also numerics-checks against the torch reference so a
bad split count cannot silently pass.
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
target = os.environ.get("KUBEAI_DECODE_SPLIT_TARGET", "2048")
print(f"== split target {target} ==", flush=True)
torch.manual_seed(0)
for B, L in ((40, 1536), (8, 8192), (256, 2048)):
    nq, nkv, hd, bs = 32, 8, 128, 16
    nb_per = (L + bs - 1) // bs
    nb = B * nb_per + 1
    kc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=dev)
    vc = torch.randn_like(kc)
    bt = torch.arange(1, nb, dtype=torch.int32, device=dev).reshape(B, nb_per)
    sl = torch.full((B,), L, dtype=torch.int32, device=dev)
    qd = torch.randn(B, nq, hd, dtype=torch.bfloat16, device=dev)
    scale = 1.0 / math.sqrt(hd)
    out = ops.paged_attention_decode(qd, kc, vc, bt, sl, scale)
    want = ref.paged_attention_decode(
        qd.float(), kc.float(), vc.float(), bt, sl, scale
    )
    err = (out.float() - want).abs().max().item()
    assert err < 0.05, f"numerics broke at target={target}: max err {err}"
    us = timeit(lambda: ops.paged_attention_decode(qd, kc, vc, bt, sl, scale))
    kv_bytes = B * L * nkv * hd * 2 * 2
    gbs = kv_bytes / (us * 1e-6) / 1e9
    print(f"B={B:4d} L={L:5d}  {us:8.1f} us  {gbs:7.0f} GB/s  err={err:.4f}",
          flush=True)
