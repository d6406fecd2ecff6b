"""Per-kernel microbenchmarks vs roofline (run on MI355X via gpurun).

Prints one line per op: time, achieved GB/s or TFLOP/s, % of the relevant
ceiling (HBM ~= 6.3 TB/s achievable, bf16 MFMA ~= 2.5 PF dense).
"""
import argparse
import math
import sys

import torch

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import kubeai_amd.ops as ops

HBM_GBS = 6300.0
MFMA_TF = 2500.0


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters * 1e3  # us


def report(name, us, bytes_moved=None, flops=None):
    parts = [f"{name:34s} {us:9.1f} us"]
    if bytes_moved:
        gbs = bytes_moved / (us * 1e-6) / 1e9
        parts.append(f"{gbs:8.0f} GB/s ({100*gbs/HBM_GBS:5.1f}% HBM)")
    if flops:
        tf = flops / (us * 1e-6) / 1e12
        parts.append(f"{tf:7.1f} TF ({100*tf/MFMA_TF:5.1f}% MFMA)")
    print("  ".join(parts), flush=True)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--quick", action="store_true")
    args = p.parse_args()
    assert torch.cuda.is_available()
    assert ops.have_hip_ext(), "HIP ext missing"
    dev = "cuda"
    torch.manual_seed(0)

    # ---- rmsnorm: T=4096 rows of H=4096 (8B shape) ----
    T, H = 4096, 4096
    x = torch.randn(T, H, dtype=torch.bfloat16, device=dev)
    res = torch.randn_like(x)
    w = torch.randn(H, dtype=torch.bfloat16, device=dev)
    us = timeit(lambda: ops.rmsnorm(x, w, 1e-5))
    report("rmsnorm 4096x4096", us, bytes_moved=2 * T * H * 2)
    us = timeit(lambda: ops.fused_add_rmsnorm(x, res, w, 1e-5))
    report("fused_add_rmsnorm 4096x4096", us, bytes_moved=4 * T * H * 2)

    # ---- silu_and_mul: 8B MLP shape ----
    xg = torch.randn(4096, 2 * 14336, dtype=torch.bfloat16, device=dev)
    us = timeit(lambda: ops.silu_and_mul(xg))
    report("silu_and_mul 4096x2x14336", us, bytes_moved=3 * 4096 * 14336 * 2)

    # ---- rope: 4096 tokens, 32+8 heads ----
    q = torch.randn(4096, 32, 128, dtype=torch.bfloat16, device=dev)
    k = torch.randn(4096, 8, 128, dtype=torch.bfloat16, device=dev)
    pos = torch.arange(4096, dtype=torch.int32, device=dev)
    from kubeai_amd.ops import ref

    cs = ref.make_cos_sin_cache(128, 8192, 500000.0).to(dev)
    us = timeit(lambda: ops.rope(q, k, pos, cs))
    report("rope 4096tok 32+8h", us, bytes_moved=2 * (q.numel() + k.numel()) * 2)

    # ---- paged decode attention: B=64, L=2048, 8B heads ----
    for B, L in ((64, 2048), (32, 512), (256, 2048)):
        if args.quick and B == 256:
            continue
        nq, nkv, hd, bs = 32, 8, 128, 16
        nb_per = (L + bs - 1) // bs
        nb = B * nb_per + 1
        kc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=dev)
        vc = torch.randn_like(kc)
        bt = torch.arange(1, nb, dtype=torch.int32, device=dev).reshape(B, nb_per)
        sl = torch.full((B,), L, dtype=torch.int32, device=dev)
        qd = torch.randn(B, nq, hd, dtype=torch.bfloat16, device=dev)
        scale = 1.0 / math.sqrt(hd)
        us = timeit(lambda: ops.paged_attention_decode(qd, kc, vc, bt, sl, scale))
        kv_bytes = B * L * nkv * hd * 2 * 2
        report(f"paged_decode B={B} L={L}", us, bytes_moved=kv_bytes)

    # ---- prefill attention: 1 seq x 4096 tokens, 8B heads ----
    for Tq, ctx in ((4096, 0), (512, 3584)):
        nq, nkv, hd, bs = 32, 8, 128, 16
        L = Tq + ctx
        nbk = (L + bs - 1) // bs
        kc = torch.randn(nbk + 1, nkv, bs, hd, dtype=torch.bfloat16, device=dev)
        vc = torch.randn_like(kc)
        bt = torch.arange(1, nbk + 1, dtype=torch.int32, device=dev).reshape(1, nbk)
        sl = torch.tensor([L], dtype=torch.int32, device=dev)
        qsl = torch.tensor([0, Tq], dtype=torch.int32, device=dev)
        qp = torch.randn(Tq, nq, hd, dtype=torch.bfloat16, device=dev)
        scale = 1.0 / math.sqrt(hd)
        us = timeit(
            lambda: ops.paged_attention_prefill(qp, kc, vc, bt, qsl, sl, scale),
            iters=20,
        )
        # causal flops: sum over q of kvlen ~ Tq*(ctx + Tq/2) per head pair
        eff_kv = ctx + Tq / 2
        flops = 2 * 2 * nq * hd * Tq * eff_kv  # QK^T + PV
        report(f"prefill Tq={Tq} ctx={ctx}", us, flops=flops)

    # ---- sampling ----
    logits = torch.randn(64, 128256, dtype=torch.float32, device=dev)
    us = timeit(lambda: ops.greedy_sample(logits))
    report("greedy_sample 64x128256", us, bytes_moved=logits.numel() * 4)
    temps = torch.full((64,), 0.8, device=dev)
    seeds = torch.arange(64, dtype=torch.int64, device=dev)
    us = timeit(lambda: ops.gumbel_sample(logits, temps, seeds, 3))
    report("gumbel_sample 64x128256", us, bytes_moved=logits.numel() * 4)

    # ---- cache write ----
    Tc = 4096
    kc = torch.zeros(Tc // 16 + 1, 8, 16, 128, dtype=torch.bfloat16, device=dev)
    vc = torch.zeros_like(kc)
    kt = torch.randn(Tc, 8, 128, dtype=torch.bfloat16, device=dev)
    vt = torch.randn_like(kt)
    slots = torch.arange(Tc, dtype=torch.int64, device=dev)
    us = timeit(lambda: ops.reshape_and_cache(kt, vt, kc, vc, slots))
    report("reshape_and_cache 4096tok", us, bytes_moved=4 * kt.numel() * 2)


if __name__ == "__main__":
    main()
