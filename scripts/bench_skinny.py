import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
os.environ["KUBEAI_SKINNY_GEMM"] = "1"
import kubeai_amd.ops as ops

def t(fn, iters=100, warm=20):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); s = torch.cuda.Event(True); e = torch.cuda.Event(True)
    s.record()
    for _ in range(iters): fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1e3

torch.manual_seed(0)
for (name, M, N, K) in (("qkv", 48, 6144, 4096), ("o", 48, 4096, 4096),
                        ("down", 48, 4096, 14336), ("gate_up", 48, 28672, 4096),
                        ("lm_head", 40, 128256, 4096), ("qkv-64", 64, 6144, 4096)):
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    us_h = t(lambda: torch.nn.functional.linear(x, w))
    us_s = t(lambda: ops.linear(x, w))
    gb = N * K * 2 / 1e9
    print(f"{name:9s} M={M:3d} N={N:6d} K={K:5d}  hipblaslt {us_h:7.1f}us ({gb/us_h*1e6:5.0f} GB/s)"
          f"  skinny {us_s:7.1f}us ({gb/us_s*1e6:5.0f} GB/s)  {us_h/us_s:4.2f}x", flush=True)
