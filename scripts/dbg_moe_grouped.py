"""Numerics + latency check: grouped fp8 expert GEMM vs _scaled_mm loop."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from kubeai_amd import _C


def main():
    dev = "cuda"
    T, K, N, E = 64, 4096, 14336, 8
    torch.manual_seed(0)
    x = torch.randn(T, K, device=dev) * 0.3
    xs = (x.abs().amax(dim=1).float() / 448.0).clamp(min=1e-8)
    xq = (x / xs[:, None]).to(torch.float8_e4m3fn)
    wq, wsl, wref = [], [], []
    for e in range(E):
        w = torch.randn(N, K, device=dev) * 0.02
        s = (w.abs().max().float() / 448.0).clamp(min=1e-8)
        wq.append((w / s).to(torch.float8_e4m3fn))
        wsl.append(s)
        wref.append(w)
    ws = torch.stack([s for s in wsl]).float().reshape(E)

    out = torch.empty(E, T, N, dtype=torch.bfloat16, device=dev)
    _C.moe_grouped_fp8(out, xq, xs, wq, ws)
    torch.cuda.synchronize()

    # reference: per-expert _scaled_mm
    for e in range(E):
        refe = torch._scaled_mm(
            xq, wq[e].t(), scale_a=xs[:, None], scale_b=ws[e].reshape(1, 1),
            out_dtype=torch.bfloat16,
        )
        cos = torch.nn.functional.cosine_similarity(
            out[e].float().flatten(), refe.float().flatten(), dim=0
        ).item()
        err = (out[e].float() - refe.float()).abs().max().item()
        print(f"expert {e}: cos={cos:.6f} max_err={err:.4f}")
        assert cos > 0.9999, "numerics mismatch"

    # latency: grouped vs loop
    def timeit(fn, n=50):
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        t0 = time.monotonic()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        return (time.monotonic() - t0) / n * 1e6

    t_g = timeit(lambda: _C.moe_grouped_fp8(out, xq, xs, wq, ws))

    def loop():
        for e in range(E):
            torch._scaled_mm(xq, wq[e].t(), scale_a=xs[:, None],
                             scale_b=ws[e].reshape(1, 1),
                             out_dtype=torch.bfloat16)

    t_l = timeit(loop)
    wbytes = E * N * K
    print(f"grouped: {t_g:.1f} us ({wbytes/t_g/1e3:.0f} GB/s weights)  "
          f"loop: {t_l:.1f} us  speedup {t_l/t_g:.2f}x")


if __name__ == "__main__":
    main()
