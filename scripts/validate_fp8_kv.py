"""Round-2 opener: device validation for the fp8(e5m2) KV cache path.

Run on a MI355X box:

    python scripts/validate_fp8_kv.py          # numerics + microbench
    python bench.py --kv-cache-dtype fp8_e5m2  # then the e2e A/B

Gates (NOTES.md): decode/prefill vs fp32 reference over an e5m2 cache
cos > 0.995 and max-err small at L=4k; cache write bit-exact vs torch's
e5m2 cast. Prints kernel timings vs the bf16-cache baseline.
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
nq, nkv, hd, bs = 32, 8, 128, 16
scale = 1.0 / math.sqrt(hd)

# ---- cache write: kernel output must equal torch's e5m2 cast ----------
T = 333
k = torch.randn(T, nkv, hd, dtype=torch.bfloat16, device=dev)
v = torch.randn_like(k)
nb = T // bs + 2
kc8 = torch.zeros(nb, nkv, bs, hd, dtype=torch.float8_e5m2, device=dev)
vc8 = torch.zeros_like(kc8)
slots = torch.arange(T, dtype=torch.int64, device=dev)
ops.reshape_and_cache(k, v, kc8, vc8, slots)
want_k = torch.zeros(nb, nkv, bs, hd, dtype=torch.float8_e5m2)
want_v = torch.zeros_like(want_k)
ref.reshape_and_cache(k.cpu(), v.cpu(), want_k, want_v, slots.cpu())
mism = (kc8.cpu().view(torch.uint8) != want_k.view(torch.uint8)).sum()
print(f"cache write: {int(mism)} byte mismatches of {kc8.numel()} "
      f"(HIP __hip_cvt vs torch RNE; a tiny count at rounding ties is ok)")

# ---- attention numerics over an fp8 cache -----------------------------
for B, L in ((4, 4096), (40, 1536)):
    nb_per = (L + bs - 1) // bs
    nb = B * nb_per + 1
    kc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=dev)
    vc = torch.randn_like(kc)
    kc8 = kc.to(torch.float8_e5m2)
    vc8 = vc.to(torch.float8_e5m2)
    bt = torch.arange(1, nb, dtype=torch.int32, device=dev).reshape(B, nb_per)
    sl = torch.full((B,), L, dtype=torch.int32, device=dev)
    qd = torch.randn(B, nq, hd, dtype=torch.bfloat16, device=dev)

    out = ops.paged_attention_decode(qd, kc8, vc8, bt, sl, scale)
    want = ref.paged_attention_decode(
        qd.float(), kc8.float(), vc8.float(), bt, sl, scale
    )
    cos = torch.nn.functional.cosine_similarity(
        out.float().flatten(), want.flatten(), dim=0
    )
    err = (out.float() - want).abs().max().item()
    status = "OK" if cos > 0.995 and err < 0.1 else "FAIL"
    print(f"decode  B={B:3d} L={L:5d}: cos={cos:.5f} max_err={err:.4f} {status}")

    us8 = timeit(lambda: ops.paged_attention_decode(qd, kc8, vc8, bt, sl, scale))
    usb = timeit(lambda: ops.paged_attention_decode(qd, kc, vc, bt, sl, scale))
    print(f"        fp8 {us8:8.1f} us vs bf16 {usb:8.1f} us "
          f"({usb / us8:.2f}x)")

# prefill over fp8 cache (single 2048-token seq)
Tq, L = 2048, 2048
nb = L // bs + 1
kc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=dev)
vc = torch.randn_like(kc)
kc8, vc8 = kc.to(torch.float8_e5m2), vc.to(torch.float8_e5m2)
bt = torch.arange(1, nb, dtype=torch.int32, device=dev).reshape(1, -1)
qsl = torch.tensor([0, Tq], dtype=torch.int32, device=dev)
sl = torch.tensor([L], dtype=torch.int32, device=dev)
qp = torch.randn(Tq, nq, hd, dtype=torch.bfloat16, device=dev)
out = ops.paged_attention_prefill(qp, kc8, vc8, bt, qsl, sl, scale)
want = ref.paged_attention_prefill(
    qp.float(), kc8.float(), vc8.float(), bt, qsl, sl, scale
)
cos = torch.nn.functional.cosine_similarity(
    out.float().flatten(), want.flatten(), dim=0
)
err = (out.float() - want).abs().max().item()
status = "OK" if cos > 0.995 and err < 0.1 else "FAIL"
print(f"prefill Tq={Tq} L={L}: cos={cos:.5f} max_err={err:.4f} {status}")
