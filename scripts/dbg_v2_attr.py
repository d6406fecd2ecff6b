import math, os, sys
sys.path.insert(0, "/root/repo")
import torch
import kubeai_amd.ops as ops

dev = "cuda"
nq, nkv, hd, bs = 32, 8, 128, 16
Tq = 8192
nb = (Tq + bs - 1) // bs + 1
kc = torch.randn(nb, nkv, bs, hd, dtype=torch.bfloat16, device=dev)
vc = torch.randn_like(kc)
bt = torch.arange(1, nb, dtype=torch.int32, device=dev).reshape(1, -1)
qsl = torch.tensor([0, Tq], dtype=torch.int32, device=dev)
sl = torch.tensor([Tq], dtype=torch.int32, device=dev)
q = torch.randn(Tq, nq, hd, dtype=torch.bfloat16, device=dev)
for _ in range(3):
    ops.paged_attention_prefill(q, kc, vc, bt, qsl, sl, 1.0 / math.sqrt(hd))
torch.cuda.synchronize()
print("ok")
