import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math, torch
import kubeai_amd.ops as ops
torch.manual_seed(0)
nq, nkv, hd, bs = 32, 8, 128, 16
Tq = 4096
nbk = Tq // bs
kc = torch.randn(nbk + 1, nkv, bs, hd, dtype=torch.bfloat16, device="cuda")
vc = torch.randn_like(kc)
bt = torch.arange(1, nbk + 1, dtype=torch.int32, device="cuda").reshape(1, nbk)
sl = torch.tensor([Tq], dtype=torch.int32, device="cuda")
qsl = torch.tensor([0, Tq], dtype=torch.int32, device="cuda")
qp = torch.randn(Tq, nq, hd, dtype=torch.bfloat16, device="cuda")
for _ in range(5):
    ops.paged_attention_prefill(qp, kc, vc, bt, qsl, sl, 1.0 / math.sqrt(hd))
torch.cuda.synchronize()
