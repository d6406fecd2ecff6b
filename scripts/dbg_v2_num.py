"""Isolate the prefill-v2 residual numeric error with targeted patterns.

Run with KUBEAI_PREFILL_V2=1 on a GPU box. Each case zeroes out parts of
the computation so the error structure identifies the broken path:
  case A: K=Q=0, patterned V  -> exercises P/V/normalize only
          (uniform softmax over the causal window)
  case B: V=1, random K/Q     -> output must be exactly 1 (softmax sums)
  case C: case A over 3 kv tiles (online-softmax rescale across tiles)
  case D: fully random, odd ctx (partial tile + prefix context)
Prints max-error grouped by q-row, head, and 16-wide d-block.
"""
import math
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import kubeai_amd.ops as ops
from kubeai_amd.ops import ref

dev = "cuda"
torch.manual_seed(0)
nq, nkv, hd, bs = 32, 8, 128, 16
scale = 1.0 / math.sqrt(hd)


def run_case(name, Tq, ctx, k_mode, q_mode, v_mode, _nq=None):
    global nq
    if _nq is not None:
        nq = _nq
    L = Tq + ctx
    nb = (L + bs - 1) // bs + 1
    shape = (nb, nkv, bs, hd)
    kc = (torch.randn(shape, device=dev) if k_mode == "rand"
          else torch.zeros(shape, device=dev)).to(torch.bfloat16)
    if v_mode == "pattern":
        vc = torch.zeros(shape, device=dev)
        for b in range(nb):
            for t in range(bs):
                tok = (b - 1) * bs + t
                vc[b, :, t, :] = (tok % 13) + torch.arange(hd, device=dev) % 7
        vc = vc.to(torch.bfloat16)
    elif v_mode == "ones":
        vc = torch.ones(shape, dtype=torch.bfloat16, device=dev)
    else:
        vc = torch.randn(shape, dtype=torch.bfloat16, device=dev)
    q = (torch.randn(Tq, nq, hd, device=dev) if q_mode == "rand"
         else torch.zeros(Tq, nq, hd, device=dev)).to(torch.bfloat16)
    bt = torch.arange(1, nb, dtype=torch.int32, device=dev).reshape(1, -1)
    sl = torch.tensor([L], dtype=torch.int32, device=dev)
    qsl = torch.tensor([0, Tq], dtype=torch.int32, device=dev)
    out = ops.paged_attention_prefill(q, kc, vc, bt, qsl, sl, scale)
    want = ref.paged_attention_prefill(
        q.float(), kc.float(), vc.float(), bt, qsl, sl, scale
    )
    err = (out.float() - want).abs()  # [Tq, nq, hd]
    print(f"== {name}: Tq={Tq} ctx={ctx} max_err={err.max().item():.4f} "
          f"mean={err.mean().item():.5f}")
    if err.max().item() < 0.05:
        print("   OK")
        return
    by_row = err.amax(dim=(1, 2))
    bad_rows = (by_row > 0.05).nonzero(as_tuple=True)[0].tolist()
    print(f"   bad q-rows ({len(bad_rows)}): {bad_rows[:24]}")
    by_head = err.amax(dim=(0, 2))
    print(f"   err by head: {[round(v, 3) for v in by_head.tolist()]}")
    by_d16 = err.view(Tq, nq, hd // 16, 16).amax(dim=(0, 1, 3))
    print(f"   err by d16-block: {[round(v, 3) for v in by_d16.tolist()]}")
    flat = err.flatten()
    top = flat.topk(8).indices
    coords = [(int(i) // (nq * hd), (int(i) // hd) % nq, int(i) % hd)
              for i in top.tolist()]
    print(f"   top-err (q,h,d): {coords}")
    if bad_rows:
        r = bad_rows[0]
        h = int(err[r].amax(dim=1).argmax())
        print(f"   sample row {r} head {h}: out={out[r, h, :8].tolist()}")
        print(f"                        want={want[r, h, :8].tolist()}")


run_case("A uniform-softmax", 64, 0, "zero", "zero", "pattern")
run_case("B softmax-sums", 64, 0, "rand", "rand", "ones")
run_case("C multi-tile", 192, 0, "zero", "zero", "pattern")
run_case("D random+ctx", 192, 37, "rand", "rand", "rand")
run_case("E random 1 tile", 64, 0, "rand", "rand", "rand")
run_case("F random multi-tile", 192, 0, "rand", "rand", "rand")
run_case("G random 1 tile + ctx", 27, 37, "rand", "rand", "rand")
run_case("H pattern + ctx", 64, 37, "zero", "zero", "pattern")
run_case("I ones + ctx (sums)", 192, 37, "rand", "rand", "ones")

# G=8 cases (the 70B head layout; residual fault seen at Tq=2048 G=8)
run_case("J G8 rand multi-tile", 96, 0, "rand", "rand", "rand", _nq=64)
run_case("K G8 rand ragged", 100, 0, "rand", "rand", "rand", _nq=64)
run_case("L G8 rand + ctx", 192, 37, "rand", "rand", "rand", _nq=64)
run_case("M G8 big", 2048, 0, "rand", "rand", "rand", _nq=64)
run_case("N G2 rand", 192, 21, "rand", "rand", "rand", _nq=16)
run_case("O G1 rand", 192, 21, "rand", "rand", "rand", _nq=8)

run_case("P G8 tiny16", 16, 0, "rand", "rand", "rand", _nq=64)
run_case("Q G1 ragged100", 100, 0, "rand", "rand", "rand", _nq=8)
run_case("R G4 big", 2048, 0, "rand", "rand", "rand", _nq=32)
run_case("S G8 big repeat", 2048, 0, "rand", "rand", "rand", _nq=64)
