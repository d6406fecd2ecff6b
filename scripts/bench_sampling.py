"""Sampling-path microbench: where does the temperature overhead go?

Times the top-k/top-p mask (fast topk path vs full sort), the Gumbel
kernel, and greedy argmax at the bench's steady-state shape.
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import kubeai_amd.ops as ops
from kubeai_amd.engine.runner import _apply_topk_topp, _sample_topk_topp

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from bench_ops import timeit  # noqa: E402

dev = "cuda"
torch.manual_seed(0)
for S, V in ((40, 128256), (256, 128256)):
    logits = torch.randn(S, V, dtype=torch.float32, device=dev)
    temps = [0.8] * S
    tps = [0.9] * S
    tks = [0] * S
    t_t = torch.full((S,), 0.8, device=dev)
    seeds = torch.arange(S, dtype=torch.int64, device=dev)

    us_mask = timeit(lambda: _apply_topk_topp(logits, tps, tks, temps))
    tp_t = torch.full((S,), 0.9, device=dev)
    tk_t = torch.zeros(S, dtype=torch.int32, device=dev)
    us_stats = timeit(lambda: ops.nucleus_stats(logits, t_t))
    m_, z_ = ops.nucleus_stats(logits, t_t)
    cand_ = ops.gumbel_sample(logits, t_t, seeds, 3)
    us_accept = timeit(
        lambda: ops.nucleus_accept(logits, cand_, m_, z_, t_t, tp_t, tk_t))
    rates = []
    for a in range(3):
        c = ops.gumbel_sample(logits, t_t, seeds, 1000 + a)
        rates.append(
            ops.nucleus_accept(logits, c, m_, z_, t_t, tp_t, tk_t)
            .float().mean().item())
    print(f"  accept rates per draw (expect ~{tps[0]}): "
          f"{[round(r, 3) for r in rates]}")

    def manual():
        m3, z3 = ops.nucleus_stats(logits, t_t)
        c1 = ops.gumbel_sample(logits, t_t, seeds, 11)
        o1 = ops.nucleus_accept(logits, c1, m3, z3, t_t, tp_t, tk_t)
        c2 = ops.gumbel_sample(logits, t_t, seeds, 12)
        o2 = ops.nucleus_accept(logits, c2, m3, z3, t_t, tp_t, tk_t)
        c3 = ops.gumbel_sample(logits, t_t, seeds, 13)
        o3 = ops.nucleus_accept(logits, c3, m3, z3, t_t, tp_t, tk_t)
        return torch.where(o1.bool(), c1,
                           torch.where(o2.bool(), c2, c3))

    us_manual = timeit(manual)
    print(f"  manual 3-draw pipeline (no fallback check): {us_manual:8.1f}us")
    # pending-rate diagnostic with the wrapper's exact step values
    step0 = 3
    oks = []
    cs = []
    for a in range(3):
        c = ops.gumbel_sample(logits, t_t, seeds, step0 + (a + 1) * 1_000_003)
        cs.append(c)
        oks.append(ops.nucleus_accept(logits, c, m_, z_, t_t, tp_t, tk_t).bool())
    pend = ~(oks[0] | oks[1] | oks[2])
    print(f"  wrapper-step pending rate: {pend.float().mean().item():.4f}  "
          f"draw-identical(c1==c2): {(cs[0] == cs[1]).float().mean().item():.3f}")
    rows = pend.nonzero(as_tuple=True)[0].tolist()
    if rows:
        # acceptance probability of the problem rows over 50 fresh draws
        accs = torch.zeros(S, device=dev)
        for a in range(50):
            c = ops.gumbel_sample(logits, t_t, seeds, 7_000_000 + a * 97)
            accs += ops.nucleus_accept(
                logits, c, m_, z_, t_t, tp_t, tk_t).float()
        print(f"  pending rows {rows[:6]} accept-prob over 50 draws: "
              f"{[round(accs[r].item() / 50, 2) for r in rows[:6]]}  "
              f"(healthy rows ~0.9; their cands: "
              f"{[int(cs[0][r]) for r in rows[:3]]})")
    _cache = {}
    us_subset = timeit(
        lambda: _sample_topk_topp(logits, tps, tks, temps, t_t, seeds, 3,
                                  cache=_cache)
    )
    print(f"  fallback rate: {_cache.get('fallbacks', 0)}"
          f"/{_cache.get('calls', 0)} calls")
    # vary step across calls (the bench reuses step=3 -> identical draws
    # every call: one unlucky batch repeats its fallback forever)
    _cache2 = {}
    _step_box = [0]

    def varied():
        _step_box[0] += 1
        return _sample_topk_topp(logits, tps, tks, temps, t_t, seeds,
                                 _step_box[0], cache=_cache2)

    us_varied = timeit(varied)
    print(f"  varied-step: {us_varied:8.1f}us  fallback "
          f"{_cache2.get('fallbacks', 0)}/{_cache2.get('calls', 0)}")

    def full_sort():
        sl, si = logits.sort(dim=-1, descending=True)
        probs = torch.softmax(sl / 0.8, dim=-1)
        cs = probs.cumsum(-1)
        keep = (cs - probs) < 0.9
        keep[:, 0] = True
        mask = torch.zeros_like(keep)
        mask.scatter_(1, si, keep)
        return logits.masked_fill(~mask, float("-inf"))

    us_sort = timeit(full_sort)
    masked = _apply_topk_topp(logits, tps, tks, temps).contiguous()
    us_gumbel = timeit(lambda: ops.gumbel_sample(masked, t_t, seeds, 3))
    us_greedy = timeit(lambda: ops.greedy_sample(logits))
    us_topk = timeit(lambda: torch.topk(logits, 1024, dim=-1))
    us_lse = timeit(lambda: torch.logsumexp(logits / 0.8, dim=-1))
    print(f"S={S:4d} V={V}: SAMPLE(rej)={us_subset:8.1f}us  stats={us_stats:6.1f}us  accept={us_accept:6.1f}us  mask(fast)={us_mask:8.1f}us  mask(sort)="
          f"{us_sort:8.1f}us  topk1024={us_topk:8.1f}us  lse={us_lse:7.1f}us"
          f"  gumbel={us_gumbel:7.1f}us  greedy={us_greedy:7.1f}us",
          flush=True)
