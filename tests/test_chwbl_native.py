"""C++ CHWBL ring vs pure-Python ring: decision-for-decision equivalence.

Same endpoints, same key stream, same held-load schedule -> identical
endpoint choices (both implement balance_chwbl.go semantics: xxhash64
vnodes, successor walk, totalLoad==0 short-circuit, bounded load
load <= (total+1)/n * factor).
"""
import asyncio
import random

import pytest

import kubeai_amd.controlplane.loadbalancer as lb_mod
from kubeai_amd.controlplane.loadbalancer import EndpointGroup


def decisions(use_native: bool, n_eps=5, n_reqs=400, replication=64):
    saved = lb_mod._native
    if not use_native:
        lb_mod._native = None
    try:
        g = EndpointGroup("m")
        addrs = [f"10.0.0.{i}:8000" for i in range(n_eps)]
        asyncio.run(
            g.reconcile(
                {a: ({"lora1"} if i % 2 == 0 else set()) for i, a in enumerate(addrs)},
                replication,
            )
        )
        rng = random.Random(7)
        held = []
        picks = []
        for r in range(n_reqs):
            key = f"conv-{rng.randrange(40)}"
            adapter = "lora1" if rng.random() < 0.3 else ""
            addr = g._chwbl_get(adapter + key, 1.25, adapter)
            picks.append(addr)
            g.endpoints[addr].in_flight += 1
            g.total_in_flight += 1
            held.append(addr)
            # release a random held request half the time
            if held and rng.random() < 0.5:
                a = held.pop(rng.randrange(len(held)))
                g.endpoints[a].in_flight -= 1
                g.total_in_flight -= 1
        return picks
    finally:
        lb_mod._native = saved


@pytest.mark.skipif(lb_mod._native is None, reason="native ext not built")
def test_native_matches_python():
    native = decisions(True)
    python = decisions(False)
    assert native == python


def test_native_xxh64_matches_python():
    if lb_mod._native is None:
        pytest.skip("native ext not built")
    from kubeai_amd import _C
    from kubeai_amd.utils.xxhash64 import xxh64

    rng = random.Random(0)
    for n in [0, 1, 3, 4, 7, 8, 15, 31, 32, 33, 63, 100]:
        data = bytes(rng.randrange(256) for _ in range(n))
        assert _C.xxh64(data, 0) == xxh64(data)
        assert _C.xxh64(data, 12345) == xxh64(data, 12345)
