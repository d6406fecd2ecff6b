"""Load balancer: strategies, adapter filtering, blocking semantics,
in-flight accounting, CHWBL affinity + bounded load, concurrency storm.

Reference parity: internal/loadbalancer/*_test.go (679 LoC of table-driven
+ 400-goroutine sweeps).
"""
import asyncio
import collections

import pytest

from kubeai_amd.controlplane.crd import (
    LoadBalancingSpec,
    Model,
    ModelSpec,
    PrefixHashSpec,
)
from kubeai_amd.controlplane.loadbalancer import EndpointGroup, LoadBalancer
from kubeai_amd.controlplane.store import Replica, ReplicaState, Store


def run(coro):
    return asyncio.run(coro)


def ll_spec():
    return LoadBalancingSpec(strategy="LeastLoad")


def ph_spec(**kw):
    return LoadBalancingSpec(strategy="PrefixHash", prefix_hash=PrefixHashSpec(**kw))


async def make_group(addrs, adapters=None):
    g = EndpointGroup("m")
    desired = {a: set(adapters.get(a, [])) if adapters else set() for a in addrs}
    await g.reconcile(desired, replication=32)
    return g


def test_least_load_picks_min():
    async def body():
        g = await make_group(["a:1", "b:1", "c:1"])
        g.endpoints["a:1"].in_flight = 5
        g.endpoints["b:1"].in_flight = 1
        g.endpoints["c:1"].in_flight = 3
        addr, done = await g.get_best_addr("", None, ll_spec())
        assert addr == "b:1"
        assert g.endpoints["b:1"].in_flight == 2
        done()
        assert g.endpoints["b:1"].in_flight == 1
        assert g.total_in_flight == 0

    run(body())


def test_adapter_filtering():
    async def body():
        g = await make_group(
            ["a:1", "b:1"], adapters={"a:1": ["lora1"], "b:1": []}
        )
        dones = []
        for _ in range(5):
            addr, done = await g.get_best_addr("lora1", None, ll_spec())
            assert addr == "a:1"  # only a:1 has the adapter
            dones.append(done)  # load grows while held, still must pick a:1
        for d in dones:
            d()

    run(body())


def test_blocks_until_endpoint_appears():
    async def body():
        g = EndpointGroup("m")

        async def add_later():
            await asyncio.sleep(0.05)
            await g.reconcile({"x:1": set()}, replication=8)

        asyncio.create_task(add_later())
        addr, done = await g.get_best_addr("", None, ll_spec(), timeout=2)
        assert addr == "x:1"
        done()

    run(body())


def test_timeout_when_no_endpoint():
    async def body():
        g = EndpointGroup("m")
        with pytest.raises(TimeoutError):
            await g.get_best_addr("", None, ll_spec(), timeout=0.05)

    run(body())


def test_chwbl_prefix_affinity():
    async def body():
        g = await make_group([f"e{i}:1" for i in range(4)])
        spec = ph_spec(mean_load_percentage=1000)  # no bound: pure affinity
        # same prefix -> same endpoint, every time
        picks = set()
        for _ in range(10):
            addr, done = await g.get_best_addr("", "conversation-42", spec)
            picks.add(addr)
            done()
        assert len(picks) == 1
        # different prefixes spread across endpoints
        spread = set()
        for i in range(64):
            addr, done = await g.get_best_addr("", f"prefix-{i}", spec)
            spread.add(addr)
            done()
        assert len(spread) == 4

    run(body())


def test_chwbl_bounded_load_spills():
    async def body():
        g = await make_group([f"e{i}:1" for i in range(4)])
        spec = ph_spec(mean_load_percentage=125)
        # hammer one prefix WITHOUT done(): bounded load must spill to
        # other endpoints once the favorite exceeds (total+1)/n * 1.25
        picks = collections.Counter()
        dones = []
        for _ in range(32):
            addr, done = await g.get_best_addr("", "hot-prefix", spec)
            picks[addr] += 1
            dones.append(done)
        assert len(picks) > 1, f"no spill: {picks}"
        favorite = picks.most_common(1)[0][1]
        assert favorite < 32
        for d in dones:
            d()

    run(body())


def test_chwbl_node_leave_rebalances_minimally():
    async def body():
        g = await make_group([f"e{i}:1" for i in range(4)])
        spec = ph_spec(mean_load_percentage=10000)
        before = {}
        for i in range(128):
            addr, done = await g.get_best_addr("", f"p{i}", spec)
            before[i] = addr
            done()
        # remove one endpoint
        await g.reconcile({f"e{i}:1": set() for i in range(3)}, replication=32)
        moved = 0
        for i in range(128):
            addr, done = await g.get_best_addr("", f"p{i}", spec)
            if addr != before[i]:
                moved += 1
            done()
        # consistent hashing: only keys on the removed node move (~1/4)
        assert moved <= 128 // 2, f"{moved} keys moved"

    run(body())


def test_concurrency_storm():
    async def body():
        g = await make_group([f"e{i}:1" for i in range(3)])
        spec = ph_spec()
        errors = []

        async def worker(wid):
            try:
                for i in range(50):
                    strat = spec if wid % 2 == 0 else ll_spec()
                    addr, done = await g.get_best_addr("", f"w{wid}-{i}", strat)
                    await asyncio.sleep(0)
                    done()
            except Exception as e:  # noqa: BLE001
                errors.append(e)

        async def churn():
            for i in range(20):
                n = 2 + (i % 3)
                await g.reconcile({f"e{k}:1": set() for k in range(n)}, 32)
                await asyncio.sleep(0.001)

        await asyncio.gather(*[worker(w) for w in range(40)], churn())
        assert not errors
        assert g.total_in_flight == 0
        for e in g.endpoints.values():
            assert e.in_flight == 0

    run(body())


def test_lb_tracks_store_replicas():
    async def body():
        store = Store()
        store.apply_model(Model(name="m", spec=ModelSpec(url="hf://x/y")))
        lb = LoadBalancer(store)
        lb.start()
        store.add_replica(
            Replica(name="r1", model="m", hash="h", state=ReplicaState.READY,
                    address="1.2.3.4:8000")
        )
        await asyncio.sleep(0.05)
        addr, done = await lb.await_best_address("m", timeout=2)
        assert addr == "1.2.3.4:8000"
        done()
        # not-ready replicas never become endpoints
        store.add_replica(
            Replica(name="r2", model="m", hash="h", state=ReplicaState.STARTING,
                    address="5.6.7.8:8000")
        )
        await asyncio.sleep(0.05)
        assert set(lb.group("m").endpoints) == {"1.2.3.4:8000"}
        await lb.stop()

    run(body())
