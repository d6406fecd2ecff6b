"""Prefix-aware load balancer: LeastLoad + CHWBL endpoint groups.

Reference parity (internal/loadbalancer/):
  - per-model endpoint groups built from ready replicas, with adapter sets
    (load_balancer.go:53-140)
  - await_best_address blocks until an endpoint exists (scale-from-zero
    hold), returns (addr, done) with in-flight accounting (group.go:53-94)
  - LeastLoad: min in-flight scan with adapter filter (balance_least_load.go)
  - CHWBL: xxhash64 ring, `replication` vnodes per endpoint, binary-search
    successor, bounded-load walk: load <= (total+1)/n * loadFactor
    (balance_chwbl.go:14-162); hash key = adapter + prefix.
"""
from __future__ import annotations

import asyncio
import bisect
import dataclasses
from typing import Optional

from kubeai_amd.utils.xxhash64 import xxh64

from . import metrics
from .crd import LoadBalancingSpec
from .store import Store

# C++ ring (csrc/chwbl.cpp) — the per-request hot loop; pure-Python path
# below implements identical semantics (tests/test_chwbl_native.py asserts
# decision-for-decision equivalence)
try:
    import torch  # noqa: F401  (loads libc10 for the extension)
    from kubeai_amd import _C as _native
except Exception:  # pragma: no cover
    _native = None


@dataclasses.dataclass
class Endpoint:
    address: str
    adapters: set[str] = dataclasses.field(default_factory=set)
    in_flight: int = 0


class EndpointGroup:
    def __init__(self, model_name: str):
        self.model_name = model_name
        self.endpoints: dict[str, Endpoint] = {}
        self.total_in_flight = 0
        self._cond = asyncio.Condition()
        # CHWBL ring: sorted list of (hash, addr)
        self._ring_hashes: list[int] = []
        self._ring_addrs: list[str] = []
        self._ring_replication = 0

    # ------------------------------------------------------------ updates
    async def reconcile(self, desired: dict[str, set[str]], replication: int) -> None:
        """desired: addr -> adapter names (from ready replicas)."""
        async with self._cond:
            changed = False
            for addr in list(self.endpoints):
                if addr not in desired:
                    self.endpoints.pop(addr)
                    changed = True
            for addr, adapters in desired.items():
                ep = self.endpoints.get(addr)
                if ep is None:
                    self.endpoints[addr] = Endpoint(addr, set(adapters))
                    changed = True
                elif ep.adapters != adapters:
                    ep.adapters = set(adapters)
                    changed = True
            if changed or replication != self._ring_replication:
                self._rebuild_ring(replication)
                self._cond.notify_all()

    def _rebuild_ring(self, replication: int) -> None:
        self._ring_replication = replication
        if _native is not None:
            self._ep_order = sorted(self.endpoints)
            self._native_ring = _native.ChwblRing()
            self._native_ring.rebuild(self._ep_order, replication)
            # keep a 1-entry marker so strategy checks see a non-empty ring
            self._ring_hashes = [0] if self.endpoints else []
            return
        self._native_ring = None
        pairs: list[tuple[int, str]] = []
        for addr in self.endpoints:
            for i in range(replication):
                # reference: balance_chwbl.go:140-150 — vnode key addr+i
                pairs.append((xxh64(f"{addr}{i}".encode()), addr))
        pairs.sort()
        self._ring_hashes = [p[0] for p in pairs]
        self._ring_addrs = [p[1] for p in pairs]

    # ------------------------------------------------------------ selection
    async def get_best_addr(
        self,
        adapter: str,
        prefix: Optional[str],
        lb: LoadBalancingSpec,
        timeout: Optional[float] = None,
    ):
        """Returns (addr, done). Blocks while no (matching) endpoint exists."""
        async with self._cond:
            def candidates():
                return [
                    e
                    for e in self.endpoints.values()
                    if not adapter or adapter in e.adapters
                ]

            if not candidates():
                try:
                    await asyncio.wait_for(
                        self._cond.wait_for(lambda: bool(candidates())), timeout
                    )
                except asyncio.TimeoutError:
                    raise TimeoutError(
                        f"no ready endpoints for model {self.model_name!r}"
                        + (f" adapter {adapter!r}" if adapter else "")
                    )
            if lb.strategy == "PrefixHash" and self._ring_hashes:
                addr = self._chwbl_get(
                    adapter + (prefix or ""),
                    lb.prefix_hash.mean_load_percentage / 100.0,
                    adapter,
                )
            else:
                addr = self._least_load(adapter)
            ep = self.endpoints[addr]
            ep.in_flight += 1
            self.total_in_flight += 1
            metrics.INFERENCE_REQUESTS_ACTIVE.labels(self.model_name).inc()

        done_called = False

        def done() -> None:
            nonlocal done_called
            if done_called:
                return
            done_called = True
            ep2 = self.endpoints.get(addr)
            if ep2 is not None:
                ep2.in_flight -= 1
            self.total_in_flight -= 1
            metrics.INFERENCE_REQUESTS_ACTIVE.labels(self.model_name).dec()

        return addr, done

    def _least_load(self, adapter: str) -> str:
        best, best_load = None, None
        for e in self.endpoints.values():
            if adapter and adapter not in e.adapters:
                continue
            if best_load is None or e.in_flight < best_load:
                best, best_load = e.address, e.in_flight
        assert best is not None
        return best

    def _chwbl_get(self, key: str, load_factor: float, adapter: str) -> str:
        if getattr(self, "_native_ring", None) is not None:
            eps = [self.endpoints[a] for a in self._ep_order]
            idx, iters, defaulted = self._native_ring.lookup(
                key,
                [e.in_flight for e in eps],
                self.total_in_flight,
                load_factor,
                [not adapter or adapter in e.adapters for e in eps],
            )
            if idx < 0:
                metrics.HASH_LOOKUP_DEFAULT.inc()
                return self._least_load(adapter)
            metrics.HASH_LOOKUP_INITIAL.inc()
            if defaulted:
                metrics.HASH_LOOKUP_DEFAULT.inc()
            else:
                metrics.HASH_LOOKUP_ITERATIONS.observe(iters)
            return self._ep_order[idx]
        h = xxh64(key.encode())
        n = len(self._ring_hashes)
        i = bisect.bisect_left(self._ring_hashes, h) % n
        i0 = i
        iterations = 0
        first = None
        while True:
            iterations += 1
            addr = self._ring_addrs[i]
            ep = self.endpoints.get(addr)
            if ep is not None and (not adapter or adapter in ep.adapters):
                if first is None:
                    first = addr
                    metrics.HASH_LOOKUP_INITIAL.inc()
                if self._load_ok(ep, load_factor):
                    metrics.HASH_LOOKUP_ITERATIONS.observe(iterations)
                    return addr
            i = (i + 1) % n
            if i == i0:
                # full loop: everything over the bound; fall back
                metrics.HASH_LOOKUP_DEFAULT.inc()
                return first if first is not None else self._least_load(adapter)

    def _load_ok(self, ep: Endpoint, load_factor: float) -> bool:
        # reference: balance_chwbl.go:152-162 chwblLoadOK — totalLoad==0 is
        # always OK; else load <= (total+1)/n * loadFactor (the +1 simulates
        # the incoming request)
        n = len(self.endpoints)
        if n == 0:
            return False
        if self.total_in_flight == 0:
            return True
        avg = (self.total_in_flight + 1) / n
        return ep.in_flight <= avg * load_factor


class LoadBalancer:
    def __init__(self, store: Store):
        self.store = store
        self.groups: dict[str, EndpointGroup] = {}
        self._task: Optional[asyncio.Task] = None

    def start(self) -> None:
        self._task = asyncio.create_task(self._loop())

    async def stop(self) -> None:
        if self._task:
            self._task.cancel()

    def group(self, model_name: str) -> EndpointGroup:
        g = self.groups.get(model_name)
        if g is None:
            g = EndpointGroup(model_name)
            self.groups[model_name] = g
        return g

    async def _loop(self) -> None:
        q = self.store.subscribe()
        while True:
            try:
                await self.reconcile_all()
            except Exception:  # noqa: BLE001 — a bad event must not kill the watch
                import traceback

                traceback.print_exc()
            # event-driven with a 1 s resync (controller-runtime style):
            # a dropped/raced event costs at most one resync period
            try:
                await asyncio.wait_for(q.get(), timeout=1.0)
            except asyncio.TimeoutError:
                continue
            while not q.empty():
                q.get_nowait()

    async def reconcile_all(self) -> None:
        by_model: dict[str, dict[str, set[str]]] = {}
        for rep in self.store.list_replicas():
            if rep.ready and rep.address:
                by_model.setdefault(rep.model, {})[rep.address] = set(rep.adapters)
        for model_name in set(by_model) | set(self.groups):
            m = self.store.get_model(model_name)
            replication = (
                m.spec.load_balancing.prefix_hash.replication if m else 256
            )
            await self.group(model_name).reconcile(
                by_model.get(model_name, {}), replication
            )

    async def await_best_address(
        self,
        model_name: str,
        adapter: str = "",
        prefix: Optional[str] = None,
        timeout: Optional[float] = 120.0,
    ):
        m = self.store.get_model(model_name)
        lb = m.spec.load_balancing if m else LoadBalancingSpec()
        return await self.group(model_name).get_best_addr(
            adapter, prefix, lb, timeout=timeout
        )
