"""Concurrency autoscaler (reference: internal/modelautoscaler/).

Leader-gated loop every `interval`: aggregate kubeai_inference_requests_active
across all control-plane replicas (self-scrape architecture,
autoscaler.go:118-137) + the per-replica engine queue depth, run the moving
average over timeWindow/interval slots, scale to ceil(avg/targetRequests)
with bounds + scale-down hysteresis, persist state for restart continuity
(state.go:32-65).
"""
from __future__ import annotations

import asyncio
import json
import math
import os
import re
from typing import Optional

import httpx

from .modelclient import ModelClient
from .movingaverage import SimpleMovingAverage
from .store import Store

_METRIC_RE = re.compile(
    r'^kubeai_inference_requests_active\{model="([^"]+)"\}\s+([0-9.eE+-]+)',
)
# only WAITING (backlog beyond engine capacity): running requests are
# already counted by the gateway's active gauge — no double count
_ENGINE_QUEUE_RE = re.compile(
    r'^kubeai_engine_num_requests_waiting\{model="([^"]+)"\}\s+([0-9.eE+-]+)',
)
_ENGINE_KV_RE = re.compile(
    r'^kubeai_engine_kv_cache_usage_perc\{model="([^"]+)"\}\s+([0-9.eE+-]+)',
)
# KV pressure: a replica whose paged KV pool is nearly full is about to
# preempt/queue even if nothing waits yet — count it as one extra
# replica's worth of load (north-star signal: queue depth + KV occupancy)
KV_PRESSURE_THRESHOLD = 0.95


class Autoscaler:
    def __init__(
        self,
        store: Store,
        model_client: ModelClient,
        interval: float = 10.0,
        time_window: float = 600.0,
        self_metric_addrs: Optional[list[str]] = None,
        state_path: Optional[str] = None,
        state_store=None,  # object with load()->dict / save(dict); K8s
        # mode passes a ConfigMap-backed store (reference state.go CM)
        is_leader=lambda: True,
        scrape_engine_queues: bool = True,
    ):
        self.store = store
        self.model_client = model_client
        self.interval = interval
        self.window_count = max(1, int(time_window / interval))
        self.self_metric_addrs = self_metric_addrs or []
        self.state_path = state_path
        self.state_store = state_store
        self.is_leader = is_leader
        self.scrape_engine_queues = scrape_engine_queues
        self.averages: dict[str, SimpleMovingAverage] = {}
        self._task: Optional[asyncio.Task] = None
        self._client = httpx.AsyncClient(timeout=5.0)
        self.last_scales: dict[str, int] = {}
        if state_store is not None:
            self._load_state()
        elif state_path and os.path.exists(state_path):
            self._load_state()

    # ------------------------------------------------------------ lifecycle
    def start(self) -> None:
        self._task = asyncio.create_task(self._loop())

    async def stop(self) -> None:
        if self._task:
            self._task.cancel()
        await self._client.aclose()

    async def _loop(self) -> None:
        while True:
            await asyncio.sleep(self.interval)
            if not self.is_leader():
                continue
            try:
                await self.tick()
            except Exception:  # noqa: BLE001
                import traceback

                traceback.print_exc()

    # ------------------------------------------------------------ one tick
    async def tick(self) -> None:
        active = await self._aggregate_active_requests()
        kv_pressured: dict[str, int] = {}
        if self.scrape_engine_queues:
            queued, kv_pressured = await self._aggregate_engine_queues()
            for m, v in queued.items():
                active[m] = active.get(m, 0.0) + v
        for model in self.model_client.list_all_models():
            if model.spec.autoscaling_disabled:
                continue
            avg = self._avg_for(model.name).next(active.get(model.name, 0.0))
            target = math.ceil(avg / max(model.spec.target_requests, 1))
            # P replicas at >=95% KV occupancy -> want at least P+1 so new
            # prefixes stop evicting hot cache (scale-down hysteresis in
            # ModelClient.scale keeps this from flapping)
            pressured = kv_pressured.get(model.name, 0)
            if pressured:
                target = max(target, pressured + 1)
            self.last_scales[model.name] = target
            self.model_client.scale(model.name, target)
        if self.state_path or self.state_store is not None:
            self._save_state()

    def _avg_for(self, name: str) -> SimpleMovingAverage:
        a = self.averages.get(name)
        if a is None:
            a = SimpleMovingAverage(self.window_count)
            self.averages[name] = a
        return a

    # ------------------------------------------------------------ scraping
    async def _aggregate_active_requests(self) -> dict[str, float]:
        """Sum kubeai_inference_requests_active across control-plane
        replicas (HA: each gateway replica holds its own in-flight counts)."""
        totals: dict[str, float] = {}
        if not self.self_metric_addrs:
            # single-process mode: read our own registry directly
            from . import metrics as cp_metrics

            for fam in cp_metrics.REGISTRY.collect():
                if fam.name == "kubeai_inference_requests_active":
                    for s in fam.samples:
                        if s.name == fam.name:
                            model = s.labels.get("model", "")
                            totals[model] = totals.get(model, 0.0) + s.value
            return totals
        for addr in self.self_metric_addrs:
            try:
                r = await self._client.get(f"http://{addr}/metrics")
            except Exception:
                continue
            for line in r.text.splitlines():
                m = _METRIC_RE.match(line)
                if m:
                    totals[m.group(1)] = totals.get(m.group(1), 0.0) + float(m.group(2))
        return totals

    async def _aggregate_engine_queues(self) -> tuple[dict, dict]:
        """Per-replica engine queue depth + KV-pressured replica counts —
        the BASELINE.json north-star signals (engine /metrics)."""
        totals: dict[str, float] = {}
        pressured: dict[str, int] = {}
        for rep in self.store.list_replicas():
            if not rep.ready or not rep.address:
                continue
            try:
                r = await self._client.get(f"http://{rep.address}/metrics")
            except Exception:
                continue
            for line in r.text.splitlines():
                m = _ENGINE_QUEUE_RE.match(line)
                if m:
                    totals[rep.model] = totals.get(rep.model, 0.0) + float(m.group(2))
                    continue
                m = _ENGINE_KV_RE.match(line)
                if m and float(m.group(2)) >= KV_PRESSURE_THRESHOLD:
                    pressured[rep.model] = pressured.get(rep.model, 0) + 1
        return totals, pressured

    # ------------------------------------------------------------ state
    def _save_state(self) -> None:
        data = {name: avg.history() for name, avg in self.averages.items()}
        if self.state_store is not None:
            try:
                self.state_store.save(data)
            except Exception:  # noqa: BLE001 — state is best-effort
                pass
            return
        tmp = self.state_path + ".tmp"
        with open(tmp, "w") as f:
            json.dump(data, f)
        os.replace(tmp, self.state_path)

    def _load_state(self) -> None:
        try:
            if self.state_store is not None:
                data = self.state_store.load()
            else:
                with open(self.state_path) as f:
                    data = json.load(f)
        except Exception:
            return
        for name, values in data.items():
            self._avg_for(name).load(values)
