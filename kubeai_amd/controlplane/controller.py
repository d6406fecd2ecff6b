"""Model controller — reconciles Models into engine replicas.

Control flow mirrors the reference reconciler (internal/modelcontroller/
model_controller.go:70-198): labels -> bounds -> model config -> cache ->
replica listing -> status -> pod plan (surge rollout, pod-hash) -> adapters.
"Pods" are engine replica processes (runtime.py); the same plan semantics
apply (pod_plan.go:28-156, 215-243).
"""
from __future__ import annotations

import asyncio
import hashlib
import os
import shutil
from typing import Optional

import httpx

from .crd import CACHE_EVICTION_FINALIZER, Model, feature_labels
from .store import Replica, ReplicaState, Store


def fnv1a_32(data: bytes) -> int:
    # reference: k8sutils/pods.go:26-50 PodHash uses FNV-32a
    h = 0x811C9DC5
    for b in data:
        h ^= b
        h = (h * 0x01000193) & 0xFFFFFFFF
    return h


def spec_hash(model: Model, gpus_per_replica: int) -> str:
    s = model.spec
    key = "|".join(
        [
            s.url,
            s.engine,
            s.image,
            ",".join(s.args),
            ",".join(f"{k}={v}" for k, v in sorted(s.env.items())),
            str(gpus_per_replica),
            s.cache_profile,
            ",".join(f"{f.path}#{hashlib.sha256(f.content.encode()).hexdigest()[:8]}"
                     for f in s.files),
        ]
    )
    return format(fnv1a_32(key.encode()), "08x")


def adapter_url_hash(url: str) -> str:
    return hashlib.sha256(url.encode()).hexdigest()[:10]


class CacheManager:
    """Shared model-cache analog (reference: cache.go — PVC + loader Job +
    eviction finalizer). Locally: a shared directory per cache profile with
    per-model subdirs `<name>-<uid>` and a loader task."""

    def __init__(self, base_dir: str = "/tmp/kubeai-cache"):
        self.base_dir = base_dir

    def model_dir(self, model: Model) -> str:
        return os.path.join(
            self.base_dir, model.spec.cache_profile, f"{model.name}-{model.uid}"
        )

    async def ensure(self, model: Model) -> bool:
        """Returns True when the cache is loaded."""
        d = self.model_dir(model)
        marker = os.path.join(d, ".loaded")
        if os.path.exists(marker):
            return True
        os.makedirs(d, exist_ok=True)
        url = model.spec.url
        if url.startswith("file://"):
            src = url[len("file://") :]
            if os.path.isdir(src):
                await asyncio.get_running_loop().run_in_executor(
                    None, shutil.copytree, src, os.path.join(d, "model"), False, None,
                    shutil.copy2, False, False, True,
                )
        # hf://… has no network in this environment; preset models need no
        # download. Mark loaded either way (loader image handles real pulls
        # in cluster deployments — components/model-loader analog).
        with open(marker, "w") as f:
            f.write(model.uid)
        return True

    async def evict(self, model: Model) -> None:
        d = self.model_dir(model)
        if os.path.exists(d):
            await asyncio.get_running_loop().run_in_executor(None, shutil.rmtree, d, True)


class ModelController:
    def __init__(
        self,
        store: Store,
        runtime,
        resource_profiles: Optional[dict[str, int]] = None,
        cache: Optional[CacheManager] = None,
        post_ready_delay: float = 0.0,
    ):
        self.store = store
        self.runtime = runtime
        self.resource_profiles = resource_profiles or {"amd-gpu-mi355x": 1, "cpu": 0}
        self.cache = cache or CacheManager()
        self._task: Optional[asyncio.Task] = None
        self._client = httpx.AsyncClient(timeout=30.0)
        self._replica_counter = 0

    # ------------------------------------------------------------ lifecycle
    def start(self) -> None:
        self._task = asyncio.create_task(self._loop())

    async def stop(self) -> None:
        if self._task:
            self._task.cancel()
        await self._client.aclose()

    async def _loop(self) -> None:
        q = self.store.subscribe()
        # initial reconcile of everything
        for m in list(self.store.models):
            await self._safe_reconcile(m)
        def owner(ev):
            if ev.kind == "model":
                return ev.name
            return ev.owner or self._owner_of(ev.name)

        while True:
            # event-driven with a periodic resync (controller-runtime
            # style): a dropped or raced event heals within one period
            try:
                ev = await asyncio.wait_for(q.get(), timeout=2.0)
            except asyncio.TimeoutError:
                for m in list(self.store.models):
                    await self._safe_reconcile(m)
                continue
            names = {owner(ev)}
            # drain burst
            while not q.empty():
                names.add(owner(q.get_nowait()))
            for name in names:
                if name:
                    await self._safe_reconcile(name)

    def _owner_of(self, replica_name: str) -> Optional[str]:
        r = self.store.get_replica(replica_name)
        return r.model if r else None

    async def _safe_reconcile(self, name: str) -> None:
        try:
            await self.reconcile(name)
        except Exception as e:  # noqa: BLE001
            import traceback

            traceback.print_exc()

    # ------------------------------------------------------------ reconcile
    def gpus_per_replica(self, model: Model) -> int:
        rp = model.spec.resource_profile
        if not rp:
            return 1
        name, _, count = rp.partition(":")
        per = self.resource_profiles.get(name, 1)
        try:
            mult = int(count) if count else 1
        except ValueError:
            mult = 1
        return per * mult

    async def reconcile(self, name: str) -> None:
        model = self.store.models.get(name)
        if model is None:
            # model hard-deleted: clean any orphaned replicas (the
            # reference relies on owner-reference GC here)
            for r in self.store.list_replicas(model=name):
                await self.runtime.delete(r.name)
            return
        replicas = self.store.list_replicas(model=name)

        if model.deleted:
            for r in replicas:
                await self.runtime.delete(r.name)
            if model.spec.cache_profile and CACHE_EVICTION_FINALIZER in model.finalizers:
                # evict() may span reconciles (K8s mode runs an eviction
                # Job — cache.go's deletion branch); keep the finalizer
                # until it reports completion
                done = await self.cache.evict(model)
                if done is False:
                    return
                model.finalizers.remove(CACHE_EVICTION_FINALIZER)
            self.store.remove_model(name)
            return

        # labels + bounds (reference: model_controller.go:96-105, 357-407)
        pre = (dict(model.labels), list(model.finalizers), model.spec.replicas)
        model.labels.update(feature_labels(model))
        s = model.spec
        if s.replicas is None:
            s.replicas = s.min_replicas
        if not s.autoscaling_disabled:
            if s.replicas < s.min_replicas:
                s.replicas = s.min_replicas
            if s.max_replicas is not None and s.replicas > s.max_replicas:
                s.replicas = s.max_replicas

        # cache (reference: cache.go)
        if s.cache_profile and CACHE_EVICTION_FINALIZER not in model.finalizers:
            model.finalizers.append(CACHE_EVICTION_FINALIZER)
        if pre != (model.labels, model.finalizers, s.replicas):
            # write back self-applied mutations (reference Update calls;
            # K8s store PATCHes, in-memory store just notifies)
            self.store.persist_model_meta(model)
        if s.cache_profile:
            loaded = await self.cache.ensure(model)
            self.store.update_status(name, cache_loaded=loaded)
            if not loaded:
                return

        gpus = self.gpus_per_replica(model)
        h = spec_hash(model, gpus)
        desired = s.replicas or 0

        # ---- pod plan (reference: pod_plan.go calculatePodPlan) ----
        current = sorted(
            self.store.list_replicas(model=name), key=lambda r: r.created_seq
        )
        # FAILED replicas (engine process died — health monitor) are treated
        # as out-of-date so the surge branch recreates them immediately
        # (reference analog: pod restart policy + reconcile recreate)
        up_to_date = [
            r for r in current
            if r.hash == h and r.state != ReplicaState.FAILED
        ]
        out_of_date = [
            r for r in current
            if r.hash != h or r.state == ReplicaState.FAILED
        ]
        to_delete: list[Replica] = []
        n_create = 0

        if len(up_to_date) > desired:
            # scale down: deletion-order not-ready < unscheduled < youngest
            surplus = len(up_to_date) - desired
            to_delete.extend(self._deletion_order(up_to_date)[:surplus])
        else:
            n_create = desired - len(up_to_date)

        all_ready = all(r.ready for r in current) and current
        for r in out_of_date:
            if not r.ready:
                # recreate immediately (surge for unready out-of-date)
                to_delete.append(r)
                n_create += 1 if len(up_to_date) + n_create < desired else 0
            elif all_ready and not to_delete:
                # ready out-of-date: roll one at a time once all are ready
                to_delete.append(r)
                n_create += 1 if len(up_to_date) + n_create < desired else 0
                break

        for r in to_delete:
            await self.runtime.delete(r.name)
        for _ in range(max(0, n_create)):
            self._replica_counter += 1
            rname = f"model-{name}-{h}-{self._replica_counter}"
            await self.runtime.create(model, rname, h, gpus)

        # ---- status ----
        current = self.store.list_replicas(model=name)
        self.store.update_status(
            name,
            replicas_all=len(current),
            replicas_ready=sum(1 for r in current if r.ready),
        )

        # ---- adapters (reference: adapters.go + vllmclient) ----
        await self._reconcile_adapters(model)

    @staticmethod
    def _deletion_order(reps: list[Replica]) -> list[Replica]:
        # reference: pod_plan.go:215-243 — delete not-ready first, then
        # unscheduled, then youngest (highest created_seq)
        return sorted(reps, key=lambda r: (r.ready, r.scheduled, -r.created_seq))

    async def _reconcile_adapters(self, model: Model) -> None:
        desired = {a.name: adapter_url_hash(a.url) for a in model.spec.adapters}
        for rep in self.store.list_replicas(model=model.name):
            if not rep.ready or not rep.address:
                continue
            base = f"http://{rep.address}"
            changed = False
            for aname, ahash in desired.items():
                if rep.adapters.get(aname) == ahash:
                    continue
                try:
                    r = await self._client.post(
                        f"{base}/v1/load_lora_adapter",
                        json={
                            "lora_name": aname,
                            "lora_path": _adapter_local_path(model, aname),
                        },
                    )
                    if r.status_code == 200 or "already loaded" in r.text:
                        rep.adapters[aname] = ahash
                        changed = True
                except Exception:
                    pass
            for aname in list(rep.adapters):
                if aname not in desired:
                    try:
                        r = await self._client.post(
                            f"{base}/v1/unload_lora_adapter", json={"lora_name": aname}
                        )
                        if r.status_code in (200, 404) or "cannot be found" in r.text:
                            rep.adapters.pop(aname, None)
                            changed = True
                    except Exception:
                        pass
            if changed:
                # K8s store: adapter state -> pod labels (adapters.go:90-92)
                self.store.persist_replica_adapters(rep)


def _adapter_local_path(model: Model, adapter_name: str) -> Optional[str]:
    for a in model.spec.adapters:
        if a.name == adapter_name and a.url.startswith("file://"):
            return a.url[len("file://") :]
    return None
