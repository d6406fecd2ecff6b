"""Kubernetes-backed Store/Runtime/Election — Models persist as CRs, engine
replicas run as Pods, leadership rides a coordination.k8s.io Lease.

This is the cluster substrate for the SAME ModelController/LoadBalancer/
Autoscaler that run against the in-memory Store (store.py): the reference
persists Models in etcd via its CRD and reconciles Pods through the API
server (internal/modelcontroller/model_controller.go:201-209); here
KubeStore keeps an informer-style local cache fed by WATCH streams and
pushes mutations through the REST client (kubeclient.py). Reads are
cache-hits; writes are synchronous API calls, like client-go's cached
reader + direct writer split.

Tests run against the in-process fake API server (fakekube.py), marking
pods Ready by PATCHing status — the reference envtest pattern
(test/integration/utils_test.go:118-132).
"""
from __future__ import annotations

import asyncio
import itertools
import json
import threading
import time
import uuid
from typing import Optional

from .crd import (ADAPTER_LABEL_PREFIX, MODEL_POD_IP_ANNOTATION,
                  MODEL_POD_PORT_ANNOTATION, POD_HASH_LABEL, POD_MODEL_LABEL,
                  Model, validate_model)
from .kubeclient import (ApiError, KubeClient, model_from_manifest,
                         model_to_manifest, pod_manifest_for)
from .store import Event, Replica, ReplicaState

MODELS = ("kubeai.org", "v1", "models")
PODS = ("", "v1", "pods")
LEASES = ("coordination.k8s.io", "v1", "leases")


class KubeStore:
    """Store interface over the Kubernetes API (models = CRs, replicas =
    Pods). Start with `await store.start()` so the informer cache is warm
    before controllers subscribe."""

    def __init__(self, client: KubeClient):
        self.kc = client
        self.models: dict[str, Model] = {}
        self.replicas: dict[str, Replica] = {}
        self._pods: dict[str, dict] = {}  # raw manifests, for translation
        self._subs: list[asyncio.Queue] = []
        self._seq = itertools.count(1)
        self._seen_order: dict[str, int] = {}  # pod name -> created_seq
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._threads: list[threading.Thread] = []
        self._stopping = threading.Event()

    # ------------------------------------------------------------ lifecycle
    async def start(self) -> None:
        self._loop = asyncio.get_running_loop()
        # prime the cache synchronously, then start watch threads
        for obj in self.kc.list(self.kc.path(*MODELS)):
            m = model_from_manifest(obj)
            self.models[m.name] = m
        for obj in self.kc.list(self.kc.path(*PODS)):
            self._ingest_pod(obj)
        for args in (
            (self.kc.path(*MODELS), self._on_model_event),
            (self.kc.path(*PODS), self._on_pod_event),
        ):
            t = threading.Thread(target=self._watch_loop, args=args, daemon=True)
            t.start()
            self._threads.append(t)

    def stop(self) -> None:
        self._stopping.set()

    def _watch_loop(self, path: str, handler) -> None:
        while not self._stopping.is_set():
            try:
                for ev in self.kc.watch(path):
                    if self._stopping.is_set():
                        return
                    handler(ev["type"], ev["object"])
            except Exception:
                time.sleep(0.2)  # re-list + re-watch

    # ------------------------------------------------------------ watch fan-out
    def subscribe(self) -> asyncio.Queue:
        q: asyncio.Queue = asyncio.Queue()
        self._subs.append(q)
        return q

    def _notify(self, kind: str, name: str, owner: Optional[str] = None) -> None:
        ev = Event(kind, name, owner)
        if self._loop is not None and not self._loop.is_closed():
            for q in self._subs:
                self._loop.call_soon_threadsafe(q.put_nowait, ev)

    # ------------------------------------------------------------ event ingestion
    def _on_model_event(self, etype: str, obj: dict) -> None:
        name = obj["metadata"]["name"]
        if etype == "DELETED":
            self.models.pop(name, None)
        else:
            self.models[name] = model_from_manifest(obj)
        self._notify("model", name)

    def _on_pod_event(self, etype: str, obj: dict) -> None:
        name = obj["metadata"]["name"]
        if etype == "DELETED":
            self._pods.pop(name, None)
            rep = self.replicas.pop(name, None)
            self._notify("replica", name, rep.model if rep else None)
        else:
            rep = self._ingest_pod(obj)
            if rep is not None:
                self._notify("replica", name, rep.model)

    def _ingest_pod(self, obj: dict) -> Optional[Replica]:
        meta = obj.get("metadata", {})
        labels = meta.get("labels") or {}
        model_name = labels.get(POD_MODEL_LABEL)
        if not model_name:
            return None  # not an engine pod
        name = meta["name"]
        self._pods[name] = obj
        if name not in self._seen_order:
            self._seen_order[name] = next(self._seq)
        rep = Replica(
            name=name,
            model=model_name,
            hash=labels.get(POD_HASH_LABEL, ""),
            created_seq=self._seen_order[name],
        )
        status = obj.get("status") or {}
        conds = {c.get("type"): c.get("status")
                 for c in (status.get("conditions") or [])}
        if meta.get("deletionTimestamp"):
            rep.state = ReplicaState.TERMINATING
        elif status.get("phase") == "Failed":
            rep.state = ReplicaState.FAILED
        elif conds.get("Ready") == "True":
            rep.state = ReplicaState.READY
        elif conds.get("PodScheduled") == "False":
            rep.state = ReplicaState.PENDING
            rep.scheduled = False
        else:
            rep.state = ReplicaState.STARTING
        # address: status.podIP + port annotation, with the reference's
        # model-pod-ip/-port dev override (adapters.go:120-141)
        model = self.models.get(model_name)
        anns = dict(meta.get("annotations") or {})
        if model is not None:
            anns.update(model.annotations)
        ip = anns.get(MODEL_POD_IP_ANNOTATION) or status.get("podIP")
        port = anns.get(MODEL_POD_PORT_ANNOTATION) or "8000"
        if ip:
            rep.address = f"{ip}:{port}"
        rep.adapters = {
            k[len(ADAPTER_LABEL_PREFIX):]: v
            for k, v in labels.items()
            if k.startswith(ADAPTER_LABEL_PREFIX)
        }
        self.replicas[name] = rep
        return rep

    # ------------------------------------------------------------ models API
    def apply_model(self, model: Model) -> Model:
        old = self.models.get(model.name)
        validate_model(model, old)
        if not model.uid:
            model.uid = old.uid if old else uuid.uuid4().hex[:8]
        manifest = model_to_manifest(model)
        path = self.kc.path(*MODELS, name=model.name)
        try:
            self.kc.get(path)
            out = self.kc.replace(path, manifest)
        except ApiError as e:
            if e.status != 404:
                raise
            out = self.kc.create(self.kc.path(*MODELS), manifest)
        got = model_from_manifest(out)
        self.models[model.name] = got
        return got

    def get_model(self, name: str) -> Optional[Model]:
        return self.models.get(name)

    def list_models(self, label_selector: Optional[dict[str, str]] = None) -> list[Model]:
        out = []
        for m in self.models.values():
            if m.deleted:
                continue
            if label_selector and any(
                m.labels.get(k) != v for k, v in label_selector.items()
            ):
                continue
            out.append(m)
        return out

    def delete_model(self, name: str) -> None:
        try:
            self.kc.delete(self.kc.path(*MODELS, name=name))
        except ApiError as e:
            if e.status != 404:
                raise
        m = self.models.get(name)
        if m is not None:
            m.deleted = True
            self._notify("model", name)

    def remove_model(self, name: str) -> None:
        """Finalizers cleared by the controller -> strip + let the API
        server complete the graceful deletion."""
        path = self.kc.path(*MODELS, name=name)
        try:
            self.kc.patch_merge(path, {"metadata": {"finalizers": None}})
            self.kc.delete(path)
        except ApiError as e:
            if e.status != 404:
                raise
        self.models.pop(name, None)
        self._notify("model", name)

    def scale_model(self, name: str, replicas: int) -> None:
        m = self.models.get(name)
        if m is None or m.deleted:
            raise KeyError(name)
        if m.spec.replicas == replicas:
            return
        # the scale subresource, like the reference (modelclient/scale.go)
        path = self.kc.path(*MODELS, name=name, subresource="scale")
        self.kc.replace(path, {"spec": {"replicas": replicas}})
        m.spec.replicas = replicas

    def update_status(self, name: str, **kw) -> None:
        m = self.models.get(name)
        if m is None:
            return
        changed = False
        for k, v in kw.items():
            if getattr(m.status, k) != v:
                setattr(m.status, k, v)
                changed = True
        if not changed:
            return
        self.kc.patch_merge(
            self.kc.path(*MODELS, name=name, subresource="status"),
            {"status": {
                "replicas": {"all": m.status.replicas_all,
                             "ready": m.status.replicas_ready},
                "cache": {"loaded": m.status.cache_loaded},
            }},
        )

    def persist_model_meta(self, model: Model) -> None:
        """Write back reconcile-path mutations: labels, finalizers and the
        bounded spec.replicas (reference applySelfLabels /
        applyAutoscalingReplicaBounds Update calls,
        model_controller.go:96-105)."""
        self.kc.patch_merge(
            self.kc.path(*MODELS, name=model.name),
            {
                "metadata": {
                    "labels": dict(model.labels),
                    "finalizers": list(model.finalizers) or None,
                },
                "spec": {"replicas": model.spec.replicas},
            },
        )

    # ------------------------------------------------------------ replicas API
    def get_replica(self, name: str) -> Optional[Replica]:
        return self.replicas.get(name)

    def list_replicas(self, model: Optional[str] = None) -> list[Replica]:
        return [r for r in self.replicas.values()
                if model is None or r.model == model]

    def update_replica(self, name: str, **kw) -> None:
        r = self.replicas.get(name)
        if r is None:
            return
        for k, v in kw.items():
            setattr(r, k, v)
        self._notify("replica", name, r.model)

    def persist_replica_adapters(self, rep: Replica) -> None:
        """Adapter state -> pod labels (reference adapters.go:90-92)."""
        pod = self._pods.get(rep.name)
        old = {
            k: None
            for k in ((pod or {}).get("metadata", {}).get("labels") or {})
            if k.startswith(ADAPTER_LABEL_PREFIX)
        }
        labels = {**old}
        for aname, ahash in rep.adapters.items():
            labels[ADAPTER_LABEL_PREFIX + aname] = ahash
        self.kc.patch_merge(
            self.kc.path(*PODS, name=rep.name),
            {"metadata": {"labels": labels}},
        )


class KubeRuntime:
    """Replica lifecycle as Pods (the reference pod-plan executor,
    pod_plan.go:171-211): create POSTs an engine pod built to the
    reference contract, delete DELETEs it; the kubelet (or a test) drives
    readiness, which flows back through the pod watch."""

    def __init__(self, store: KubeStore, image: str = "kubeai-amd-engine:latest",
                 gpu_resource: str = "amd.com/gpu",
                 engine_images: Optional[dict] = None):
        self.store = store
        self.image = image
        self.gpu_resource = gpu_resource
        # per-engine image defaults (chart modelServers analog,
        # config/system.go:222-231)
        self.engine_images = engine_images or {}

    async def create(self, model: Model, name: str, spec_hash: str,
                     n_gpus: int) -> None:
        kc = self.store.kc
        manifest = pod_manifest_for(
            model, name, spec_hash, n_gpus,
            image=(model.spec.image
                   or self.engine_images.get(model.spec.engine)
                   or self.image),
            namespace=kc.namespace,
            gpu_resource=self.gpu_resource,
        )
        try:
            obj = kc.create(kc.path(*PODS), manifest)
        except ApiError as e:
            if e.status == 409:
                return
            raise
        self.store._ingest_pod(obj)

    async def delete(self, name: str) -> None:
        kc = self.store.kc
        try:
            kc.delete(kc.path(*PODS, name=name))
        except ApiError as e:
            if e.status != 404:
                raise
        self.store._pods.pop(name, None)
        rep = self.store.replicas.pop(name, None)
        self.store._notify("replica", name, rep.model if rep else None)


class LeaseElection:
    """Leader election over a coordination.k8s.io Lease (reference:
    internal/leader/election.go — client-go LeaseLock). Same is_leader()
    interface as the file-lock Election (leader.py)."""

    def __init__(self, client: KubeClient, lease_name: str = "kubeai.org",
                 identity: Optional[str] = None,
                 lease_duration: float = 15.0):
        self.kc = client
        self.lease_name = lease_name
        self.identity = identity or f"kubeai-{uuid.uuid4().hex[:8]}"
        self.lease_duration = lease_duration
        self._is_leader = False
        self._task: Optional[asyncio.Task] = None

    def is_leader(self) -> bool:
        return self._is_leader

    def start(self) -> None:
        self._task = asyncio.create_task(self._loop())

    async def stop(self) -> None:
        if self._task:
            self._task.cancel()
        if self._is_leader:
            try:
                self.kc.patch_merge(
                    self.kc.path(*LEASES, name=self.lease_name),
                    {"spec": {"holderIdentity": None}},
                )
            except Exception:
                pass
        self._is_leader = False

    async def _loop(self) -> None:
        loop = asyncio.get_running_loop()
        while True:
            try:
                await loop.run_in_executor(None, self.tick)
            except Exception:
                self._is_leader = False
            await asyncio.sleep(self.lease_duration / 3)

    def tick(self) -> bool:
        """One acquire/renew attempt; returns (and records) leadership."""
        self._is_leader = self._tick()
        return self._is_leader

    def _tick(self) -> bool:
        path = self.kc.path(*LEASES, name=self.lease_name)
        now = time.time()
        spec = {
            "holderIdentity": self.identity,
            "leaseDurationSeconds": int(self.lease_duration),
            "renewTime": _micro_time(now),
        }
        try:
            obj = self.kc.get(path)
        except ApiError as e:
            if e.status != 404:
                raise
            try:
                self.kc.create(
                    self.kc.path(*LEASES),
                    {"apiVersion": "coordination.k8s.io/v1", "kind": "Lease",
                     "metadata": {"name": self.lease_name},
                     "spec": {**spec, "acquireTime": _micro_time(now)}},
                )
                return True
            except ApiError as e2:
                if e2.status == 409:
                    return False
                raise
        holder = (obj.get("spec") or {}).get("holderIdentity")
        renew = _parse_micro_time((obj.get("spec") or {}).get("renewTime"))
        if holder == self.identity:
            self.kc.patch_merge(path, {"spec": spec})
            return True
        if holder and renew is not None and now - renew < self.lease_duration:
            return False  # held by a live peer
        # expired or released: take over
        self.kc.patch_merge(
            path, {"spec": {**spec, "acquireTime": _micro_time(now)}}
        )
        return True


def _micro_time(t: float) -> str:
    base = time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime(t))
    return f"{base}.{int((t % 1) * 1e6):06d}Z"


def _parse_micro_time(s: Optional[str]) -> Optional[float]:
    if not s:
        return None
    import calendar

    head, _, frac = s.rstrip("Z").partition(".")
    t = calendar.timegm(time.strptime(head, "%Y-%m-%dT%H:%M:%S"))
    return t + (float("0." + frac) if frac else 0.0)


class ConfigMapStateStore:
    """Autoscaler moving-average state in a ConfigMap (reference
    internal/modelautoscaler/state.go: state persisted to a CM each tick
    and preloaded on boot, so leader failover keeps the windows)."""

    def __init__(self, kc: KubeClient, name: str = "kubeai-autoscaler-state"):
        self.kc = kc
        self.name = name
        self._path = (
            f"/api/v1/namespaces/{kc.namespace}/configmaps/{name}"
        )

    def load(self) -> dict:
        cm = self.kc.get_opt(self._path)
        if cm is None:
            return {}
        raw = (cm.get("data") or {}).get("state", "{}")
        return json.loads(raw)

    def save(self, data: dict) -> None:
        body = {
            "apiVersion": "v1",
            "kind": "ConfigMap",
            "metadata": {"name": self.name,
                         "namespace": self.kc.namespace},
            "data": {"state": json.dumps(data)},
        }
        if self.kc.get_opt(self._path) is None:
            self.kc.create(self._path.rsplit("/", 1)[0], body)
        else:
            self.kc.patch_merge(self._path, {"data": body["data"]})
