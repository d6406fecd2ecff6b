"""In-process object store with watch semantics — the control plane's
API-server analog.

The reference is a Kubernetes operator: controllers watch the API server
and reconcile. Here the same controllers run against
this store; in a cluster deployment the identical Model schema ships as a
CRD (deploy/crds/) and the store is replaced by the K8s API. Keeping the
watch/reconcile shape means the controller logic is identical in both.

Single-event-loop asyncio design: mutations enqueue events to subscriber
queues; controllers consume and reconcile.
"""
from __future__ import annotations

import asyncio
import dataclasses
import enum
import itertools
import uuid
from typing import Optional

from .crd import Model, validate_model


class ReplicaState(str, enum.Enum):
    PENDING = "Pending"
    STARTING = "Starting"
    READY = "Ready"
    FAILED = "Failed"
    TERMINATING = "Terminating"


@dataclasses.dataclass
class Replica:
    """Engine replica — the Pod analog (one engine server process)."""

    name: str
    model: str  # POD_MODEL_LABEL
    hash: str  # POD_HASH_LABEL
    state: ReplicaState = ReplicaState.PENDING
    address: Optional[str] = None  # "ip:port" once serving
    adapters: dict[str, str] = dataclasses.field(default_factory=dict)  # name->urlhash
    gpu_ids: list[int] = dataclasses.field(default_factory=list)
    created_seq: int = 0
    scheduled: bool = True

    @property
    def ready(self) -> bool:
        return self.state == ReplicaState.READY


@dataclasses.dataclass
class Event:
    kind: str  # "model" | "replica"
    name: str
    owner: Optional[str] = None  # owning model (replica events)


class Store:
    def __init__(self):
        self.models: dict[str, Model] = {}
        self.replicas: dict[str, Replica] = {}
        self._subs: list[asyncio.Queue] = []
        self._seq = itertools.count(1)

    # ---------------------------------------------------------- watch
    def subscribe(self) -> asyncio.Queue:
        q: asyncio.Queue = asyncio.Queue()
        self._subs.append(q)
        return q

    def _notify(self, kind: str, name: str, owner: Optional[str] = None) -> None:
        for q in self._subs:
            q.put_nowait(Event(kind, name, owner))

    # ---------------------------------------------------------- models
    def apply_model(self, model: Model) -> Model:
        old = self.models.get(model.name)
        validate_model(model, old)
        model.generation = (old.generation + 1) if old else 1
        if not model.uid:
            model.uid = old.uid if old else uuid.uuid4().hex[:8]
        self.models[model.name] = model
        self._notify("model", model.name)
        return model

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
        m = self.models.get(name)
        if m is None:
            return
        m.deleted = True
        self._notify("model", name)

    def remove_model(self, name: str) -> None:
        """Finalizers cleared -> actually remove."""
        self.models.pop(name, None)
        self._notify("model", name)

    def scale_model(self, name: str, replicas: int) -> None:
        """The scale-subresource analog (reference: modelclient/scale.go)."""
        m = self.models.get(name)
        if m is None or m.deleted:
            raise KeyError(name)
        if m.spec.replicas != replicas:
            m.spec.replicas = replicas
            self._notify("model", name)

    def update_status(self, name: str, **kw) -> None:
        m = self.models.get(name)
        if m is None:
            return
        changed = False
        for k, v in kw.items():
            if getattr(m.status, k) != v:
                setattr(m.status, k, v)
                changed = True
        if changed:
            self._notify("model", name)

    # write-back hooks: the in-memory store shares live objects with the
    # controller, so reconcile-path mutations (labels/finalizers/bounds,
    # adapter state) are already visible; the K8s-backed store overrides
    # these to PATCH the API server (kubestore.py)
    def persist_model_meta(self, model: Model) -> None:
        self._notify("model", model.name)

    def persist_replica_adapters(self, rep: "Replica") -> None:
        self._notify("replica", rep.name, rep.model)

    # ---------------------------------------------------------- replicas
    def add_replica(self, rep: Replica) -> Replica:
        rep.created_seq = next(self._seq)
        self.replicas[rep.name] = rep
        self._notify("replica", rep.name, rep.model)
        return rep

    def get_replica(self, name: str) -> Optional[Replica]:
        return self.replicas.get(name)

    def list_replicas(self, model: Optional[str] = None) -> list[Replica]:
        return [
            r
            for r in self.replicas.values()
            if model is None or r.model == model
        ]

    def update_replica(self, name: str, **kw) -> None:
        r = self.replicas.get(name)
        if r is None:
            return
        changed = False
        for k, v in kw.items():
            if getattr(r, k) != v:
                setattr(r, k, v)
                changed = True
        if changed:
            self._notify("replica", name, r.model)

    def remove_replica(self, name: str) -> None:
        r = self.replicas.pop(name, None)
        if r is not None:
            self._notify("replica", name, r.model)
