"""Model lookup + scaling client (reference: internal/modelclient/).

Scale() applies min/max bounds and the consecutive-scale-down hysteresis
(scale.go:43-100); scale_at_least_one_replica() is the 0->1 fast path the
proxy hits before awaiting an endpoint (scale.go:14-39).
"""
from __future__ import annotations

from typing import Optional

from .crd import Model
from .store import Store


class ModelClient:
    def __init__(
        self,
        store: Store,
        required_consecutive_scale_downs: int = 1,
        autoscaling_interval: float = 10.0,
    ):
        self.store = store
        self.required_consecutive_scale_downs = required_consecutive_scale_downs
        self.autoscaling_interval = autoscaling_interval
        self._consecutive_scale_downs: dict[str, int] = {}

    def lookup_model(
        self, name: str, selectors: Optional[dict[str, str]] = None
    ) -> Optional[Model]:
        m = self.store.get_model(name)
        if m is None or m.deleted:
            return None
        if selectors and any(m.labels.get(k) != v for k, v in selectors.items()):
            return None
        return m

    def list_all_models(self) -> list[Model]:
        return self.store.list_models()

    def scale_at_least_one_replica(self, name: str) -> None:
        m = self.store.get_model(name)
        if m is None or m.deleted or m.spec.autoscaling_disabled:
            return
        if (m.spec.replicas or 0) == 0 and (m.spec.max_replicas or 1) > 0:
            self.store.scale_model(name, 1)

    def scale(self, name: str, replicas: int) -> None:
        m = self.store.get_model(name)
        if m is None or m.deleted:
            return
        s = m.spec
        lo = s.min_replicas
        hi = s.max_replicas if s.max_replicas is not None else replicas
        target = max(lo, min(replicas, hi))
        current = s.replicas or 0
        if target < current:
            # scale-down hysteresis: require N consecutive ticks; N derives
            # from the per-model scaleDownDelaySeconds (reference:
            # model_types.go:118-121 + config.RequiredConsecutiveScaleDowns)
            per_model = max(
                1, int(s.scale_down_delay_seconds / max(self.autoscaling_interval, 1e-9))
            )
            required = max(self.required_consecutive_scale_downs, per_model)
            n = self._consecutive_scale_downs.get(name, 0) + 1
            self._consecutive_scale_downs[name] = n
            if n < required:
                return
        else:
            self._consecutive_scale_downs[name] = 0
        if target != current:
            self.store.scale_model(name, target)
            self._consecutive_scale_downs[name] = 0
