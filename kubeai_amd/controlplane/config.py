"""System configuration (reference: internal/config/system.go).

Single YAML file -> SystemConfig with defaulting + validation; shipped as a
ConfigMap in cluster deployments (deploy/kubeai/templates/configmap.yaml).
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import yaml


def parse_duration(v) -> float:
    """Accept Go-style duration strings ("10s", "10m", "1h30m") or numbers
    (reference: config durations are metav1.Duration)."""
    if isinstance(v, (int, float)):
        return float(v)
    import re

    total = 0.0
    for num, unit in re.findall(r"([0-9.]+)(ms|s|m|h)", str(v)):
        total += float(num) * {"ms": 0.001, "s": 1, "m": 60, "h": 3600}[unit]
    if total == 0.0 and str(v).strip():
        try:
            return float(v)
        except ValueError:
            raise ValueError(f"invalid duration: {v!r}")
    return total


@dataclasses.dataclass
class AutoscalingConfig:
    # reference: system.go:119-146
    interval_seconds: float = 10.0
    time_window_seconds: float = 600.0
    state_path: Optional[str] = "/tmp/kubeai-amd-autoscaler-state.json"

    @property
    def required_consecutive_scale_downs(self) -> int:
        # ceil(scaleDownDelaySeconds? — derived per model); system-level:
        # reference derives from time window; per-model delay handled in
        # modelclient. Here: one full averaging window must agree.
        return max(1, int(self.time_window_seconds / self.interval_seconds / 10))


@dataclasses.dataclass
class MessagingStream:
    requests_url: str = "mem://requests"
    responses_url: str = "mem://responses"
    max_handlers: int = 8


@dataclasses.dataclass
class SystemConfig:
    # resourceProfiles: name -> GPUs per unit (reference: system.go:191-200)
    resource_profiles: dict = dataclasses.field(
        default_factory=lambda: {"amd-gpu-mi355x": 1, "cpu": 0}
    )
    cache_dir: str = "/tmp/kubeai-cache"
    autoscaling: AutoscalingConfig = dataclasses.field(default_factory=AutoscalingConfig)
    messaging: list[MessagingStream] = dataclasses.field(default_factory=list)
    api_port: int = 8000
    metrics_port: int = 8080
    fixed_self_metric_addrs: list[str] = dataclasses.field(default_factory=list)
    leader_lock_path: str = "/tmp/kubeai-amd-leader.lock"
    n_gpus: Optional[int] = None  # None -> detect
    # priorityClasses: name -> importance value (k8s PriorityClass analog,
    # HIGHER = more important). A Model's priorityClassName maps every
    # request for that model onto the engine's admission priority.
    priority_classes: dict = dataclasses.field(default_factory=dict)
    # cacheProfiles: name -> {"sharedFilesystem": {"storageClassName":
    # ..., "persistentVolumeName": ...}} (reference config/system.go:202-
    # 208); consumed by the K8s-mode PVC/Job cache machinery
    cacheProfiles: dict = dataclasses.field(default_factory=dict)
    # kubernetes: when set, Models persist as CRs and replicas run as
    # Pods against this API server (kubestore.py). Keys: apiUrl (empty =
    # in-cluster), namespace, engineImage, gpuResource, leaseName.
    kubernetes: Optional[dict] = None

    def validate(self) -> None:
        if self.autoscaling.interval_seconds <= 0:
            raise ValueError("autoscaling.interval_seconds must be > 0")
        if self.autoscaling.time_window_seconds < self.autoscaling.interval_seconds:
            raise ValueError("time window must be >= interval")
        for name, gpus in self.resource_profiles.items():
            if gpus < 0:
                raise ValueError(f"resource profile {name}: negative GPU count")


def load_config(path: Optional[str]) -> SystemConfig:
    if not path:
        cfg = SystemConfig()
        cfg.validate()
        return cfg
    with open(path) as f:
        raw = yaml.safe_load(f) or {}
    auto = raw.get("modelAutoscaling", {})
    cfg = SystemConfig(
        resource_profiles={
            k: int(v.get("gpus", v) if isinstance(v, dict) else v)
            for k, v in raw.get("resourceProfiles", {"amd-gpu-mi355x": 1, "cpu": 0}).items()
        },
        cache_dir=raw.get("cacheDir", "/tmp/kubeai-cache"),
        autoscaling=AutoscalingConfig(
            interval_seconds=parse_duration(auto.get("interval", 10)),
            time_window_seconds=parse_duration(auto.get("timeWindow", 600)),
            state_path=auto.get("statePath", "/tmp/kubeai-amd-autoscaler-state.json"),
        ),
        messaging=[
            MessagingStream(
                requests_url=s.get("requestsURL", "mem://requests"),
                responses_url=s.get("responsesURL", "mem://responses"),
                max_handlers=int(s.get("maxHandlers", 8)),
            )
            for s in raw.get("messaging", {}).get("streams", [])
        ],
        api_port=int(raw.get("apiPort", 8000)),
        metrics_port=int(raw.get("metricsPort", 8080)),
        fixed_self_metric_addrs=list(raw.get("fixedSelfMetricAddrs", [])),
        n_gpus=raw.get("nGPUs"),
        priority_classes={
            str(k): int(v) for k, v in raw.get("priorityClasses", {}).items()
        },
        kubernetes=raw.get("kubernetes"),
        cacheProfiles=raw.get("cacheProfiles") or {},
    )
    cfg.validate()
    return cfg
