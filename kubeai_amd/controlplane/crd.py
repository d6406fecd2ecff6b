"""Model resource types — the control plane's CRD analog.

Field-for-field parity with the reference CRD (reference:
api/k8s/v1/model_types.go:37-249), including the CEL validation rules
(model_types.go:27-35) re-expressed in validate(). The same schema ships as
a real CRD manifest in deploy/crds/ for cluster use; in-process the spec is
a plain dataclass living in the object store.
"""
from __future__ import annotations

import dataclasses
import re
from typing import Optional

# metadata constants (reference: api/k8s/v1/metadata.go:3-31)
POD_MODEL_LABEL = "model"
POD_HASH_LABEL = "pod-hash"
FEATURE_LABEL_DOMAIN = "features.kubeai.org"
MODEL_POD_IP_ANNOTATION = "model-pod-ip"
MODEL_POD_PORT_ANNOTATION = "model-pod-port"
ADAPTER_LABEL_PREFIX = "adapter.kubeai.org/"
CACHE_EVICTION_FINALIZER = "kubeai.org/cache-eviction"

FEATURES = ("TextGeneration", "TextEmbedding", "Reranking", "SpeechToText")
ENGINES = ("KubeAIEngine", "OLlama", "VLLM", "FasterWhisper", "Infinity")
LB_STRATEGIES = ("LeastLoad", "PrefixHash")

_URL_RE = re.compile(r"^(hf|pvc|ollama|s3|gs|oss|file)://")


@dataclasses.dataclass
class PrefixHashSpec:
    # reference: model_types.go:185-209
    mean_load_percentage: int = 125
    replication: int = 256
    prefix_char_length: int = 100


@dataclasses.dataclass
class LoadBalancingSpec:
    strategy: str = "LeastLoad"
    prefix_hash: PrefixHashSpec = dataclasses.field(default_factory=PrefixHashSpec)


@dataclasses.dataclass
class AdapterSpec:
    # reference: model_types.go:163-171
    name: str = ""
    url: str = ""


@dataclasses.dataclass
class FileSpec:
    # reference: model_types.go:211-225
    path: str = ""
    content: str = ""


@dataclasses.dataclass
class ModelSpec:
    url: str = ""
    features: list[str] = dataclasses.field(default_factory=lambda: ["TextGeneration"])
    engine: str = "KubeAIEngine"
    adapters: list[AdapterSpec] = dataclasses.field(default_factory=list)
    resource_profile: str = ""  # "name:count"
    cache_profile: str = ""  # immutable once set
    image: str = ""
    args: list[str] = dataclasses.field(default_factory=list)
    env: dict[str, str] = dataclasses.field(default_factory=dict)
    replicas: Optional[int] = None
    min_replicas: int = 0
    max_replicas: Optional[int] = None
    autoscaling_disabled: bool = False
    target_requests: int = 100  # reference default, model_types.go:112-116
    scale_down_delay_seconds: int = 30
    load_balancing: LoadBalancingSpec = dataclasses.field(
        default_factory=LoadBalancingSpec
    )
    files: list[FileSpec] = dataclasses.field(default_factory=list)
    priority_class_name: str = ""
    owner: str = ""


@dataclasses.dataclass
class ModelStatus:
    replicas_all: int = 0
    replicas_ready: int = 0
    cache_loaded: bool = False


@dataclasses.dataclass
class Model:
    name: str
    spec: ModelSpec
    status: ModelStatus = dataclasses.field(default_factory=ModelStatus)
    labels: dict[str, str] = dataclasses.field(default_factory=dict)
    annotations: dict[str, str] = dataclasses.field(default_factory=dict)
    finalizers: list[str] = dataclasses.field(default_factory=list)
    generation: int = 0
    deleted: bool = False
    uid: str = ""


class ValidationError(ValueError):
    pass


def validate_model(m: Model, old: Optional[Model] = None) -> None:
    """CEL-rule analogs (reference: model_types.go:27-35 + field markers)."""
    s = m.spec
    if not _URL_RE.match(s.url):
        raise ValidationError(f"invalid model url scheme: {s.url!r}")
    for f in s.features:
        if f not in FEATURES:
            raise ValidationError(f"unknown feature {f!r}")
    if s.engine not in ENGINES:
        raise ValidationError(f"unknown engine {s.engine!r}")
    if s.load_balancing.strategy not in LB_STRATEGIES:
        raise ValidationError(f"unknown LB strategy {s.load_balancing.strategy!r}")
    if s.min_replicas < 0:
        raise ValidationError("minReplicas must be >= 0")
    if s.max_replicas is not None and s.min_replicas > s.max_replicas:
        # CEL: "minReplicas should be less than or equal to maxReplicas"
        raise ValidationError("minReplicas should be less than or equal to maxReplicas")
    if s.replicas is not None and s.replicas < 0:
        raise ValidationError("replicas must be >= 0")
    if s.target_requests < 1:
        raise ValidationError("targetRequests must be >= 1")
    if s.resource_profile and ":" not in s.resource_profile:
        raise ValidationError("resourceProfile must be of the form <name>:<count>")
    for a in s.adapters:
        if not re.fullmatch(r"[a-z0-9]([a-z0-9\-._]*[a-z0-9])?", a.name or ""):
            raise ValidationError(f"invalid adapter name {a.name!r}")
        if "_" in a.name:
            raise ValidationError("adapter name must not contain '_'")
    for fl in s.files:
        if not fl.path.startswith("/") or ".." in fl.path:
            raise ValidationError(f"invalid file path {fl.path!r}")
    if old is not None:
        # CEL: cacheProfile is immutable; url immutable when cacheProfile set
        if old.spec.cache_profile and s.cache_profile != old.spec.cache_profile:
            raise ValidationError("cacheProfile is immutable")
        if old.spec.cache_profile and s.url != old.spec.url:
            raise ValidationError("url is immutable when cacheProfile is set")


def feature_labels(m: Model) -> dict[str, str]:
    # reference: model_controller.go:374-407 applySelfLabels
    return {f"{FEATURE_LABEL_DOMAIN}/{f}": "true" for f in m.spec.features}
