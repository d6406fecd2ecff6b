"""Control-plane metrics (reference: internal/metrics/metrics.go:16-79).

kubeai_inference_requests_active is THE autoscaling signal: the autoscaler
scrapes it from every control-plane replica and sums (autoscaler.go:118-137).
"""
import prometheus_client as prom

REGISTRY = prom.CollectorRegistry()

INFERENCE_REQUESTS_ACTIVE = prom.Gauge(
    "kubeai_inference_requests_active",
    "active inference requests by model",
    ["model"],
    registry=REGISTRY,
)
HASH_LOOKUP_ITERATIONS = prom.Histogram(
    "kubeai_inference_requests_hash_lookup_iterations",
    "CHWBL ring-walk iterations",
    registry=REGISTRY,
    buckets=(1, 2, 4, 8, 16, 32, 64, 128, 256),
)
HASH_LOOKUP_INITIAL = prom.Counter(
    "kubeai_inference_requests_hash_lookup_initial",
    "CHWBL lookups that found an initial candidate",
    registry=REGISTRY,
)
HASH_LOOKUP_DEFAULT = prom.Counter(
    "kubeai_inference_requests_hash_lookup_default",
    "CHWBL lookups that fell back to default selection",
    registry=REGISTRY,
)


def render() -> bytes:
    return prom.generate_latest(REGISTRY)
