"""Model validation matrix (reference: test/integration/model_validation_test.go
+ CEL rules in api/k8s/v1/model_types.go:27-35)."""
import dataclasses

import pytest

from kubeai_amd.controlplane.crd import (
    AdapterSpec,
    Model,
    ModelSpec,
    ValidationError,
    validate_model,
)


def mk(**kw):
    spec = ModelSpec(url="hf://meta-llama/Llama-3-8B", **kw)
    return Model(name="m", spec=spec)


def test_valid_model():
    validate_model(mk())


@pytest.mark.parametrize(
    "url", ["http://x", "ftp://y", "", "hf:/missing-slash", "s3:bad"]
)
def test_invalid_url_scheme(url):
    m = mk()
    m.spec.url = url
    with pytest.raises(ValidationError):
        validate_model(m)


@pytest.mark.parametrize("url", ["hf://a/b", "pvc://claim/path", "ollama://qwen2:0.5b", "s3://b/k", "gs://b/k", "oss://b/k"])
def test_valid_url_schemes(url):
    m = mk()
    m.spec.url = url
    validate_model(m)


def test_min_max_replicas():
    m = mk(min_replicas=3, max_replicas=2)
    with pytest.raises(ValidationError, match="minReplicas"):
        validate_model(m)
    validate_model(mk(min_replicas=2, max_replicas=2))


def test_bad_feature_engine_strategy():
    m = mk()
    m.spec.features = ["Nope"]
    with pytest.raises(ValidationError):
        validate_model(m)
    m = mk()
    m.spec.engine = "wat"
    with pytest.raises(ValidationError):
        validate_model(m)
    m = mk()
    m.spec.load_balancing.strategy = "RoundRobin"
    with pytest.raises(ValidationError):
        validate_model(m)


def test_adapter_name_rules():
    m = mk(adapters=[AdapterSpec(name="ok-adapter.1", url="hf://x/y")])
    validate_model(m)
    m = mk(adapters=[AdapterSpec(name="has_underscore", url="hf://x/y")])
    with pytest.raises(ValidationError):
        validate_model(m)
    m = mk(adapters=[AdapterSpec(name="-bad", url="hf://x/y")])
    with pytest.raises(ValidationError):
        validate_model(m)


def test_cache_profile_immutable():
    old = mk(cache_profile="fast")
    new = mk(cache_profile="slow")
    with pytest.raises(ValidationError, match="immutable"):
        validate_model(new, old)
    # url immutable when cacheProfile set
    new2 = mk(cache_profile="fast")
    new2.spec.url = "hf://other/model"
    with pytest.raises(ValidationError, match="immutable"):
        validate_model(new2, old)


def test_file_path_rules():
    from kubeai_amd.controlplane.crd import FileSpec

    m = mk(files=[FileSpec(path="relative.txt", content="x")])
    with pytest.raises(ValidationError):
        validate_model(m)
    m = mk(files=[FileSpec(path="/etc/../evil", content="x")])
    with pytest.raises(ValidationError):
        validate_model(m)
    validate_model(mk(files=[FileSpec(path="/config/f.json", content="x")]))
