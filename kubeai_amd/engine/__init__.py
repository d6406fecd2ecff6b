"""Engine package. Lazy re-exports (PEP 562) so that importing a submodule
(e.g. engine.batch from the model definitions) does not pull the whole
engine stack and create an import cycle."""

_EXPORTS = {
    "ForwardBatch": ("kubeai_amd.engine.batch", "ForwardBatch"),
    "LLMEngine": ("kubeai_amd.engine.engine", "LLMEngine"),
    "EngineConfig": ("kubeai_amd.engine.engine", "EngineConfig"),
    "Request": ("kubeai_amd.engine.scheduler", "Request"),
    "SamplingParams": ("kubeai_amd.engine.scheduler", "SamplingParams"),
    "RequestStatus": ("kubeai_amd.engine.scheduler", "RequestStatus"),
}

__all__ = list(_EXPORTS)


def __getattr__(name):
    try:
        mod_name, attr = _EXPORTS[name]
    except KeyError:
        raise AttributeError(name) from None
    import importlib

    return getattr(importlib.import_module(mod_name), attr)
