from .batch import ForwardBatch
from .engine import LLMEngine, EngineConfig
from .scheduler import Request, SamplingParams, RequestStatus

__all__ = [
    "ForwardBatch",
    "LLMEngine",
    "EngineConfig",
    "Request",
    "SamplingParams",
    "RequestStatus",
]
