"""W3C TraceContext tracing skeleton.

The reference installs OTel propagators + an (exporter-less) trace
provider and wraps handlers in otelhttp route tags
(internal/manager/otel.go:40-97, openaiserver/handler.go:28-46) — traces
propagate even with no backend attached. Same posture here without the
OTel SDK dependency: parse/generate `traceparent`, record a server span
per gateway request (route tag, status, duration), inject the child
context into the forwarded engine request, and keep a bounded in-memory
ring of recent spans served at `/debug/traces` (the no-backend
inspection surface).
"""
from __future__ import annotations

import dataclasses
import secrets
import threading
import time
from collections import deque
from typing import Optional


@dataclasses.dataclass
class SpanContext:
    trace_id: str  # 32 hex chars
    span_id: str   # 16 hex chars
    sampled: bool = True

    def traceparent(self) -> str:
        flags = "01" if self.sampled else "00"
        return f"00-{self.trace_id}-{self.span_id}-{flags}"


def parse_traceparent(header: Optional[str]) -> Optional[SpanContext]:
    if not header:
        return None
    parts = header.strip().split("-")
    if len(parts) != 4 or len(parts[1]) != 32 or len(parts[2]) != 16:
        return None
    if parts[1] == "0" * 32 or parts[2] == "0" * 16:
        return None
    try:
        flags = int(parts[3], 16)
    except ValueError:
        return None
    return SpanContext(parts[1], parts[2], bool(flags & 1))


def new_context(parent: Optional[SpanContext] = None) -> SpanContext:
    return SpanContext(
        trace_id=parent.trace_id if parent else secrets.token_hex(16),
        span_id=secrets.token_hex(8),
        sampled=parent.sampled if parent else True,
    )


@dataclasses.dataclass
class Span:
    name: str
    trace_id: str
    span_id: str
    parent_span_id: Optional[str]
    start: float
    duration_ms: float = 0.0
    attributes: dict = dataclasses.field(default_factory=dict)

    def to_dict(self) -> dict:
        return {
            "name": self.name,
            "traceId": self.trace_id,
            "spanId": self.span_id,
            "parentSpanId": self.parent_span_id,
            "startTimeUnixNano": int(self.start * 1e9),
            "durationMs": round(self.duration_ms, 3),
            "attributes": self.attributes,
        }


class Tracer:
    """Bounded ring of completed spans; thread-safe."""

    def __init__(self, capacity: int = 512):
        self._spans: deque[Span] = deque(maxlen=capacity)
        self._lock = threading.Lock()

    def start_span(self, name: str,
                   parent: Optional[SpanContext] = None) -> "_Live":
        ctx = new_context(parent)
        return _Live(self, name, ctx, parent.span_id if parent else None)

    def _record(self, span: Span) -> None:
        with self._lock:
            self._spans.append(span)

    def recent(self, limit: int = 100) -> list[dict]:
        with self._lock:
            out = list(self._spans)[-limit:]
        return [s.to_dict() for s in reversed(out)]


class _Live:
    def __init__(self, tracer: Tracer, name: str, ctx: SpanContext,
                 parent_span_id: Optional[str]):
        self.tracer = tracer
        self.ctx = ctx
        self.span = Span(
            name=name, trace_id=ctx.trace_id, span_id=ctx.span_id,
            parent_span_id=parent_span_id, start=time.time(),
        )
        self._t0 = time.monotonic()

    def set(self, key: str, value) -> None:
        self.span.attributes[key] = value

    def __enter__(self) -> "_Live":
        return self

    def __exit__(self, exc_type, exc, tb) -> None:
        self.end(error=exc is not None)

    def end(self, error: bool = False) -> None:
        self.span.duration_ms = (time.monotonic() - self._t0) * 1e3
        if error:
            self.span.attributes.setdefault("error", True)
        self.tracer._record(self.span)


TRACER = Tracer()  # process-wide default, like the global OTel provider
