"""Distributed tracing (reference: observability/otel_trace.rs — OTLP batch
exporter, W3C TraceContext propagated into engines (http/router.rs:466
inject_trace_context_http); structured events events.rs).

The image has no opentelemetry package, so spans are W3C-correct and exported
to the structured log (and an in-memory ring for tests); the OTLP exporter
slot is a hook for deployments that install one.
"""
from __future__ import annotations

import collections
import contextvars
import logging
import secrets
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

log = logging.getLogger("smg.trace")

_current_span: contextvars.ContextVar[Optional["Span"]] = contextvars.ContextVar("smg_span", default=None)


@dataclass
class Span:
    name: str
    trace_id: str
    span_id: str
    parent_id: Optional[str] = None
    start: float = field(default_factory=time.time)
    end: Optional[float] = None
    attributes: Dict[str, object] = field(default_factory=dict)
    events: List[dict] = field(default_factory=list)

    def set(self, key: str, value) -> None:
        self.attributes[key] = value

    def add_event(self, name: str, **attrs) -> None:
        self.events.append({"name": name, "ts": time.time(), **attrs})

    def traceparent(self) -> str:
        return f"00-{self.trace_id}-{self.span_id}-01"


class Tracer:
    def __init__(self, enabled: bool = True, ring_size: int = 1024, otlp_exporter=None):
        self.enabled = enabled
        self.finished: collections.deque = collections.deque(maxlen=ring_size)
        self.otlp_exporter = otlp_exporter

    def start_span(self, name: str, traceparent: Optional[str] = None, **attrs) -> Span:
        parent = _current_span.get()
        if traceparent and parent is None:
            parts = traceparent.split("-")
            trace_id = parts[1] if len(parts) >= 3 else secrets.token_hex(16)
            parent_id = parts[2] if len(parts) >= 3 else None
        elif parent is not None:
            trace_id, parent_id = parent.trace_id, parent.span_id
        else:
            trace_id, parent_id = secrets.token_hex(16), None
        span = Span(name, trace_id, secrets.token_hex(8), parent_id, attributes=attrs)
        return span

    def end_span(self, span: Span) -> None:
        span.end = time.time()
        if not self.enabled:
            return
        self.finished.append(span)
        log.debug(
            "span %s trace=%s dur_ms=%.2f attrs=%s",
            span.name, span.trace_id, (span.end - span.start) * 1e3, span.attributes,
        )
        if self.otlp_exporter is not None:
            try:
                self.otlp_exporter(span)
            except Exception:
                pass

    class _SpanCtx:
        def __init__(self, tracer, span):
            self.tracer, self.span = tracer, span

        def __enter__(self):
            self._token = _current_span.set(self.span)
            return self.span

        def __exit__(self, *exc):
            _current_span.reset(self._token)
            self.tracer.end_span(self.span)

    def span(self, name: str, traceparent: Optional[str] = None, **attrs) -> "Tracer._SpanCtx":
        return Tracer._SpanCtx(self, self.start_span(name, traceparent, **attrs))


GLOBAL_TRACER = Tracer(enabled=False)


def inject_trace_context(headers: Dict[str, str]) -> None:
    """Stamp the current span's W3C traceparent onto outbound worker headers
    (reference inject_trace_context_http)."""
    span = _current_span.get()
    if span is not None:
        headers["traceparent"] = span.traceparent()


class InFlightTracker:
    """Age-bucketed in-flight request gauges (reference inflight_tracker.rs:22)."""

    BUCKETS = (1, 5, 30, 120, 600)

    def __init__(self):
        self._inflight: Dict[str, float] = {}

    def start(self, request_id: str) -> None:
        self._inflight[request_id] = time.monotonic()

    def finish(self, request_id: str) -> None:
        self._inflight.pop(request_id, None)

    def age_histogram(self) -> Dict[str, int]:
        now = time.monotonic()
        out = {f"<{b}s": 0 for b in self.BUCKETS}
        out["older"] = 0
        for t0 in self._inflight.values():
            age = now - t0
            for b in self.BUCKETS:
                if age < b:
                    out[f"<{b}s"] += 1
                    break
            else:
                out["older"] += 1
        return out

    def __len__(self) -> int:
        return len(self._inflight)
