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


class OtlpHttpExporter:
    """Wire-level OTLP/HTTP JSON span exporter (reference otel_trace.rs's
    batch exporter; endpoint = `--otlp-traces-endpoint`, path /v1/traces).
    Spans batch until `max_batch` or `flush_interval_s`, then POST as the
    OTLP JSON mapping (resourceSpans -> scopeSpans -> spans with hex ids and
    unix-nano timestamps) — consumable by any OTLP collector, no otel SDK
    needed."""

    def __init__(self, endpoint: str, service_name: str = "smg-amd",
                 max_batch: int = 64, flush_interval_s: float = 5.0):
        self.endpoint = endpoint.rstrip("/")
        if not self.endpoint.endswith("/v1/traces"):
            self.endpoint += "/v1/traces"
        self.service_name = service_name
        self.max_batch = max_batch
        self.flush_interval_s = flush_interval_s
        self._buf: List[Span] = []
        self._last_flush = time.time()
        self._task = None
        self.exported = 0
        self.export_errors = 0

    # sync hook called from Tracer.end_span
    def __call__(self, span: Span) -> None:
        self._buf.append(span)
        if len(self._buf) >= self.max_batch or (
            time.time() - self._last_flush > self.flush_interval_s and self._buf
        ):
            self.schedule_flush()

    def schedule_flush(self) -> None:
        import asyncio

        try:
            loop = asyncio.get_running_loop()
        except RuntimeError:
            return  # no loop (sync test context); flush() can be awaited manually
        if self._task is None or self._task.done():
            self._task = loop.create_task(self.flush())

    @staticmethod
    def _span_json(s: Span) -> dict:
        to_nano = lambda t: str(int(t * 1e9))
        return {
            "traceId": s.trace_id,
            "spanId": s.span_id,
            "parentSpanId": s.parent_id or "",
            "name": s.name,
            "kind": 2,  # SERVER
            "startTimeUnixNano": to_nano(s.start),
            "endTimeUnixNano": to_nano(s.end or s.start),
            "attributes": [
                {"key": k, "value": {"stringValue": str(v)}} for k, v in s.attributes.items()
            ],
            "events": [
                {"timeUnixNano": to_nano(e.get("ts", s.start)), "name": e.get("name", "")}
                for e in s.events
            ],
        }

    def payload(self, spans: List[Span]) -> dict:
        return {
            "resourceSpans": [{
                "resource": {"attributes": [
                    {"key": "service.name", "value": {"stringValue": self.service_name}}
                ]},
                "scopeSpans": [{
                    "scope": {"name": "smg_amd.tracing"},
                    "spans": [self._span_json(s) for s in spans],
                }],
            }]
        }

    async def flush(self) -> int:
        if not self._buf:
            return 0
        batch, self._buf = self._buf, []
        self._last_flush = time.time()
        try:
            import aiohttp

            async with aiohttp.ClientSession() as session:
                async with session.post(
                    self.endpoint, json=self.payload(batch),
                    timeout=aiohttp.ClientTimeout(total=10),
                ) as resp:
                    if resp.status // 100 == 2:
                        self.exported += len(batch)
                        return len(batch)
                    self.export_errors += 1
        except Exception:
            self.export_errors += 1
        return 0


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
