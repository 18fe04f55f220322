"""Lightweight span tracing.

Role parity with the reference's OTel wiring (reference:
pkg/observability/tracing.go:29-71 — gated StartSpan that is a no-op when
disabled; trace context persisted into run status trace_types.go).  Spans
collect into an in-process ring buffer exportable as JSON; on a GPU box the
same span names appear as rocprof ranges when torch profiling is active.
"""
from __future__ import annotations

import contextlib
import json
import threading
import time
import typing as _t
import uuid
from collections import deque
from dataclasses import dataclass, field


@dataclass
class Span:
    name: str
    trace_id: str
    span_id: str
    parent_id: str = ""
    start: float = 0.0
    end: float = 0.0
    attributes: dict = field(default_factory=dict)
    error: str = ""

    @property
    def duration(self) -> float:
        return self.end - self.start


class Tracer:
    def __init__(self, enabled: bool = False, capacity: int = 4096):
        self.enabled = enabled
        self._spans: deque = deque(maxlen=capacity)
        self._lock = threading.Lock()
        self._local = threading.local()

    @contextlib.contextmanager
    def span(self, name: str, trace_id: str = "", **attributes):
        if not self.enabled:
            yield None
            return
        parent = getattr(self._local, "current", None)
        sp = Span(
            name=name,
            trace_id=trace_id or (parent.trace_id if parent else uuid.uuid4().hex),
            span_id=uuid.uuid4().hex[:16],
            parent_id=parent.span_id if parent else "",
            start=time.time(),
            attributes=attributes,
        )
        self._local.current = sp
        try:
            yield sp
        except Exception as exc:
            sp.error = str(exc)
            raise
        finally:
            sp.end = time.time()
            self._local.current = parent
            with self._lock:
                self._spans.append(sp)

    def spans(self, name: _t.Optional[str] = None) -> _t.List[Span]:
        with self._lock:
            items = list(self._spans)
        if name is not None:
            items = [s for s in items if s.name == name]
        return items

    def export_json(self) -> str:
        return json.dumps(
            [
                {
                    "name": s.name,
                    "traceId": s.trace_id,
                    "spanId": s.span_id,
                    "parentId": s.parent_id,
                    "start": s.start,
                    "durationMs": s.duration * 1000.0,
                    "attributes": s.attributes,
                    "error": s.error,
                }
                for s in self.spans()
            ],
            indent=2,
        )


GLOBAL = Tracer(enabled=False)


# ---------------------------------------------------------------------------
# OTLP export (role parity: reference pkg/observability/exporter.go — OTLP
# exporter wired at startup, flushed LIFO at shutdown, cmd/main.go:253-288).
# Spans serialize to the OTLP/JSON ResourceSpans shape; the sink is a file
# path or an OTLP/HTTP endpoint (http(s)://host:4318/v1/traces).
# ---------------------------------------------------------------------------


def _otlp_attr(key: str, value) -> dict:
    if isinstance(value, bool):
        v = {"boolValue": value}
    elif isinstance(value, int):
        v = {"intValue": str(value)}
    elif isinstance(value, float):
        v = {"doubleValue": value}
    else:
        v = {"stringValue": str(value)}
    return {"key": key, "value": v}


def spans_to_otlp(spans: _t.Iterable[Span], service_name: str = "bobrapet-amd") -> dict:
    otlp_spans = []
    for s in spans:
        otlp_spans.append(
            {
                "traceId": s.trace_id[:32].ljust(32, "0"),
                "spanId": s.span_id[:16].ljust(16, "0"),
                "parentSpanId": s.parent_id[:16].ljust(16, "0") if s.parent_id else "",
                "name": s.name,
                "kind": 1,  # SPAN_KIND_INTERNAL
                "startTimeUnixNano": str(int(s.start * 1e9)),
                "endTimeUnixNano": str(int(s.end * 1e9)),
                "attributes": [_otlp_attr(k, v) for k, v in s.attributes.items()],
                "status": {"code": 2, "message": s.error} if s.error else {"code": 1},
            }
        )
    return {
        "resourceSpans": [
            {
                "resource": {"attributes": [_otlp_attr("service.name", service_name)]},
                "scopeSpans": [
                    {"scope": {"name": "bobrapet_amd.tracer"}, "spans": otlp_spans}
                ],
            }
        ]
    }


class OTLPExporter:
    """Flush tracer spans as OTLP/JSON to a file or an OTLP/HTTP endpoint.

    Endpoint forms: a filesystem path (append one JSON document per flush,
    newline-delimited) or http(s)://host:port/v1/traces (POST, stdlib
    urllib — no extra deps; failures are counted, never raised into the
    engine)."""

    def __init__(self, endpoint: str, service_name: str = "bobrapet-amd"):
        self.endpoint = endpoint
        self.service_name = service_name
        self.exported = 0
        self.errors = 0
        self._cursor = 0

    def flush(self, tracer: "Tracer") -> int:
        spans = tracer.spans()
        fresh = spans[self._cursor :]
        self._cursor = len(spans)
        if not fresh:
            return 0
        doc = spans_to_otlp(fresh, self.service_name)
        payload = json.dumps(doc, separators=(",", ":"))
        try:
            if self.endpoint.startswith(("http://", "https://")):
                import urllib.request

                req = urllib.request.Request(
                    self.endpoint,
                    data=payload.encode(),
                    headers={"Content-Type": "application/json"},
                    method="POST",
                )
                urllib.request.urlopen(req, timeout=5.0).read()
            else:
                with open(self.endpoint, "a", encoding="utf-8") as fh:
                    fh.write(payload + "\n")
            self.exported += len(fresh)
            return len(fresh)
        except Exception:
            self.errors += 1
            return 0
