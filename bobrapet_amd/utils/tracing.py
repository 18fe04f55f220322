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
