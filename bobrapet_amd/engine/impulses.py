"""Impulse runtime: always-on trigger workloads.

Role parity with the reference's Impulse materialization
(reference: internal/controller/impulse_controller.go:78-2217 — the
controller deploys trigger containers with the BUBU_* env contract and
delivery/throttle policy): here impulses are in-process handlers started
by the ImpulseRuntime; each emitted event goes through StoryTrigger
admission (dedupe/throttle — engine/triggers.py) and, for streaming
stories, packets stream into the live pipeline (the gRPC-ingress role of
BASELINE config #4 is the HTTP ingress in serve_http()).
"""
from __future__ import annotations

import itertools
import threading
import time
import typing as _t
from dataclasses import dataclass, field

from ..engrams.base import ImpulseHandler
from ..enums import StoryPattern, TriggerDecision
from ..specs import types as T
from ..templating import Evaluator
from .triggers import StoryTrigger

if _t.TYPE_CHECKING:
    from .engine import RunEngine


_IMPULSE_REGISTRY: _t.Dict[str, _t.Callable[[], ImpulseHandler]] = {}


def register_impulse(name: str, factory: _t.Callable[[], ImpulseHandler]) -> None:
    _IMPULSE_REGISTRY[name] = factory


class IntervalImpulse(ImpulseHandler):
    """Emits a payload every `intervalMs` (config), up to `count` times."""

    name = "interval"

    def __init__(self, interval_ms: float = 100.0, count: _t.Optional[int] = None):
        self.interval_ms = interval_ms
        self.count = count
        self._stop = threading.Event()
        self._thread: _t.Optional[threading.Thread] = None

    def configure(self, config: dict) -> None:
        self.interval_ms = float(config.get("intervalMs", self.interval_ms))
        if config.get("count") is not None:
            self.count = int(config["count"])

    def start(self, emit: _t.Callable[[dict], _t.Any]) -> None:
        def loop():
            n = 0
            while not self._stop.is_set():
                if self.count is not None and n >= self.count:
                    return
                emit({"tick": n, "ts": time.time()})
                n += 1
                self._stop.wait(self.interval_ms / 1000.0)

        self._thread = threading.Thread(target=loop, daemon=True, name="impulse-interval")
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2)


class ManualImpulse(ImpulseHandler):
    """Programmatic ingress: call .emit(payload) from application code (the
    in-process equivalent of the HTTP/gRPC connector)."""

    name = "manual"

    def __init__(self):
        self._emit: _t.Optional[_t.Callable] = None

    def configure(self, config: dict) -> None:
        pass

    def start(self, emit: _t.Callable[[dict], _t.Any]) -> None:
        self._emit = emit

    def emit(self, payload: dict):
        if self._emit is None:
            raise RuntimeError("impulse not started")
        return self._emit(payload)


register_impulse("interval", IntervalImpulse)
register_impulse("manual", ManualImpulse)


@dataclass
class LiveImpulse:
    impulse: T.Impulse
    handler: ImpulseHandler
    emitted: int = 0
    decisions: _t.Dict[str, int] = field(default_factory=dict)
    stream_key: _t.Optional[str] = None


class ImpulseRuntime:
    def __init__(self, engine: "RunEngine"):
        self.engine = engine
        self.live: _t.Dict[str, LiveImpulse] = {}
        self._seq = itertools.count()
        self._lock = threading.Lock()
        self._evaluator = Evaluator()
        # idempotent trigger-counting tokens (reference:
        # trigger_annotations.go) — survive impulse restarts
        self._counted_tokens: _t.Dict[str, set] = {}

    def start(self, impulse: _t.Union[T.Impulse, str]) -> LiveImpulse:
        """Materialize one Impulse: resolve its template to a builtin handler
        and start emitting."""
        if isinstance(impulse, str):
            ns, _, nm = impulse.rpartition("/")
            with self.engine.registry._lock:
                impulse = self.engine.registry.impulses[f"{ns or 'default'}/{nm}"]
        tpl = self.engine.registry.impulse_template(impulse.template_ref.name)
        impl_name = tpl.implementation
        factory = _IMPULSE_REGISTRY.get(impl_name)
        if factory is None:
            raise KeyError(
                f"no builtin impulse implementation {impl_name!r} "
                f"(known: {sorted(_IMPULSE_REGISTRY)})"
            )
        handler = factory()
        if hasattr(handler, "configure"):
            handler.configure(dict(impulse.with_ or {}))
        live = LiveImpulse(impulse=impulse, handler=handler)
        with self._lock:
            self.live[impulse.key] = live
        handler.start(lambda payload: self._on_event(live, payload))
        return live

    def stop(self, impulse_key: str) -> None:
        with self._lock:
            live = self.live.pop(impulse_key, None)
        if live is not None:
            live.handler.stop()

    def stop_all(self) -> None:
        with self._lock:
            keys = list(self.live)
        for k in keys:
            self.stop(k)

    def status(self, impulse_key: str, max_scan: int = 2000) -> dict:
        return aggregate_impulse_stats(self, impulse_key, max_scan=max_scan)

    # ------------------------------------------------------------------

    def _on_event(self, live: LiveImpulse, payload: dict):
        """One trigger event → mapped inputs → StoryTrigger admission; for
        streaming stories packets flow into the live pipeline."""
        eng = self.engine
        imp = live.impulse
        live.emitted += 1
        inputs = payload
        if imp.mapping is not None and imp.mapping.inputs is not None:
            inputs = self._evaluator.resolve_value(imp.mapping.inputs, {"event": payload})

        story_ns = imp.story_ref.resolve_namespace(imp.namespace)
        story = eng.registry.story(imp.story_ref.name, story_ns)

        if story.pattern == StoryPattern.STREAMING:
            # PerStoryRun pipeline: one live run per impulse; packets stream in
            if live.stream_key is None or eng.stream_of(live.stream_key) is None:
                stream = eng.submit_stream(
                    story, inputs={"impulse": imp.name}, namespace=imp.namespace
                )
                live.stream_key = stream.run.key
            stream = eng.stream_of(live.stream_key)
            stream.push(inputs)
            return {"streamed": True, "run": live.stream_key}

        dedupe_key = None
        if (
            imp.delivery is not None
            and imp.delivery.dedupe is not None
            and imp.delivery.dedupe.key_template
        ):
            dedupe_key = str(
                self._evaluator.resolve_string(
                    imp.delivery.dedupe.key_template, {"event": payload}
                )
            )
        trig = StoryTrigger(
            submission_id=f"{imp.name}-{next(self._seq)}",
            story_name=imp.story_ref.name,
            story_namespace=story_ns,
            namespace=imp.namespace,
            key=dedupe_key,
            inputs=inputs,
            impulse=imp.name,
        )
        result = eng.triggers.submit(trig, throttle=imp.throttle)
        live.decisions[str(result.decision)] = live.decisions.get(str(result.decision), 0) + 1
        eng.metrics.inc("impulse_events_total", impulse=imp.name, decision=str(result.decision))
        return result


def build_http_app(engine: "RunEngine"):
    """HTTP ingress (the reference's gRPC/webhook connector role): POST
    /impulses/{ns}/{name} emits an event; GET /healthz, /metrics."""
    from fastapi import FastAPI, HTTPException
    from fastapi.responses import PlainTextResponse

    app = FastAPI(title="bobrapet_amd ingress")
    runtime: ImpulseRuntime = engine.impulses

    @app.post("/impulses/{ns}/{name}")
    async def trigger(ns: str, name: str, payload: dict):
        live = runtime.live.get(f"{ns}/{name}")
        if live is None:
            raise HTTPException(404, f"impulse {ns}/{name} not running")
        handler = live.handler
        if isinstance(handler, ManualImpulse):
            result = handler.emit(payload)
        else:
            result = runtime._on_event(live, payload)
        if isinstance(result, StoryTrigger):
            return {
                "decision": str(result.decision),
                "storyRun": result.story_run_ref,
                "message": result.message,
            }
        return result

    # ---- run control plane (the kube-apiserver role for runs) ----

    def _run_view(run) -> dict:
        return {
            "name": run.name,
            "namespace": run.namespace,
            "story": f"{run.story_namespace}/{run.story_name}",
            "phase": str(run.phase),
            "execPhase": run.exec_phase,
            "steps": {
                k: {
                    "phase": str(st.phase),
                    "retries": st.retries,
                    "message": st.message,
                    "error": st.error.message if st.error else None,
                }
                for k, st in run.step_states.items()
            },
            "output": run.output,
            "error": run.error.to_dict() if run.error else None,
            "degraded": run.degraded,
        }

    def _get_run(ns: str, name: str):
        run = engine.store.try_get_story_run(f"{ns}/{name}")
        if run is None:
            raise HTTPException(404, f"run {ns}/{name} not found")
        return run

    @app.post("/resources")
    async def apply_resources(body: dict):
        """Apply CRD-style YAML documents (the admission-webhook surface:
        invalid specs are rejected with the validation errors)."""
        text = body.get("yaml")
        if not isinstance(text, str):
            raise HTTPException(400, "body must be {yaml: '<documents>'}")
        try:
            applied = engine.apply_yaml(text)
        except ValueError as exc:
            raise HTTPException(422, str(exc))
        return {"applied": len(applied)}

    @app.get("/stories")
    async def list_stories():
        return {
            "stories": [
                {"key": key, **engine.registry.story_status(key)}
                for key in sorted(engine.registry.stories)
            ]
        }

    @app.post("/stories/{ns}/{name}/runs")
    async def submit(ns: str, name: str, body: _t.Optional[dict] = None):
        body = body or {}
        try:
            run = engine.submit_run(
                f"{ns}/{name}", body.get("inputs") or {}, name=body.get("runName")
            )
        except KeyError:
            raise HTTPException(404, f"story {ns}/{name} not found")
        except ValueError as exc:
            raise HTTPException(422, str(exc))
        if body.get("wait"):
            run = engine.wait(run, timeout=float(body.get("timeout", 300.0)))
        return _run_view(run)

    @app.get("/runs")
    async def list_runs(phase: _t.Optional[str] = None, limit: int = 100):
        runs = []
        for run in engine.store.all_runs():
            if phase and str(run.phase) != phase:
                continue
            runs.append(
                {
                    "name": run.name,
                    "namespace": run.namespace,
                    "story": f"{run.story_namespace}/{run.story_name}",
                    "phase": str(run.phase),
                }
            )
            if len(runs) >= max(1, min(limit, 1000)):
                break
        return {"runs": runs}

    @app.get("/runs/{ns}/{name}")
    async def run_status(ns: str, name: str):
        return _run_view(_get_run(ns, name))

    @app.get("/runs/{ns}/{name}/trace")
    async def run_trace(ns: str, name: str):
        """Span trace for one run (the reference persists TraceInfo on run
        status; spans here come from the in-process tracer ring)."""
        run = _get_run(ns, name)
        spans = [
            {
                "name": sp.name,
                "start": sp.start,
                "end": sp.end,
                "durationMs": round(sp.duration * 1e3, 3),
                "attributes": sp.attributes,
                "error": sp.error,
            }
            for sp in engine.tracer.spans()
            if sp.attributes.get("run") == run.name or sp.trace_id == run.trace.trace_id
        ]
        return {"traceId": run.trace.trace_id, "spans": spans}

    @app.post("/runs/{ns}/{name}/cancel")
    async def cancel_run(ns: str, name: str, body: _t.Optional[dict] = None):
        run = _get_run(ns, name)
        engine.cancel(run, graceful=bool((body or {}).get("graceful", True)))
        return {"ok": True, "phase": str(run.phase)}

    @app.post("/runs/{ns}/{name}/redrive")
    async def redrive_run(ns: str, name: str, body: _t.Optional[dict] = None):
        run = _get_run(ns, name)
        step = (body or {}).get("fromStep")
        try:
            if step:
                engine.redrive_from_step(run, step)
            else:
                engine.redrive(run)
        except ValueError as exc:
            raise HTTPException(409, str(exc))
        return {"ok": True}

    @app.post("/runs/{ns}/{name}/gates/{step}")
    async def decide_gate(ns: str, name: str, step: str, body: dict):
        run = _get_run(ns, name)
        approve = bool(body.get("approve", False))
        decided_by = str(body.get("decidedBy", "http"))
        if approve:
            engine.approve_gate(run, step, decided_by)
        else:
            engine.reject_gate(run, step, decided_by)
        return {"ok": True, "state": "Approved" if approve else "Rejected"}

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    @app.get("/metrics", response_class=PlainTextResponse)
    async def metrics():
        return engine.metrics.export_text()

    return app


def serve_http(engine: "RunEngine", host: str = "127.0.0.1", port: int = 8080):
    import uvicorn

    uvicorn.run(build_http_app(engine), host=host, port=port, log_level="warning")


# ---------------------------------------------------------------------------
# trigger-stats aggregation + bounded backfill (reference:
# internal/controller/impulse_controller.go:1151-1233 trigger stats from
# StoryRuns, trigger_annotations.go idempotent counting tokens with
# bounded backfill loops / bobrapet_trigger_backfill_* metrics)
# ---------------------------------------------------------------------------


def aggregate_impulse_stats(
    runtime: "ImpulseRuntime",
    impulse_key: str,
    max_scan: int = 2000,
):
    """Status of one Impulse: live emit/decision counters plus a BOUNDED
    backfill over existing StoryRuns labelled with this impulse — each
    run's trigger token is counted exactly once (idempotent tokens kept on
    the runtime, so repeated aggregation and impulse restarts never
    double-count, like the reference's counting annotations)."""
    eng = runtime.engine
    ns, _, name = impulse_key.rpartition("/")
    live = runtime.live.get(impulse_key)
    counted = runtime._counted_tokens.setdefault(impulse_key, set())

    scanned = 0
    backfilled = 0
    active = 0
    phases: _t.Dict[str, int] = {}
    capped = False
    for run in eng.store.all_runs():
        if scanned >= max_scan:
            capped = True
            break
        scanned += 1
        if run.labels.get("impulse") != name:
            continue
        if not run.is_terminal:
            active += 1
        phases[str(run.phase)] = phases.get(str(run.phase), 0) + 1
        for tok in run.trigger_tokens:
            if tok not in counted:
                counted.add(tok)
                backfilled += 1
    eng.metrics.inc("trigger_backfill_scans_total", impulse=name)
    if backfilled:
        eng.metrics.inc("trigger_backfill_counted_total", impulse=name, n=backfilled)

    return {
        "impulse": impulse_key,
        "running": live is not None,
        "emitted": live.emitted if live else 0,
        "decisions": dict(live.decisions) if live else {},
        "triggers": len(counted),
        "backfilled": backfilled,
        "activeRuns": active,
        "runPhases": phases,
        "scanCapped": capped,
    }
