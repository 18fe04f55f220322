"""RunEngine: the event-driven heart of bobrapet_amd.

This replaces the reference's controller-runtime manager + reconcile loops
(reference: cmd/main.go:113-335, storyrun_controller.go:216-291): one
event-loop thread processes run submissions, step completions, timers and
gate decisions; DAG ticks are in-memory graph passes (engine/dag.py); engram
steps run on (gpu, stream) worker slots (engine/workers.py).  The
reference's requeue-bound 5-30 s step-to-step latency becomes event-bound
microseconds.
"""
from __future__ import annotations

import heapq
import itertools
import queue
import threading
import typing as _t

from ..enums import ErrorType, ExitClass, Phase, classify_exit_code
from ..engrams import registry as engram_registry
from ..engrams.base import EngramContext, EngramFailure, EngramResult
from ..specs import types as T
from ..specs.validation import (
    validate_engram,
    validate_engram_template,
    validate_impulse,
    validate_impulse_template,
    validate_story,
    validate_transport,
)
from ..storage.manager import StorageManager
from ..templating import EvalConfig, Evaluator
from ..utils import metrics as metrics_mod
from ..utils import tracing as tracing_mod
from ..utils.jsonschema import apply_defaults, validate_instance
from .cache import StepCache
from .config import EngineConfig, ExecutionConfigResolver
from .dag import DAGReconciler
from .effects import EffectLedger
from .executor import StepExecutor, finish_step_run
from .records import (
    StepRun,
    StoryRun,
    StructuredError,
    input_hash,
    monotonic_now,
)
from .retry import compute_retry_delay, should_retry
from .store import NotFound, ResourceRegistry, RunStore
from .workers import WorkerPool

_UNKNOWN_RETRY_ABS_CAP = 50  # safety net so budget-free UNKNOWN retries terminate


_RUN_SEQ = __import__("itertools").count()  # GIL-atomic uniqueness suffix


class RunEngine:
    def __init__(
        self,
        config: _t.Optional[EngineConfig] = None,
        storage: _t.Optional[StorageManager] = None,
        device_count: _t.Optional[int] = None,
        device_ids: _t.Optional[_t.List[int]] = None,
        metrics=None,
        tracer=None,
    ):
        self.config = config or EngineConfig()
        self.registry = ResourceRegistry()
        self.store = RunStore()
        self.storage = storage or StorageManager(max_inline_size=self.config.max_inline_size)
        self.evaluator = Evaluator(
            EvalConfig(
                deterministic=self.config.template_deterministic,
                max_output_bytes=self.config.max_output_bytes,
                max_ops=self.config.template_max_ops,
                offloaded_policy=self.config.offloaded_data_policy,
            ),
            hydrator=self.storage.resolve_ref,
        )
        self.resolver = ExecutionConfigResolver(self.config)
        self.cache = StepCache(evaluator=self.evaluator)
        self.effects = EffectLedger()
        self.metrics = metrics if metrics is not None else metrics_mod.MetricsRegistry()
        self.evaluator.metrics = self.metrics
        from ..utils.logging import StructuredLogger

        self.log = StructuredLogger("engine")
        self.tracer = tracer if tracer is not None else tracing_mod.Tracer(enabled=False)
        self.workers = WorkerPool(
            device_count=device_count,
            device_ids=device_ids,
            workers_per_device=self.config.workers_per_device,
            cpu_workers=self.config.cpu_workers,
        )
        self.executor = StepExecutor(self)
        self.dag = DAGReconciler(self)
        from .triggers import TriggerAdmission

        self.triggers = TriggerAdmission(self)
        from .impulses import ImpulseRuntime

        self.impulses = ImpulseRuntime(self)

        self._events: "queue.Queue" = queue.Queue()
        self._timers: _t.List[_t.Tuple[float, int, str, str]] = []
        self._timer_seq = itertools.count()
        self._run_done: _t.Dict[str, threading.Event] = {}
        self._scheduling: _t.List[str] = []  # run keys waiting for admission
        self._lock = threading.RLock()
        self._running = False
        self._thread: _t.Optional[threading.Thread] = None
        self._retry_pending: _t.Dict[str, float] = {}  # steprun key → next_retry_at
        self._streams: _t.Dict[str, object] = {}  # run key → StreamingRun

    # ------------------------------------------------------------------
    # lifecycle
    # ------------------------------------------------------------------

    def start(self) -> "RunEngine":
        if self._running:
            return self
        self._running = True
        self._thread = threading.Thread(target=self._loop, name="run-engine", daemon=True)
        self._thread.start()
        if self.config.checkpoint_path:
            self.set_timer(
                monotonic_now() + self.config.checkpoint_interval_seconds, "", "checkpoint"
            )
        if self.config.otlp_endpoint:
            # span export (reference: observability.InitTracerProvider —
            # exporter started with the manager, flushed at shutdown)
            self.tracer.enabled = True
            self._otlp = tracing_mod.OTLPExporter(self.config.otlp_endpoint)
            self.set_timer(
                monotonic_now() + self.config.otlp_flush_interval_seconds, "", "otlp"
            )
        self.log.debug("engine started", feature="engine")
        return self

    def stop(self) -> None:
        if not self._running:
            return
        self.impulses.stop_all()
        for stream in list(self._streams.values()):
            stream.cancel()
        self._running = False
        self._events.put(("__stop__",))
        if self._thread is not None:
            self._thread.join(timeout=5)
        self.workers.shutdown()
        if getattr(self, "_otlp", None) is not None:
            self._otlp.flush(self.tracer)  # LIFO shutdown flush

    def __enter__(self) -> "RunEngine":
        return self.start()

    def __exit__(self, *exc) -> None:
        self.stop()

    # ------------------------------------------------------------------
    # resource application (admission-webhook parity on apply)
    # ------------------------------------------------------------------

    def apply(self, obj) -> None:
        if isinstance(obj, T.Story):
            validate_story(obj).raise_if_invalid()
        elif isinstance(obj, T.Engram):
            tpl = None
            if obj.template_ref is not None:
                try:
                    tpl = self.registry.engram_template(obj.template_ref.name)
                except NotFound:
                    tpl = None
            validate_engram(obj, tpl).raise_if_invalid()
        elif isinstance(obj, T.Impulse):
            tpl = None
            if obj.template_ref is not None:
                try:
                    tpl = self.registry.impulse_template(obj.template_ref.name)
                except NotFound:
                    tpl = None
            validate_impulse(obj, tpl).raise_if_invalid()
        elif isinstance(obj, T.EngramTemplate):
            validate_engram_template(obj).raise_if_invalid()
        elif isinstance(obj, T.ImpulseTemplate):
            validate_impulse_template(obj).raise_if_invalid()
        elif isinstance(obj, T.Transport):
            validate_transport(obj).raise_if_invalid()
        self.registry.apply(obj)
        # watch-driven wakeup: Blocked runs may now unblock
        # (reference: mapEngramToStepRuns steprun_controller.go:5710)
        if isinstance(obj, (T.Engram, T.EngramTemplate, T.Story)):
            for run in self.store.all_runs():
                if not run.is_terminal:
                    self._post(("tick", run.key))

    def apply_yaml(self, text: str) -> _t.List[object]:
        from ..specs import load_yaml

        objs = load_yaml(text)
        for obj in objs:
            self.apply(obj)
        return objs

    # ------------------------------------------------------------------
    # run submission / control
    # ------------------------------------------------------------------

    def submit_run(
        self,
        story: _t.Union[T.Story, str],
        inputs=None,
        name: _t.Optional[str] = None,
        namespace: _t.Optional[str] = None,
        parent_run: _t.Optional[str] = None,
        parent_step: _t.Optional[str] = None,
        recursion_depth: int = 0,
        trigger_token: _t.Optional[str] = None,
        labels: _t.Optional[dict] = None,
        _trusted: bool = False,
    ) -> StoryRun:
        if isinstance(story, str):
            ns, _, nm = story.rpartition("/")
            story = self.registry.story(nm, ns or "default")
        namespace = namespace or story.namespace
        if name is None:
            # timestamp for readability + atomic counter for uniqueness:
            # concurrent submitters in the same clock tick must never
            # collide (a collision silently adopts the other run)
            name = (
                f"{story.name}-{monotonic_now():.6f}-{next(_RUN_SEQ)}".replace(".", "-")[:63]
            )

        # guards (reference: storyrun_controller.go:981-1045)
        inputs = self._prepare_inputs(story, inputs, trusted=_trusted)

        queue_name = self.config.default_queue
        priority = 0
        if story.policy is not None:
            queue_name = story.policy.queue or queue_name
            priority = story.policy.priority or 0

        run = StoryRun(
            name=name,
            namespace=namespace,
            story_name=story.name,
            story_namespace=story.namespace,
            inputs=inputs,
            parent_run=parent_run,
            parent_step=parent_step,
            recursion_depth=recursion_depth,
            story_generation=story.generation,
            input_hash=input_hash(inputs),
            queue=queue_name,
            priority=priority,
            labels=dict(labels or {}),
        )
        if trigger_token:
            run.trigger_tokens.append(trigger_token)
        # schema pins (reference: ensureStoryRunSchemaRefs / bubu:// refs)
        if story.inputs_schema is not None:
            run.annotations["schema/inputs"] = f"bubu://story/{story.namespace}/{story.name}/inputs@g{story.generation}"
        if story.outputs_schema is not None:
            run.annotations["schema/outputs"] = f"bubu://story/{story.namespace}/{story.name}/outputs@g{story.generation}"
        self.store.create_story_run(run)
        self._run_done.setdefault(run.key, threading.Event())
        self.metrics.inc("storyruns_total", phase="submitted")
        self._post(("admit", run.key))
        return run

    def _prepare_inputs(self, story: T.Story, inputs, trusted: bool = False):
        """Oversized-input offload + schema defaults + validation
        (reference: storyrun_controller.go:981-1045, pkg/runs/inputs).
        User-submitted inputs may not carry $storageRef markers (spoofing
        rejection — reference: storyrun_webhook.go:389-423)."""
        if inputs is None:
            inputs = {}
        if not trusted and self.storage.contains_refs(inputs):
            # engine-created refs arrive only via trusted submitters
            # (executor sub-stories); user-facing paths reject them
            raise ValueError(
                "storyrun inputs may not contain $storageRef/$envRef/$fileRef values"
            )
        if story.inputs_schema is not None:
            inputs = apply_defaults(inputs, story.inputs_schema)
            errs = validate_instance(inputs, story.inputs_schema)
            if errs:
                raise ValueError(f"storyrun inputs invalid: {'; '.join(errs[:5])}")
        size = self.storage._json_size(inputs)
        if size > self.config.max_storyrun_input_bytes:
            inputs = self.storage.dehydrate(inputs, self.storage.input_prefix)
        return inputs

    def wait(self, run: _t.Union[StoryRun, str], timeout: _t.Optional[float] = None) -> StoryRun:
        key = run.key if isinstance(run, StoryRun) else run
        ev = self._run_done.setdefault(key, threading.Event())
        current = self.store.try_get_story_run(key)
        if current is not None and current.is_terminal:
            return current
        ev.wait(timeout=timeout)
        return self.store.get_story_run(key)

    def run_story(
        self, story: _t.Union[T.Story, str], inputs=None, timeout: _t.Optional[float] = 60.0, **kw
    ) -> StoryRun:
        run = self.submit_run(story, inputs, **kw)
        return self.wait(run, timeout=timeout)

    def cancel(self, run: _t.Union[StoryRun, str], graceful: bool = True) -> None:
        key = run.key if isinstance(run, StoryRun) else run
        stream = self._streams.get(key)
        if stream is not None:
            stream.cancel()
            stream.finish(timeout=5.0)
            return
        r = self.store.get_story_run(key)
        r.cancel_requested = True
        if not graceful:
            r.timers["canceldrain"] = 0.0
            r.cancel_observed_at = monotonic_now()
        self._post(("tick", key))

    def approve_gate(self, run: _t.Union[StoryRun, str], step: str, decided_by: str = "") -> None:
        self._decide_gate(run, step, "Approved", decided_by)

    def reject_gate(self, run: _t.Union[StoryRun, str], step: str, decided_by: str = "") -> None:
        self._decide_gate(run, step, "Rejected", decided_by)

    def _decide_gate(self, run, step: str, state: str, decided_by: str) -> None:
        from .records import GateStatus

        key = run.key if isinstance(run, StoryRun) else run
        r = self.store.get_story_run(key)
        gate = r.gates.get(step)
        if gate is None:
            gate = GateStatus(step=step)
            r.gates[step] = gate
        gate.state = state
        gate.decided_by = decided_by
        gate.decided_at = monotonic_now()
        self._post(("tick", key))

    def redrive(self, run: _t.Union[StoryRun, str]) -> None:
        """Re-run a terminal StoryRun from scratch (reference:
        storyrun_controller.go:295-557 redrive)."""
        key = run.key if isinstance(run, StoryRun) else run
        r = self.store.get_story_run(key)
        if not r.is_terminal:
            raise ValueError("redrive requires a terminal run")
        self.store.delete_steps_of(key)
        self._reset_run(r)
        self._post(("admit", key))

    def redrive_from_step(self, run: _t.Union[StoryRun, str], step: str) -> None:
        """Reset one step + its downstream closure and re-execute
        (reference: resolveRedriveFromStepSet storyrun_controller.go:535-558
        — BFS over dependents)."""
        key = run.key if isinstance(run, StoryRun) else run
        r = self.store.get_story_run(key)
        story = self.registry.story(r.story_name, r.story_namespace)
        cs = self.dag.compiled(story)
        closure = {step}
        frontier = [step]
        while frontier:
            cur = frontier.pop()
            for dep in cs.dependents.get(cur, ()):
                if dep not in closure:
                    closure.add(dep)
                    frontier.append(dep)
        for name in closure:
            r.step_states.pop(name, None)
            # stale delegated-evaluation results must not survive a redrive
            # (upstream outputs in the closure may change)
            r.materialized.pop(name, None)
            for sr in self.store.step_runs_of(key):
                if sr.spec.step_name == name or sr.spec.step_name.startswith(name + "/"):
                    self.store.delete_step_run(sr.key)
            for tag in list(r.timers):
                if tag.endswith(f":{name}"):
                    del r.timers[tag]
            r.primitive_children.pop(name, None)
            r.gates.pop(name, None)
        r.materialized.pop("__output__", None)
        for sr in self.store.step_runs_of(key):
            if sr.spec.step_name == "__output__/materialize":
                self.store.delete_step_run(sr.key)
        r.phase = Phase.RUNNING
        r.exec_phase = "main"
        r.failure_cause = None
        r.output = None
        r.error = None
        r.finished_at = None
        r.redrive_count += 1
        ev = self._run_done.get(key)
        if ev is not None:
            ev.clear()
        self._post(("tick", key))

    def _reset_run(self, r: StoryRun) -> None:
        r.step_states.clear()
        r.materialized.clear()
        r.gates.clear()
        r.primitive_children.clear()
        r.timers.clear()
        r.annotations.pop("stop-phase", None)
        r.annotations.pop("stop-message", None)
        r.phase = Phase.PENDING
        r.exec_phase = "main"
        r.failure_cause = None
        r.output = None
        r.error = None
        r.degraded = False
        r.cancel_requested = False
        r.cancel_observed_at = None
        r.started_at = None
        r.finished_at = None
        r.redrive_count += 1
        ev = self._run_done.get(r.key)
        if ev is not None:
            ev.clear()

    # ------------------------------------------------------------------
    # streaming (PerStoryRun pipelines — engine/streaming.py)
    # ------------------------------------------------------------------

    def submit_stream(
        self,
        story,
        inputs=None,
        name: _t.Optional[str] = None,
        namespace: _t.Optional[str] = None,
    ):
        """Materialize a streaming Story as a live pipeline; returns the
        StreamingRun handle (push/finish/cancel)."""
        from ..enums import StoryPattern
        from .streaming import StreamingRun

        if isinstance(story, str):
            ns, _, nm = story.rpartition("/")
            story = self.registry.story(nm, ns or "default")
        if story.pattern != StoryPattern.STREAMING:
            raise ValueError(f"story {story.key} is not a streaming story")
        namespace = namespace or story.namespace
        if name is None:
            # timestamp for readability + atomic counter for uniqueness:
            # concurrent submitters in the same clock tick must never
            # collide (a collision silently adopts the other run)
            name = (
                f"{story.name}-{monotonic_now():.6f}-{next(_RUN_SEQ)}".replace(".", "-")[:63]
            )
        run = StoryRun(
            name=name,
            namespace=namespace,
            story_name=story.name,
            story_namespace=story.namespace,
            inputs=inputs or {},
            story_generation=story.generation,
            input_hash=input_hash(inputs or {}),
        )
        self.store.create_story_run(run)
        self._run_done.setdefault(run.key, threading.Event())
        stream = StreamingRun(self, run, story)
        self._streams[run.key] = stream
        self.metrics.inc("storyruns_total", phase="streaming")
        return stream

    def stream_of(self, run_key: str):
        return self._streams.get(run_key)

    # ------------------------------------------------------------------
    # checkpoint/resume (reference: "state IS the checkpoint" SURVEY §5.4)
    # ------------------------------------------------------------------

    def save_state(self, path: str) -> None:
        from . import snapshot

        snapshot.save_state(self, path)

    def _checkpoint_tick(self) -> None:
        """Periodic durability snapshot (atomic rename) + reschedule."""
        path = self.config.checkpoint_path
        if not path:
            return
        import os as _os

        tmp = f"{path}.tmp"
        try:
            self.save_state(tmp)
            _os.replace(tmp, path)
            self.metrics.inc("checkpoints_total")
        except Exception as exc:  # durability must never kill the loop
            self.metrics.inc("checkpoint_errors_total")
            self.log.error("checkpoint failed", error=str(exc))
        self.set_timer(
            monotonic_now() + self.config.checkpoint_interval_seconds, "", "checkpoint"
        )

    def load_state(self, path: str) -> int:
        from . import snapshot

        return snapshot.load_state(self, path)

    # ------------------------------------------------------------------
    # scope building (reference: getPriorStepOutputs dag.go:2083-2597)
    # ------------------------------------------------------------------

    def build_scope(self, run: StoryRun, story: T.Story) -> dict:
        steps_view: _t.Dict[str, _t.Any] = {}
        for name, state in run.step_states.items():
            entry = {
                "phase": str(state.phase),
                "output": state.output,
                "error": state.error.to_dict() if state.error else None,
                "retries": state.retries,
            }
            steps_view[name] = entry
        # alias keys (underscored) point at the same entries (dag.go:2597)
        for s in story.all_steps():
            if s.alias != s.name and s.name in steps_view:
                steps_view.setdefault(s.alias, steps_view[s.name])
        # merge step signals seq-ordered into the view (dag.go:2289-2481)
        for sr in self.store.step_runs_of(run.key):
            if sr.status.signals and sr.spec.step_name in steps_view:
                sigs = sorted(sr.status.signals, key=lambda s: s.seq)
                steps_view[sr.spec.step_name]["signals"] = [
                    {"name": s.name, "payload": s.payload, "seq": s.seq} for s in sigs
                ]
        return {
            "inputs": run.inputs,
            "steps": steps_view,
            "run": {
                "name": run.name,
                "namespace": run.namespace,
                "phase": str(run.phase),
                "redrives": run.redrive_count,
            },
            "story": {"name": story.name, "namespace": story.namespace, "version": story.version},
        }

    # ------------------------------------------------------------------
    # engram step launching + completion (worker side)
    # ------------------------------------------------------------------

    def launch_engram_step(self, run: StoryRun, sr: StepRun, template, resolved_cfg) -> None:
        impl_name = template.implementation if template is not None else None
        if impl_name is None and sr.spec.engram:
            impl_name = sr.spec.engram.split("/")[-1]
        run_key = run.key
        sr_key = sr.key
        timeout = sr.spec.timeout_seconds
        if timeout is not None:
            self.set_timer(monotonic_now() + timeout, run_key, f"steptimeout:{sr_key}")

        def body(slot) -> None:
            self._execute_engram_body(run_key, sr_key, impl_name, template, resolved_cfg, slot)

        self.workers.submit(body, device=sr.spec.placement_gpu)

    def _execute_engram_body(self, run_key, sr_key, impl_name, template, resolved_cfg, slot) -> None:
        sr = self.store.try_get_step_run(sr_key)
        run = self.store.try_get_story_run(run_key)
        if sr is None or run is None or sr.is_terminal:
            return
        started = monotonic_now()
        ctx = EngramContext(
            story_name=run.story_name,
            story_run=run.name,
            step_name=sr.spec.step_name,
            step_run=sr.name,
            namespace=sr.namespace,
            input=self.storage.hydrate(sr.spec.input),
            config=sr.spec.config,
            runtime=sr.spec.runtime,
            execution_mode=sr.spec.mode,
            max_inline_size=resolved_cfg.max_inline_size,
            timeout_seconds=sr.spec.timeout_seconds,
            max_recursion_depth=self.config.max_recursion_depth,
            device=slot.device,
            stream=slot.stream,
            storage=self.storage,
            trace_id=run.trace.trace_id,
            cancel_check=lambda: sr.cancel_requested,
            effect_guard=lambda key, desc: self.effects.acquire(
                f"{run.name}/{sr.spec.step_name}/{key}", sr.name, description=desc
            )[1],
        )
        try:
            with self.tracer.span(
                "engram.run", run=run.name, step=sr.spec.step_name, impl=impl_name or ""
            ):
                if template is not None and getattr(template, "command", None):
                    from ..engrams.process import ProcessEngram

                    secrets = {}
                    if sr.spec.engram and "/" in sr.spec.engram:
                        e_ns, e_name = sr.spec.engram.split("/", 1)
                        eng_obj = self.registry.try_engram(e_name, e_ns)
                        if eng_obj is not None:
                            secrets = eng_obj.secrets
                    impl = ProcessEngram(
                        template.command,
                        secret_defs=getattr(template, "secrets", None),
                        secrets=secrets,
                    )
                else:
                    impl = engram_registry.resolve(impl_name)
                if impl.wants_gpu and slot.device is None:
                    import torch

                    if torch.cuda.is_available():
                        raise EngramFailure(
                            f"engram {impl_name!r} requires GPU placement", exit_code=2
                        )
                result = impl.run(ctx)
            if not isinstance(result, EngramResult):
                result = EngramResult(output=result)
            exit_code = result.exit_code
            output = result.output
            error = None
        except EngramFailure as exc:
            exit_code = exc.exit_code
            output = None
            error = exc.to_structured()
        except Exception as exc:  # unexpected engram crash → retryable
            exit_code = 1
            output = None
            error = StructuredError(
                type=ErrorType.EXECUTION, message=f"{type(exc).__name__}: {exc}", retryable=True
            )
        # SDK-side status patch equivalent: record signals/effects/logs
        sr.status.signals.extend(ctx.signals)
        sr.status.effects.extend(ctx.effects)
        sr.status.logs.extend(ctx.logs)
        for eff in ctx.effects:
            self.effects.complete(
                f"{run.name}/{sr.spec.step_name}/{eff.idempotency_key}", sr.name
            )
        sr.status.worker = slot.name if hasattr(slot, "name") else ""
        duration = monotonic_now() - started
        self.metrics.observe("steprun_duration_seconds", duration, engram=impl_name or "?")
        self._post(("engram_done", sr_key, run_key, exit_code, output, error))

    # ------------------------------------------------------------------
    # event loop
    # ------------------------------------------------------------------

    def _post(self, event: tuple) -> None:
        self._events.put(event)

    def set_timer(self, deadline: float, run_key: str, tag: str) -> None:
        with self._lock:
            heapq.heappush(self._timers, (deadline, next(self._timer_seq), run_key, tag))
        self._events.put(("__wake__",))

    def _loop(self) -> None:
        while self._running:
            timeout = None
            now = monotonic_now()
            with self._lock:
                while self._timers and self._timers[0][0] <= now:
                    _, _, run_key, tag = heapq.heappop(self._timers)
                    self._events.put(("timer", run_key, tag))
                if self._timers:
                    timeout = max(self._timers[0][0] - now, 0.0005)
            try:
                event = self._events.get(timeout=timeout)
            except queue.Empty:
                continue
            if event[0] in ("__stop__", "__wake__"):
                continue
            try:
                self._handle(event)
            except Exception:
                import traceback

                traceback.print_exc()

    def _handle(self, event: tuple) -> None:
        kind = event[0]
        if kind == "tick":
            run = self.store.try_get_story_run(event[1])
            if run is not None:
                self._tick(run)
        elif kind == "admit":
            self._admit_runs(event[1] if len(event) > 1 else None)
        elif kind == "timer":
            self._on_timer(event[1], event[2])
        elif kind == "engram_done":
            self._on_engram_done(*event[1:])

    def _tick(self, run: StoryRun) -> None:
        with self.tracer.span("dag.tick", run=run.name):
            t0 = monotonic_now()
            self.dag.tick(run)
            self.metrics.observe("dag_tick_seconds", monotonic_now() - t0)

    # -- admission (queue/global concurrency + priority aging;
    #    reference: dag.go:1801-1961, controller_config.go:527-544) --------

    def _admit_runs(self, new_key: _t.Optional[str]) -> None:
        if new_key is not None and new_key not in self._scheduling:
            self._scheduling.append(new_key)
        if not self._scheduling:
            return
        running = [r for r in self.store.all_runs() if r.phase == Phase.RUNNING or (r.phase == Phase.PENDING and r.started_at)]
        global_running = len([r for r in running if not r.is_terminal])
        per_queue: _t.Dict[str, int] = {}
        per_story: _t.Dict[str, int] = {}
        for r in running:
            per_queue[r.queue] = per_queue.get(r.queue, 0) + 1
            per_story[f"{r.story_namespace}/{r.story_name}"] = (
                per_story.get(f"{r.story_namespace}/{r.story_name}", 0) + 1
            )
        now = monotonic_now()

        def effective_priority(key: str) -> float:
            r = self.store.try_get_story_run(key)
            if r is None:
                return -1e18
            qc = self.config.queue(r.queue)
            aging = qc.priority_aging_seconds or 60.0
            return r.priority + (now - r.created_at) / aging

        self._scheduling.sort(key=effective_priority, reverse=True)
        # queue observability (reference: storyrun_queue_depth /
        # storyrun_queue_age_seconds pkg/metrics)
        depth: _t.Dict[str, int] = {}
        for key in self._scheduling:
            r = self.store.try_get_story_run(key)
            if r is not None:
                depth[r.queue or "default"] = depth.get(r.queue or "default", 0) + 1
        for q, d in depth.items():
            self.metrics.set_gauge("storyrun_queue_depth", d, queue=q)
        admitted = []
        for key in list(self._scheduling):
            r = self.store.try_get_story_run(key)
            if r is None or r.is_terminal:
                admitted.append(key)
                continue
            if self.config.global_concurrency and global_running >= self.config.global_concurrency:
                r.phase = Phase.SCHEDULING
                continue
            qc = self.config.queue(r.queue)
            if qc.concurrency and per_queue.get(r.queue, 0) >= qc.concurrency:
                r.phase = Phase.SCHEDULING
                continue
            story_key = f"{r.story_namespace}/{r.story_name}"
            limit = None
            try:
                story = self.registry.story(r.story_name, r.story_namespace)
                if story.policy is not None:
                    limit = story.policy.concurrency
            except NotFound:
                pass
            if limit and per_story.get(story_key, 0) >= limit:
                r.phase = Phase.SCHEDULING
                continue
            # admitted
            admitted.append(key)
            self.metrics.observe(
                "storyrun_queue_age_seconds", now - r.created_at, queue=r.queue or "default"
            )
            global_running += 1
            per_queue[r.queue] = per_queue.get(r.queue, 0) + 1
            per_story[story_key] = per_story.get(story_key, 0) + 1
            r.phase = Phase.RUNNING
            self._tick(r)
        for key in admitted:
            self._scheduling.remove(key)

    # -- timers ----------------------------------------------------------

    def _on_timer(self, run_key: str, tag: str) -> None:
        if tag == "otlp":
            exporter = getattr(self, "_otlp", None)
            if exporter is not None:
                exporter.flush(self.tracer)
                self.set_timer(
                    monotonic_now() + self.config.otlp_flush_interval_seconds, "", "otlp"
                )
            return
        if tag == "checkpoint":
            self._checkpoint_tick()
            return
        run = self.store.try_get_story_run(run_key)
        if run is None:
            return
        if tag.startswith("steptimeout:"):
            self._on_step_timeout(run, tag.split(":", 1)[1])
            return
        if tag.startswith("branchsleep:"):
            sr = self.store.try_get_step_run(tag.split(":", 1)[1])
            if sr is not None and not sr.is_terminal:
                finish_step_run(sr, Phase.SUCCEEDED, output={"slept": True})
            self._tick(run)
            return
        if tag.startswith("retry:"):
            self._on_retry_due(run, tag.split(":", 1)[1])
            return
        if tag == "ttl" or tag == "retention":
            self._on_cleanup_timer(run, tag)
            return
        if run.is_terminal:
            return
        self._tick(run)

    def _on_step_timeout(self, run: StoryRun, sr_key: str) -> None:
        sr = self.store.try_get_step_run(sr_key)
        if sr is None or sr.is_terminal:
            return
        sr.cancel_requested = True
        finish_step_run(
            sr,
            Phase.TIMEOUT,
            error=StructuredError(type=ErrorType.TIMEOUT, message="step timeout"),
            exit_class=ExitClass.TERMINAL,
        )
        self.metrics.inc("stepruns_total", result="timeout")
        self._tick(run)

    # -- engram completion + retry engine --------------------------------

    def _on_engram_done(self, sr_key, run_key, exit_code, output, error) -> None:
        sr = self.store.try_get_step_run(sr_key)
        run = self.store.try_get_story_run(run_key)
        if sr is None or run is None:
            return
        if sr.is_terminal:  # timeout/cancel won the race; terminal wins
            self._tick(run)
            return
        exit_class = classify_exit_code(exit_code)
        if exit_class == ExitClass.SUCCESS:
            self._complete_engram_success(run, sr, output)
            if sr.spec.step_name.endswith("/materialize") and sr.status.phase == Phase.SUCCEEDED:
                base = sr.spec.step_name.rsplit("/", 1)[0]
                # the completion path may have re-offloaded the whole
                # {"result": ...} envelope — unwrap it, then re-offload just
                # the result so run.materialized stays marker-sized
                out = self.storage.hydrate(sr.status.output)
                result = out.get("result") if isinstance(out, dict) else None
                run.materialized[base] = self.storage.dehydrate_document(result)
        else:
            self._handle_engram_failure(run, sr, exit_code, exit_class, error)
        self._tick(run)

    def _complete_engram_success(self, run: StoryRun, sr: StepRun, output) -> None:
        story = None
        step = None
        try:
            story = self.registry.story(run.story_name, run.story_namespace)
            step = story.step(sr.spec.step_name)
        except NotFound:
            pass
        # output schema + declared keys + postExecution checks
        # (reference: steprun_controller.go:2050-2124, 947)
        if step is not None and step.ref is not None:
            template = self._template_of(step, story)
            if template is not None:
                if template.output_schema is not None:
                    errs = validate_instance(output, template.output_schema)
                    if errs:
                        finish_step_run(
                            sr,
                            Phase.FAILED,
                            error=StructuredError(
                                type=ErrorType.VALIDATION,
                                message=f"output schema: {'; '.join(errs[:5])}",
                            ),
                            exit_code=2,
                            exit_class=ExitClass.TERMINAL,
                        )
                        return
                if template.declared_output_keys and isinstance(output, dict):
                    missing = [k for k in template.declared_output_keys if k not in output]
                    if missing:
                        finish_step_run(
                            sr,
                            Phase.FAILED,
                            error=StructuredError(
                                type=ErrorType.VALIDATION,
                                message=f"declared output keys missing: {missing}",
                            ),
                            exit_code=2,
                            exit_class=ExitClass.TERMINAL,
                        )
                        return
            if step.post_execution is not None and story is not None:
                scope = self.build_scope(run, story)
                scope["output"] = output
                ok = False
                try:
                    ok = self.evaluator.evaluate_condition(step.post_execution.condition, scope)
                except Exception:
                    ok = False
                if not ok:
                    finish_step_run(
                        sr,
                        Phase.FAILED,
                        error=StructuredError(
                            type=ErrorType.VALIDATION,
                            message=step.post_execution.failure_message
                            or "postExecution condition failed",
                        ),
                        exit_code=2,
                        exit_class=ExitClass.TERMINAL,
                    )
                    return
        # offload oversized outputs (reference: step_executor.go:662-737)
        output = self.storage.dehydrate_document(output)
        finish_step_run(sr, Phase.SUCCEEDED, output=output, exit_code=0, exit_class=ExitClass.SUCCESS)
        self.metrics.inc("stepruns_total", result="succeeded")
        # cache write
        if step is not None and step.ref is not None and story is not None:
            engram = self.registry.try_engram(
                step.ref.name, step.ref.resolve_namespace(story.namespace)
            )
            template = self._template_of(step, story)
            cfg = self.resolver.resolve(step=step, story=story, engram=engram, template=template)
            if cfg.cache_enabled and cfg.cache_mode.writes:
                self.cache.write(step, cfg, sr.spec.input, output)

    def _template_of(self, step: T.Step, story: _t.Optional[T.Story]):
        if step.ref is None or story is None:
            return None
        engram = self.registry.try_engram(step.ref.name, step.ref.resolve_namespace(story.namespace))
        if engram is None or engram.template_ref is None:
            return None
        try:
            return self.registry.engram_template(engram.template_ref.name)
        except NotFound:
            return None

    def _handle_engram_failure(self, run, sr: StepRun, exit_code, exit_class, error) -> None:
        story = None
        step = None
        try:
            story = self.registry.story(run.story_name, run.story_namespace)
            step = story.step(sr.spec.step_name)
        except NotFound:
            pass
        cfg = None
        if step is not None and story is not None:
            engram = (
                self.registry.try_engram(step.ref.name, step.ref.resolve_namespace(story.namespace))
                if step.ref is not None
                else None
            )
            template = self._template_of(step, story)
            cfg = self.resolver.resolve(step=step, story=story, engram=engram, template=template)
        if cfg is not None and should_retry(cfg, exit_class, sr.status.retries) and (
            exit_class.consumes_retry_budget or sr.status.retries < _UNKNOWN_RETRY_ABS_CAP
        ) and not sr.cancel_requested:
            if exit_class.consumes_retry_budget:
                sr.status.retries += 1
            attempt = sr.status.retries if exit_class.consumes_retry_budget else sr.status.retries + 1
            delay = compute_retry_delay(cfg, max(attempt, 1), exit_class)
            sr.status.next_retry_at = monotonic_now() + delay
            sr.status.error = error
            sr.status.exit_code = exit_code
            sr.status.exit_class = exit_class
            sr.status.phase = Phase.PENDING  # waiting for retry
            self.metrics.inc("steprun_retries_total")
            self.set_timer(sr.status.next_retry_at, run.key, f"retry:{sr.key}")
            # keep the run-level state Running while the retry waits
            state = run.step_state(sr.spec.step_name)
            state.phase = Phase.RUNNING
            state.retries = sr.status.retries
            return
        finish_step_run(
            sr,
            Phase.FAILED,
            error=error,
            exit_code=exit_code,
            exit_class=exit_class,
        )
        self.metrics.inc("stepruns_total", result="failed")

    def _on_retry_due(self, run: StoryRun, sr_key: str) -> None:
        sr = self.store.try_get_step_run(sr_key)
        if sr is None or sr.is_terminal or run.is_terminal or sr.cancel_requested:
            return
        story = self.registry.story(run.story_name, run.story_namespace)
        step = story.step(sr.spec.step_name)
        template = self._template_of(step, story) if step is not None else None
        engram = (
            self.registry.try_engram(step.ref.name, step.ref.resolve_namespace(story.namespace))
            if step is not None and step.ref is not None
            else None
        )
        cfg = self.resolver.resolve(step=step, story=story, engram=engram, template=template)
        sr.status.phase = Phase.RUNNING
        sr.status.next_retry_at = None
        self.launch_engram_step(run, sr, template, cfg)

    # ------------------------------------------------------------------
    # terminal handling + retention
    # ------------------------------------------------------------------

    def on_run_terminal(self, run: StoryRun) -> None:
        self.log.debug(
            "run terminal", feature="runs", run=run.name, phase=str(run.phase),
            story=f"{run.story_namespace}/{run.story_name}",
        )
        if run.phase in (Phase.FAILED, Phase.TIMEOUT):
            self.log.warn(
                "run failed", run=run.name,
                error=(run.error.message if run.error else run.failure_cause),
            )
        self.metrics.inc("storyruns_total", phase=str(run.phase))
        if run.started_at is not None and run.finished_at is not None:
            self.metrics.observe(
                "storyrun_duration_seconds", run.finished_at - run.started_at
            )
        ev = self._run_done.setdefault(run.key, threading.Event())
        ev.set()
        # wake the parent (executeStory join) and admission
        if run.parent_run is not None:
            self._post(("tick", f"{run.namespace}/{run.parent_run}"))
        self._post(("admit",))
        # two-phase cleanup (reference: handleTerminalStoryRun
        # storyrun_controller.go:1811-2057): children after TTL, the record
        # after retention
        ttl = self.config.child_ttl_seconds
        if ttl is not None and ttl >= 0:
            self.set_timer(monotonic_now() + ttl, run.key, "ttl")
        retention = self.config.storyrun_retention_seconds
        if retention is not None and retention >= 0:
            self.set_timer(monotonic_now() + max(retention, ttl or 0), run.key, "retention")

    def _on_cleanup_timer(self, run: StoryRun, tag: str) -> None:
        if not run.is_terminal:
            return
        if tag == "ttl" and run.children_cleaned_at is None:
            self.store.delete_steps_of(run.key)
            run.children_cleaned_at = monotonic_now()
            self.metrics.inc("resource_cleanup_total", kind="children")
        elif tag == "retention":
            if run.children_cleaned_at is None:
                self.store.delete_steps_of(run.key)
            self.store.delete_story_run(run.key)
            self._run_done.pop(run.key, None)
            # orphan effect claims: the run (and its StepRuns) are gone, so
            # the run-scoped claims can never be referenced again
            pruned = self.effects.prune_prefix(f"{run.name}/")
            if pruned:
                self.metrics.inc("resource_cleanup_total", kind="effectclaims")
            self.metrics.inc("resource_cleanup_total", kind="storyrun")
