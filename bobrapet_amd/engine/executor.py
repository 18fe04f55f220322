"""Step executor: launches one ready step.

Role parity with the reference's StepExecutor
(reference: internal/controller/runs/step_executor.go:132-185 dispatch;
createEngramStepRun 360-513 — deterministic naming, merged with,
template resolution, idempotency key, offload; executeParallelStep
740-811; executeStoryStep 1132-1230; stop 1081-1130) and with the engram
half of the StepRun controller (reference: steprun_controller.go —
cache lookup 3106-3477, timeout computation, requires-context guard
5523-5582, output validation 2050-2124).

MI355X redesign: instead of materializing a Job, an engram step binds to a
(gpu, stream) worker slot and runs in-process; completion is a callback
into the engine loop (no watch/requeue).
"""
from __future__ import annotations

import typing as _t

from ..enums import ErrorType, ExitClass, Phase, StepType, StopMode
from ..specs import types as T
from ..templating import OffloadedDataUsage, TemplateError
from ..utils.durations import DurationError, parse_duration
from ..utils.jsonschema import validate_instance
from .records import (
    StepRun,
    StepRunSpec,
    StepState,
    StoryRun,
    StructuredError,
    compose_name,
    monotonic_now,
)

if _t.TYPE_CHECKING:
    from .engine import RunEngine


class StepExecutor:
    def __init__(self, engine: "RunEngine"):
        self.engine = engine

    # ------------------------------------------------------------------

    def execute(self, run: StoryRun, story: T.Story, step: T.Step, scope: dict) -> None:
        """Launch one ready step; updates the run's StepState in place.
        Dispatch by ref/type (reference: step_executor.go:132-185)."""
        state = run.step_state(step.name)
        state.started_at = state.started_at or monotonic_now()
        try:
            if step.ref is not None:
                self._execute_engram(run, story, step, scope, state)
            elif step.type == StepType.CONDITION:
                self._execute_condition(run, step, scope, state)
            elif step.type == StepType.SLEEP:
                self._execute_sleep(run, step, scope, state)
            elif step.type == StepType.STOP:
                self._execute_stop(run, step, scope, state)
            elif step.type == StepType.WAIT:
                self._execute_wait(run, step, scope, state)
            elif step.type == StepType.GATE:
                self._execute_gate(run, step, scope, state)
            elif step.type == StepType.PARALLEL:
                self._execute_parallel(run, story, step, scope, state)
            elif step.type == StepType.EXECUTE_STORY:
                self._execute_story(run, story, step, scope, state)
            else:
                raise ValueError(f"unknown step type {step.type}")
        except (TemplateError, DurationError, ValueError) as exc:
            state.phase = Phase.FAILED
            state.finished_at = monotonic_now()
            state.error = StructuredError(
                type=ErrorType.VALIDATION
                if isinstance(exc, (DurationError, ValueError))
                else ErrorType.EXECUTION,
                message=str(exc),
            )
            state.message = str(exc)

    # -- engram ---------------------------------------------------------

    def _execute_engram(
        self, run: StoryRun, story: T.Story, step: T.Step, scope: dict, state: StepState
    ) -> None:
        eng = self.engine
        ref = step.ref
        ns = ref.resolve_namespace(story.namespace)
        engram = eng.registry.try_engram(ref.name, ns)
        if engram is None:
            # Blocked, not Failed: watch-driven wakeup when the Engram appears
            # (reference: steprun_controller.go:319-327)
            state.phase = Phase.BLOCKED
            state.message = f"engram {ns}/{ref.name} not found"
            return
        if ns != story.namespace and not eng.registry.allows_cross_namespace(
            "Story", story.namespace, "Engram", ns, ref.name
        ):
            state.phase = Phase.FAILED
            state.finished_at = monotonic_now()
            state.error = StructuredError(
                type=ErrorType.VALIDATION,
                message=f"cross-namespace engram reference {ns}/{ref.name} not granted",
            )
            return
        template = None
        if engram.template_ref is not None:
            try:
                template = eng.registry.engram_template(engram.template_ref.name)
            except KeyError:
                state.phase = Phase.BLOCKED
                state.message = f"engram template {engram.template_ref.name} not found"
                return

        resolved_cfg = eng.resolver.resolve(step=step, story=story, engram=engram, template=template)

        # merged `with` = engram.with ⊕ step.with, template-resolved
        # (reference: step_executor.go:360-513 resolveTemplateWith)
        try:
            step_with = eng.evaluator.resolve_value(step.with_, scope) if step.with_ is not None else None
        except OffloadedDataUsage:
            # policy=block: never evaluate over offloaded data on the engine
            # loop — delegate to the materialize engram on a worker slot
            # (reference: materialize.go resolveMaterialize:326)
            if step.name in run.materialized:
                step_with = run.materialized[step.name]
            else:
                self._delegate_materialize(run, story, step, scope, state)
                return
        merged_input = step_with
        config = engram.with_

        # input schema validation (template.inputSchema)
        if template is not None and template.input_schema is not None:
            errs = validate_instance(merged_input or {}, template.input_schema)
            if errs:
                state.phase = Phase.FAILED
                state.finished_at = monotonic_now()
                state.error = StructuredError(
                    type=ErrorType.VALIDATION,
                    message=f"input schema: {'; '.join(errs[:5])}",
                )
                return

        idem_key = None
        if step.idempotency_key_template:
            idem_key = str(eng.evaluator.resolve_string(step.idempotency_key_template, scope))

        sr_name = compose_name(run.name, step.name)
        spec = StepRunSpec(
            story_run=run.name,
            step_name=step.name,
            engram=f"{ns}/{ref.name}",
            input=merged_input,
            config=config,
            runtime=step.runtime,
            idempotency_key=idem_key,
            timeout_seconds=resolved_cfg.timeout_seconds,
            template_generation=template.generation if template else 0,
            placement_gpu=self._place(run, step, resolved_cfg),
            mode=resolved_cfg.mode,
        )
        sr = StepRun(name=sr_name, namespace=run.namespace, spec=spec)
        sr, created = eng.store.create_or_get_step_run(sr)
        if not created and sr.is_terminal:
            # adopt a finished attempt (idempotent replay)
            state.phase = sr.status.phase
            state.output = sr.status.output
            state.error = sr.status.error
            state.finished_at = sr.status.finished_at
            return

        # cache lookup (reference: tryCacheHit steprun_controller.go:3346)
        if resolved_cfg.cache_enabled and resolved_cfg.cache_mode.reads:
            hit = eng.cache.lookup(step, resolved_cfg, merged_input, scope)
            if hit is not None:
                sr.status.phase = Phase.SUCCEEDED
                sr.status.output = hit
                sr.status.cache_hit = True
                sr.status.finished_at = monotonic_now()
                if hit not in (None, {}, []):
                    sr.status.last_output_at = sr.status.finished_at
                state.phase = Phase.SUCCEEDED
                state.output = hit
                state.finished_at = sr.status.finished_at
                eng.metrics.inc("steprun_cache_lookups_total", result="hit")
                return
            eng.metrics.inc("steprun_cache_lookups_total", result="miss")

        state.phase = Phase.RUNNING
        sr.status.phase = Phase.RUNNING
        sr.status.started_at = monotonic_now()
        eng.launch_engram_step(run, sr, template, resolved_cfg)

    def _delegate_materialize(
        self, run: StoryRun, story: T.Story, step: T.Step, scope: dict, state: StepState
    ) -> None:
        """Create (or observe) the aux materialize StepRun for a step whose
        `with` templates touch `$storageRef` data under policy=block
        (reference: ensureMaterializeStepRun materialize.go:142-240).  The
        aux run's step_name carries a "/" so DAG state sync ignores it; its
        result lands in run.materialized[step] and the step re-dispatches."""
        eng = self.engine
        sr_name = compose_name(run.name, f"{step.name}-mat")
        sr_key = f"{run.namespace}/{sr_name}"
        existing = eng.store.try_get_step_run(sr_key)
        if existing is not None:
            if existing.status.phase == Phase.SUCCEEDED:
                # completion hook normally fills run.materialized; recover here
                # after a snapshot restore where only the StepRun survived
                out = eng.storage.hydrate(existing.status.output)
                result = out.get("result") if isinstance(out, dict) else None
                run.materialized[step.name] = eng.storage.dehydrate_document(result)
                state.message = ""
                return
            if existing.is_terminal:  # failed/canceled materialization
                state.phase = Phase.FAILED
                state.finished_at = monotonic_now()
                state.error = existing.status.error or StructuredError(
                    type=ErrorType.EXECUTION, message="materialize step failed"
                )
            else:
                state.message = "materializing offloaded inputs"
            return
        resolved_cfg = eng.resolver.resolve(step=step, story=story)
        spec = StepRunSpec(
            story_run=run.name,
            step_name=f"{step.name}/materialize",
            engram="builtin/materialize",
            input={"mode": "value", "template": step.with_, "vars": dict(scope)},
            timeout_seconds=resolved_cfg.timeout_seconds,
            placement_gpu=self._place(run, step, resolved_cfg),
        )
        sr = StepRun(name=sr_name, namespace=run.namespace, spec=spec)
        sr, created = eng.store.create_or_get_step_run(sr)
        sr.status.phase = Phase.RUNNING
        sr.status.started_at = monotonic_now()
        state.message = "materializing offloaded inputs"
        eng.metrics.inc("materialize_runs_total")
        eng.launch_engram_step(run, sr, None, resolved_cfg)

    def delegate_output_materialize(self, run: StoryRun, story: T.Story, scope: dict) -> str:
        """Same delegation for the run-level output template
        (reference: finalizeSuccessfulRun resolving output over offloaded
        step data).  Returns "pending" while the aux run is in flight,
        "failed" if it terminally failed; on success run.materialized
        holds "__output__" and the caller re-finalizes."""
        eng = self.engine
        sr_name = compose_name(run.name, "output-mat")
        sr_key = f"{run.namespace}/{sr_name}"
        existing = eng.store.try_get_step_run(sr_key)
        if existing is not None:
            if existing.status.phase == Phase.SUCCEEDED:
                out = eng.storage.hydrate(existing.status.output)
                result = out.get("result") if isinstance(out, dict) else None
                run.materialized["__output__"] = eng.storage.dehydrate_document(result)
                return "pending"  # caller re-enters finalize next tick
            return "failed" if existing.is_terminal else "pending"
        resolved_cfg = eng.resolver.resolve(story=story)
        spec = StepRunSpec(
            story_run=run.name,
            step_name="__output__/materialize",
            engram="builtin/materialize",
            input={"mode": "value", "template": story.output, "vars": dict(scope)},
            timeout_seconds=resolved_cfg.timeout_seconds,
        )
        sr = StepRun(name=sr_name, namespace=run.namespace, spec=spec)
        sr, _created = eng.store.create_or_get_step_run(sr)
        sr.status.phase = Phase.RUNNING
        sr.status.started_at = monotonic_now()
        eng.metrics.inc("materialize_runs_total")
        eng.launch_engram_step(run, sr, None, resolved_cfg)
        return "pending"

    def _place(self, run: StoryRun, step: T.Step, cfg) -> _t.Optional[int]:
        """Deterministic (gpu) placement for the step (SURVEY.md §2.6:
        the DAG scheduler places each StepRun on one of the GPUs)."""
        if cfg.placement_gpu is not None:
            return cfg.placement_gpu
        pool = self.engine.workers
        if pool.device_count <= 0:
            return None
        allowed = cfg.placement_gpus or list(pool.device_ids)
        # stable hash spread of (run, step) over allowed devices
        h = hash((run.name, step.name)) & 0x7FFFFFFF
        return allowed[h % len(allowed)]

    # -- primitives ------------------------------------------------------

    def _with(self, step: T.Step, scope: dict) -> dict:
        w = step.with_ if isinstance(step.with_, dict) else {}
        return self.engine.evaluator.resolve_value(w, scope) if w else {}

    def _execute_condition(self, run: StoryRun, step: T.Step, scope: dict, state: StepState) -> None:
        w = step.with_ if isinstance(step.with_, dict) else {}
        expr = w.get("expression") or w.get("if") or step.if_ or "true"
        result = self.engine.evaluator.evaluate_condition(str(expr), scope)
        state.phase = Phase.SUCCEEDED
        state.output = {"result": bool(result)}
        state.finished_at = monotonic_now()

    def _execute_sleep(self, run: StoryRun, step: T.Step, scope: dict, state: StepState) -> None:
        """with.duration (reference: dag.go:1549-1567)."""
        w = self._with(step, scope)
        seconds = parse_duration(w.get("duration"))
        if seconds is None:
            raise ValueError(f"sleep step {step.name!r}: with.duration required")
        state.phase = Phase.RUNNING
        deadline = monotonic_now() + max(seconds, 0.0)
        run.timers[f"sleep:{step.name}"] = deadline  # durable (dag.go:64-191)
        self.engine.set_timer(deadline, run.key, f"sleep:{step.name}")

    def _execute_stop(self, run: StoryRun, step: T.Step, scope: dict, state: StepState) -> None:
        """with.{phase|mode, message} (reference: step_executor.go:1081-1130)."""
        w = self._with(step, scope)
        phase_name = w.get("phase")
        if phase_name is None and w.get("mode"):
            phase_name = str(StopMode(w["mode"]).terminal_phase)
        stop_phase = Phase(phase_name) if phase_name else Phase.SUCCEEDED
        message = w.get("message", "")
        state.phase = Phase.SUCCEEDED
        state.output = {"phase": str(stop_phase), "message": message}
        state.finished_at = monotonic_now()
        run.annotations["stop-phase"] = str(stop_phase)
        if message:
            run.annotations["stop-message"] = message

    def _execute_wait(self, run: StoryRun, step: T.Step, scope: dict, state: StepState) -> None:
        """with.{until(required), timeout, pollInterval, onTimeout}
        (reference: dag.go:1569-1606)."""
        w = step.with_ if isinstance(step.with_, dict) else {}
        if "until" not in w:
            raise ValueError(f"wait step {step.name!r}: with.until required")
        state.phase = Phase.RUNNING
        now = monotonic_now()
        timeout = parse_duration(w.get("timeout"))
        if timeout is not None:
            run.timers[f"waitdeadline:{step.name}"] = now + timeout
        poll = parse_duration(w.get("pollInterval")) or self.engine.config.default_wait_poll_interval
        poll = max(poll, self.engine.config.min_poll_interval)
        run.timers[f"waitpoll:{step.name}"] = poll  # interval, not deadline
        # first check happens on this tick (dag sync); schedule the next poll
        self.engine.set_timer(now + poll, run.key, f"wait:{step.name}")

    def _execute_gate(self, run: StoryRun, step: T.Step, scope: dict, state: StepState) -> None:
        """Manual approval; decision read from run.gates[step]
        (reference: dag.go:1455-1547)."""
        from .records import GateStatus

        w = step.with_ if isinstance(step.with_, dict) else {}
        state.phase = Phase.PAUSED
        if step.name not in run.gates:
            run.gates[step.name] = GateStatus(step=step.name)
        timeout = parse_duration(w.get("timeout"))
        if timeout is not None:
            deadline = monotonic_now() + timeout
            run.timers[f"gatedeadline:{step.name}"] = deadline
            self.engine.set_timer(deadline, run.key, f"gate:{step.name}")

    def _execute_parallel(
        self, run: StoryRun, story: T.Story, step: T.Step, scope: dict, state: StepState
    ) -> None:
        """Fan out branch StepRuns; join handled by the DAG sync
        (reference: step_executor.go:740-811, join dag.go:1112-1194)."""
        w = step.with_ if isinstance(step.with_, dict) else {}
        branches = w.get("steps") or []
        children: _t.List[str] = []
        state.phase = Phase.RUNNING
        for raw in branches:
            branch = T._step_from_dict(dict(raw))
            child_name = compose_name(compose_name(run.name, step.name), branch.name)
            children.append(f"{run.namespace}/{child_name}")
            self._launch_branch(run, story, step, branch, child_name, scope)
        run.primitive_children[step.name] = children

    def _launch_branch(
        self,
        run: StoryRun,
        story: T.Story,
        parent: T.Step,
        branch: T.Step,
        child_name: str,
        scope: dict,
    ) -> None:
        eng = self.engine
        branch_scope = dict(scope)
        branch_scope["branch"] = {"name": branch.name, "parent": parent.name}
        eng.metrics.inc("child_stepruns_created_total")
        spec = StepRunSpec(story_run=run.name, step_name=f"{parent.name}/{branch.name}")
        sr = StepRun(name=child_name, namespace=run.namespace, spec=spec)
        sr, created = eng.store.create_or_get_step_run(sr)
        if not created:
            return
        allow_failure = bool(branch.allow_failure)
        sr.status.message = "allowFailure" if allow_failure else ""
        try:
            if branch.ref is not None:
                ns = branch.ref.resolve_namespace(story.namespace)
                engram = eng.registry.try_engram(branch.ref.name, ns)
                if engram is None:
                    raise ValueError(f"engram {ns}/{branch.ref.name} not found")
                template = (
                    eng.registry.engram_template(engram.template_ref.name)
                    if engram.template_ref is not None
                    else None
                )
                cfg = eng.resolver.resolve(step=branch, story=story, engram=engram, template=template)
                sr.spec.engram = f"{ns}/{branch.ref.name}"
                sr.spec.input = (
                    eng.evaluator.resolve_value(branch.with_, branch_scope)
                    if branch.with_ is not None
                    else None
                )
                sr.spec.config = engram.with_
                sr.spec.runtime = branch.runtime
                sr.spec.timeout_seconds = cfg.timeout_seconds
                sr.spec.placement_gpu = self._place(run, branch, cfg)
                sr.status.phase = Phase.RUNNING
                sr.status.started_at = monotonic_now()
                eng.launch_engram_step(run, sr, template, cfg)
            elif branch.type == StepType.CONDITION:
                w = branch.with_ if isinstance(branch.with_, dict) else {}
                expr = w.get("expression") or w.get("if") or "true"
                result = eng.evaluator.evaluate_condition(str(expr), branch_scope)
                sr.status.phase = Phase.SUCCEEDED
                sr.status.output = {"result": bool(result)}
                sr.status.finished_at = monotonic_now()
            elif branch.type == StepType.SLEEP:
                w = eng.evaluator.resolve_value(
                    branch.with_ if isinstance(branch.with_, dict) else {}, branch_scope
                )
                seconds = parse_duration(w.get("duration")) or 0.0
                sr.status.phase = Phase.RUNNING
                eng.set_timer(
                    monotonic_now() + seconds, run.key, f"branchsleep:{sr.key}"
                )
            else:
                raise ValueError(
                    f"parallel branch {branch.name!r}: unsupported branch type "
                    f"{branch.type} (engram ref, condition and sleep branches are supported)"
                )
        except (TemplateError, ValueError, KeyError) as exc:
            sr.status.phase = Phase.FAILED
            sr.status.finished_at = monotonic_now()
            sr.status.error = StructuredError(type=ErrorType.EXECUTION, message=str(exc))

    def _execute_story(
        self, run: StoryRun, story: T.Story, step: T.Step, scope: dict, state: StepState
    ) -> None:
        """Sub-story (reference: step_executor.go:1132-1230, ensureSubStoryRun
        1407-1510): target story's policy.with merged under step inputs."""
        eng = self.engine
        w = step.with_ if isinstance(step.with_, dict) else {}
        target_name = w.get("storyRef") or w.get("story")
        target_ns = w.get("namespace") or story.namespace
        if isinstance(target_name, dict):
            target_ns = target_name.get("namespace") or target_ns
            target_name = target_name.get("name")
        if run.recursion_depth + 1 > eng.config.max_recursion_depth:
            raise ValueError(
                f"executeStory exceeds max recursion depth {eng.config.max_recursion_depth}"
            )
        try:
            target = eng.registry.story(target_name, target_ns)
        except KeyError:
            state.phase = Phase.BLOCKED
            state.message = f"story {target_ns}/{target_name} not found"
            return
        if target_ns != story.namespace and not eng.registry.allows_cross_namespace(
            "Story", story.namespace, "Story", target_ns, target_name
        ):
            raise ValueError(f"cross-namespace story reference {target_ns}/{target_name} not granted")

        sub_inputs = eng.evaluator.resolve_value(w.get("with") or {}, scope)
        if target.policy is not None and isinstance(target.policy.with_, dict):
            merged = dict(target.policy.with_)
            merged.update(sub_inputs if isinstance(sub_inputs, dict) else {})
            sub_inputs = merged

        child_name = compose_name(run.name, step.name)
        wait = w.get("waitForCompletion", True)
        existing = eng.store.try_get_story_run(f"{run.namespace}/{child_name}")
        if existing is None:
            eng.submit_run(
                story=target,
                inputs=sub_inputs,
                name=child_name,
                namespace=run.namespace,
                parent_run=run.name,
                parent_step=step.name,
                recursion_depth=run.recursion_depth + 1,
                _trusted=True,  # engine-resolved inputs may carry $storageRef
            )
        state.output = {"storyRun": f"{run.namespace}/{child_name}"}
        if wait:
            state.phase = Phase.RUNNING  # join happens in DAG sync
        else:
            state.phase = Phase.SUCCEEDED
            state.finished_at = monotonic_now()


_finish_lock = __import__("threading").Lock()


def finish_step_run(
    sr: StepRun,
    phase: Phase,
    output=None,
    error: _t.Optional[StructuredError] = None,
    exit_code: _t.Optional[int] = None,
    exit_class: _t.Optional[ExitClass] = None,
) -> bool:
    """Terminal-phase-wins write to a StepRun status (SDK-race discipline,
    reference: stepStatusPatchedBySDK steprun_controller.go:2031).  Returns
    False when an earlier terminal phase already won.  The engine loop is
    the only writer in normal operation; the lock makes the first-terminal-
    wins guarantee hold even for out-of-band writers (tests, future
    multi-loop setups) — the critical section is a few field writes."""
    with _finish_lock:
        if sr.status.phase.is_terminal:
            return False
        sr.status.phase = phase
        sr.status.output = output if output is not None else sr.status.output
        if output not in (None, {}, []):
            sr.status.last_output_at = monotonic_now()
        sr.status.error = error
        sr.status.exit_code = exit_code
        sr.status.exit_class = exit_class
        sr.status.finished_at = monotonic_now()
        return True
