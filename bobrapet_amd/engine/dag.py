"""The DAG engine: per-tick reconciliation of one StoryRun.

Role parity with the reference's DAGReconciler
(reference: internal/controller/runs/dag.go:306-542 — sync StepRun phases
into StepStates, collect prior outputs, iterate ≤ len(steps)+1 times, the
3-phase machine main→compensation→finally at 482-511, findReadySteps
2631-2848 with needs + template-implied deps + `if` + requires guards,
buildDependencyGraphs 3024-3074, cycle check 3076, finalize 693-754 /
2871-2994 with the 1 MiB output cap → Degraded, story timeout 544-578).

MI355X redesign: the tick is an in-memory graph pass driven by completion
events (no requeue cadence); a full tick on an idle run is microseconds, so
step-to-step latency is event-bound, not poll-bound.
"""
from __future__ import annotations

import typing as _t
from dataclasses import dataclass, field

from ..enums import OnTimeout, Phase, StepType
from ..specs import types as T
from ..templating import OffloadedDataUsage, TemplateError, deps as tdeps
from ..utils.durations import parse_duration
from .records import StepState, StoryRun, StructuredError, monotonic_now
from ..enums import ErrorType

if _t.TYPE_CHECKING:
    from .engine import RunEngine

HARD_FAIL_PHASES = frozenset(
    {Phase.FAILED, Phase.TIMEOUT, Phase.ABORTED, Phase.CANCELED}
)


@dataclass
class CompiledStory:
    """Cached per-(story, generation) dependency structure."""

    story: T.Story
    generation: int
    deps: _t.Dict[str, _t.Set[str]] = field(default_factory=dict)  # step → its deps
    dependents: _t.Dict[str, _t.Set[str]] = field(default_factory=dict)
    alias_to_real: _t.Dict[str, str] = field(default_factory=dict)
    comp_deps: _t.Dict[str, _t.Set[str]] = field(default_factory=dict)
    fin_deps: _t.Dict[str, _t.Set[str]] = field(default_factory=dict)
    cycle_error: _t.Optional[str] = None


def compile_story(story: T.Story) -> CompiledStory:
    """Build explicit + template-implied dependency graphs
    (reference: dag.go:3024-3074) and run the runtime cycle check (3076)."""
    cs = CompiledStory(story=story, generation=story.generation)
    cs.alias_to_real = {s.alias: s.name for s in story.all_steps() if s.alias != s.name}
    cs.deps = _graph_of(story.steps, cs.alias_to_real)
    cs.comp_deps = _graph_of(story.compensations, cs.alias_to_real)
    cs.fin_deps = _graph_of(story.finally_, cs.alias_to_real)
    for name, ds in cs.deps.items():
        for d in ds:
            cs.dependents.setdefault(d, set()).add(name)
    cs.cycle_error = _cycle_check(cs.deps, {s.name for s in story.steps})
    return cs


def _graph_of(steps: _t.List[T.Step], aliases: _t.Mapping[str, str]) -> _t.Dict[str, _t.Set[str]]:
    names = {s.name for s in steps}
    graph: _t.Dict[str, _t.Set[str]] = {}
    for s in steps:
        ds: _t.Set[str] = set(s.needs)
        implied: _t.Set[str] = set()
        if s.if_:
            implied |= tdeps.extract_referenced_steps(s.if_)
        if s.with_ is not None and (s.ref is not None or s.type == StepType.EXECUTE_STORY):
            implied |= tdeps.referenced_steps_of_value(s.with_)
        for r in s.requires:
            parts = r.split(".")
            if parts[0] == "steps" and len(parts) >= 2:
                implied.add(parts[1])
        ds |= {d for d in tdeps.resolve_aliases(implied, aliases) if d in names and d != s.name}
        graph[s.name] = ds
    return graph


def _cycle_check(graph: _t.Mapping[str, _t.Set[str]], names: _t.Set[str]) -> _t.Optional[str]:
    indeg = {n: 0 for n in names}
    dependents: _t.Dict[str, _t.Set[str]] = {}
    for n, ds in graph.items():
        local = {d for d in ds if d in names}
        indeg[n] = len(local)
        for d in local:
            dependents.setdefault(d, set()).add(n)
    queue = [n for n, d in indeg.items() if d == 0]
    seen = 0
    while queue:
        cur = queue.pop()
        seen += 1
        for nxt in dependents.get(cur, ()):
            indeg[nxt] -= 1
            if indeg[nxt] == 0:
                queue.append(nxt)
    if seen < len(names):
        stuck = sorted(n for n, d in indeg.items() if d > 0)
        return f"runtime dependency cycle among {stuck} (template-implied deps included)"
    return None


class DAGReconciler:
    def __init__(self, engine: "RunEngine"):
        self.engine = engine
        self._compiled: _t.Dict[_t.Tuple[str, int], CompiledStory] = {}

    def compiled(self, story: T.Story) -> CompiledStory:
        key = (story.key, story.generation)
        cs = self._compiled.get(key)
        if cs is None:
            cs = compile_story(story)
            self._compiled[key] = cs
        return cs

    # ------------------------------------------------------------------

    def tick(self, run: StoryRun) -> None:
        """One reconcile pass (reference: dag.go:306-378 Prepare/Ensure/
        Finalize)."""
        eng = self.engine
        if run.is_terminal:
            return
        try:
            story = eng.registry.story(run.story_name, run.story_namespace)
        except KeyError:
            # requeue-with-jitter in the reference; here the apply() of the
            # story re-ticks the run (watch-driven)
            run.phase = Phase.BLOCKED
            return
        if run.phase == Phase.BLOCKED:
            run.phase = Phase.RUNNING
        cs = self.compiled(story)
        if cs.cycle_error:
            self._fail_run(run, cs.cycle_error)
            return

        if run.started_at is None:
            run.started_at = monotonic_now()
            eng.metrics.inc("storyruns_total", phase="started")

        self._sync_from_step_runs(run)
        self._sync_primitives(run, story, cs)

        if run.cancel_requested and self._handle_cancel(run, story):
            return

        if self._enforce_story_timeout(run, story):
            return

        # iterate: launching a step may immediately finish it (condition/
        # stop/cache hits), unlocking dependents in the same tick
        # (reference: runDagIterations dag.go:381-542)
        for _ in range(len(story.all_steps()) + 1):
            progressed = self._phase_pass(run, story, cs)
            self._sync_primitives(run, story, cs)
            if not progressed:
                break

        self._maybe_finalize(run, story, cs)

    # ------------------------------------------------------------------

    def _sync_from_step_runs(self, run: StoryRun) -> None:
        """Merge StepRun statuses into StepStates, terminal-wins
        (reference: syncStateFromStepRuns dag.go:965)."""
        for sr in self.engine.store.step_runs_of(run.key):
            if "/" in sr.spec.step_name:
                continue  # parallel branch children join via primitive sync
            state = run.step_state(sr.spec.step_name)
            incoming = StepState(
                name=sr.spec.step_name,
                phase=sr.status.phase,
                output=sr.status.output,
                error=sr.status.error,
                retries=sr.status.retries,
                started_at=sr.status.started_at,
                finished_at=sr.status.finished_at,
                message=sr.status.message,
            )
            state.merge_from(incoming)

    # ------------------------------------------------------------------

    def _sync_primitives(self, run: StoryRun, story: T.Story, cs: CompiledStory) -> None:
        """Gates, sleeps, waits, parallels, sub-stories
        (reference: checkSync* dag.go:1112-1606)."""
        now = monotonic_now()
        for step in story.all_steps():
            state = run.step_states.get(step.name)
            if state is None or state.phase.is_terminal:
                continue
            if step.type == StepType.SLEEP and state.phase == Phase.RUNNING:
                deadline = run.timers.get(f"sleep:{step.name}")
                if deadline is not None and now >= deadline:
                    state.phase = Phase.SUCCEEDED
                    state.output = {"slept": True}
                    state.finished_at = now
                    run.timers.pop(f"sleep:{step.name}", None)
            elif step.type == StepType.WAIT and state.phase == Phase.RUNNING:
                self._check_wait(run, story, step, state, now)
            elif step.type == StepType.GATE and state.phase == Phase.PAUSED:
                self._check_gate(run, step, state, now)
            elif step.type == StepType.PARALLEL and state.phase == Phase.RUNNING:
                self._join_parallel(run, step, state)
            elif step.type == StepType.EXECUTE_STORY and state.phase == Phase.RUNNING:
                self._join_substory(run, step, state)

    def _check_wait(self, run: StoryRun, story: T.Story, step: T.Step, state: StepState, now: float) -> None:
        w = step.with_ if isinstance(step.with_, dict) else {}
        scope = self.engine.build_scope(run, story)
        try:
            done = self.engine.evaluator.evaluate_condition(str(w.get("until")), scope)
        except TemplateError as exc:
            state.phase = Phase.FAILED
            state.error = StructuredError(type=ErrorType.EXECUTION, message=str(exc))
            state.finished_at = now
            return
        if done:
            state.phase = Phase.SUCCEEDED
            state.output = {"waited": True}
            state.finished_at = now
            run.timers.pop(f"waitdeadline:{step.name}", None)
            return
        deadline = run.timers.get(f"waitdeadline:{step.name}")
        if deadline is not None and now >= deadline:
            behavior = OnTimeout(w.get("onTimeout", "fail"))
            state.phase = behavior.timeout_phase
            state.message = "wait timed out"
            state.finished_at = now
            return
        poll = run.timers.get(f"waitpoll:{step.name}", self.engine.config.default_wait_poll_interval)
        self.engine.set_timer(now + poll, run.key, f"wait:{step.name}")

    def _check_gate(self, run: StoryRun, step: T.Step, state: StepState, now: float) -> None:
        gate = run.gates.get(step.name)
        if gate is None:
            return
        if gate.state == "Approved":
            state.phase = Phase.SUCCEEDED
            state.output = {"approved": True, "decidedBy": gate.decided_by}
            state.finished_at = now
            run.timers.pop(f"gatedeadline:{step.name}", None)
        elif gate.state == "Rejected":
            state.phase = Phase.FAILED
            state.output = {"approved": False, "decidedBy": gate.decided_by}
            state.error = StructuredError(type=ErrorType.EXECUTION, message="gate rejected")
            state.finished_at = now
            run.timers.pop(f"gatedeadline:{step.name}", None)
        else:
            deadline = run.timers.get(f"gatedeadline:{step.name}")
            if deadline is not None and now >= deadline:
                w = step.with_ if isinstance(step.with_, dict) else {}
                behavior = OnTimeout(w.get("onTimeout", "fail"))
                state.phase = behavior.timeout_phase
                state.message = "gate timed out"
                state.finished_at = now

    def _join_parallel(self, run: StoryRun, step: T.Step, state: StepState) -> None:
        """All children terminal ⇒ parent terminal; per-branch allowFailure
        honored (reference: dag.go:1112-1194)."""
        children = run.primitive_children.get(step.name, [])
        outputs: _t.Dict[str, _t.Any] = {}
        failed: _t.List[str] = []
        details: _t.Dict[str, str] = {}
        for key in children:
            sr = self.engine.store.try_get_step_run(key)
            if sr is None or not sr.is_terminal:
                return  # still joining
            branch_name = sr.spec.step_name.split("/", 1)[-1]
            outputs[branch_name] = sr.status.output
            if sr.status.phase in HARD_FAIL_PHASES and sr.status.message != "allowFailure":
                failed.append(branch_name)
                if sr.status.error is not None:
                    details[branch_name] = sr.status.error.message
        now = monotonic_now()
        state.output = {"branches": outputs}
        state.finished_at = now
        if failed:
            state.phase = Phase.FAILED
            state.error = StructuredError(
                type=ErrorType.EXECUTION,
                message=f"parallel branches failed: {failed}",
                details=details or None,
            )
        else:
            state.phase = Phase.SUCCEEDED

    def _join_substory(self, run: StoryRun, step: T.Step, state: StepState) -> None:
        ref = (state.output or {}).get("storyRun")
        if not ref:
            return
        child = self.engine.store.try_get_story_run(ref)
        if child is None or not child.is_terminal:
            return
        now = monotonic_now()
        state.finished_at = now
        state.output = {"storyRun": ref, "output": child.output, "phase": str(child.phase)}
        if child.phase == Phase.SUCCEEDED:
            state.phase = Phase.SUCCEEDED
        else:
            state.phase = Phase.FAILED
            state.error = StructuredError(
                type=ErrorType.EXECUTION,
                message=f"sub-story finished {child.phase}",
            )

    # ------------------------------------------------------------------

    def _enforce_story_timeout(self, run: StoryRun, story: T.Story) -> bool:
        """(reference: enforceStoryTimeout dag.go:544-578)."""
        timeout = None
        if story.policy is not None and story.policy.timeouts is not None:
            timeout = parse_duration(story.policy.timeouts.story)
        if timeout is None:
            timeout = self.engine.config.default_story_timeout
        if timeout is None or run.started_at is None:
            return False
        deadline = run.started_at + timeout
        if monotonic_now() >= deadline:
            for state in run.step_states.values():
                if not state.phase.is_terminal:
                    state.phase = Phase.TIMEOUT
                    state.finished_at = monotonic_now()
            self._cancel_outstanding_step_runs(run)
            run.phase = Phase.TIMEOUT
            run.finished_at = monotonic_now()
            run.error = StructuredError(type=ErrorType.TIMEOUT, message="story timeout")
            self.engine.on_run_terminal(run)
            return True
        self.engine.set_timer(deadline, run.key, "storytimeout")
        return False

    # ------------------------------------------------------------------

    def _phase_pass(self, run: StoryRun, story: T.Story, cs: CompiledStory) -> bool:
        """Select the execution phase and launch ready steps; returns True
        when anything changed (reference: dag.go:482-511 + findAndLaunch)."""
        exec_phase, steps, deps = self._select_phase(run, story, cs)
        if exec_phase is None:
            return False
        if exec_phase != run.exec_phase:
            run.exec_phase = exec_phase
        fail_fast = self._fail_fast(story)

        failure_present = run.failure_cause is not None or self._has_failure(run, story)
        progressed = False
        launched = 0
        scope = None
        for step in steps:
            state = run.step_states.get(step.name)
            if state is not None and state.phase != Phase.PENDING and state.phase != Phase.BLOCKED:
                continue
            # a stop step earlier in this very pass may have set the directive
            if "stop-phase" in run.annotations and exec_phase == "main":
                st = run.step_state(step.name)
                st.phase = Phase.SKIPPED
                st.message = "skipped by stop"
                st.finished_at = monotonic_now()
                progressed = True
                continue
            verdict = self._readiness(run, story, step, deps, exec_phase, failure_present, fail_fast)
            if verdict == "skip":
                st = run.step_state(step.name)
                st.phase = Phase.SKIPPED
                st.finished_at = monotonic_now()
                progressed = True
                continue
            if verdict != "ready":
                continue
            if not self._admit_step(run, story):
                continue
            if scope is None:
                scope = self.engine.build_scope(run, story)
            st = run.step_state(step.name)
            if st.phase == Phase.BLOCKED:
                st.phase = Phase.PENDING
            self.engine.executor.execute(run, story, step, scope)
            self.engine.metrics.inc("dag_steps_launched_total")
            launched += 1
            progressed = True
            scope = None if st.phase.is_terminal else scope  # refresh after immediate completion
        if launched:
            # (reference: bobrapet_dag_iteration_steps histogram)
            self.engine.metrics.observe("dag_iteration_steps", launched)
        active = sum(1 for st in run.step_states.values() if st.phase == Phase.RUNNING)
        done = sum(1 for st in run.step_states.values() if st.phase.is_terminal)
        self.engine.metrics.set_gauge("storyrun_steps_active", active)
        self.engine.metrics.set_gauge("storyrun_steps_completed", done)
        return progressed

    def _select_phase(
        self, run: StoryRun, story: T.Story, cs: CompiledStory
    ) -> _t.Tuple[_t.Optional[str], _t.List[T.Step], _t.Dict[str, _t.Set[str]]]:
        """main → compensation (on failure) → finally (always)."""
        main_done = self._steps_settled(run, story.steps, run.failure_cause is not None or self._has_failure(run, story), self._fail_fast(story))
        if not main_done:
            return "main", story.steps, cs.deps
        failure = self._has_failure(run, story) or run.cancel_requested
        if failure and story.compensations:
            comp_done = self._steps_settled(run, story.compensations, False, False)
            if not comp_done:
                return "compensation", story.compensations, cs.comp_deps
        if story.finally_:
            fin_done = self._steps_settled(run, story.finally_, False, False)
            if not fin_done:
                return "finally", story.finally_, cs.fin_deps
        return None, [], {}

    def _steps_settled(
        self, run: StoryRun, steps: _t.List[T.Step], failure_present: bool, fail_fast: bool
    ) -> bool:
        """A phase is settled when every step is terminal, or — under
        fail-fast with a failure present — every non-terminal step is still
        unstarted (those get skipped by the phase transition)."""
        for s in steps:
            state = run.step_states.get(s.name)
            if state is None or not state.phase.is_terminal:
                if failure_present and fail_fast:
                    if state is not None and state.phase in (Phase.RUNNING, Phase.PAUSED):
                        return False
                    # unstarted step under fail-fast: mark skipped
                    st = run.step_state(s.name)
                    if not st.phase.is_terminal:
                        st.phase = Phase.SKIPPED
                        st.message = "skipped by fail-fast"
                        st.finished_at = monotonic_now()
                    continue
                return False
        return True

    def _has_failure(self, run: StoryRun, story: T.Story) -> bool:
        for s in story.steps:
            state = run.step_states.get(s.name)
            if state is None:
                continue
            if state.phase in HARD_FAIL_PHASES and not s.allow_failure:
                if run.failure_cause is None:
                    run.failure_cause = s.name
                return True
        return run.failure_cause is not None

    @staticmethod
    def _fail_fast(story: T.Story) -> bool:
        """(reference: shouldFailFast dag.go:3504)."""
        if story.policy is not None and story.policy.retries is not None:
            cont = story.policy.retries.continue_on_step_failure
            if cont is not None:
                return not cont
        return True

    def _readiness(
        self,
        run: StoryRun,
        story: T.Story,
        step: T.Step,
        deps: _t.Mapping[str, _t.Set[str]],
        exec_phase: str,
        failure_present: bool,
        fail_fast: bool,
    ) -> str:
        """'ready' | 'skip' | 'wait' (reference: findReadySteps dag.go:2631)."""
        if exec_phase == "main" and failure_present and fail_fast:
            return "skip"
        for dep in deps.get(step.name, ()):
            dep_state = run.step_states.get(dep)
            if dep_state is None or not dep_state.phase.is_terminal:
                return "wait"
            dep_spec = story.step(dep)
            dep_allow = dep_spec is not None and bool(dep_spec.allow_failure)
            if dep_state.phase == Phase.SKIPPED:
                return "skip"
            if dep_state.phase in HARD_FAIL_PHASES and not dep_allow:
                return "skip"
        if step.if_:
            scope = self.engine.build_scope(run, story)
            try:
                if not self.engine.evaluator.evaluate_condition(step.if_, scope):
                    return "skip"
            except TemplateError:
                return "skip"
        if step.requires:
            scope = self.engine.build_scope(run, story)
            for path in step.requires:
                try:
                    value = self.engine.evaluator.resolve_string("{{ " + path + " }}", scope)
                except TemplateError:
                    return "skip"
                if value is None:
                    return "skip"
        return "ready"

    def _admit_step(self, run: StoryRun, story: T.Story) -> bool:
        """Per-run step concurrency (reference: enforceStoryConcurrency
        dag.go:1780; the queue/global limits act at run admission —
        engine._admit_runs)."""
        limit = None
        if story.policy is not None:
            limit = story.policy.concurrency
        if not limit:
            return True
        running = sum(
            1 for s in run.step_states.values() if s.phase in (Phase.RUNNING, Phase.PAUSED)
        )
        return running < limit

    # ------------------------------------------------------------------

    def _maybe_finalize(self, run: StoryRun, story: T.Story, cs: CompiledStory) -> None:
        exec_phase, _steps, _deps = self._select_phase(run, story, cs)
        if exec_phase is not None:
            return
        # nothing left to run anywhere → terminal
        for state in run.step_states.values():
            if not state.phase.is_terminal:
                return  # still waiting on a running step
        self._finalize(run, story)

    def _finalize(self, run: StoryRun, story: T.Story) -> None:
        """(reference: finalizeStoryRun dag.go:693-754 /
        finalizeSuccessfulRun 2871-2994)."""
        failure = self._has_failure(run, story)
        comp_ok = all(
            run.step_states.get(s.name) is not None
            and run.step_states[s.name].phase in (Phase.SUCCEEDED, Phase.SKIPPED)
            for s in story.compensations
        ) if story.compensations else False
        fin_failed = any(
            (st := run.step_states.get(s.name)) is not None
            and st.phase in HARD_FAIL_PHASES
            and not s.allow_failure
            for s in story.finally_
        )

        stop_phase = run.annotations.get("stop-phase")
        if run.cancel_requested:
            run.phase = Phase.CANCELED
        elif stop_phase:
            run.phase = Phase(stop_phase)
        elif failure and story.compensations and comp_ok:
            run.phase = Phase.COMPENSATED
        elif failure or fin_failed:
            run.phase = Phase.FAILED
        else:
            timed_out = any(
                st.phase == Phase.TIMEOUT for st in run.step_states.values()
            )
            run.phase = Phase.TIMEOUT if timed_out and failure else Phase.SUCCEEDED

        if run.phase == Phase.SUCCEEDED and story.output is not None:
            try:
                scope = self.engine.build_scope(run, story)
                if "__output__" in run.materialized:
                    out = run.materialized["__output__"]
                else:
                    out = self.engine.evaluator.resolve_value(story.output, scope)
                size = _json_size(out)
                if size > self.engine.config.max_output_bytes:
                    run.degraded = True
                    run.output = None
                    run.conditions.set(
                        "Degraded", True, "OutputTooLarge",
                        f"story output {size} bytes exceeds the cap",
                    )
                else:
                    run.output = out
            except OffloadedDataUsage:
                # policy=block: delegate the output template to the
                # materialize engram and finalize on its completion tick
                status = self.engine.executor.delegate_output_materialize(run, story, scope)
                if status == "pending":
                    run.phase = Phase.RUNNING
                    return
                run.phase = Phase.FAILED
                run.error = StructuredError(
                    type=ErrorType.EXECUTION, message="output template materialization failed"
                )
            except TemplateError as exc:
                run.phase = Phase.FAILED
                run.error = StructuredError(
                    type=ErrorType.EXECUTION, message=f"output template: {exc}"
                )
        if failure and run.error is None and run.failure_cause:
            st = run.step_states.get(run.failure_cause)
            run.error = st.error if st is not None else None

        run.finished_at = monotonic_now()
        self.engine.on_run_terminal(run)

    def _handle_cancel(self, run: StoryRun, story: T.Story) -> bool:
        """Graceful cancel with drain deadline (reference:
        storyrun_controller.go:1517-1796)."""
        now = monotonic_now()
        eng = self.engine
        if run.cancel_observed_at is None:
            run.cancel_observed_at = now
            grace = eng.config.default_graceful_shutdown
            if story.policy is not None and story.policy.timeouts is not None:
                g = parse_duration(story.policy.timeouts.graceful_shutdown_timeout)
                if g is not None:
                    grace = g
            run.timers["canceldrain"] = now + grace
            eng.set_timer(now + grace, run.key, "canceldrain")
            for sr in eng.store.step_runs_of(run.key):
                if not sr.is_terminal:
                    sr.cancel_requested = True
        # still-running steps drain; skip everything unstarted
        pending_settled = True
        for state in run.step_states.values():
            if state.phase in (Phase.PENDING, Phase.BLOCKED):
                state.phase = Phase.SKIPPED
                state.message = "canceled"
                state.finished_at = now
            elif not state.phase.is_terminal:
                pending_settled = False
        deadline = run.timers.get("canceldrain", 0)
        if not pending_settled and now < deadline:
            return True  # wait for drain or the timer
        if not pending_settled:
            # force: mark outstanding steps Canceled
            for state in run.step_states.values():
                if not state.phase.is_terminal:
                    state.phase = Phase.CANCELED
                    state.finished_at = now
            self._cancel_outstanding_step_runs(run)
        # run finally steps before going terminal? The reference drains and
        # finishes; compensations are not run on cancel. Finalize directly.
        self._finalize(run, self.engine.registry.story(run.story_name, run.story_namespace))
        return True

    def _cancel_outstanding_step_runs(self, run: StoryRun) -> None:
        for sr in self.engine.store.step_runs_of(run.key):
            if not sr.is_terminal:
                sr.cancel_requested = True
                sr.status.phase = Phase.CANCELED
                sr.status.finished_at = monotonic_now()

    def _fail_run(self, run: StoryRun, message: str) -> None:
        run.phase = Phase.FAILED
        run.error = StructuredError(type=ErrorType.VALIDATION, message=message)
        run.finished_at = monotonic_now()
        self.engine.on_run_terminal(run)


def _json_size(value) -> int:
    import json

    try:
        return len(json.dumps(value, separators=(",", ":"), default=str))
    except (TypeError, ValueError):
        return 0
