"""Native fast path: compile Stories onto the bobraccel C++ DAG core.

The C++ core (csrc/core) owns the run state machine (the role of the
reference's DAGReconciler — internal/controller/runs/dag.go:306-542 —
including the 3-phase main→compensation→finally machine of dag.go:482-511,
postExecution checks of steprun_controller.go:2050-2124, timers, retry
classes, and template evaluation); engram steps are dispatched to the same
(gpu, stream) worker slots the Python engine uses.  The full BATCH surface
runs natively; streaming-pattern Stories route to the streaming runtime —
``story_supported()`` gates the fast path.
"""
from __future__ import annotations

import typing as _t

from ..engrams import registry as engram_registry
from ..engrams.base import Engram, EngramContext, EngramFailure, EngramResult
from ..enums import StepType, StopMode, StoryPattern
from ..specs import types as T
from ..templating import is_template, parse_expression, parse_template
from ..templating.parser import CompiledTemplate
from ..utils.durations import parse_duration

_KIND = {
    None: 0,  # engram
    StepType.CONDITION: 1,
    StepType.SLEEP: 2,
    StepType.STOP: 3,
    StepType.WAIT: 4,
    StepType.GATE: 5,
    StepType.PARALLEL: 6,
    StepType.EXECUTE_STORY: 7,
}

_BACKOFF = {"exponential": 0, "linear": 1, "constant": 2}


class NativeCompileError(ValueError):
    pass


def load_core():
    try:
        from bobrapet_amd import _core

        return _core
    except ImportError as exc:
        raise RuntimeError(
            "bobrapet_amd._core is not built; run `python -m bobrapet_amd.csrc.build`"
        ) from exc


def core_available() -> bool:
    try:
        load_core()
        return True
    except RuntimeError:
        return False


# ---------------------------------------------------------------------------
# template/expression compilation to the C++ tuple IR
# ---------------------------------------------------------------------------


def compile_template_value(value):
    """JSON-like value with embedded {{ }} strings → tnode tuples."""
    if isinstance(value, str):
        if is_template(value):
            tpl: CompiledTemplate = parse_template(value)
            if tpl.single:
                return ("expr", tpl.parts[0][1])
            parts = []
            for kind, part in tpl.parts:
                if kind == "lit":
                    parts.append((part, None))
                else:
                    parts.append(("", part))
            return ("parts", parts)
        return ("lit", value)
    if isinstance(value, dict):
        return ("obj", [(k, compile_template_value(v)) for k, v in value.items()])
    if isinstance(value, list):
        return ("arr", [compile_template_value(v) for v in value])
    return ("lit", value)


def compile_condition(src: str):
    src = src.strip()
    if is_template(src):
        tpl = parse_template(src)
        if tpl.single:
            return tpl.parts[0][1]
        raise NativeCompileError(f"condition {src!r} must be a single expression")
    return parse_expression(src)


def _requires_expr(path: str):
    return parse_expression(path)


# ---------------------------------------------------------------------------
# story → plan dict
# ---------------------------------------------------------------------------


def story_supported(story: T.Story) -> _t.Optional[str]:
    """None if the native fast path can run this story; else the reason."""
    if story.pattern != StoryPattern.BATCH:
        return "streaming stories run on the streaming runtime"
    for s in story.all_steps():
        if s.type == StepType.PARALLEL:
            w = s.with_ if isinstance(s.with_, dict) else {}
            for raw in w.get("steps", []):
                b = T._step_from_dict(dict(raw))
                if b.type not in (None, StepType.CONDITION, StepType.SLEEP):
                    return f"parallel branch type {b.type} unsupported natively"
    return None


def compile_story_plan(
    story: T.Story,
    resolver,
    registry,
    plan_ids: _t.Mapping[str, int],
    native_kinds: _t.Optional[_t.Mapping[str, int]] = None,
) -> dict:
    """Build the plan dict the C++ core ingests."""
    from ..engine.dag import compile_story as py_compile

    reason = story_supported(story)
    if reason is not None:
        raise NativeCompileError(reason)
    cs = py_compile(story)
    if cs.cycle_error:
        raise NativeCompileError(cs.cycle_error)
    # global step index: main ++ compensations ++ finally (the core's
    # 3-phase layout; comp/fin needs may reference main steps by name)
    ordered = list(story.steps) + list(story.compensations) + list(story.finally_)
    index = {s.name: i for i, s in enumerate(ordered)}
    dep_graphs = (
        [(s, cs.deps) for s in story.steps]
        + [(s, cs.comp_deps) for s in story.compensations]
        + [(s, cs.fin_deps) for s in story.finally_]
    )

    steps = []
    for s, graph in dep_graphs:
        steps.append(
            _compile_step(s, story, resolver, registry, index, graph, plan_ids,
                          native_kinds)
        )

    plan = {
        "name": story.key,
        "steps": steps,
        "failFast": True,
        "concurrency": 0,
        "nMain": len(story.steps),
        "nComp": len(story.compensations),
        "nFin": len(story.finally_),
    }
    if story.policy is not None:
        if story.policy.retries is not None and story.policy.retries.continue_on_step_failure:
            plan["failFast"] = False
        if story.policy.timeouts is not None and story.policy.timeouts.story:
            plan["storyTimeout"] = parse_duration(story.policy.timeouts.story)
    if story.output is not None:
        plan["output"] = compile_template_value(story.output)
    return plan


def _compile_step(s: T.Step, story, resolver, registry, index, dep_graph, plan_ids,
                  native_kinds=None) -> dict:
    d: dict = {"name": s.name, "kind": _KIND[s.type]}
    deps = sorted(dep_graph.get(s.name, set()))
    by_name = {x.name: x for x in story.all_steps()}
    d["deps"] = [index[x] for x in deps if x in index]
    d["depAllowFailure"] = [
        bool(by_name[x].allow_failure) if x in by_name else False
        for x in deps
        if x in index
    ]
    if s.if_:
        d["if"] = compile_condition(s.if_)
    if s.requires:
        d["requires"] = [_requires_expr(p) for p in s.requires]
    if s.allow_failure:
        d["allowFailure"] = True

    w = s.with_ if isinstance(s.with_, dict) else {}
    if s.type is None:  # engram
        engram = None
        template = None
        if s.ref is not None:
            ns = s.ref.resolve_namespace(story.namespace)
            d["engram"] = f"{ns}/{s.ref.name}"
            engram = registry.try_engram(s.ref.name, ns)
            if engram is not None and engram.template_ref is not None:
                try:
                    template = registry.engram_template(engram.template_ref.name)
                except KeyError:
                    template = None
        cfg = resolver.resolve(step=s, story=story, engram=engram, template=template)
        if native_kinds and template is not None and getattr(template, "builtin", None):
            kind = native_kinds.get(str(template.builtin))
            if kind:
                # GIL-free lane dispatch (csrc/hip/native_engrams.cpp); the
                # lane falls back to the Python launcher per-input
                d["nativeKind"] = int(kind)
                d["nativeCfg"] = dict(engram.with_ or {}) if engram is not None else {}
        d["retry"] = {
            "maxRetries": cfg.max_retries,
            "delay": cfg.retry_delay,
            "maxDelay": cfg.retry_max_delay,
            "jitterPct": cfg.retry_jitter_pct,
            "backoff": _BACKOFF.get(str(cfg.backoff), 0),
        }
        if cfg.timeout_seconds:
            d["timeout"] = cfg.timeout_seconds
        if s.with_ is not None:
            d["with"] = compile_template_value(s.with_)
        if s.post_execution is not None:
            d["postExec"] = compile_condition(s.post_execution.condition)
            if s.post_execution.failure_message:
                d["postExecMsg"] = s.post_execution.failure_message
    elif s.type == StepType.CONDITION:
        expr = w.get("expression") or w.get("if") or s.if_ or "true"
        d["if"] = compile_condition(str(expr))
        d.pop("requires", None)
    elif s.type == StepType.SLEEP:
        dur = w.get("duration")
        if isinstance(dur, str) and is_template(dur):
            d["with"] = compile_template_value({"duration": dur})
            d["sleepDuration"] = 0.0
        else:
            d["sleepDuration"] = parse_duration(dur) or 0.0
    elif s.type == StepType.STOP:
        phase = w.get("phase")
        if phase is None and w.get("mode"):
            phase = str(StopMode(w["mode"]).terminal_phase)
        d["stopPhase"] = phase or "Succeeded"
    elif s.type == StepType.WAIT:
        d["until"] = compile_condition(str(w.get("until", "false")))
        if w.get("timeout"):
            d["waitTimeout"] = parse_duration(w["timeout"])
        d["pollInterval"] = parse_duration(w.get("pollInterval")) or 0.02
        d["onTimeoutSkip"] = w.get("onTimeout") == "skip"
    elif s.type == StepType.GATE:
        if w.get("timeout"):
            d["waitTimeout"] = parse_duration(w["timeout"])
        d["onTimeoutSkip"] = w.get("onTimeout") == "skip"
    elif s.type == StepType.PARALLEL:
        branches = []
        for raw in w.get("steps", []):
            b = T._step_from_dict(dict(raw))
            branches.append(
                _compile_step(b, story, resolver, registry, {}, {}, plan_ids,
                              native_kinds)
            )
        d["branches"] = branches
    elif s.type == StepType.EXECUTE_STORY:
        target = w.get("storyRef") or w.get("story")
        target_ns = w.get("namespace") or story.namespace
        key = f"{target_ns}/{target}"
        if key not in plan_ids:
            raise NativeCompileError(f"executeStory target {key} not compiled")
        d["targetPlan"] = plan_ids[key]
        if w.get("with") is not None:
            d["with"] = compile_template_value({"with": w["with"]})
    return d


# ---------------------------------------------------------------------------
# runner
# ---------------------------------------------------------------------------


class NativeRunner:
    """Owns a NativeEngine; engram dispatch goes to the shared worker pool.

    Borrow the resource registry / storage / workers from a RunEngine (or
    construct standalone with registry+workers)."""

    def __init__(self, registry, resolver, storage, workers, metrics=None):
        core = load_core()
        self.engine = core.NativeEngine()
        self.registry = registry
        self.resolver = resolver
        self.storage = storage
        self.workers = workers
        self.metrics = metrics
        self.plan_ids: _t.Dict[str, int] = {}
        self._plan_gen: _t.Dict[str, int] = {}
        self._engram_cache: _t.Dict[str, tuple] = {}
        self._engram_cache_gen = -1
        self.engine.set_launcher(self._launch)
        self.native_kinds: _t.Dict[str, int] = {}
        self._setup_native_lane()
        # expression-level hydration of offloaded payloads (markers passed
        # THROUGH stay zero-copy; only consuming expressions materialize)
        if self.storage is not None and hasattr(core, "set_expr_hydrator"):
            core.set_expr_hydrator(self.storage.hydrate)
        self.engine.start()

    def _setup_native_lane(self) -> None:
        """Register the GIL-free built-in engram lane when a GPU and the
        HIP extension are present (embed always; the local join only at
        world_size 1 — multi-rank joins all-gather over RCCL in Python)."""
        try:
            import torch

            if not torch.cuda.is_available():
                return
            from bobrapet_amd import _hipops

            if not hasattr(_hipops, "native_lane_capsule") or not hasattr(
                self.engine, "set_native_lane"
            ):
                return
            self.engine.set_native_lane(_hipops.native_lane_capsule())
            devs = list(getattr(self.workers, "device_ids", None) or [0])
            self.engine.set_devices([int(d) for d in devs])
            self.native_kinds = {"embed": 1}
            import torch.distributed as dist

            if not (dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1):
                self.native_kinds["allgather-join"] = 2
        except Exception:
            self.native_kinds = {}

    @classmethod
    def from_run_engine(cls, eng) -> "NativeRunner":
        return cls(eng.registry, eng.resolver, eng.storage, eng.workers, eng.metrics)

    def stop(self) -> None:
        self.engine.stop()

    # ------------------------------------------------------------------

    def compile(self, story: T.Story) -> int:
        key = story.key
        if key in self.plan_ids and self._plan_gen.get(key) == story.generation:
            return self.plan_ids[key]
        # compile executeStory targets first (cycles guarded by recursion cap)
        for s in story.steps:
            if s.type == StepType.EXECUTE_STORY and isinstance(s.with_, dict):
                target = s.with_.get("storyRef") or s.with_.get("story")
                tns = s.with_.get("namespace") or story.namespace
                if f"{tns}/{target}" not in self.plan_ids:
                    self.compile(self.registry.story(target, tns))
        plan = compile_story_plan(
            story, self.resolver, self.registry, self.plan_ids,
            native_kinds=self.native_kinds or None,
        )
        pid = self.engine.register_plan(plan)
        self.plan_ids[key] = pid
        self._plan_gen[key] = story.generation
        return pid

    def submit(self, story: _t.Union[T.Story, str], inputs=None) -> int:
        if isinstance(story, str):
            ns, _, nm = story.rpartition("/")
            story = self.registry.story(nm, ns or "default")
        pid = self.compile(story)
        return self.engine.submit(pid, inputs if inputs is not None else {})

    def wait(self, run_id: int, timeout: float = 60.0) -> dict:
        ok = self.engine.wait(run_id, timeout)
        if not ok:
            raise TimeoutError(f"native run {run_id} did not finish in {timeout}s")
        return self.engine.run_status(run_id)

    def run_story(self, story, inputs=None, timeout: float = 60.0, gc: bool = True) -> dict:
        run_id = self.submit(story, inputs)
        status = self.wait(run_id, timeout)
        if gc:
            # reclaim the run record (+ terminal executeStory descendants)
            # once its status is in hand — the fast path's retention
            self.engine.gc_run(run_id)
        return status

    def decide_gate(self, run_id: int, step_index: int, approved: bool) -> None:
        self.engine.decide_gate(run_id, step_index, approved)

    def cancel(self, run_id: int) -> None:
        self.engine.cancel(run_id)

    # ------------------------------------------------------------------

    def _resolve_engram(self, engram_key: str):
        """(impl_name, config) for an engram key — resolved once per key and
        cached: the launcher runs on the core loop thread per step body, so
        registry/template lookups there are pure per-step overhead."""
        gen = getattr(self.registry, "mutations", 0)
        if gen != self._engram_cache_gen:
            self._engram_cache.clear()
            self._engram_cache_gen = gen
        hit = self._engram_cache.get(engram_key)
        if hit is not None:
            return hit
        engram = None
        if "/" in engram_key:
            ns, name = engram_key.split("/", 1)
            engram = self.registry.try_engram(name, ns)
        if engram is None:
            return None
        template = None
        if engram.template_ref is not None:
            try:
                template = self.registry.engram_template(engram.template_ref.name)
            except KeyError:
                pass
        if template is not None and getattr(template, "command", None):
            from ..engrams.process import ProcessEngram

            hit = (
                ProcessEngram(
                    template.command,
                    secret_defs=getattr(template, "secrets", None),
                    secrets=engram.secrets,
                ),
                engram.with_,
            )
            self._engram_cache[engram_key] = hit
            return hit
        impl_name = template.implementation if template is not None else engram_key.split("/")[-1]
        hit = (impl_name, engram.with_)
        self._engram_cache[engram_key] = hit
        return hit

    def _launch(self, run_id, step, branch, attempt, engram_key, step_name, resolved_input):
        """Engram launcher (called from the core's loop thread, GIL held):
        enqueue onto the worker pool and return immediately."""
        hit = self._resolve_engram(engram_key)
        if hit is None:
            self.engine.complete_engram(
                run_id, step, branch, attempt, 2, None, f"engram {engram_key} not found"
            )
            return
        impl_name, config = hit
        device = None
        if self.workers.device_count > 0:
            ids = self.workers.device_ids
            device = ids[(hash((run_id, step, branch)) & 0x7FFFFFFF) % len(ids)]

        impl = (
            impl_name
            if isinstance(impl_name, Engram)
            else engram_registry.resolve(impl_name)
        )

        def body(slot):
            try:
                ctx = EngramContext(
                    story_run=str(run_id),
                    step_name=step_name,
                    input=self.storage.hydrate(resolved_input),
                    config=config,
                    device=slot.device,
                    stream=slot.stream,
                    storage=self.storage,
                )
                result = impl.run(ctx)
                if not isinstance(result, EngramResult):
                    result = EngramResult(output=result)
                output = self.storage.dehydrate_document(result.output)
                self.engine.complete_engram(
                    run_id, step, branch, attempt, result.exit_code, output, ""
                )
            except EngramFailure as exc:
                self.engine.complete_engram(
                    run_id, step, branch, attempt, exc.exit_code, None, str(exc)
                )
            except Exception as exc:  # crash → retryable
                self.engine.complete_engram(
                    run_id, step, branch, attempt, 1, None,
                    f"{type(exc).__name__}: {exc}",
                )

        self.workers.submit(body, device=device)
