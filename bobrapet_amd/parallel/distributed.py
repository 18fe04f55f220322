"""Distributed story execution: one Story's steps spread across ranks.

The MI355X replacement for the reference's cross-pod step placement
(SURVEY.md §2.6): every rank runs the same engine (SPMD); step→rank
placement is a deterministic function of the DAG, so no control plane is
needed; step outputs cross ranks as RCCL broadcasts over xGMI — JSON parts
as objects, tensor parts as device broadcasts (payloads never serialize
through pickle).

Execution model: topological levels.  Steps in one level run concurrently,
each on its placed rank (on that rank's GPU streams); at the level
boundary every step's output is broadcast from its owner.  This keeps all
ranks' scope views identical by construction (the same determinism that
lets the reference dedupe by name).
"""
from __future__ import annotations

import typing as _t

import torch
import torch.distributed as dist

from ..enums import Phase, StepType
from ..specs import types as T
from . import group

_DTYPES = {
    "torch.bfloat16": torch.bfloat16,
    "torch.float16": torch.float16,
    "torch.float32": torch.float32,
    "torch.int32": torch.int32,
    "torch.int64": torch.int64,
}


def place_steps(story: T.Story, world: int) -> _t.Dict[str, int]:
    """Deterministic step→rank placement: explicit placement.gpu pins win;
    otherwise round-robin in declaration order (identical on every rank)."""
    placement: _t.Dict[str, int] = {}
    nxt = 0
    for s in story.steps:
        pin = None
        if s.execution is not None and s.execution.placement is not None:
            pin = s.execution.placement.gpu
        if pin is not None:
            placement[s.name] = pin % world
        else:
            placement[s.name] = nxt % world
            nxt += 1
    return placement


def topo_levels(story: T.Story) -> _t.List[_t.List[str]]:
    from ..engine.dag import compile_story

    cs = compile_story(story)
    indeg = {s.name: len(cs.deps.get(s.name, ())) for s in story.steps}
    levels: _t.List[_t.List[str]] = []
    remaining = {s.name for s in story.steps}
    while remaining:
        level = sorted(n for n in remaining if indeg[n] == 0)
        if not level:
            raise ValueError("dependency cycle in distributed story")
        levels.append(level)
        for n in level:
            remaining.discard(n)
            for d in cs.dependents.get(n, ()):
                indeg[d] -= 1
    return levels


# ---------------------------------------------------------------------------
# tensor-aware output broadcast
# ---------------------------------------------------------------------------


def _extract_tensors(value, out: _t.List[torch.Tensor], path=""):
    """Replace tensors with {"$tensorBcast": {idx, shape, dtype}} markers."""
    if torch.is_tensor(value):
        idx = len(out)
        out.append(value)
        return {
            "$tensorBcast": {
                "idx": idx,
                "shape": list(value.shape),
                "dtype": str(value.dtype),
            }
        }
    if isinstance(value, dict):
        return {k: _extract_tensors(v, out, path) for k, v in value.items()}
    if isinstance(value, list):
        return [_extract_tensors(v, out) for v in value]
    return value


def _restore_tensors(value, tensors: _t.List[torch.Tensor]):
    if isinstance(value, dict):
        if "$tensorBcast" in value and isinstance(value["$tensorBcast"], dict):
            return tensors[value["$tensorBcast"]["idx"]]
        return {k: _restore_tensors(v, tensors) for k, v in value.items()}
    if isinstance(value, list):
        return [_restore_tensors(v, tensors) for v in value]
    return value


def broadcast_step_output(output, src: int, device=None):
    """Broadcast one step's (possibly tensor-bearing) output from its owner."""
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return output
    rank = dist.get_rank()
    tensors: _t.List[torch.Tensor] = []
    meta: _t.List = [None]
    if rank == src:
        meta = [_extract_tensors(output, tensors)]
    dist.broadcast_object_list(meta, src=src)
    skeleton = meta[0]
    # rebuild tensor list from markers on non-owner ranks
    markers: _t.List[dict] = []

    def collect(v):
        if isinstance(v, dict):
            if "$tensorBcast" in v:
                markers.append(v["$tensorBcast"])
            else:
                for x in v.values():
                    collect(x)
        elif isinstance(v, list):
            for x in v:
                collect(x)

    collect(skeleton)
    markers.sort(key=lambda m: m["idx"])
    dev = device
    if dev is None:
        dev = f"cuda:{torch.cuda.current_device()}" if torch.cuda.is_available() else "cpu"
    bufs: _t.List[torch.Tensor] = []
    for i, m in enumerate(markers):
        if rank == src:
            t = tensors[i]
            if torch.cuda.is_available() and not t.is_cuda:
                t = t.to(dev)
        else:
            t = torch.empty(m["shape"], dtype=_DTYPES.get(m["dtype"], torch.float32), device=dev)
        dist.broadcast(t, src=src)
        bufs.append(t)
    return _restore_tensors(skeleton, bufs)


# ---------------------------------------------------------------------------
# distributed run
# ---------------------------------------------------------------------------


class DistributedStoryError(RuntimeError):
    pass


def run_story_distributed(
    engine,
    story: _t.Union[T.Story, str],
    inputs=None,
    timeout: float = 300.0,
) -> dict:
    """Execute one Story with steps placed across all ranks.

    Every rank calls this with identical arguments (SPMD).  Returns
    {phase, steps: {name: {phase, output}}, output} — identical on all
    ranks.  Primitives run on every rank (they are deterministic);
    engram steps run on their placed rank only, outputs broadcast."""
    if isinstance(story, str):
        ns, _, nm = story.rpartition("/")
        story = engine.registry.story(nm, ns or "default")
    world = group.world_size()
    rank = group.rank()
    placement = place_steps(story, world)
    levels = topo_levels(story)
    by_name = {s.name: s for s in story.steps}

    scope = {
        "inputs": inputs or {},
        "steps": {},
        "run": {"name": f"dist-{story.name}", "world": world, "rank": rank},
        "story": {"name": story.name, "namespace": story.namespace},
    }
    states: _t.Dict[str, dict] = {}
    failure = False

    for level in levels:
        # launch all owned steps of the level concurrently
        local: _t.Dict[str, _t.Any] = {}
        for name in level:
            step = by_name[name]
            skip = failure or _deps_skip(step, states, by_name)
            if not skip and step.if_:
                try:
                    skip = not engine.evaluator.evaluate_condition(step.if_, scope)
                except Exception:
                    skip = True
            if skip:
                states[name] = {"phase": "Skipped", "output": None}
                continue
            if step.type is not None and step.type != StepType.PARALLEL:
                # primitives are deterministic → evaluate on every rank
                states[name] = _run_primitive(engine, step, scope)
                continue
            if placement[name] == rank:
                local[name] = _run_engram_local(engine, story, step, scope)
            else:
                states[name] = None  # to be received

        # broadcast outputs in deterministic order
        for name in sorted(n for n in level if states.get(n) is None or n in local):
            owner = placement[name]
            payload = local.get(name)
            received = broadcast_step_output(payload, src=owner)
            states[name] = received
        # sync scope
        for name in level:
            st = states[name]
            scope["steps"][name] = {
                "phase": st["phase"],
                "output": st.get("output"),
                "error": st.get("error"),
            }
            alias = by_name[name].alias
            if alias != name:
                scope["steps"][alias] = scope["steps"][name]
            if st["phase"] in ("Failed", "Timeout") and not by_name[name].allow_failure:
                failure = True

    phase = Phase.FAILED if failure else Phase.SUCCEEDED
    output = None
    if phase == Phase.SUCCEEDED and story.output is not None:
        output = engine.evaluator.resolve_value(story.output, scope)
    return {"phase": str(phase), "steps": states, "output": output}


def _deps_skip(step: T.Step, states, by_name) -> bool:
    for dep in step.needs:
        st = states.get(dep)
        if st is None:
            continue
        if st["phase"] == "Skipped":
            return True
        if st["phase"] in ("Failed", "Timeout") and not by_name[dep].allow_failure:
            return True
    return False


def _run_primitive(engine, step: T.Step, scope) -> dict:
    import time

    w = step.with_ if isinstance(step.with_, dict) else {}
    if step.type == StepType.CONDITION:
        expr = w.get("expression") or w.get("if") or step.if_ or "true"
        try:
            result = engine.evaluator.evaluate_condition(str(expr), scope)
        except Exception as exc:
            return {"phase": "Failed", "output": None, "error": str(exc)}
        return {"phase": "Succeeded", "output": {"result": bool(result)}}
    if step.type == StepType.SLEEP:
        from ..utils.durations import parse_duration

        time.sleep(parse_duration(w.get("duration")) or 0.0)
        return {"phase": "Succeeded", "output": {"slept": True}}
    return {
        "phase": "Failed",
        "output": None,
        "error": f"primitive {step.type} unsupported in distributed mode",
    }


def _run_engram_local(engine, story: T.Story, step: T.Step, scope) -> dict:
    from ..engrams import registry as engram_registry
    from ..engrams.base import EngramContext, EngramFailure, EngramResult

    ns = step.ref.resolve_namespace(story.namespace)
    engram = engine.registry.try_engram(step.ref.name, ns)
    if engram is None:
        return {"phase": "Failed", "output": None, "error": f"engram {ns}/{step.ref.name} not found"}
    template = None
    if engram.template_ref is not None:
        try:
            template = engine.registry.engram_template(engram.template_ref.name)
        except KeyError:
            pass
    impl_name = template.implementation if template else step.ref.name
    resolved = (
        engine.evaluator.resolve_value(step.with_, scope) if step.with_ is not None else None
    )
    device = torch.cuda.current_device() if torch.cuda.is_available() else None
    ctx = EngramContext(
        story_name=story.name,
        step_name=step.name,
        input=engine.storage.hydrate(resolved),
        config=engram.with_,
        runtime=step.runtime,
        device=device,
        storage=engine.storage,
    )
    try:
        result = engram_registry.resolve(impl_name).run(ctx)
        if not isinstance(result, EngramResult):
            result = EngramResult(output=result)
        # hydrate tensor refs so the broadcast moves raw tensors, not refs
        output = engine.storage.hydrate(result.output)
        return {"phase": "Succeeded", "output": output}
    except EngramFailure as exc:
        return {"phase": "Failed", "output": None, "error": str(exc)}
    except Exception as exc:
        return {"phase": "Failed", "output": None, "error": f"{type(exc).__name__}: {exc}"}
