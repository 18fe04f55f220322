"""CPU utility engrams: filter / transform / echo / fail.

These are the workflow-plumbing workers the north star names
("filter/transform" — BASELINE.json); echo/fail/sleepy exist for tests and
benchmarks (the reference exercises these paths by hand-patching StepRun
status in envtest — SURVEY.md §4.2; here they are real engrams).
"""
from __future__ import annotations

import time
import typing as _t

from ..templating import EvalConfig, Evaluator
from .base import Engram, EngramContext, EngramFailure, EngramResult
from .registry import register_class


@register_class
class EchoEngram(Engram):
    """Returns its input as output (optionally merged with config)."""

    name = "echo"

    def run(self, ctx: EngramContext) -> EngramResult:
        out = ctx.input
        if isinstance(ctx.input, dict) and isinstance(ctx.config, dict):
            out = {**ctx.config, **ctx.input}
        return EngramResult(output=out)


@register_class
class FailEngram(Engram):
    """Fails with a configurable exit code; succeeds after `succeedAfter`
    attempts (input key "attempt" carries the retry count). Test/benchmark
    engram for the retry machinery."""

    name = "fail"

    def __init__(self):
        self.calls: _t.Dict[str, int] = {}

    def run(self, ctx: EngramContext) -> EngramResult:
        cfg = ctx.input if isinstance(ctx.input, dict) else {}
        merged = dict(ctx.config or {})
        merged.update(cfg)
        key = f"{ctx.story_run}/{ctx.step_name}"
        self.calls[key] = self.calls.get(key, 0) + 1
        succeed_after = merged.get("succeedAfter")
        if succeed_after is not None and self.calls[key] > int(succeed_after):
            return EngramResult(output={"attempts": self.calls[key]})
        raise EngramFailure(
            merged.get("message", "fail engram failing"),
            exit_code=int(merged.get("exitCode", 1)),
        )


@register_class
class SleepyEngram(Engram):
    """Sleeps for input.seconds then echoes; cancellable."""

    name = "sleepy"

    def run(self, ctx: EngramContext) -> EngramResult:
        cfg = ctx.input if isinstance(ctx.input, dict) else {}
        seconds = float(cfg.get("seconds", 0.01))
        deadline = time.monotonic() + seconds
        while time.monotonic() < deadline:
            if ctx.canceled:
                raise EngramFailure("canceled during sleep", exit_code=2)
            time.sleep(min(0.005, max(deadline - time.monotonic(), 0)))
        return EngramResult(output={"slept": seconds})


@register_class
class FilterEngram(Engram):
    """Filters input.items by a template predicate over {item, index}.

    config/input: {items: [...], where: "{{ item.score > 0.5 }}"}"""

    name = "filter"

    def __init__(self):
        self._ev = Evaluator(EvalConfig(deterministic=True))

    def run(self, ctx: EngramContext) -> EngramResult:
        cfg = dict(ctx.config or {})
        if isinstance(ctx.input, dict):
            cfg.update(ctx.input)
        items = cfg.get("items")
        if not isinstance(items, list):
            raise EngramFailure("filter: input.items must be a list", exit_code=2)
        rt = ctx.runtime if isinstance(ctx.runtime, dict) else {}
        where = rt.get("where") or cfg.get("where")
        if not where:
            return EngramResult(output={"items": items, "count": len(items)})
        kept = []
        for i, item in enumerate(items):
            if self._ev.evaluate_condition(where, {"item": item, "index": i}):
                kept.append(item)
        return EngramResult(output={"items": kept, "count": len(kept)})


@register_class
class TransformEngram(Engram):
    """Maps input.items through a template over {item, index}.

    config/input: {items: [...], map: {id: "{{ item.id }}", double: "{{ item.x * 2 }}"}}"""

    name = "transform"

    def __init__(self):
        self._ev = Evaluator(EvalConfig(deterministic=True))

    def run(self, ctx: EngramContext) -> EngramResult:
        cfg = dict(ctx.config or {})
        if isinstance(ctx.input, dict):
            cfg.update(ctx.input)
        items = cfg.get("items")
        rt = ctx.runtime if isinstance(ctx.runtime, dict) else {}
        mapping = rt.get("map") or cfg.get("map")
        if not isinstance(items, list):
            raise EngramFailure("transform: input.items must be a list", exit_code=2)
        if mapping is None:
            return EngramResult(output={"items": items})
        out = [
            self._ev.resolve_value(mapping, {"item": item, "index": i})
            for i, item in enumerate(items)
        ]
        return EngramResult(output={"items": out, "count": len(out)})
