"""Materialize engram: delegated template evaluation over offloaded data.

Reference: internal/controller/runs/materialize.go (resolveMaterialize:326,
ensureMaterializeStepRun:142-240) — under
`templating.offloaded-data-policy: block` the controller never hydrates
`$storageRef` data in-process; evaluation is delegated to a dedicated
materialize Engram running as a StepRun.  Here the "worker" is a pool slot
(GPU + HIP stream): ctx.input arrives with `$storageRef` values hydrated to
live tensors by the normal engram-input hydration, evaluation runs off the
engine loop, and tensor-bearing results are re-offloaded to HBM by the normal
StepRun completion path, so the engine only ever moves markers.

Input contract (reference builds {mode, template, vars} at
materialize.go:125-141): ``{"mode": "value"|"condition", "template": <raw
template JSON>, "vars": {<scope>}}`` → output ``{"result": <resolved>}``.
"""
from __future__ import annotations

from ..templating import EvalConfig, Evaluator
from .base import Engram, EngramContext, EngramFailure, EngramResult
from .registry import register_class


@register_class
class MaterializeEngram(Engram):
    name = "materialize"

    def run(self, ctx: EngramContext) -> EngramResult:
        spec = ctx.input if isinstance(ctx.input, dict) else {}
        mode = spec.get("mode", "value")
        template = spec.get("template")
        vars_ = spec.get("vars") or {}
        # vars arrive hydrated; inject policy lets any stray nested marker
        # pass through untouched rather than blocking again.
        ev = Evaluator(EvalConfig())
        try:
            if mode == "condition":
                result = ev.evaluate_condition(template, vars_)
            else:
                result = ev.resolve_value(template, vars_)
        except Exception as exc:  # evaluation errors are terminal, not retryable
            raise EngramFailure(f"materialize: {exc}", exit_code=2)
        return EngramResult(output={"result": result})
