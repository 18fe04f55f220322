"""Run-state snapshot save/restore.

Role parity with the reference's durability model ("state
IS the checkpoint" — every piece of progress persisted in CR status
survives operator restarts).  The in-process engine persists the same
state — run records, step states, timers, gates, trigger tokens — to a
JSON snapshot; tensor payloads stay in HBM and are recorded as refs
(re-hydration after a real process restart re-runs the producing steps via
redrive, like the reference re-runs lost pods).
"""
from __future__ import annotations

import json
import typing as _t

from ..enums import EffectClaimPhase, ExitClass, Phase, classify_exit_code
from .effects import EffectClaim
from .records import (
    GateStatus,
    StepRun,
    StepRunSpec,
    StepRunStatus,
    StepState,
    StoryRun,
    StructuredError,
)
from ..enums import ErrorType


def _err_to_dict(e: _t.Optional[StructuredError]):
    return e.to_dict() if e is not None else None


def _err_from_dict(d):
    if not d:
        return None
    return StructuredError(
        type=ErrorType(d.get("type", "unknown")),
        message=d.get("message", ""),
        retryable=bool(d.get("retryable")),
        details=d.get("details"),
    )


def _clean(value):
    """Strip non-JSON payload parts (tensors become notes)."""
    t = type(value)
    if t.__module__ == "torch":
        return {"$lostTensor": True}
    if isinstance(value, dict):
        return {k: _clean(v) for k, v in value.items()}
    if isinstance(value, list):
        return [_clean(v) for v in value]
    return value


def _restore_exit_class(sd: dict):
    """Restore the recorded exit class; fall back to classifying the exit
    code (older snapshots recorded only exitCode)."""
    ec = sd.get("exitClass")
    if ec:
        return ExitClass(ec)
    code = sd.get("exitCode")
    if code is None:
        return None
    return classify_exit_code(int(code))


def dump_state(engine) -> dict:
    runs = []
    for run in engine.store.all_runs():
        runs.append(
            {
                "name": run.name,
                "namespace": run.namespace,
                "story": [run.story_name, run.story_namespace],
                "inputs": _clean(run.inputs),
                "phase": str(run.phase),
                "execPhase": run.exec_phase,
                "failureCause": run.failure_cause,
                "stepStates": {
                    k: {
                        "phase": str(v.phase),
                        "output": _clean(v.output),
                        "error": _err_to_dict(v.error),
                        "retries": v.retries,
                        "startedAt": v.started_at,
                        "finishedAt": v.finished_at,
                        "message": v.message,
                    }
                    for k, v in run.step_states.items()
                },
                "gates": {
                    k: {"state": g.state, "decidedBy": g.decided_by}
                    for k, g in run.gates.items()
                },
                "primitiveChildren": run.primitive_children,
                "materialized": _clean(run.materialized),
                "triggerTokens": run.trigger_tokens,
                "timers": run.timers,
                "output": _clean(run.output),
                "error": _err_to_dict(run.error),
                "annotations": run.annotations,
                "labels": run.labels,
                "queue": run.queue,
                "priority": run.priority,
                "inputHash": run.input_hash,
                "redrives": run.redrive_count,
                "createdAt": run.created_at,
                "startedAt": run.started_at,
                "finishedAt": run.finished_at,
            }
        )
    steps = []
    for run in engine.store.all_runs():
        for sr in engine.store.step_runs_of(run.key):
            steps.append(
                {
                    "name": sr.name,
                    "namespace": sr.namespace,
                    "storyRun": sr.spec.story_run,
                    "stepName": sr.spec.step_name,
                    "engram": sr.spec.engram,
                    "input": _clean(sr.spec.input),
                    "idempotencyKey": sr.spec.idempotency_key,
                    "phase": str(sr.status.phase),
                    "output": _clean(sr.status.output),
                    "error": _err_to_dict(sr.status.error),
                    "retries": sr.status.retries,
                    "exitCode": sr.status.exit_code,
                    "exitClass": str(sr.status.exit_class) if sr.status.exit_class is not None else None,
                    "cacheHit": sr.status.cache_hit,
                }
            )
    # EffectClaim ledger: without this, completed side effects re-run after
    # a restore (restored non-terminal steps re-execute against an empty
    # ledger) — the reference persists claims durably as the EffectClaim CRD.
    effects = []
    with engine.effects._lock:
        for claim in engine.effects._claims.values():
            effects.append(
                {
                    "key": claim.key,
                    "holder": claim.holder,
                    "phase": str(claim.phase),
                    "leaseDuration": claim.lease_duration,
                    "acquiredAt": claim.acquired_at,
                    "renewedAt": claim.renewed_at,
                    "takeovers": claim.takeovers,
                    "description": claim.description,
                }
            )
    return {"version": 1, "storyRuns": runs, "stepRuns": steps, "effectClaims": effects}


def save_state(engine, path: str) -> None:
    with open(path, "w", encoding="utf-8") as fh:
        json.dump(dump_state(engine), fh, default=str)


def load_state(engine, path: str) -> int:
    """Restore run records into a fresh engine's store; non-terminal runs
    are re-ticked (their finished step states are preserved — the engine
    resumes from the checkpoint like the reference resumes from CR status).
    Returns the number of restored StoryRuns."""
    with open(path, "r", encoding="utf-8") as fh:
        data = json.load(fh)
    n = 0
    for rd in data.get("storyRuns", []):
        run = StoryRun(
            name=rd["name"],
            namespace=rd["namespace"],
            story_name=rd["story"][0],
            story_namespace=rd["story"][1],
            inputs=rd.get("inputs"),
            phase=Phase(rd["phase"]),
            exec_phase=rd.get("execPhase", "main"),
            failure_cause=rd.get("failureCause"),
            primitive_children={k: list(v) for k, v in rd.get("primitiveChildren", {}).items()},
            trigger_tokens=list(rd.get("triggerTokens", [])),
            timers=dict(rd.get("timers", {})),
            output=rd.get("output"),
            error=_err_from_dict(rd.get("error")),
            annotations=dict(rd.get("annotations", {})),
            labels=dict(rd.get("labels", {})),
            queue=rd.get("queue", "default"),
            priority=int(rd.get("priority", 0)),
            input_hash=rd.get("inputHash", ""),
            redrive_count=int(rd.get("redrives", 0)),
        )
        run.created_at = rd.get("createdAt") or run.created_at
        run.started_at = rd.get("startedAt")
        run.finished_at = rd.get("finishedAt")
        for k, sd in rd.get("stepStates", {}).items():
            st = StepState(
                name=k,
                phase=Phase(sd["phase"]),
                output=sd.get("output"),
                error=_err_from_dict(sd.get("error")),
                retries=int(sd.get("retries", 0)),
                started_at=sd.get("startedAt"),
                finished_at=sd.get("finishedAt"),
                message=sd.get("message", ""),
            )
            # in-flight work died with the old process: mark for re-execution
            if not st.phase.is_terminal:
                st.phase = Phase.PENDING
                st.started_at = None
            run.step_states[k] = st
        run.materialized = rd.get("materialized") or {}
        for k, gd in rd.get("gates", {}).items():
            run.gates[k] = GateStatus(step=k, state=gd.get("state", "Pending"), decided_by=gd.get("decidedBy", ""))
        engine.store.create_story_run(run)
        n += 1
    for sd in data.get("stepRuns", []):
        phase = Phase(sd["phase"])
        if not phase.is_terminal:
            continue  # lost with the process; the run re-executes the step
        sr = StepRun(
            name=sd["name"],
            namespace=sd["namespace"],
            spec=StepRunSpec(
                story_run=sd["storyRun"],
                step_name=sd["stepName"],
                engram=sd.get("engram"),
                input=sd.get("input"),
                idempotency_key=sd.get("idempotencyKey"),
            ),
            status=StepRunStatus(
                phase=phase,
                output=sd.get("output"),
                error=_err_from_dict(sd.get("error")),
                retries=int(sd.get("retries", 0)),
                exit_code=sd.get("exitCode"),
                exit_class=_restore_exit_class(sd),
                cache_hit=bool(sd.get("cacheHit")),
            ),
        )
        engine.store.create_or_get_step_run(sr)
    # restore the EffectClaim ledger BEFORE re-ticking runs, so side effects
    # already Completed pre-restart are not performed again (exactly-once)
    with engine.effects._lock:
        for cd in data.get("effectClaims", []):
            claim = EffectClaim(
                key=cd["key"],
                holder=cd.get("holder", ""),
                phase=EffectClaimPhase(cd.get("phase", "Reserved")),
                lease_duration=float(cd.get("leaseDuration", 60.0)),
                takeovers=int(cd.get("takeovers", 0)),
                description=cd.get("description", ""),
            )
            claim.acquired_at = float(cd.get("acquiredAt", claim.acquired_at))
            claim.renewed_at = float(cd.get("renewedAt", claim.renewed_at))
            engine.effects._claims.setdefault(claim.key, claim)
    # resume every non-terminal run
    for run in engine.store.all_runs():
        if not run.is_terminal:
            engine._post(("tick", run.key))
    return n
