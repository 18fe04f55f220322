"""Engram execution contract.

Role parity with the reference's SDK env contract (SURVEY.md §2.5 —
the BUBU_* environment the controller passes to every worker container:
story/run/step identity, resolved input, step config, execution mode,
storage config, timeout, debug).  Here the engram runs in-process: the
EngramContext object IS the contract, and instead of patching
StepRun.status over the apiserver, the engram returns an EngramResult
(or raises) and may emit signals/effects through the context.
"""
from __future__ import annotations

import typing as _t
from dataclasses import dataclass, field

from ..enums import ErrorType

if _t.TYPE_CHECKING:
    from ..engine.records import EffectRecord, SignalEvent, StructuredError


class EngramFailure(Exception):
    """Raise from an engram to fail the step with a structured error and a
    chosen exit code (0=success, 1=retry, 2=terminal, 3=rateLimited)."""

    def __init__(
        self,
        message: str,
        exit_code: int = 1,
        error_type: ErrorType = ErrorType.EXECUTION,
        details: _t.Optional[dict] = None,
    ):
        super().__init__(message)
        self.exit_code = exit_code
        self.error_type = error_type
        self.details = details

    def to_structured(self) -> "StructuredError":
        from ..engine.records import StructuredError

        return StructuredError(
            type=self.error_type,
            message=str(self),
            retryable=self.exit_code in (1, 3),
            details=self.details,
        )


@dataclass
class EngramResult:
    output: _t.Any = None
    exit_code: int = 0
    logs: _t.List[str] = field(default_factory=list)


@dataclass
class EngramContext:
    """What a step execution sees (the in-process BUBU_* contract)."""

    story_name: str = ""
    story_version: str = ""
    story_run: str = ""
    step_name: str = ""
    step_run: str = ""
    namespace: str = "default"
    input: _t.Any = None  # resolved step input (template-evaluated `with`)
    config: _t.Any = None  # engram instance `with` config
    runtime: _t.Any = None  # raw per-item templates (Step.runtime, unresolved)
    execution_mode: str = "job"
    max_inline_size: int = 8 << 10
    timeout_seconds: _t.Optional[float] = None
    max_recursion_depth: int = 8
    debug: bool = False
    device: _t.Optional[int] = None  # GPU ordinal this step is placed on
    stream: _t.Any = None  # torch.cuda.Stream bound to the slot
    storage: _t.Any = None  # StorageManager
    trace_id: str = ""
    # collected during execution --------------------------------------------
    signals: _t.List["SignalEvent"] = field(default_factory=list)
    effects: _t.List["EffectRecord"] = field(default_factory=list)
    logs: _t.List[str] = field(default_factory=list)
    cancel_check: _t.Optional[_t.Callable[[], bool]] = None
    effect_guard: _t.Optional[_t.Callable[[str, str], bool]] = None
    _signal_seq: int = 0

    def log(self, message: str) -> None:
        self.logs.append(message)

    def emit_signal(self, name: str, payload=None) -> None:
        """Ordered signal (reference: steprun_types.go SignalEvent; merged
        into prior outputs seq-ordered, dag.go:2289-2481)."""
        from ..engine.records import SignalEvent

        self._signal_seq += 1
        self.signals.append(SignalEvent(seq=self._signal_seq, name=name, payload=payload))

    def record_effect(self, idempotency_key: str, description: str = "") -> bool:
        """Exactly-once side-effect ledger entry. Returns False when the
        effect was already claimed (EffectClaim held by an earlier attempt)
        and the side effect must NOT run again."""
        if self.effect_guard is not None:
            fresh = self.effect_guard(idempotency_key, description)
            if not fresh:
                return False
        from ..engine.records import EffectRecord

        self.effects.append(EffectRecord(idempotency_key=idempotency_key, description=description))
        return True

    @property
    def canceled(self) -> bool:
        return bool(self.cancel_check and self.cancel_check())


class Engram:
    """Base class for built-in engram implementations.

    The reference's engrams are external container images implementing the
    SDK contract; here the library is in-process, with
    GPU engrams launching HIP kernels on their placed (device, stream)."""

    #: registry name; EngramTemplate.builtin (or image) resolves to this
    name: str = ""
    #: set False for CPU-only engrams
    wants_gpu: bool = False

    def run(self, ctx: EngramContext) -> _t.Union[EngramResult, _t.Any]:
        raise NotImplementedError

    def warmup(self, ctx: EngramContext) -> None:
        """Optional: preload weights / compile before first use."""


class ImpulseHandler:
    """Base class for built-in impulse (trigger) implementations."""

    name: str = ""

    def start(self, emit: _t.Callable[[dict], _t.Any]) -> None:
        raise NotImplementedError

    def stop(self) -> None:
        pass
