"""Run-state records: StoryRun, StepRun, StepState, conditions.

Role parity with the reference's run CRDs
(reference: api/runs/v1alpha1/storyrun_types.go:54-298,
steprun_types.go:61-374, structured_error_types.go:22-83,
trace_types.go:19-30).  These are in-process records held by the run-state
store (engine/store.py) instead of etcd; every field that the reference
persists for checkpoint/resume is here.
"""
from __future__ import annotations

import hashlib
import itertools
import threading
import time
import typing as _t
from dataclasses import dataclass, field

from ..enums import ErrorType, ExitClass, Phase

JSON = _t.Any

_seq = itertools.count(1)
_seq_lock = threading.Lock()


def _next_seq() -> int:
    with _seq_lock:
        return next(_seq)


def monotonic_now() -> float:
    return time.time()


@dataclass
class Condition:
    """Status condition (reference: pkg/conditions/conditions.go)."""

    type: str
    status: bool
    reason: str = ""
    message: str = ""
    last_transition: float = field(default_factory=monotonic_now)


class ConditionManager:
    """Tracks Ready/Progressing/Degraded-style conditions on a record."""

    def __init__(self):
        self._conditions: _t.Dict[str, Condition] = {}

    def set(self, ctype: str, status: bool, reason: str = "", message: str = "") -> None:
        cur = self._conditions.get(ctype)
        if cur and cur.status == status and cur.reason == reason:
            cur.message = message
            return
        self._conditions[ctype] = Condition(ctype, status, reason, message)

    def get(self, ctype: str) -> _t.Optional[Condition]:
        return self._conditions.get(ctype)

    def as_list(self) -> _t.List[Condition]:
        return sorted(self._conditions.values(), key=lambda c: c.type)


@dataclass
class StructuredError:
    """Versioned error contract (reference: structured_error_types.go)."""

    type: ErrorType = ErrorType.UNKNOWN
    message: str = ""
    retryable: bool = False
    version: str = "v1"
    details: _t.Optional[dict] = None

    def to_dict(self) -> dict:
        out = {
            "version": self.version,
            "type": str(self.type),
            "message": self.message,
            "retryable": self.retryable,
        }
        if self.details:
            out["details"] = self.details
        return out


@dataclass
class TraceInfo:
    """Trace context persisted in run status (reference: trace_types.go)."""

    trace_id: str = ""
    span_id: str = ""


@dataclass
class SignalEvent:
    """Ordered signal emitted by a step (reference: steprun_types.go SignalEvent);
    merged into prior outputs in sequence order (dag.go:2289-2481)."""

    seq: int
    name: str
    payload: JSON = None
    ts: float = field(default_factory=monotonic_now)


@dataclass
class EffectRecord:
    """Append-only external side-effect ledger entry
    (reference: steprun_types.go:298-305)."""

    idempotency_key: str
    description: str = ""
    ts: float = field(default_factory=monotonic_now)


@dataclass
class StepState:
    """Per-step state mirrored into StoryRun.status.stepStates
    (reference: storyrun_types.go StepState)."""

    name: str
    phase: Phase = Phase.PENDING
    output: JSON = None
    error: _t.Optional[StructuredError] = None
    retries: int = 0
    started_at: _t.Optional[float] = None
    finished_at: _t.Optional[float] = None
    message: str = ""

    def merge_from(self, other: "StepState") -> None:
        """Terminal-phase-wins merge discipline (reference: step_state.go:36,
        persistMergedStates dag.go:780-792): never overwrite a terminal phase
        with a non-terminal one."""
        if self.phase.is_terminal and not other.phase.is_terminal:
            return
        self.phase = other.phase
        if other.output is not None:
            self.output = other.output
        if other.error is not None:
            self.error = other.error
        self.retries = max(self.retries, other.retries)
        self.started_at = self.started_at or other.started_at
        self.finished_at = other.finished_at or self.finished_at
        if other.message:
            self.message = other.message


@dataclass
class GateStatus:
    """Manual-approval gate state (reference: storyrun_types.go GateStatus)."""

    step: str
    state: str = "Pending"  # Pending | Approved | Rejected
    decided_by: str = ""
    decided_at: _t.Optional[float] = None


@dataclass
class StoryRun:
    """Execution instance of a Story (reference: storyrun_types.go:54-298)."""

    name: str
    namespace: str = "default"
    story_name: str = ""
    story_namespace: str = "default"
    inputs: JSON = None
    # status ---------------------------------------------------------------
    phase: Phase = Phase.PENDING
    exec_phase: str = "main"  # main | compensation | finally (dag.go:482-511)
    failure_cause: _t.Optional[str] = None
    step_states: _t.Dict[str, StepState] = field(default_factory=dict)
    gates: _t.Dict[str, GateStatus] = field(default_factory=dict)
    primitive_children: _t.Dict[str, _t.List[str]] = field(default_factory=dict)
    # policy=block: delegated template results keyed by step name
    # (reference: materialize.go readMaterializeResult 304-318)
    materialized: _t.Dict[str, _t.Any] = field(default_factory=dict)
    trigger_tokens: _t.List[str] = field(default_factory=list)
    timers: _t.Dict[str, float] = field(default_factory=dict)  # durable deadlines
    output: JSON = None
    error: _t.Optional[StructuredError] = None
    conditions: ConditionManager = field(default_factory=ConditionManager)
    trace: TraceInfo = field(default_factory=TraceInfo)
    cancel_requested: bool = False
    cancel_observed_at: _t.Optional[float] = None
    degraded: bool = False
    labels: _t.Dict[str, str] = field(default_factory=dict)
    annotations: _t.Dict[str, str] = field(default_factory=dict)
    parent_run: _t.Optional[str] = None  # parent StoryRun name (executeStory)
    parent_step: _t.Optional[str] = None
    recursion_depth: int = 0
    story_generation: int = 0
    input_hash: str = ""
    queue: str = ""
    priority: int = 0
    created_at: float = field(default_factory=monotonic_now)
    started_at: _t.Optional[float] = None
    finished_at: _t.Optional[float] = None
    children_cleaned_at: _t.Optional[float] = None
    redrive_count: int = 0
    resource_version: int = 0

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.name}"

    @property
    def is_terminal(self) -> bool:
        return self.phase.is_terminal

    def step_state(self, name: str) -> StepState:
        st = self.step_states.get(name)
        if st is None:
            st = StepState(name=name)
            self.step_states[name] = st
        return st

    def active_steps(self) -> _t.List[str]:
        return [n for n, s in self.step_states.items() if s.phase == Phase.RUNNING]

    def completed_steps(self) -> _t.List[str]:
        return [n for n, s in self.step_states.items() if s.phase.is_terminal]


@dataclass
class StepRunSpec:
    """Resolved spec of one step execution (reference: steprun_types.go:77-135)."""

    story_run: str = ""
    step_name: str = ""
    engram: _t.Optional[str] = None  # ns/name of the Engram
    input: JSON = None
    config: JSON = None  # merged engram.with ⊕ step.with, template-resolved
    runtime: JSON = None  # per-packet/per-item templates, passed RAW (Step.runtime)
    idempotency_key: _t.Optional[str] = None
    timeout_seconds: _t.Optional[float] = None
    template_generation: int = 0
    placement_gpu: _t.Optional[int] = None
    mode: str = "job"


@dataclass
class StepRunStatus:
    """Execution status (reference: steprun_types.go:196-306)."""

    phase: Phase = Phase.PENDING
    output: JSON = None
    error: _t.Optional[StructuredError] = None
    exit_code: _t.Optional[int] = None
    exit_class: _t.Optional[ExitClass] = None
    retries: int = 0
    next_retry_at: _t.Optional[float] = None
    signals: _t.List[SignalEvent] = field(default_factory=list)
    effects: _t.List[EffectRecord] = field(default_factory=list)
    logs: _t.List[str] = field(default_factory=list)
    cache_hit: bool = False
    # time of the most recent non-empty output write (cache hits included,
    # EMPTY outputs excluded — reference: applyCacheHit / LastOutputAt,
    # steprun_last_output_at_test.go behavior)
    last_output_at: _t.Optional[float] = None
    started_at: _t.Optional[float] = None
    finished_at: _t.Optional[float] = None
    message: str = ""
    worker: str = ""  # which executor slot ran it (gpu:stream)


@dataclass
class StepRun:
    """Atomic step execution record (reference: steprun_types.go:61-374)."""

    name: str
    namespace: str = "default"
    spec: StepRunSpec = field(default_factory=StepRunSpec)
    status: StepRunStatus = field(default_factory=StepRunStatus)
    created_at: float = field(default_factory=monotonic_now)
    resource_version: int = 0
    cancel_requested: bool = False

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.name}"

    @property
    def is_terminal(self) -> bool:
        return self.status.phase.is_terminal


def compose_name(parent: str, child: str, max_len: int = 63) -> str:
    """Deterministic child naming (reference: pkg/kubeutil/naming.go
    ComposeName): parent-child, hash-suffixed when truncation is needed, so
    create-or-adopt idempotency holds everywhere."""
    base = f"{parent}-{child}"
    if len(base) <= max_len:
        return base
    digest = hashlib.sha256(base.encode()).hexdigest()[:8]
    keep = max_len - len(digest) - 1
    return f"{base[:keep]}-{digest}"


def canonical_json(value) -> str:
    import json

    return json.dumps(value, sort_keys=True, separators=(",", ":"), default=str)


def input_hash(inputs) -> str:
    """sha256 over canonical inputs (reference:
    storytrigger_controller.go:225 resolveTriggerInputHash)."""
    return hashlib.sha256(canonical_json(inputs).encode()).hexdigest()


def derive_story_run_name(namespace: str, story: str, token: str) -> str:
    """Deterministic StoryRun name from the trigger identity
    (reference: pkg/runs/identity/storyrun_trigger.go:35-67) ⇒ dedupe by name."""
    digest = hashlib.sha256(f"{namespace}/{story}/{token}".encode()).hexdigest()[:10]
    base = f"{story}-{digest}"
    return base[:63]
