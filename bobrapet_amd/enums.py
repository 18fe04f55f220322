"""Typed string enumerations used across the bobrapet_amd API surface.

Vocabulary parity with the reference workflow engine's enums
(reference: pkg/enums/enums.go:36-306): the same phase set, terminality
rules, step types, exit classes, stop modes and backoff strategies, so
Story YAML written for the reference validates unchanged here.
"""
from __future__ import annotations

import enum


class Phase(str, enum.Enum):
    """Execution phase of a run resource (reference: pkg/enums/enums.go:36-115)."""

    PENDING = "Pending"
    RUNNING = "Running"
    SUCCEEDED = "Succeeded"
    FAILED = "Failed"
    FINISHED = "Finished"
    CANCELED = "Canceled"
    COMPENSATED = "Compensated"
    PAUSED = "Paused"
    BLOCKED = "Blocked"
    SCHEDULING = "Scheduling"
    TIMEOUT = "Timeout"
    ABORTED = "Aborted"
    SKIPPED = "Skipped"

    @property
    def is_terminal(self) -> bool:
        return self in _TERMINAL_PHASES

    def __str__(self) -> str:  # YAML/JSON friendliness
        return self.value


_TERMINAL_PHASES = frozenset(
    {
        Phase.SUCCEEDED,
        Phase.FAILED,
        Phase.FINISHED,
        Phase.CANCELED,
        Phase.COMPENSATED,
        Phase.TIMEOUT,
        Phase.ABORTED,
        Phase.SKIPPED,
    }
)


class StopMode(str, enum.Enum):
    """How a `stop` primitive terminates the run (reference: enums.go:119-140)."""

    SUCCESS = "success"
    FAILURE = "failure"
    CANCEL = "cancel"

    @property
    def terminal_phase(self) -> Phase:
        return {
            StopMode.SUCCESS: Phase.SUCCEEDED,
            StopMode.FAILURE: Phase.FAILED,
            StopMode.CANCEL: Phase.FINISHED,
        }[self]

    def __str__(self) -> str:
        return self.value


class StepType(str, enum.Enum):
    """Built-in primitive step types (reference: enums.go:142-183)."""

    CONDITION = "condition"
    PARALLEL = "parallel"
    SLEEP = "sleep"
    STOP = "stop"
    WAIT = "wait"
    EXECUTE_STORY = "executeStory"
    GATE = "gate"

    def __str__(self) -> str:
        return self.value


#: Primitives only legal in batch-pattern Stories (reference:
#: internal/webhook/v1alpha1/story_webhook.go:564-576 rejects wait/gate in realtime).
BATCH_ONLY_STEP_TYPES = frozenset({StepType.WAIT, StepType.GATE})


class StoryPattern(str, enum.Enum):
    """Execution pattern of a Story: one-shot batch DAG vs long-lived streaming
    pipeline (reference Story.spec.pattern)."""

    BATCH = "batch"
    STREAMING = "streaming"

    def __str__(self) -> str:
        return self.value


class WorkloadMode(str, enum.Enum):
    """How an engram step is materialized (reference: enums.go:196-218).

    The MI355X engine maps these onto in-process execution models:
    ``job`` = one-shot step on a (gpu, stream) slot; ``deployment`` /
    ``statefulset`` = persistent streaming worker (hipGraph-captured stage).
    """

    JOB = "job"
    DEPLOYMENT = "deployment"
    STATEFULSET = "statefulset"

    @property
    def is_realtime(self) -> bool:
        return self is not WorkloadMode.JOB

    def __str__(self) -> str:
        return self.value


class BackoffStrategy(str, enum.Enum):
    """Retry backoff strategies (reference: enums.go:221-246)."""

    EXPONENTIAL = "exponential"
    LINEAR = "linear"
    CONSTANT = "constant"

    def __str__(self) -> str:
        return self.value


class ValidationStatus(str, enum.Enum):
    """Spec validation states (reference: enums.go:262-283)."""

    VALID = "valid"
    INVALID = "invalid"
    UNKNOWN = "unknown"
    PENDING = "pending"

    def __str__(self) -> str:
        return self.value


class ExitClass(str, enum.Enum):
    """Interpretation of a step's exit code, driving retry logic
    (reference: enums.go:280-306).

    UNKNOWN (infrastructure loss, indeterminate state) is retryable but does
    NOT consume the retry budget.
    """

    SUCCESS = "success"
    RETRY = "retry"
    TERMINAL = "terminal"
    RATE_LIMITED = "rateLimited"
    UNKNOWN = "unknown"

    @property
    def is_retryable(self) -> bool:
        return self in (ExitClass.RETRY, ExitClass.RATE_LIMITED, ExitClass.UNKNOWN)

    @property
    def consumes_retry_budget(self) -> bool:
        return self is not ExitClass.UNKNOWN

    def __str__(self) -> str:
        return self.value


def classify_exit_code(code: int) -> ExitClass:
    """Map a step exit code to its ExitClass.

    Same contract as the reference's classifier
    (internal/controller/runs/steprun_controller.go:4815):
    0=success, 1=retry, 2=terminal, 3=rateLimited, anything else unknown.
    """
    return {
        0: ExitClass.SUCCESS,
        1: ExitClass.RETRY,
        2: ExitClass.TERMINAL,
        3: ExitClass.RATE_LIMITED,
    }.get(code, ExitClass.UNKNOWN)


class TransportMode(str, enum.Enum):
    """How a transport is used in a Story (reference: enums.go:185-194)."""

    HOT = "hot"
    FALLBACK = "fallback"

    def __str__(self) -> str:
        return self.value


class TriggerDecision(str, enum.Enum):
    """Outcome of StoryTrigger admission
    (reference: api/runs/v1alpha1/storytrigger_types.go:51-58)."""

    PENDING = "Pending"
    CREATED = "Created"
    REUSED = "Reused"
    REJECTED = "Rejected"

    def __str__(self) -> str:
        return self.value


class EffectClaimPhase(str, enum.Enum):
    """Lease phases of an exactly-once side-effect claim
    (reference: api/runs/v1alpha1/effectclaim_types.go:34-42)."""

    RESERVED = "Reserved"
    COMPLETED = "Completed"
    RELEASED = "Released"
    ABANDONED = "Abandoned"

    def __str__(self) -> str:
        return self.value


class ErrorType(str, enum.Enum):
    """StructuredError taxonomy
    (reference: api/runs/v1alpha1/structured_error_types.go:22-83)."""

    TIMEOUT = "timeout"
    STORAGE = "storage"
    SERIALIZATION = "serialization"
    VALIDATION = "validation"
    INITIALIZATION = "initialization"
    EXECUTION = "execution"
    UNKNOWN = "unknown"

    def __str__(self) -> str:
        return self.value


class OffloadedDataPolicy(str, enum.Enum):
    """What to do when a template references offloaded (`$storageRef`) data
    (reference: internal/config/controller_config.go:142-144 —
    ignore | inject | block)."""

    IGNORE = "ignore"
    INJECT = "inject"
    BLOCK = "block"

    def __str__(self) -> str:
        return self.value


class CacheMode(str, enum.Enum):
    """Step output cache modes (reference: shared_types.go:266-276)."""

    READ = "read"
    WRITE = "write"
    READ_WRITE = "readWrite"

    @property
    def reads(self) -> bool:
        return self in (CacheMode.READ, CacheMode.READ_WRITE)

    @property
    def writes(self) -> bool:
        return self in (CacheMode.WRITE, CacheMode.READ_WRITE)

    def __str__(self) -> str:
        return self.value


class OnTimeout(str, enum.Enum):
    """Timeout behavior for wait/gate primitives
    (reference: internal/controller/runs/dag.go:1655-1668)."""

    FAIL = "fail"
    SKIP = "skip"

    @property
    def timeout_phase(self) -> Phase:
        return Phase.TIMEOUT if self is OnTimeout.FAIL else Phase.SKIPPED

    def __str__(self) -> str:
        return self.value
