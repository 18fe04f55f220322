"""Enum vocabulary + terminality + exit-class semantics (reference parity:
pkg/enums/enums.go)."""
from bobrapet_amd.enums import (
    ExitClass,
    OnTimeout,
    Phase,
    StepType,
    StopMode,
    classify_exit_code,
)


def test_phase_values_cover_reference_vocabulary():
    expected = {
        "Pending",
        "Running",
        "Succeeded",
        "Failed",
        "Finished",
        "Canceled",
        "Compensated",
        "Paused",
        "Blocked",
        "Scheduling",
        "Timeout",
        "Aborted",
        "Skipped",
    }
    assert {p.value for p in Phase} == expected


def test_terminal_phases():
    terminal = {p for p in Phase if p.is_terminal}
    assert terminal == {
        Phase.SUCCEEDED,
        Phase.FAILED,
        Phase.FINISHED,
        Phase.CANCELED,
        Phase.COMPENSATED,
        Phase.TIMEOUT,
        Phase.ABORTED,
        Phase.SKIPPED,
    }
    assert not Phase.RUNNING.is_terminal
    assert not Phase.PAUSED.is_terminal
    assert not Phase.BLOCKED.is_terminal


def test_exit_code_classification():
    assert classify_exit_code(0) == ExitClass.SUCCESS
    assert classify_exit_code(1) == ExitClass.RETRY
    assert classify_exit_code(2) == ExitClass.TERMINAL
    assert classify_exit_code(3) == ExitClass.RATE_LIMITED
    assert classify_exit_code(137) == ExitClass.UNKNOWN
    assert classify_exit_code(-1) == ExitClass.UNKNOWN


def test_unknown_exit_class_does_not_burn_retry_budget():
    assert ExitClass.UNKNOWN.is_retryable
    assert not ExitClass.UNKNOWN.consumes_retry_budget
    assert ExitClass.RETRY.consumes_retry_budget
    assert not ExitClass.TERMINAL.is_retryable


def test_stop_mode_phases():
    assert StopMode.SUCCESS.terminal_phase == Phase.SUCCEEDED
    assert StopMode.FAILURE.terminal_phase == Phase.FAILED
    assert StopMode.CANCEL.terminal_phase == Phase.FINISHED


def test_step_types():
    assert {s.value for s in StepType} == {
        "condition",
        "parallel",
        "sleep",
        "stop",
        "wait",
        "executeStory",
        "gate",
    }


def test_on_timeout_phases():
    assert OnTimeout.FAIL.timeout_phase == Phase.TIMEOUT
    assert OnTimeout.SKIP.timeout_phase == Phase.SKIPPED
