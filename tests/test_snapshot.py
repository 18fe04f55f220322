"""Checkpoint/resume: run-state snapshot save + restore into a fresh engine
(reference durability model — SURVEY §5.4: state IS the checkpoint)."""
import time

import pytest

from bobrapet_amd.engine import EngineConfig, RunEngine
from bobrapet_amd.enums import Phase

RESOURCES = """
kind: EngramTemplate
metadata: {name: echo-tpl}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: echoer}
spec: {templateRef: {name: echo-tpl}}
---
kind: Story
metadata: {name: snap}
spec:
  steps:
    - {name: first, ref: {name: echoer}, with: {v: "{{ inputs.x }}"}}
    - {name: approval, type: gate, needs: [first]}
    - {name: last, ref: {name: echoer}, needs: [approval], with: {v: "{{ steps.first.output.v + 1 }}"}}
  output: {v: "{{ steps.last.output.v }}"}
"""


def test_snapshot_roundtrip_resumes_run(tmp_path):
    path = str(tmp_path / "state.json")
    eng1 = RunEngine(EngineConfig(cpu_workers=2)).start()
    try:
        eng1.apply_yaml(RESOURCES)
        run = eng1.submit_run("default/snap", {"x": 41})
        deadline = time.time() + 5
        while time.time() < deadline:
            r = eng1.store.get_story_run(run.key)
            if r.step_states.get("approval") and r.step_states["approval"].phase == Phase.PAUSED:
                break
            time.sleep(0.01)
        # also a finished run in the store
        done = eng1.run_story("default/snap", {"x": 1}, timeout=5)  # waits at gate
        eng1.approve_gate(done.key, "approval")
        done = eng1.wait(done.key, timeout=5)
        assert done.phase == Phase.SUCCEEDED
        eng1.save_state(path)
    finally:
        eng1.stop()

    # fresh engine: restore, approve the gate, run completes with the
    # already-finished step preserved
    eng2 = RunEngine(EngineConfig(cpu_workers=2)).start()
    try:
        eng2.apply_yaml(RESOURCES)
        n = eng2.load_state(path)
        assert n == 2
        r = eng2.store.get_story_run(run.key)
        assert r.step_states["first"].phase == Phase.SUCCEEDED
        assert r.step_states["first"].output == {"v": 41}
        eng2.approve_gate(run.key, "approval")
        final = eng2.wait(run.key, timeout=10)
        assert final.phase == Phase.SUCCEEDED
        assert final.output == {"v": 42}
        # the terminal run restored terminal
        done2 = eng2.store.get_story_run(done.key)
        assert done2.phase == Phase.SUCCEEDED
    finally:
        eng2.stop()


def test_in_flight_engram_step_reexecutes(tmp_path):
    path = str(tmp_path / "state.json")
    eng1 = RunEngine(EngineConfig(cpu_workers=2)).start()
    try:
        eng1.apply_yaml(RESOURCES)
        eng1.apply_yaml(
            """
kind: EngramTemplate
metadata: {name: sleepy-tpl}
spec: {builtin: sleepy}
---
kind: Engram
metadata: {name: sleeper}
spec: {templateRef: {name: sleepy-tpl}}
---
kind: Story
metadata: {name: snap-slow}
spec:
  steps:
    - {name: slow, ref: {name: sleeper}, with: {seconds: 30}}
"""
        )
        run = eng1.submit_run("default/snap-slow", {})
        time.sleep(0.1)  # step running
        eng1.save_state(path)
    finally:
        eng1.stop()

    eng2 = RunEngine(EngineConfig(cpu_workers=2)).start()
    try:
        eng2.apply_yaml(RESOURCES)
        eng2.apply_yaml(
            """
kind: EngramTemplate
metadata: {name: sleepy-tpl}
spec: {builtin: sleepy}
---
kind: Engram
metadata: {name: sleeper}
spec: {templateRef: {name: sleepy-tpl}}
---
kind: Story
metadata: {name: snap-slow}
spec:
  steps:
    - {name: slow, ref: {name: sleeper}, with: {seconds: 0.01}}
"""
        )
        eng2.load_state(path)
        # the in-flight step was reset to Pending and re-executes
        final = eng2.wait(run.key, timeout=10)
        assert final.phase == Phase.SUCCEEDED
    finally:
        eng2.stop()


def test_spoofed_storage_ref_rejected():
    eng = RunEngine(EngineConfig(cpu_workers=1)).start()
    try:
        eng.apply_yaml(RESOURCES)
        with pytest.raises(ValueError, match="storageRef"):
            eng.submit_run(
                "default/snap",
                {"x": 1, "sneaky": {"$storageRef": {"key": "outputs/x", "kind": "json"}}},
            )
    finally:
        eng.stop()


def test_effect_ledger_and_exit_class_persist(tmp_path):
    """completed EffectClaims must survive a restore (else
    re-executed steps repeat already-performed side effects), and failed
    steps keep their exit class."""
    from bobrapet_amd.enums import EffectClaimPhase, ExitClass
    from bobrapet_amd.engine import snapshot

    path = str(tmp_path / "state.json")
    eng1 = RunEngine(EngineConfig(cpu_workers=1)).start()
    try:
        eng1.apply_yaml(RESOURCES)
        done = eng1.submit_run("default/snap", {"x": 1})
        time.sleep(0.2)
        claim, fresh = eng1.effects.acquire("run/step/send-email", "worker-1")
        assert fresh
        eng1.effects.complete("run/step/send-email", "worker-1")
        state = snapshot.dump_state(eng1)
        assert any(c["key"] == "run/step/send-email" for c in state["effectClaims"])
        eng1.save_state(path)
    finally:
        eng1.stop()

    eng2 = RunEngine(EngineConfig(cpu_workers=1)).start()
    try:
        eng2.apply_yaml(RESOURCES)
        eng2.load_state(path)
        restored = eng2.effects.get("run/step/send-email")
        assert restored is not None
        assert restored.phase == EffectClaimPhase.COMPLETED
        # a re-executing step must observe fresh=False (skip the effect)
        _, fresh = eng2.effects.acquire("run/step/send-email", "worker-2")
        assert not fresh
    finally:
        eng2.stop()

    # exit-class restore fallback: classify from exit code when absent
    from bobrapet_amd.engine.snapshot import _restore_exit_class

    assert _restore_exit_class({"exitClass": "terminal"}) == ExitClass.TERMINAL
    assert _restore_exit_class({"exitCode": 3}) == ExitClass.RATE_LIMITED
    assert _restore_exit_class({"exitCode": None}) is None


class TestAutoCheckpoint:
    """Periodic durability snapshots (EngineConfig.checkpoint_path) — the
    opt-in analog of the reference's always-durable etcd state."""

    def test_periodic_checkpoint_and_restore(self, tmp_path):
        import os
        import time

        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.enums import Phase

        ckpt = str(tmp_path / "state.json")
        eng = RunEngine(
            EngineConfig(cpu_workers=2, checkpoint_path=ckpt, checkpoint_interval_seconds=0.1)
        ).start()
        story = """
kind: Story
metadata: {name: s}
spec:
  steps: [{name: a, type: sleep, with: {duration: 0ms}}]
"""
        try:
            eng.apply_yaml(story)
            run = eng.run_story("default/s", {}, timeout=10)
            assert run.phase == Phase.SUCCEEDED
            deadline = time.time() + 5
            while not os.path.exists(ckpt) and time.time() < deadline:
                time.sleep(0.05)
            assert os.path.exists(ckpt)
            assert eng.metrics.counter_value("checkpoints_total") >= 1
        finally:
            eng.stop()
        eng2 = RunEngine(EngineConfig(cpu_workers=2)).start()
        try:
            eng2.apply_yaml(story)
            assert eng2.load_state(ckpt) >= 1
            restored = eng2.store.try_get_story_run(run.key)
            assert restored is not None and restored.phase == Phase.SUCCEEDED
        finally:
            eng2.stop()
