"""bobraccel native DAG core tests: the C++ engine must match the Python
engine's state-machine semantics (same stories, same outcomes)."""
import time

import pytest

from bobrapet_amd.engine import EngineConfig, RunEngine

native_mod = pytest.importorskip("bobrapet_amd._core")

from bobrapet_amd.runtime.native import (  # noqa: E402
    NativeCompileError,
    NativeRunner,
    story_supported,
)


RESOURCES = """
kind: EngramTemplate
metadata: {name: echo-tpl}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: echoer}
spec: {templateRef: {name: echo-tpl}}
---
kind: EngramTemplate
metadata: {name: fail-tpl}
spec: {builtin: fail}
---
kind: Engram
metadata: {name: failer}
spec: {templateRef: {name: fail-tpl}}
"""


@pytest.fixture()
def rig():
    eng = RunEngine(EngineConfig(cpu_workers=4)).start()
    eng.apply_yaml(RESOURCES)
    import bobrapet_amd.engrams.registry as reg

    reg.reset_instances()
    nr = NativeRunner.from_run_engine(eng)
    yield eng, nr
    nr.stop()
    eng.stop()


def _apply(eng, yaml_text):
    return eng.apply_yaml(yaml_text)[0]


class TestNativeExpressions:
    def test_eval_expression(self):
        from bobrapet_amd.templating import parse_expression

        ev = native_mod.eval_expression
        scope = {"inputs": {"x": 3, "items": [1, 2, 3]}, "steps": {}}
        assert ev(parse_expression("inputs.x + 1"), scope) == 4
        assert ev(parse_expression("inputs.items[2] * 2"), scope) == 6
        assert ev(parse_expression("size(inputs.items)"), scope) == 3
        assert ev(parse_expression("inputs.x > 2 && inputs.x < 5"), scope) is True
        assert ev(parse_expression("inputs.missing"), scope) is None
        assert ev(parse_expression("'a' + 'b'"), scope) == "ab"
        assert ev(parse_expression("inputs.x == 3 ? 'yes' : 'no'"), scope) == "yes"
        assert ev(parse_expression("2 in inputs.items"), scope) is True
        assert ev(parse_expression("default(inputs.missing, 9)"), scope) == 9
        assert ev(parse_expression("inputs.items.size()"), scope) == 3

    def test_python_and_native_evaluators_agree(self):
        from bobrapet_amd.templating import Evaluator, parse_expression

        pyev = Evaluator()
        scope = {
            "inputs": {"a": 2, "b": "xy", "l": [1, 2], "m": {"k": 5}},
            "steps": {"s-1": {"output": {"v": 7}, "phase": "Succeeded"}},
        }
        exprs = [
            "inputs.a * 3 - 1",
            "inputs.b + '!'",
            "steps.s_1.output.v >= 7",
            "inputs.m.k % 3",
            "size(inputs.l) == 2 || false",
            "!has(inputs, 'zz')",
            "min(inputs.l)",
            "upper(inputs.b)",
            "join(inputs.l, '-')",
        ]
        for src in exprs:
            py = pyev.resolve_string("{{ " + src + " }}", scope)
            nat = native_mod.eval_expression(parse_expression(src), scope)
            assert py == nat, (src, py, nat)


class TestNativeRuns:
    def test_basic_dag(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n1}
spec:
  steps:
    - {name: pause, type: sleep, with: {duration: 1ms}}
    - name: check
      type: condition
      needs: [pause]
      with: {expression: "{{ steps.pause.phase == 'Succeeded' }}"}
    - {name: work, ref: {name: echoer}, needs: [check], with: {v: "{{ inputs.x * 2 }}"}}
  output: {v: "{{ steps.work.output.v }}"}
""",
        )
        st = nr.run_story("default/n1", {"x": 21}, timeout=10)
        assert st["phase"] == "Succeeded"
        assert st["output"] == {"v": 42}

    def test_if_and_requires_skip(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n2}
spec:
  steps:
    - {name: a, ref: {name: echoer}, with: {v: 1}}
    - {name: b, ref: {name: echoer}, if: "{{ steps.a.output.v > 5 }}", with: {v: 2}}
    - {name: c, ref: {name: echoer}, requires: ["steps.a.output.missing"], with: {v: 3}}
    - {name: d, ref: {name: echoer}, needs: [b], with: {v: 4}}
""",
        )
        st = nr.run_story("default/n2", {}, timeout=10)
        assert st["phase"] == "Succeeded"
        assert st["steps"]["b"]["phase"] == "Skipped"
        assert st["steps"]["c"]["phase"] == "Skipped"
        assert st["steps"]["d"]["phase"] == "Skipped"  # cascade

    def test_failure_fail_fast(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n3}
spec:
  steps:
    - {name: boom, ref: {name: failer}, with: {exitCode: 2}}
    - {name: after, ref: {name: echoer}, needs: [boom], with: {v: 1}}
""",
        )
        st = nr.run_story("default/n3", {}, timeout=10)
        assert st["phase"] == "Failed"
        assert st["steps"]["after"]["phase"] == "Skipped"

    def test_allow_failure(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n4}
spec:
  steps:
    - {name: boom, ref: {name: failer}, allowFailure: true, with: {exitCode: 2}}
    - {name: after, ref: {name: echoer}, needs: [boom], with: {v: 1}}
""",
        )
        st = nr.run_story("default/n4", {}, timeout=10)
        assert st["phase"] == "Succeeded"
        assert st["steps"]["after"]["phase"] == "Succeeded"

    def test_retry_until_success(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n5}
spec:
  steps:
    - name: flaky
      ref: {name: failer}
      with: {succeedAfter: 2, exitCode: 1}
      execution:
        retry: {maxRetries: 5, delay: 5ms, jitter: 0, backoff: constant}
""",
        )
        st = nr.run_story("default/n5", {}, timeout=10)
        assert st["phase"] == "Succeeded", st
        assert st["steps"]["flaky"]["retries"] == 2

    def test_stop_primitive(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n6}
spec:
  steps:
    - {name: halt, type: stop, with: {mode: failure}}
    - {name: never, ref: {name: echoer}, needs: [halt], with: {v: 1}}
""",
        )
        st = nr.run_story("default/n6", {}, timeout=10)
        assert st["phase"] == "Failed"
        assert st["steps"]["never"]["phase"] == "Skipped"

    def test_wait_and_timeout(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n7}
spec:
  steps:
    - name: watch
      type: wait
      with: {until: "{{ false }}", pollInterval: 5ms, timeout: 40ms, onTimeout: skip}
""",
        )
        st = nr.run_story("default/n7", {}, timeout=10)
        assert st["phase"] == "Succeeded"
        assert st["steps"]["watch"]["phase"] == "Skipped"

    def test_wait_until_dep(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n8}
spec:
  steps:
    - {name: slow, type: sleep, with: {duration: 30ms}}
    - name: watch
      type: wait
      with: {until: "{{ steps.slow.phase == 'Succeeded' }}", pollInterval: 5ms, timeout: 5s}
""",
        )
        st = nr.run_story("default/n8", {}, timeout=10)
        assert st["phase"] == "Succeeded"
        assert st["steps"]["watch"]["phase"] == "Succeeded"

    def test_wait_event_driven_wakeup(self, rig):
        """A wait over a completing dependency resolves on the dependency's
        own tick, not on the next poll timer: with a 5 s pollInterval the run
        still finishes in milliseconds (engine.cpp sync_primitives Wait)."""
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n8b}
spec:
  steps:
    - {name: slow, type: sleep, with: {duration: 30ms}}
    - name: watch
      type: wait
      with: {until: "{{ steps.slow.phase == 'Succeeded' }}", pollInterval: 5s, timeout: 30s}
""",
        )
        t0 = time.monotonic()
        st = nr.run_story("default/n8b", {}, timeout=10)
        elapsed = time.monotonic() - t0
        assert st["phase"] == "Succeeded"
        assert st["steps"]["watch"]["phase"] == "Succeeded"
        assert elapsed < 2.0, f"wait waited for the poll timer: {elapsed:.3f}s"

    def test_wait_timeout_with_event_wakeup(self, rig):
        """The event-driven re-check must not defeat wait timeouts: a
        condition that never becomes true still times out (tag-6 timer)
        even though other steps complete and tick the run."""
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n8c}
spec:
  steps:
    - {name: busy, type: sleep, with: {duration: 20ms}}
    - name: watch
      type: wait
      with: {until: "{{ false }}", pollInterval: 5s, timeout: 60ms, onTimeout: skip}
""",
        )
        st = nr.run_story("default/n8c", {}, timeout=10)
        assert st["phase"] == "Succeeded"
        assert st["steps"]["watch"]["phase"] == "Skipped"
        assert st["steps"]["busy"]["phase"] == "Succeeded"

    def test_gate_approved(self, rig):
        eng, nr = rig
        story = _apply(
            eng,
            """
kind: Story
metadata: {name: n9}
spec:
  steps:
    - {name: approval, type: gate}
    - {name: after, ref: {name: echoer}, needs: [approval], with: {v: 1}}
""",
        )
        rid = nr.submit(story, {})
        time.sleep(0.05)
        assert nr.engine.run_status(rid)["steps"]["approval"]["phase"] == "Paused"
        nr.decide_gate(rid, 0, True)
        st = nr.wait(rid, timeout=10)
        assert st["phase"] == "Succeeded"

    def test_parallel_branches(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n10}
spec:
  steps:
    - name: fan
      type: parallel
      with:
        steps:
          - {name: b1, ref: {name: echoer}, with: {v: 1}}
          - {name: b2, ref: {name: echoer}, with: {v: 2}}
          - {name: b3, type: condition, with: {expression: "true"}}
    - name: use
      ref: {name: echoer}
      needs: [fan]
      with: {got: "{{ steps.fan.output.branches.b2.v }}"}
""",
        )
        st = nr.run_story("default/n10", {}, timeout=10)
        assert st["phase"] == "Succeeded", st
        assert st["steps"]["use"]["output"] == {"got": 2}

    def test_execute_story_nested(self, rig):
        eng, nr = rig
        eng.apply_yaml(
            """
kind: Story
metadata: {name: inner-n}
spec:
  steps:
    - {name: calc, ref: {name: echoer}, with: {v: "{{ inputs.a + 1 }}"}}
  output: {v: "{{ steps.calc.output.v }}"}
---
kind: Story
metadata: {name: outer-n}
spec:
  steps:
    - name: sub
      type: executeStory
      with: {storyRef: inner-n, with: {a: 9}}
  output: {v: "{{ steps.sub.output.output.v }}"}
"""
        )
        st = nr.run_story("default/outer-n", {}, timeout=10)
        assert st["phase"] == "Succeeded", st
        assert st["output"] == {"v": 10}

    def test_cancel(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n11}
spec:
  steps:
    - {name: slow, type: sleep, with: {duration: 10s}}
""",
        )
        rid = nr.submit("default/n11", {})
        time.sleep(0.02)
        nr.cancel(rid)
        st = nr.wait(rid, timeout=5)
        assert st["phase"] == "Canceled"

    def test_story_timeout(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n12}
spec:
  policy:
    timeouts: {story: 50ms}
  steps:
    - {name: slow, type: sleep, with: {duration: 10s}}
""",
        )
        st = nr.run_story("default/n12", {}, timeout=10)
        assert st["phase"] == "Timeout"

    def test_step_timeout(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: n13}
spec:
  steps:
    - name: slow
      ref: {name: echoer}
      with: {v: 1}
      execution: {timeout: 1ms}
""",
        )
        # echo finishes fast; use the sleepy engram instead
        eng.apply_yaml(
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
metadata: {name: n13b}
spec:
  steps:
    - name: slow
      ref: {name: sleeper}
      with: {seconds: 5}
      execution: {timeout: 50ms}
"""
        )
        st = nr.run_story("default/n13b", {}, timeout=10)
        assert st["phase"] == "Failed"
        assert st["steps"]["slow"]["phase"] == "Timeout"


class TestSupportGating:
    def test_streaming_not_supported(self, rig):
        eng, nr = rig
        from bobrapet_amd.specs import load_yaml

        (story,) = load_yaml(
            """
kind: Story
metadata: {name: s}
spec:
  pattern: streaming
  steps:
    - {name: a, ref: {name: x}}
"""
        )
        assert story_supported(story) is not None

    def test_compensations_now_supported(self, rig):
        from bobrapet_amd.specs import load_yaml

        (story,) = load_yaml(
            """
kind: Story
metadata: {name: s}
spec:
  steps:
    - {name: a, ref: {name: x}}
  compensations:
    - {name: undo, ref: {name: x}}
"""
        )
        assert story_supported(story) is None  # 3-phase machine is native now


class TestThroughput:
    def test_native_faster_than_python(self, rig):
        eng, nr = rig
        _apply(
            eng,
            """
kind: Story
metadata: {name: perf}
spec:
  steps:
    - {name: pause, type: sleep, with: {duration: 0ms}}
    - name: check
      type: condition
      needs: [pause]
      with: {expression: "{{ steps.pause.phase == 'Succeeded' }}"}
""",
        )
        pid = nr.compile(eng.registry.story("perf"))
        n = 300
        t0 = time.monotonic()
        ids = [nr.engine.submit(pid, {"i": i}) for i in range(n)]
        for rid in ids:
            assert nr.engine.wait(rid, 30.0)
        native_dt = time.monotonic() - t0
        for rid in ids:
            nr.engine.gc_run(rid)
        assert n / native_dt > 2000, f"native only {n/native_dt:.0f} runs/s"


class TestNativeThreePhase:
    """bobraccel 3-phase machine: main → compensation (on failure) →
    finally (always) — previously Python-engine-only."""

    YAML = """
kind: EngramTemplate
metadata: {name: echo}
spec: {builtin: echo}
---
kind: EngramTemplate
metadata: {name: fail}
spec: {builtin: fail}
---
kind: Engram
metadata: {name: e}
spec: {templateRef: {name: echo}}
---
kind: Engram
metadata: {name: f}
spec: {templateRef: {name: fail}}
---
kind: Story
metadata: {name: comp}
spec:
  steps:
    - {name: work, ref: {name: f}, with: {succeedAfter: 99}}
    - {name: after, ref: {name: e}, needs: [work], with: {x: 1}}
  compensations:
    - {name: undo, ref: {name: e}, with: {undid: "{{ steps.work.phase }}"}}
  finally:
    - {name: report, ref: {name: e}, with: {done: true}}
---
kind: Story
metadata: {name: ok}
spec:
  steps:
    - {name: work, ref: {name: e}, with: {v: 7}}
  compensations:
    - {name: undo, ref: {name: e}, with: {nope: 1}}
  finally:
    - {name: report, ref: {name: e}, with: {done: true}}
"""

    def _runner(self):
        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.runtime.native import NativeRunner

        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        eng.apply_yaml(self.YAML)
        return eng, NativeRunner.from_run_engine(eng)

    def test_failure_compensates_then_finally(self):
        eng, nr = self._runner()
        try:
            res = nr.run_story("default/comp", {}, timeout=30)
            assert res["phase"] == "Compensated", res
            ph = {k: v["phase"] for k, v in res["steps"].items()}
            assert ph == {
                "work": "Failed", "after": "Skipped",
                "undo": "Succeeded", "report": "Succeeded",
            }
            # compensation scope sees main-step state
            assert res["steps"]["undo"]["output"]["undid"] == "Failed"
        finally:
            eng.stop()

    def test_success_skips_compensations_runs_finally(self):
        eng, nr = self._runner()
        try:
            res = nr.run_story("default/ok", {}, timeout=30)
            assert res["phase"] == "Succeeded", res
            # never-launched compensations have no state surface (reference
            # parity: unreconciled steps don't appear in stepStates)
            assert "undo" not in res["steps"]
            assert res["steps"]["report"]["phase"] == "Succeeded"
        finally:
            eng.stop()

    def test_parity_with_python_engine(self):
        """Same stories through the Python engine produce the same phases."""
        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.enums import Phase

        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        try:
            eng.apply_yaml(self.YAML)
            run = eng.run_story("default/comp", {}, timeout=30)
            assert run.phase == Phase.COMPENSATED
            assert run.step_states["undo"].phase == Phase.SUCCEEDED
            assert run.step_states["report"].phase == Phase.SUCCEEDED
        finally:
            eng.stop()


class TestNativePostExecution:
    """postExecution checks evaluated by the C++ expression VM on engram
    completion (reference: steprun_controller.go:2050-2124)."""

    def test_post_execution_native(self):
        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.runtime.native import NativeRunner, story_supported

        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        try:
            eng.apply_yaml(
                """
kind: EngramTemplate
metadata: {name: echo}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: e}
spec: {templateRef: {name: echo}}
---
kind: Story
metadata: {name: pe}
spec:
  steps:
    - name: a
      ref: {name: e}
      with: {v: 3}
      postExecution: {condition: "{{ output.v > 5 }}", failureMessage: "v too small"}
"""
            )
            assert story_supported(eng.registry.story("pe", "default")) is None
            nr = NativeRunner.from_run_engine(eng)
            res = nr.run_story("default/pe", {}, timeout=30)
            assert res["phase"] == "Failed"
            assert "v too small" in res["steps"]["a"]["error"]
        finally:
            eng.stop()


class TestNativeProcessEngram:
    def test_command_engram_through_native_core(self):
        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.runtime.native import NativeRunner

        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        try:
            eng.apply_yaml(
                """
kind: EngramTemplate
metadata: {name: sh}
spec:
  command: [python3, -c, "import os, json; print(json.dumps({'via': 'native', 'step': os.environ['BUBU_STEP_NAME']}))"]
---
kind: Engram
metadata: {name: sheller}
spec: {templateRef: {name: sh}}
---
kind: Story
metadata: {name: extn}
spec:
  steps:
    - {name: go, ref: {name: sheller}, with: {v: 1}}
"""
            )
            nr = NativeRunner.from_run_engine(eng)
            res = nr.run_story("default/extn", {}, timeout=40)
            assert res["phase"] == "Succeeded", res
            assert res["steps"]["go"]["output"] == {"via": "native", "step": "go"}
        finally:
            eng.stop()


class TestNativeOffloadedSubPath:
    """Templates indexing THROUGH an offloaded output on the native core:
    the C++ VM derives `$storageRef` sub-path markers (storage path DSL)
    instead of mis-reading marker dicts — worker-side hydration resolves
    them, so the engine loop never touches payload bytes."""

    def test_sub_path_through_marker_matches_python(self):
        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.enums import Phase
        from bobrapet_amd.runtime.native import NativeRunner

        eng = RunEngine(EngineConfig(cpu_workers=2, max_inline_size=64)).start()
        try:
            eng.apply_yaml(
                """
kind: EngramTemplate
metadata: {name: echo}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: e}
spec: {templateRef: {name: echo}}
---
kind: Story
metadata: {name: sub}
spec:
  steps:
    - name: big
      ref: {name: e}
      with: {data: {x: 41, arr: [7, 8, 9], pad: "%s"}}
    - name: use
      ref: {name: e}
      needs: [big]
      with:
        got: "{{ steps.big.output.data.x }}"
        second: "{{ steps.big.output.data.arr[1] }}"
"""
                % ("x" * 90)
            )
            run = eng.run_story("default/sub", {}, timeout=30)
            assert run.phase == Phase.SUCCEEDED
            py_out = run.step_states["use"].output
            res = NativeRunner.from_run_engine(eng).run_story("default/sub", {}, timeout=30)
            assert res["phase"] == "Succeeded"
            assert res["steps"]["use"]["output"] == {"got": 41, "second": 8}
            assert py_out == {"got": 41, "second": 8}
        finally:
            eng.stop()

    def test_string_splice_hydrates(self):
        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.runtime.native import NativeRunner

        eng = RunEngine(EngineConfig(cpu_workers=2, max_inline_size=48)).start()
        try:
            eng.apply_yaml(
                """
kind: EngramTemplate
metadata: {name: echo}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: e}
spec: {templateRef: {name: echo}}
---
kind: Story
metadata: {name: splice}
spec:
  steps:
    - {name: big, ref: {name: e}, with: {data: {id: "ORD-7", pad: "%s"}}}
    - {name: use, ref: {name: e}, needs: [big], with: {msg: "order={{ steps.big.output.data.id }}!"}}
"""
                % ("x" * 60)
            )
            res = NativeRunner.from_run_engine(eng).run_story("default/splice", {}, timeout=30)
            assert res["phase"] == "Succeeded"
            assert res["steps"]["use"]["output"] == {"msg": "order=ORD-7!"}
        finally:
            eng.stop()

    def test_consuming_expressions_hydrate(self):
        """Conditions/arithmetic CONSUMING offloaded values hydrate through
        the expression-level hydrator (pass-through stays marker-based)."""
        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.runtime.native import NativeRunner

        eng = RunEngine(EngineConfig(cpu_workers=2, max_inline_size=64)).start()
        try:
            eng.apply_yaml(
                """
kind: EngramTemplate
metadata: {name: echo}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: e}
spec: {templateRef: {name: echo}}
---
kind: Story
metadata: {name: consume}
spec:
  steps:
    - name: big
      ref: {name: e}
      with: {data: {x: 41, arr: [7, 8, 9], pad: "%s"}}
    - name: gatekeep
      type: condition
      needs: [big]
      with: {expression: "{{ steps.big.output.data.x > 40 && size(steps.big.output.data.arr) == 3 }}"}
    - name: use
      ref: {name: e}
      needs: [gatekeep]
      if: "{{ steps.big.output.data.arr[2] == 9 }}"
      with: {sum: "{{ steps.big.output.data.x + 1 }}"}
"""
                % ("x" * 90)
            )
            res = NativeRunner.from_run_engine(eng).run_story("default/consume", {}, timeout=30)
            assert res["phase"] == "Succeeded", res
            assert res["steps"]["use"]["output"] == {"sum": 42}
        finally:
            eng.stop()


class TestNativeRunRetention:
    def test_run_story_reclaims_records_including_substories(self):
        """The fast path's retention: run_story gc's the terminal run and
        its executeStory descendants (no unbounded run-map growth)."""
        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.runtime.native import NativeRunner

        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        try:
            eng.apply_yaml(
                """
kind: Story
metadata: {name: inner}
spec:
  steps: [{name: a, type: sleep, with: {duration: 0ms}}]
---
kind: Story
metadata: {name: outer}
spec:
  steps:
    - {name: sub, type: executeStory, with: {storyRef: inner}}
"""
            )
            nr = NativeRunner.from_run_engine(eng)
            for _ in range(10):
                assert nr.run_story("default/outer", {}, timeout=30)["phase"] == "Succeeded"
            assert nr.engine.run_count() == 0
            # opt-out keeps the record for later inspection
            res = nr.run_story("default/outer", {}, timeout=30, gc=False)
            assert res["phase"] == "Succeeded"
            assert nr.engine.run_count() >= 1
        finally:
            eng.stop()
