"""End-to-end engine tests: the DAG machine, primitives, retry, cancel,
gates, redrive, sub-stories, triggers, effects, cache.

These mirror the reference's envtest suites — but the
actors the reference must simulate (SDK patches, kubelet) are real here:
engrams execute in-process on worker slots.
"""
import time

import pytest

from bobrapet_amd.engine import EngineConfig, RunEngine
from bobrapet_amd.enums import Phase, TriggerDecision


BASE_RESOURCES = """
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
---
kind: EngramTemplate
metadata: {name: sleepy-tpl}
spec: {builtin: sleepy}
---
kind: Engram
metadata: {name: sleeper}
spec: {templateRef: {name: sleepy-tpl}}
"""


@pytest.fixture()
def eng():
    engine = RunEngine(EngineConfig(cpu_workers=4, child_ttl_seconds=3600)).start()
    engine.apply_yaml(BASE_RESOURCES)
    import bobrapet_amd.engrams.registry as reg

    reg.reset_instances()  # fresh fail-engram counters per test
    yield engine
    engine.stop()


def _story(eng, yaml_text):
    return eng.apply_yaml(yaml_text)[0]


class TestBasicFlows:
    def test_two_step_sleep_condition(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: two-step}
spec:
  steps:
    - {name: pause, type: sleep, with: {duration: 5ms}}
    - name: check
      type: condition
      needs: [pause]
      with: {expression: "{{ steps.pause.phase == 'Succeeded' }}"}
  output: {ok: "{{ steps.check.output.result }}"}
""",
        )
        run = eng.run_story("default/two-step", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.output == {"ok": True}
        assert run.step_states["pause"].phase == Phase.SUCCEEDED

    def test_engram_chain_with_templates(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: chain}
spec:
  steps:
    - {name: a, ref: {name: echoer}, with: {v: "{{ inputs.x }}"}}
    - {name: b, ref: {name: echoer}, with: {v: "{{ steps.a.output.v + 1 }}"}}
  output: {v: "{{ steps.b.output.v }}"}
""",
        )
        run = eng.run_story("default/chain", {"x": 41}, timeout=10)
        assert run.phase == Phase.SUCCEEDED, (run.error, run.step_states)
        assert run.output == {"v": 42}

    def test_implicit_dependency_orders_steps(self, eng):
        # no `needs`: the template reference alone must order b after a
        _story(
            eng,
            """
kind: Story
metadata: {name: implicit}
spec:
  steps:
    - {name: second, ref: {name: echoer}, with: {v: "{{ steps.first.output.v }}"}}
    - {name: first, ref: {name: echoer}, with: {v: 7}}
""",
        )
        run = eng.run_story("default/implicit", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["second"].output == {"v": 7}

    def test_independent_steps_run_concurrently(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: conc}
spec:
  steps:
    - {name: s1, ref: {name: sleeper}, with: {seconds: 0.15}}
    - {name: s2, ref: {name: sleeper}, with: {seconds: 0.15}}
    - {name: s3, ref: {name: sleeper}, with: {seconds: 0.15}}
""",
        )
        t0 = time.monotonic()
        run = eng.run_story("default/conc", {}, timeout=10)
        elapsed = time.monotonic() - t0
        assert run.phase == Phase.SUCCEEDED
        assert elapsed < 0.40, f"steps did not overlap: {elapsed:.2f}s"

    def test_if_condition_skips(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: iffy}
spec:
  steps:
    - {name: a, ref: {name: echoer}, with: {v: 1}}
    - {name: b, ref: {name: echoer}, if: "{{ steps.a.output.v > 100 }}", with: {v: 2}}
    - {name: c, ref: {name: echoer}, if: "{{ steps.a.output.v < 100 }}", with: {v: 3}}
""",
        )
        run = eng.run_story("default/iffy", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["b"].phase == Phase.SKIPPED
        assert run.step_states["c"].phase == Phase.SUCCEEDED

    def test_skip_cascades_to_dependents(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: cascade}
spec:
  steps:
    - {name: a, type: condition, with: {expression: "false"}}
    - {name: b, ref: {name: echoer}, if: "{{ steps.a.output.result }}", with: {v: 1}}
    - {name: c, ref: {name: echoer}, needs: [b], with: {v: 2}}
""",
        )
        run = eng.run_story("default/cascade", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["b"].phase == Phase.SKIPPED
        assert run.step_states["c"].phase == Phase.SKIPPED

    def test_requires_guard(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: reqy}
spec:
  steps:
    - {name: a, ref: {name: echoer}, with: {v: 1}}
    - {name: b, ref: {name: echoer}, requires: ["steps.a.output.missing"], with: {v: 2}}
    - {name: c, ref: {name: echoer}, requires: ["steps.a.output.v"], with: {v: 3}}
""",
        )
        run = eng.run_story("default/reqy", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["b"].phase == Phase.SKIPPED
        assert run.step_states["c"].phase == Phase.SUCCEEDED

    def test_inputs_schema_defaults_and_validation(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: schema-story}
spec:
  inputsSchema:
    type: object
    required: [name]
    properties:
      name: {type: string}
      count: {type: integer, default: 3}
  steps:
    - {name: a, ref: {name: echoer}, with: {n: "{{ inputs.count }}"}}
  output: {n: "{{ steps.a.output.n }}"}
""",
        )
        run = eng.run_story("default/schema-story", {"name": "x"}, timeout=10)
        assert run.output == {"n": 3}
        with pytest.raises(ValueError):
            eng.submit_run("default/schema-story", {"count": 1})

    def test_inputs_schema_defaults_do_not_mutate_submitted_dict(self, eng):
        # resolving inputs (schema defaults injected) must not write back
        # into the caller's payload (reference behavior:
        # steprun_input_exposure_test.go — resolved bytes carry defaults,
        # the persisted spec input stays as submitted)
        _story(
            eng,
            """
kind: Story
metadata: {name: schema-story2}
spec:
  inputsSchema:
    type: object
    properties:
      secret: {type: string}
      mode: {type: string, default: safe}
  steps:
    - {name: a, ref: {name: echoer}, with: {m: "{{ inputs.mode }}"}}
  output: {m: "{{ steps.a.output.m }}"}
""",
        )
        submitted = {"secret": "value"}
        run = eng.run_story("default/schema-story2", submitted, timeout=10)
        assert run.output == {"m": "safe"}
        assert submitted == {"secret": "value"}  # caller dict untouched
        assert run.inputs == {"secret": "value", "mode": "safe"}


class TestFailureMachinery:
    def test_fail_fast_skips_and_fails(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: failing}
spec:
  steps:
    - {name: boom, ref: {name: failer}, with: {exitCode: 2}}
    - {name: after, ref: {name: echoer}, needs: [boom], with: {v: 1}}
""",
        )
        run = eng.run_story("default/failing", {}, timeout=10)
        assert run.phase == Phase.FAILED
        assert run.step_states["boom"].phase == Phase.FAILED
        assert run.step_states["after"].phase == Phase.SKIPPED
        assert run.failure_cause == "boom"

    def test_allow_failure(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: allowed}
spec:
  steps:
    - {name: boom, ref: {name: failer}, allowFailure: true, with: {exitCode: 2}}
    - {name: after, ref: {name: echoer}, needs: [boom], with: {v: 1}}
""",
        )
        run = eng.run_story("default/allowed", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["after"].phase == Phase.SUCCEEDED

    def test_continue_on_step_failure(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: continue-on-fail}
spec:
  policy:
    retries: {continueOnStepFailure: true}
  steps:
    - {name: boom, ref: {name: failer}, with: {exitCode: 2}}
    - {name: dependent, ref: {name: echoer}, needs: [boom], with: {v: 1}}
    - {name: independent, ref: {name: echoer}, with: {v: 2}}
""",
        )
        run = eng.run_story("default/continue-on-fail", {}, timeout=10)
        assert run.phase == Phase.FAILED
        assert run.step_states["independent"].phase == Phase.SUCCEEDED
        assert run.step_states["dependent"].phase == Phase.SKIPPED

    def test_retry_until_success(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: retrying}
spec:
  steps:
    - name: flaky
      ref: {name: failer}
      with: {succeedAfter: 2, exitCode: 1}
      execution:
        retry: {maxRetries: 5, delay: 10ms, jitter: 0, backoff: constant}
""",
        )
        run = eng.run_story("default/retrying", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED, (run.error, run.step_states["flaky"].error)
        assert run.step_states["flaky"].retries == 2
        assert run.step_states["flaky"].output == {"attempts": 3}

    def test_terminal_exit_code_does_not_retry(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: no-retry}
spec:
  steps:
    - name: fatal
      ref: {name: failer}
      with: {exitCode: 2}
      execution:
        retry: {maxRetries: 5, delay: 5ms}
""",
        )
        run = eng.run_story("default/no-retry", {}, timeout=10)
        assert run.phase == Phase.FAILED
        assert run.step_states["fatal"].retries == 0

    def test_retry_budget_exhaustion(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: exhausted}
spec:
  steps:
    - name: always-fails
      ref: {name: failer}
      with: {exitCode: 1}
      execution:
        retry: {maxRetries: 2, delay: 5ms, jitter: 0}
""",
        )
        run = eng.run_story("default/exhausted", {}, timeout=10)
        assert run.phase == Phase.FAILED
        assert run.step_states["always-fails"].retries == 2

    def test_compensations_run_on_failure(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: saga}
spec:
  steps:
    - {name: work, ref: {name: failer}, with: {exitCode: 2}}
  compensations:
    - {name: undo, ref: {name: echoer}, with: {undone: true}}
""",
        )
        run = eng.run_story("default/saga", {}, timeout=10)
        assert run.phase == Phase.COMPENSATED
        assert run.step_states["undo"].phase == Phase.SUCCEEDED

    def test_finally_always_runs(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: fin-ok}
spec:
  steps:
    - {name: work, ref: {name: echoer}, with: {v: 1}}
  finally:
    - {name: report, ref: {name: echoer}, with: {done: true}}
""",
        )
        run = eng.run_story("default/fin-ok", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["report"].phase == Phase.SUCCEEDED

        _story(
            eng,
            """
kind: Story
metadata: {name: fin-fail}
spec:
  steps:
    - {name: work, ref: {name: failer}, with: {exitCode: 2}}
  finally:
    - {name: report, ref: {name: echoer}, with: {done: true}}
""",
        )
        run = eng.run_story("default/fin-fail", {}, timeout=10)
        assert run.phase == Phase.FAILED
        assert run.step_states["report"].phase == Phase.SUCCEEDED

    def test_step_timeout(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: slow-step}
spec:
  steps:
    - name: slow
      ref: {name: sleeper}
      with: {seconds: 5}
      execution: {timeout: 100ms}
""",
        )
        t0 = time.monotonic()
        run = eng.run_story("default/slow-step", {}, timeout=10)
        assert run.phase == Phase.FAILED
        assert run.step_states["slow"].phase == Phase.TIMEOUT
        assert time.monotonic() - t0 < 3.0

    def test_story_timeout(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: slow-story}
spec:
  policy:
    timeouts: {story: 100ms}
  steps:
    - {name: slow, ref: {name: sleeper}, with: {seconds: 5}}
""",
        )
        run = eng.run_story("default/slow-story", {}, timeout=10)
        assert run.phase == Phase.TIMEOUT


class TestPrimitives:
    def test_stop_success(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: stopper}
spec:
  steps:
    - {name: halt, type: stop, with: {mode: success, message: done early}}
    - {name: never, ref: {name: echoer}, needs: [halt], with: {v: 1}}
""",
        )
        run = eng.run_story("default/stopper", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["never"].phase == Phase.SKIPPED

    def test_stop_failure_mode(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: stop-fail}
spec:
  steps:
    - {name: halt, type: stop, with: {mode: failure}}
""",
        )
        run = eng.run_story("default/stop-fail", {}, timeout=10)
        assert run.phase == Phase.FAILED

    def test_wait_until_condition(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: waiter}
spec:
  steps:
    - {name: slow, ref: {name: sleeper}, with: {seconds: 0.1}}
    - name: watch
      type: wait
      with:
        until: "{{ steps.slow.phase == 'Succeeded' }}"
        pollInterval: 10ms
        timeout: 5s
""",
        )
        run = eng.run_story("default/waiter", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["watch"].phase == Phase.SUCCEEDED

    def test_wait_timeout_skip(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: wait-skip}
spec:
  steps:
    - name: watch
      type: wait
      with: {until: "{{ false }}", pollInterval: 10ms, timeout: 50ms, onTimeout: skip}
""",
        )
        run = eng.run_story("default/wait-skip", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["watch"].phase == Phase.SKIPPED

    def test_wait_timeout_fail(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: wait-fail}
spec:
  steps:
    - name: watch
      type: wait
      with: {until: "{{ false }}", pollInterval: 10ms, timeout: 50ms, onTimeout: fail}
""",
        )
        run = eng.run_story("default/wait-fail", {}, timeout=10)
        assert run.phase == Phase.FAILED
        assert run.step_states["watch"].phase == Phase.TIMEOUT

    def test_gate_approved(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: gated}
spec:
  steps:
    - {name: approval, type: gate}
    - {name: after, ref: {name: echoer}, needs: [approval], with: {v: 1}}
""",
        )
        run = eng.submit_run("default/gated", {})
        time.sleep(0.1)
        r = eng.store.get_story_run(run.key)
        assert r.step_states["approval"].phase == Phase.PAUSED
        eng.approve_gate(run.key, "approval", decided_by="tester")
        run = eng.wait(run.key, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["approval"].output["approved"] is True

    def test_gate_rejected(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: gated-no}
spec:
  steps:
    - {name: approval, type: gate}
""",
        )
        run = eng.submit_run("default/gated-no", {})
        time.sleep(0.05)
        eng.reject_gate(run.key, "approval")
        run = eng.wait(run.key, timeout=10)
        assert run.phase == Phase.FAILED

    def test_gate_timeout(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: gate-to}
spec:
  steps:
    - {name: approval, type: gate, with: {timeout: 50ms, onTimeout: skip}}
""",
        )
        run = eng.run_story("default/gate-to", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["approval"].phase == Phase.SKIPPED

    def test_parallel_fanout_join(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: fan}
spec:
  steps:
    - name: fanout
      type: parallel
      with:
        steps:
          - {name: b1, ref: {name: echoer}, with: {v: 1}}
          - {name: b2, ref: {name: echoer}, with: {v: 2}}
          - {name: b3, ref: {name: echoer}, with: {v: 3}}
    - name: join
      ref: {name: echoer}
      needs: [fanout]
      with: {got: "{{ steps.fanout.output.branches.b2.v }}"}
""",
        )
        run = eng.run_story("default/fan", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["join"].output == {"got": 2}

    def test_parallel_branch_failure(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: fan-fail}
spec:
  steps:
    - name: fanout
      type: parallel
      with:
        steps:
          - {name: ok, ref: {name: echoer}, with: {v: 1}}
          - {name: bad, ref: {name: failer}, with: {exitCode: 2}}
""",
        )
        run = eng.run_story("default/fan-fail", {}, timeout=10)
        assert run.phase == Phase.FAILED
        assert run.step_states["fanout"].phase == Phase.FAILED

    def test_parallel_branch_allow_failure(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: fan-allow}
spec:
  steps:
    - name: fanout
      type: parallel
      with:
        steps:
          - {name: ok, ref: {name: echoer}, with: {v: 1}}
          - {name: bad, ref: {name: failer}, allowFailure: true, with: {exitCode: 2}}
""",
        )
        run = eng.run_story("default/fan-allow", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED

    def test_execute_story_nested(self, eng):
        eng.apply_yaml(
            """
kind: Story
metadata: {name: inner}
spec:
  policy:
    with: {base: 10}
  steps:
    - {name: calc, ref: {name: echoer}, with: {v: "{{ inputs.base + inputs.add }}"}}
  output: {v: "{{ steps.calc.output.v }}"}
---
kind: Story
metadata: {name: outer}
spec:
  steps:
    - name: sub
      type: executeStory
      with:
        storyRef: inner
        with: {add: 5}
    - name: use
      ref: {name: echoer}
      needs: [sub]
      with: {got: "{{ steps.sub.output.output.v }}"}
  output: {v: "{{ steps.use.output.got }}"}
"""
        )
        run = eng.run_story("default/outer", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED, (run.error, {k: (v.phase, v.error) for k, v in run.step_states.items()})
        assert run.output == {"v": 15}

    def test_execute_story_failure_propagates(self, eng):
        eng.apply_yaml(
            """
kind: Story
metadata: {name: inner-bad}
spec:
  steps:
    - {name: boom, ref: {name: failer}, with: {exitCode: 2}}
---
kind: Story
metadata: {name: outer-bad}
spec:
  steps:
    - {name: sub, type: executeStory, with: {storyRef: inner-bad}}
"""
        )
        run = eng.run_story("default/outer-bad", {}, timeout=10)
        assert run.phase == Phase.FAILED

    def test_recursion_depth_cap(self, eng):
        # DIRECT self-reference is rejected at apply time (webhook parity);
        # the runtime depth cap guards INDIRECT cycles (A -> B -> A)
        eng.apply_yaml(
            """
kind: Story
metadata: {name: recur}
spec:
  steps:
    - {name: hop, type: executeStory, with: {storyRef: recur2}}
---
kind: Story
metadata: {name: recur2}
spec:
  steps:
    - {name: back, type: executeStory, with: {storyRef: recur}}
"""
        )
        run = eng.run_story("default/recur", {}, timeout=20)
        assert run.phase == Phase.FAILED

    def test_execute_story_self_reference_rejected_at_apply(self, eng):
        from bobrapet_amd.specs.validation import SpecValidationError

        with pytest.raises(SpecValidationError):
            eng.apply_yaml(
                """
kind: Story
metadata: {name: selfref}
spec:
  steps:
    - {name: again, type: executeStory, with: {storyRef: selfref}}
"""
            )

    def test_execute_story_cross_namespace_rejected_at_apply(self, eng):
        from bobrapet_amd.specs.validation import SpecValidationError

        with pytest.raises(SpecValidationError):
            eng.apply_yaml(
                """
kind: Story
metadata: {name: xns}
spec:
  steps:
    - name: other
      type: executeStory
      with: {storyRef: {name: elsewhere, namespace: other-ns}}
"""
            )


class TestControl:
    def test_graceful_cancel(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: cancelme}
spec:
  steps:
    - {name: slow, ref: {name: sleeper}, with: {seconds: 10}}
""",
        )
        run = eng.submit_run("default/cancelme", {})
        time.sleep(0.1)
        eng.cancel(run.key)
        run = eng.wait(run.key, timeout=10)
        assert run.phase == Phase.CANCELED

    def test_redrive(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: redrive-me}
spec:
  steps:
    - name: flaky
      ref: {name: failer}
      with: {succeedAfter: 1, exitCode: 1}
""",
        )
        run = eng.run_story("default/redrive-me", {}, timeout=10)
        assert run.phase == Phase.FAILED  # no retries configured, first call fails
        eng.redrive(run.key)
        run = eng.wait(run.key, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.redrive_count == 1

    def test_redrive_from_step(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: partial-redrive}
spec:
  steps:
    - {name: a, ref: {name: echoer}, with: {v: 1}}
    - {name: b, ref: {name: failer}, needs: [a], with: {succeedAfter: 1, exitCode: 1}}
    - {name: c, ref: {name: echoer}, needs: [b], with: {v: 3}}
""",
        )
        run = eng.run_story("default/partial-redrive", {}, timeout=10)
        assert run.phase == Phase.FAILED
        a_finished = run.step_states["a"].finished_at
        eng.redrive_from_step(run.key, "b")
        run = eng.wait(run.key, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert run.step_states["c"].phase == Phase.SUCCEEDED
        # step a was NOT re-executed
        assert run.step_states["a"].finished_at == a_finished

    def test_story_concurrency_limit(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: limited}
spec:
  policy: {concurrency: 1}
  steps:
    - {name: slow, ref: {name: sleeper}, with: {seconds: 0.2}}
""",
        )
        runs = [eng.submit_run("default/limited", {"i": i}) for i in range(3)]
        time.sleep(0.1)
        phases = [eng.store.get_story_run(r.key).phase for r in runs]
        assert Phase.SCHEDULING in phases  # at least one is queued
        for r in runs:
            final = eng.wait(r.key, timeout=15)
            assert final.phase == Phase.SUCCEEDED


class TestTriggers:
    def test_trigger_created_then_reused(self, eng):
        from bobrapet_amd.engine.triggers import StoryTrigger

        _story(
            eng,
            """
kind: Story
metadata: {name: triggered}
spec:
  steps:
    - {name: a, ref: {name: echoer}, with: {v: 1}}
""",
        )
        t1 = eng.triggers.submit(
            StoryTrigger(submission_id="s1", story_name="triggered", inputs={"x": 1})
        )
        assert t1.decision == TriggerDecision.CREATED
        t2 = eng.triggers.submit(
            StoryTrigger(submission_id="s1", story_name="triggered", inputs={"x": 1})
        )
        assert t2.decision == TriggerDecision.REUSED
        assert t2.story_run_ref == t1.story_run_ref
        t3 = eng.triggers.submit(
            StoryTrigger(submission_id="s2", story_name="triggered", inputs={"x": 2})
        )
        assert t3.decision == TriggerDecision.CREATED
        assert t3.story_run_ref != t1.story_run_ref

    def test_trigger_missing_story_rejected(self, eng):
        from bobrapet_amd.engine.triggers import StoryTrigger

        t = eng.triggers.submit(StoryTrigger(submission_id="s1", story_name="ghost"))
        assert t.decision == TriggerDecision.REJECTED

    def test_trigger_identity_required(self, eng):
        from bobrapet_amd.engine.triggers import StoryTrigger

        t = eng.triggers.submit(StoryTrigger(submission_id="", story_name="x"))
        assert t.decision == TriggerDecision.REJECTED


class TestCacheAndEffects:
    def test_step_cache_hits(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: cached}
spec:
  steps:
    - name: work
      ref: {name: failer}
      with: {succeedAfter: 0}
      execution:
        cache: {enabled: true, mode: readWrite}
""",
        )
        r1 = eng.run_story("default/cached", {}, timeout=10)
        assert r1.phase == Phase.SUCCEEDED
        # second run with identical inputs: cache hit, fail engram NOT called
        r2 = eng.submit_run("default/cached", {})
        r2 = eng.wait(r2.key, timeout=10)
        assert r2.phase == Phase.SUCCEEDED
        assert r2.step_states["work"].output == r1.step_states["work"].output
        assert eng.metrics.counter_value("steprun_cache_lookups_total", result="hit") == 1
        # cache hit sets lastOutputAt (reference: applyCacheHit /
        # steprun_last_output_at_test.go); non-empty output required
        sr2 = [s2 for s2 in eng.store.step_runs_of(r2.key) if s2.spec.step_name == "work"][0]
        assert sr2.status.cache_hit
        assert sr2.status.last_output_at is not None

    def test_effect_ledger_exactly_once(self, eng):
        from bobrapet_amd.engine.effects import EffectLedger
        from bobrapet_amd.enums import EffectClaimPhase

        ledger = EffectLedger()
        claim, fresh = ledger.acquire("charge-42", holder="attempt-1")
        assert fresh
        ledger.complete("charge-42", "attempt-1")
        claim2, fresh2 = ledger.acquire("charge-42", holder="attempt-2")
        assert not fresh2
        assert claim2.phase == EffectClaimPhase.COMPLETED

    def test_effect_stale_takeover(self, eng):
        from bobrapet_amd.engine.effects import EffectLedger

        ledger = EffectLedger()
        claim, fresh = ledger.acquire("x", holder="h1", lease_duration=0.01)
        assert fresh
        time.sleep(0.05)
        claim2, fresh2 = ledger.acquire("x", holder="h2", lease_duration=60)
        assert fresh2
        assert claim2.holder == "h2"
        assert claim2.takeovers == 1

    def test_metrics_exported(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: metered}
spec:
  steps:
    - {name: a, ref: {name: echoer}, with: {v: 1}}
""",
        )
        eng.run_story("default/metered", {}, timeout=10)
        text = eng.metrics.export_text()
        assert "bobrapet_amd_storyruns_total" in text
        assert "bobrapet_amd_stepruns_total" in text
        assert "bobrapet_amd_storyrun_duration_seconds" in text


class TestBlockedWakeup:
    def test_run_blocked_until_engram_applied(self, eng):
        _story(
            eng,
            """
kind: Story
metadata: {name: needs-worker}
spec:
  steps:
    - {name: a, ref: {name: late-engram}, with: {v: 1}}
""",
        )
        run = eng.submit_run("default/needs-worker", {})
        time.sleep(0.1)
        r = eng.store.get_story_run(run.key)
        assert r.step_states["a"].phase == Phase.BLOCKED
        eng.apply_yaml(
            """
kind: Engram
metadata: {name: late-engram}
spec: {templateRef: {name: echo-tpl}}
"""
        )
        run = eng.wait(run.key, timeout=10)
        assert run.phase == Phase.SUCCEEDED


class TestConcurrentWriterRaces:
    """Race-shaped tests (reference: steprun_sdk_race_test.go,
    persistMergedStates dag.go:774-793): terminal-phase-wins under
    concurrent completion attempts, and cancel racing completion."""

    def test_finish_step_run_terminal_wins_concurrent(self):
        import threading

        from bobrapet_amd.engine.executor import finish_step_run
        from bobrapet_amd.engine.records import StepRun, StepRunSpec
        from bobrapet_amd.enums import ExitClass, Phase

        sr = StepRun(name="r-a", spec=StepRunSpec(story_run="r", step_name="a"))
        wins = []
        barrier = threading.Barrier(8)

        def writer(phase, code):
            barrier.wait()
            if finish_step_run(sr, phase, output={"p": str(phase)}, exit_code=code):
                wins.append(phase)

        threads = [
            threading.Thread(
                target=writer,
                args=(Phase.SUCCEEDED if i % 2 == 0 else Phase.FAILED, i % 2),
            )
            for i in range(8)
        ]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        # exactly one writer won; status matches the winner and never flips
        assert len(wins) == 1
        assert sr.status.phase == wins[0]
        assert sr.status.output == {"p": str(wins[0])}

    def test_step_state_merge_preserves_terminal(self):
        from bobrapet_amd.engine.records import StepState
        from bobrapet_amd.enums import Phase

        dst = StepState(name="a", phase=Phase.SUCCEEDED, output={"v": 1})
        src = StepState(name="a", phase=Phase.RUNNING, output=None)
        dst.merge_from(src)
        assert dst.phase == Phase.SUCCEEDED
        assert dst.output == {"v": 1}

    def test_cancel_races_completion(self):
        """Cancel posted while engrams are mid-flight must converge to a
        terminal run without deadlock, whichever side wins each step."""
        from bobrapet_amd.engine import EngineConfig, RunEngine

        eng = RunEngine(EngineConfig(cpu_workers=4)).start()
        try:
            eng.apply_yaml(
                """
kind: EngramTemplate
metadata: {name: sleepy}
spec: {builtin: sleepy}
---
kind: Engram
metadata: {name: s}
spec: {templateRef: {name: sleepy}}
---
kind: Story
metadata: {name: race}
spec:
  steps:
    - {name: a, ref: {name: s}, with: {seconds: 0.05}}
    - {name: b, ref: {name: s}, needs: [a], with: {seconds: 0.05}}
"""
            )
            for _ in range(5):
                run = eng.submit_run("default/race", {})
                import time as _time

                _time.sleep(0.02)
                eng.cancel(run, graceful=False)
                run = eng.wait(run, timeout=20)
                assert run.is_terminal
        finally:
            eng.stop()


class TestAdmissionStress:
    def test_concurrent_submitters_all_complete(self):
        """200 runs from 8 threads through queue admission (concurrency 16,
        priority aging) — every run completes, none lost or duplicated."""
        import concurrent.futures as cf

        from bobrapet_amd.engine import EngineConfig, QueueConfig, RunEngine
        from bobrapet_amd.enums import Phase

        eng = RunEngine(
            EngineConfig(
                cpu_workers=8,
                global_concurrency=16,
                queues={"default": QueueConfig(concurrency=16)},
                child_ttl_seconds=60.0,
            )
        ).start()
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
metadata: {name: stress}
spec:
  steps:
    - {name: a, ref: {name: e}, with: {v: "{{ inputs.i }}"}}
    - {name: b, type: condition, needs: [a], with: {expression: "{{ steps.a.output.v >= 0 }}"}}
"""
            )

            def one(i: int) -> str:
                run = eng.run_story("default/stress", {"i": i}, timeout=60)
                assert run.phase == Phase.SUCCEEDED, (i, run.error)
                assert run.step_states["a"].output["v"] == i
                return run.name

            with cf.ThreadPoolExecutor(max_workers=8) as ex:
                names = list(ex.map(one, range(200)))
            assert len(set(names)) == 200
        finally:
            eng.stop()


class TestImpulseTriggerStats:
    """Trigger-stats aggregation + bounded backfill (reference:
    impulse_controller.go:1151-1233, trigger_annotations.go)."""

    YAML = """
kind: EngramTemplate
metadata: {name: echo-tpl-ts}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: e-ts}
spec: {templateRef: {name: echo-tpl-ts}}
---
kind: Story
metadata: {name: ts-story}
spec:
  steps: [{name: a, ref: {name: e-ts}, with: {v: "{{ inputs.n }}"}}]
---
kind: ImpulseTemplate
metadata: {name: manual-ts}
spec: {builtin: manual}
---
kind: Impulse
metadata: {name: poker}
spec:
  templateRef: {name: manual-ts}
  storyRef: {name: ts-story}
  mapping: {n: "{{ event.n }}"}
"""

    def test_stats_and_idempotent_backfill(self):
        from bobrapet_amd.engine import EngineConfig, RunEngine

        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        try:
            eng.apply_yaml(self.YAML)
            live = eng.impulses.start("default/poker")
            for n in range(5):
                r = live.handler.emit({"n": n})
                eng.wait(r.story_run_ref, timeout=10)
            st = eng.impulses.status("default/poker")
            assert st["emitted"] == 5
            assert st["decisions"].get("Created") == 5
            assert st["triggers"] == 5
            assert st["backfilled"] == 5
            # second aggregation: tokens already counted → no double count
            st2 = eng.impulses.status("default/poker")
            assert st2["triggers"] == 5 and st2["backfilled"] == 0
            # restart the impulse: counted tokens survive (idempotent)
            eng.impulses.stop("default/poker")
            eng.impulses.start("default/poker")
            st3 = eng.impulses.status("default/poker")
            assert st3["triggers"] == 5 and st3["backfilled"] == 0
            # bounded scan reports the cap
            st4 = eng.impulses.status("default/poker", max_scan=2)
            assert st4["scanCapped"] is True
        finally:
            eng.stop()


class TestTriggerConflicts:
    """Trigger admission edge cases (reference:
    storytrigger_controller_test.go: conflicting existing run → Rejected;
    differing impulse provenance → no reuse; concurrent duplicates
    collapse to one StoryRun)."""

    STORY = """
kind: Story
metadata: {name: trig-tgt}
spec:
  steps:
    - {name: a, ref: {name: echoer}, with: {v: "{{ inputs.x }}"}}
  output: {v: "{{ steps.a.output.v }}"}
"""

    def test_conflicting_inputs_rejected(self, eng):
        from bobrapet_amd.engine.triggers import StoryTrigger
        from bobrapet_amd.enums import TriggerDecision

        _story(eng, self.STORY)
        t1 = eng.triggers.submit(StoryTrigger(
            submission_id="s1", key="same-key", story_name="trig-tgt",
            inputs={"x": 1}))
        assert t1.decision == TriggerDecision.CREATED
        eng.wait(t1.story_run_ref, timeout=10)
        t2 = eng.triggers.submit(StoryTrigger(
            submission_id="s2", key="same-key", story_name="trig-tgt",
            inputs={"x": 2}))
        assert t2.decision == TriggerDecision.REJECTED
        assert "identity collision" in t2.message

    def test_impulse_provenance_blocks_reuse(self, eng):
        from bobrapet_amd.engine.triggers import StoryTrigger
        from bobrapet_amd.enums import TriggerDecision

        _story(eng, self.STORY)
        t1 = eng.triggers.submit(StoryTrigger(
            submission_id="p1", key="prov-key", story_name="trig-tgt",
            inputs={"x": 1}, impulse="default/imp-a"))
        assert t1.decision == TriggerDecision.CREATED
        # same identity+inputs but a DIFFERENT impulse: not reused
        t2 = eng.triggers.submit(StoryTrigger(
            submission_id="p1", key="prov-key", story_name="trig-tgt",
            inputs={"x": 1}, impulse="default/imp-b"))
        assert t2.decision == TriggerDecision.REJECTED

    def test_concurrent_duplicates_collapse(self, eng):
        import concurrent.futures as cf

        from bobrapet_amd.engine.triggers import StoryTrigger
        from bobrapet_amd.enums import TriggerDecision

        _story(eng, self.STORY)
        before = len(eng.store.all_runs())

        def fire(i):
            return eng.triggers.submit(StoryTrigger(
                submission_id="dup", key="dup-key", story_name="trig-tgt",
                inputs={"x": 9}))

        with cf.ThreadPoolExecutor(max_workers=8) as ex:
            results = list(ex.map(fire, range(16)))
        created = [r for r in results if r.decision == TriggerDecision.CREATED]
        reused = [r for r in results if r.decision == TriggerDecision.REUSED]
        assert len(created) == 1, [str(r.decision) for r in results]
        assert len(created) + len(reused) == 16
        assert len(eng.store.all_runs()) == before + 1
        refs = {r.story_run_ref for r in results}
        assert len(refs) == 1


def test_retention_prunes_orphan_effect_claims():
    """Reference: effectclaim_controller_test.go — orphan claims whose
    StepRun is gone are deleted.  Claims are run-scoped here, so retention
    cleanup of the run prunes them."""
    from bobrapet_amd.engine import EngineConfig, RunEngine

    from bobrapet_amd.engrams import registry as ereg
    from bobrapet_amd.engrams.base import Engram, EngramResult

    class Effectful(Engram):
        builtin = "effectful"

        def run(self, ctx):
            key = ((ctx.config or {}).get("effectKey")
                   or (ctx.input or {}).get("effectKey", "k"))
            ctx.record_effect(key, "test effect")
            return EngramResult(output={"done": True})

    ereg.register("effectful", Effectful)
    eng = RunEngine(EngineConfig(
        cpu_workers=1, child_ttl_seconds=0.05, storyrun_retention_seconds=0.1,
    )).start()
    try:
        eng.apply_yaml("""
kind: EngramTemplate
metadata: {name: fx-tpl}
spec: {builtin: effectful}
---
kind: Engram
metadata: {name: fx}
spec: {templateRef: {name: fx-tpl}}
---
kind: Story
metadata: {name: fx-story}
spec:
  steps:
    - {name: charge, ref: {name: fx}, with: {effectKey: pay-1}}
""")
        run = eng.run_story("default/fx-story", {}, timeout=10)
        assert run.phase == Phase.SUCCEEDED
        assert len(eng.effects) >= 1
        deadline = time.time() + 10
        while time.time() < deadline and eng.store.try_get_story_run(run.key) is not None:
            time.sleep(0.02)
        assert eng.store.try_get_story_run(run.key) is None
        assert len(eng.effects) == 0, "claims must be pruned with the run"
    finally:
        eng.stop()


def test_story_status_warns_unknown_engrams_in_all_phases():
    """Engram-reference scan covers main, compensation, finally AND
    parallel branches (reference: ValidateEngramReferencesIncludes
    Compensation/Finally)."""
    from bobrapet_amd.engine import EngineConfig, RunEngine

    eng = RunEngine(EngineConfig(cpu_workers=1)).start()
    try:
        eng.apply_yaml("""
kind: EngramTemplate
metadata: {name: e-tpl}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: known}
spec: {templateRef: {name: e-tpl}}
---
kind: Story
metadata: {name: scan-story}
spec:
  steps:
    - {name: a, ref: {name: known}, with: {v: 1}}
    - name: fan
      type: parallel
      with:
        steps:
          - {name: b1, ref: {name: ghost-branch}, with: {v: 2}}
  compensations:
    - {name: undo, ref: {name: ghost-comp}, with: {v: 3}}
  finally:
    - {name: always, ref: {name: ghost-finally}, with: {v: 4}}
""")
        st = eng.registry.story_status("default/scan-story")
        joined = " ".join(st["validationWarnings"])
        for ghost in ("ghost-branch", "ghost-comp", "ghost-finally"):
            assert ghost in joined, (ghost, st["validationWarnings"])
        assert "engram default/known" not in joined
    finally:
        eng.stop()
