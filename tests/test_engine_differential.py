"""Differential property test: the Python engine and the bobraccel C++ core
must agree on final phases and outputs for randomly generated batch stories
("two engines, one semantics" — ARCHITECTURE.md).

Random DAGs of echo / fail / condition / sleep steps with random needs
edges, if-templates, allowFailure flags, and a compensation + finally
tail — run through both engines, compare run phase + every step phase.
"""
import random

import pytest

from bobrapet_amd.engine import EngineConfig, RunEngine
from bobrapet_amd.runtime.native import NativeRunner, story_supported

RESOURCES = """
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
"""


def _random_story(rng: random.Random, idx: int) -> str:
    n = rng.randint(2, 7)
    lines = [f"kind: Story", f"metadata: {{name: diff-{idx}}}", "spec:", "  steps:"]
    for i in range(n):
        name = f"s{i}"
        deps = [f"s{j}" for j in range(i) if rng.random() < 0.4]
        needs = f" needs: [{', '.join(deps)}]," if deps else ""
        kind = rng.random()
        extra = " allowFailure: true," if rng.random() < 0.25 else ""
        if kind < 0.1:
            step = (
                f"    - {{name: {name},{needs}{extra} ref: {{name: f}}, "
                f"with: {{succeedAfter: 99}}}}"
            )
        elif kind < 0.18:
            # fails once then succeeds — exercises retry parity (the fail
            # engram counts attempts per (storyRun, step))
            step = (
                f"    - {{name: {name},{needs}{extra} ref: {{name: f}}, "
                f"retry: {{maxRetries: 2, delay: 1ms}}, "
                f"with: {{succeedAfter: 1}}}}"
            )
        elif kind < 0.22 and i == n - 1:
            step = f"    - {{name: {name},{needs} type: stop, with: {{phase: Succeeded}}}}"
        elif kind < 0.27:
            nb = rng.randint(2, 3)
            branches = ", ".join(
                f"{{name: b{j}, ref: {{name: e}}, with: {{j: {j}}}}}" for j in range(nb)
            )
            step = (
                f"    - {{name: {name},{needs}{extra} type: parallel, "
                f"with: {{steps: [{branches}]}}}}"
            )
        elif kind < 0.3 and deps:
            cond = f"steps.{deps[0]}.phase == 'Succeeded'"
            step = (
                f"    - {{name: {name},{needs}{extra} type: condition, "
                f"with: {{expression: \"{{{{ {cond} }}}}\"}}}}"
            )
        elif kind < 0.4:
            step = f"    - {{name: {name},{needs}{extra} type: sleep, with: {{duration: 1ms}}}}"
        else:
            iff = ""
            if deps and rng.random() < 0.3:
                iff = f" if: \"{{{{ steps.{deps[0]}.phase != 'Failed' }}}}\","
            step = (
                f"    - {{name: {name},{needs}{extra}{iff} ref: {{name: e}}, "
                f"with: {{i: {i}, v: \"{{{{ inputs.x + {i} }}}}\"}}}}"
            )
        lines.append(step)
    if rng.random() < 0.5:
        lines.append("  compensations:")
        lines.append("    - {name: undo, ref: {name: e}, with: {undo: true}}")
    if rng.random() < 0.5:
        lines.append("  finally:")
        lines.append("    - {name: fin, ref: {name: e}, with: {fin: true}}")
    return "\n".join(lines)


@pytest.mark.timeout(300)
def test_python_and_native_engines_agree():
    rng = random.Random(20260914)
    eng = RunEngine(EngineConfig(cpu_workers=4, default_max_retries=0)).start()
    try:
        eng.apply_yaml(RESOURCES)
        nr = NativeRunner.from_run_engine(eng)
        checked = 0
        for idx in range(100):
            yaml_text = _random_story(rng, idx)
            try:
                eng.apply_yaml(yaml_text)
            except ValueError:
                continue  # generator made an invalid story; skip
            story = eng.registry.story(f"diff-{idx}", "default")
            if story_supported(story) is not None:
                continue
            run = eng.run_story(f"default/diff-{idx}", {"x": 10}, timeout=60)
            py_phases = {k: str(v.phase) for k, v in run.step_states.items()}
            res = nr.run_story(f"default/diff-{idx}", {"x": 10}, timeout=60)
            nat_phases = {k: v["phase"] for k, v in res["steps"].items()}
            assert str(run.phase) == res["phase"], (
                idx, yaml_text, str(run.phase), res["phase"], py_phases, nat_phases,
            )
            run_failed = res["phase"] in ("Failed", "Compensated", "Canceled")
            stop_fired = "type: stop" in yaml_text
            for k in set(py_phases) | set(nat_phases):
                a, b = py_phases.get(k), nat_phases.get(k)
                if a == b:
                    continue
                # fail-fast race window: a step whose deps were satisfied may
                # launch (and then succeed OR fail on its own) before an
                # unrelated failure lands, or be skipped after — all legal,
                # in the reference too (findReadySteps skips PENDING steps
                # only); the RUN phase above is still compared strictly
                if run_failed and {a, b} <= {"Succeeded", "Skipped", "Failed", None}:
                    continue
                # stop race window: a stop directive races in-flight steps
                # and pending retries — their final states are timing-defined
                if stop_fired and {a, b} <= {"Succeeded", "Skipped", "Failed", None}:
                    continue
                raise AssertionError((idx, yaml_text, k, a, b, py_phases, nat_phases))
            checked += 1
        assert checked >= 70, f"only {checked} stories compared"
    finally:
        eng.stop()
