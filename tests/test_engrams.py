"""Direct unit tests for the builtin engram library (CPU paths) + the CLI."""
import pytest

from bobrapet_amd.engrams.base import EngramContext, EngramFailure
from bobrapet_amd.engrams import registry


def _ctx(**kw):
    return EngramContext(**kw)


class TestFilterTransform:
    def test_filter_where(self):
        impl = registry.resolve("filter")
        out = impl.run(
            _ctx(
                input={"items": [{"s": 1}, {"s": 5}, {"s": 3}]},
                runtime={"where": "{{ item.s >= 3 }}"},
            )
        ).output
        assert out == {"items": [{"s": 5}, {"s": 3}], "count": 2}

    def test_transform_map(self):
        impl = registry.resolve("transform")
        out = impl.run(
            _ctx(
                input={"items": [{"x": 2}, {"x": 3}]},
                runtime={"map": {"y": "{{ item.x * item.x }}", "i": "{{ index }}"}},
            )
        ).output
        assert out["items"] == [{"y": 4, "i": 0}, {"y": 9, "i": 1}]

    def test_filter_requires_list(self):
        impl = registry.resolve("filter")
        with pytest.raises(EngramFailure):
            impl.run(_ctx(input={"items": "nope"}))


class TestEchoFail:
    def test_echo_merges_config(self):
        impl = registry.resolve("echo")
        out = impl.run(_ctx(input={"a": 1}, config={"b": 2, "a": 0})).output
        assert out == {"a": 1, "b": 2}

    def test_fail_succeed_after(self):
        impl = registry.resolve("fail")
        ctx = _ctx(input={"succeedAfter": 1}, story_run="r", step_name="s")
        with pytest.raises(EngramFailure):
            impl.run(ctx)
        assert impl.run(ctx).output == {"attempts": 2}

    def test_unknown_engram(self):
        with pytest.raises(registry.UnknownEngram):
            registry.resolve("no-such-engram")


class TestEngramContextContract:
    def test_signals_are_seq_ordered(self):
        ctx = _ctx()
        ctx.emit_signal("a", 1)
        ctx.emit_signal("b", 2)
        assert [s.seq for s in ctx.signals] == [1, 2]
        assert [s.name for s in ctx.signals] == ["a", "b"]

    def test_effect_guard_blocks_duplicates(self):
        calls = []

        def guard(key, desc):
            calls.append(key)
            return len(calls) == 1  # only the first acquire is fresh

        ctx = _ctx(effect_guard=guard)
        assert ctx.record_effect("k1") is True
        assert ctx.record_effect("k1") is False
        assert len(ctx.effects) == 1

    def test_cancel_check(self):
        ctx = _ctx(cancel_check=lambda: True)
        assert ctx.canceled


class TestCli:
    def test_validate_and_run(self, tmp_path):
        from typer.testing import CliRunner

        from bobrapet_amd.cli import app

        runner = CliRunner()
        res = runner.invoke(app, ["validate", "-f", "examples/order-flow.yaml"])
        assert res.exit_code == 0, res.output
        res = runner.invoke(
            app,
            ["run", "-f", "examples/order-flow.yaml", "--inputs", '{"orderId": "o-9"}', "--json"],
        )
        assert res.exit_code == 0, res.output
        assert '"confirmed": true' in res.output

    def test_validate_rejects_bad_story(self, tmp_path):
        from typer.testing import CliRunner

        from bobrapet_amd.cli import app

        bad = tmp_path / "bad.yaml"
        bad.write_text(
            """
kind: Story
metadata: {name: bad}
spec:
  steps:
    - {name: a, type: sleep, needs: [ghost], with: {duration: 1s}}
"""
        )
        runner = CliRunner()
        res = runner.invoke(app, ["validate", "-f", str(bad)])
        assert res.exit_code == 1
        assert "unknown step" in res.output

    def test_engrams_list(self):
        from typer.testing import CliRunner

        from bobrapet_amd.cli import app

        res = CliRunner().invoke(app, ["engrams"])
        assert "llm-infer" in res.output and "embed" in res.output


class TestProcessEngram:
    """External-process engrams: the bubu-sdk BUBU_* env contract as real
    env vars (reference: steprun_controller.go:1692-1732 env build +
    exit-code classes)."""

    YAML = """
kind: EngramTemplate
metadata: {name: sh}
spec:
  command: [python3, -c, "%s"]
---
kind: Engram
metadata: {name: sheller}
spec: {templateRef: {name: sh}}
---
kind: Story
metadata: {name: ext}
spec:
  steps:
    - {name: go, ref: {name: sheller}, with: {v: 42}}
  output: {out: "{{ steps.go.output }}"}
"""

    def _run(self, script, expect_phase="Succeeded", retries=0):
        from bobrapet_amd.engine import EngineConfig, RunEngine

        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        try:
            eng.apply_yaml(self.YAML % script)
            run = eng.run_story("default/ext", {}, timeout=40)
            srs = list(eng.store.step_runs_of(run.key))
            return run, srs[0] if srs else None
        finally:
            eng.stop()

    def test_env_contract_and_json_output(self):
        run, sr = self._run(
            "import os, json; print('log line');"
            " print(json.dumps({'step': os.environ['BUBU_STEP_NAME'],"
            " 'v': json.loads(os.environ['BUBU_TRIGGER_DATA'])['v']}))"
        )
        assert str(run.phase) == "Succeeded", run.error
        assert run.output["out"] == {"step": "go", "v": 42}
        assert any("log line" in l for l in sr.status.logs)

    def test_exit_two_is_terminal_no_retry(self):
        run, sr = self._run("import sys; sys.exit(2)")
        assert str(run.phase) == "Failed"
        assert sr.status.retries == 0
        assert "exited 2" in (sr.status.error.message or "")

    def test_non_json_stdout_wrapped(self):
        run, _sr = self._run("print('plain text result')")
        assert str(run.phase) == "Succeeded"
        assert run.output["out"] == {"stdout": "plain text result"}

    def test_process_timeout_is_retryable(self):
        """A process exceeding its step timeout maps to exit class retry
        (the reference kills the Job pod and retries per policy)."""
        from bobrapet_amd.engrams.base import EngramContext, EngramFailure
        from bobrapet_amd.engrams.process import ProcessEngram

        impl = ProcessEngram(["python3", "-c", "import time; time.sleep(30)"])
        ctx = EngramContext(input={}, timeout_seconds=0.5)
        import pytest as _p

        with _p.raises(EngramFailure) as e:
            impl.run(ctx)
        assert e.value.exit_code == 1  # retryable

    def test_missing_binary_is_terminal(self):
        from bobrapet_amd.engrams.base import EngramContext, EngramFailure
        from bobrapet_amd.engrams.process import ProcessEngram

        impl = ProcessEngram(["/no/such/binary"])
        import pytest as _p

        with _p.raises(EngramFailure) as e:
            impl.run(EngramContext(input={}))
        assert e.value.exit_code == 2  # terminal


class TestSecretDelivery:
    """Secret artifacts for process engrams (reference:
    pkg/podspec/secrets.go + SecretDefinition shared_types.go:296-322):
    env-prefix and file mounts per the template's SecretDefinitions, with
    required-secret enforcement."""

    YAML = """
kind: EngramTemplate
metadata: {name: sec-tpl}
spec:
  command: [python3, -c, "%s"]
  secrets:
    - {name: api-key, required: true, mountType: env}
    - {name: cert, mountType: file}
    - {name: optional-thing, mountType: env}
---
kind: Engram
metadata: {name: secret-user}
spec:
  templateRef: {name: sec-tpl}
  secrets: {api-key: "sk-12345", cert: "---CERT---"}
---
kind: Story
metadata: {name: sec-story}
spec:
  steps:
    - {name: go, ref: {name: secret-user}}
  output: {out: "{{ steps.go.output }}"}
"""

    def _run(self, script, yaml=None):
        from bobrapet_amd.engine import EngineConfig, RunEngine

        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        try:
            eng.apply_yaml((yaml or self.YAML) % script)
            return eng.run_story("default/sec-story", {}, timeout=40)
        finally:
            eng.stop()

    def test_env_and_file_mounts(self):
        run = self._run(
            "import os, json;"
            " print(json.dumps({'key': os.environ['BUBU_SECRET_API_KEY'],"
            " 'cert': open(os.environ['BUBU_SECRET_FILE_CERT']).read(),"
            " 'missing': 'BUBU_SECRET_OPTIONAL_THING' in os.environ}))"
        )
        assert str(run.phase) == "Succeeded", run.error
        assert run.output["out"] == {
            "key": "sk-12345", "cert": "---CERT---", "missing": False,
        }

    def test_secret_file_cleaned_up(self):
        run = self._run(
            "import os, json; print(json.dumps({'path': os.environ['BUBU_SECRET_FILE_CERT']}))"
        )
        import os
        assert not os.path.exists(run.output["out"]["path"])

    def test_missing_required_secret_is_terminal(self):
        yaml = self.YAML.replace('secrets: {api-key: "sk-12345", cert: "---CERT---"}',
                                 'secrets: {cert: "x"}')
        run = self._run("print('never runs')", yaml=yaml)
        assert str(run.phase) == "Failed"
        assert "api-key" in (run.error.message if run.error else str(run))
