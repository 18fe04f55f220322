"""Utility layer tests: durations + minimal JSON Schema."""
import pytest

from bobrapet_amd.utils.durations import DurationError, format_duration, parse_duration
from bobrapet_amd.utils.jsonschema import apply_defaults, validate_instance


class TestDurations:
    def test_basic(self):
        assert parse_duration("1s") == 1.0
        assert parse_duration("300ms") == 0.3
        assert parse_duration("2m30s") == 150.0
        assert parse_duration("1.5h") == 5400.0
        assert parse_duration("1d") == 86400.0
        assert parse_duration("0") == 0.0

    def test_numbers_pass_through(self):
        assert parse_duration(5) == 5.0
        assert parse_duration(0.25) == 0.25
        assert parse_duration("42") == 42.0

    def test_none(self):
        assert parse_duration(None) is None

    def test_negative(self):
        assert parse_duration("-5s") == -5.0

    def test_malformed(self):
        with pytest.raises(DurationError):
            parse_duration("5x")
        with pytest.raises(DurationError):
            parse_duration("s5")

    def test_format(self):
        assert format_duration(0) == "0s"
        assert format_duration(150) == "2m30s"
        assert format_duration(5400) == "1h30m"
        assert parse_duration(format_duration(3723.5)) == pytest.approx(3723.5)


class TestJsonSchema:
    SCHEMA = {
        "type": "object",
        "required": ["name"],
        "properties": {
            "name": {"type": "string", "minLength": 1},
            "count": {"type": "integer", "minimum": 0, "default": 1},
            "tags": {"type": "array", "items": {"type": "string"}, "maxItems": 3},
            "mode": {"enum": ["a", "b"]},
        },
    }

    def test_valid(self):
        assert validate_instance({"name": "x", "count": 2, "tags": ["t"]}, self.SCHEMA) == []

    def test_missing_required(self):
        errs = validate_instance({}, self.SCHEMA)
        assert any("missing required" in e for e in errs)

    def test_wrong_type(self):
        errs = validate_instance({"name": 5}, self.SCHEMA)
        assert any("expected type" in e for e in errs)

    def test_enum(self):
        errs = validate_instance({"name": "x", "mode": "c"}, self.SCHEMA)
        assert any("enum" in e for e in errs)

    def test_bounds(self):
        errs = validate_instance({"name": "x", "count": -1}, self.SCHEMA)
        assert any("minimum" in e for e in errs)
        errs = validate_instance({"name": "x", "tags": ["a", "b", "c", "d"]}, self.SCHEMA)
        assert any("more than" in e for e in errs)

    def test_bool_is_not_integer(self):
        errs = validate_instance({"name": "x", "count": True}, self.SCHEMA)
        assert any("expected type" in e for e in errs)

    def test_defaults(self):
        assert apply_defaults({"name": "x"}, self.SCHEMA) == {"name": "x", "count": 1}
        # existing values win
        assert apply_defaults({"name": "x", "count": 7}, self.SCHEMA)["count"] == 7

    def test_nested_defaults(self):
        schema = {
            "type": "object",
            "properties": {
                "outer": {
                    "type": "object",
                    "properties": {"inner": {"default": "d"}},
                }
            },
        }
        assert apply_defaults({"outer": {}}, schema) == {"outer": {"inner": "d"}}


class TestMetricsParitySeries:
    """The reference's meaningful bobrapet_* series have counterparts
    (reference: pkg/metrics/controller_metrics.go — queue depth/age,
    dag_iteration_steps, steps active/completed, cel eval counters,
    child stepruns, cleanup)."""

    def test_series_emitted_by_engine_run(self):
        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.enums import Phase

        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        try:
            eng.apply_yaml(
                """
kind: EngramTemplate
metadata: {name: echo}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: echoer}
spec: {templateRef: {name: echo}}
---
kind: Story
metadata: {name: m}
spec:
  steps:
    - name: par
      type: parallel
      with:
        steps:
          - {name: a, ref: {name: echoer}, with: {v: "{{ inputs.x }}"}}
          - {name: b, ref: {name: echoer}, with: {v: 2}}
"""
            )
            run = eng.run_story("default/m", {"x": 1}, timeout=30)
            assert run.phase == Phase.SUCCEEDED, run.error
            text = eng.metrics.export_text()
            for series in (
                "storyruns_total",
                "stepruns_total",
                "dag_steps_launched_total",
                "dag_iteration_steps",
                "storyrun_steps_completed",
                "storyrun_queue_depth",
                "storyrun_queue_age_seconds",
                "template_evaluations_total",
                "child_stepruns_created_total",
            ):
                assert series in text, f"missing series {series}\n{text[:800]}"
        finally:
            eng.stop()


class TestStructuredLogging:
    def test_json_lines_and_feature_gating(self):
        import io
        import json as _json

        from bobrapet_amd.utils.logging import (
            ContractLogger,
            StructuredLogger,
            enable_feature,
        )

        buf = io.StringIO()
        log = StructuredLogger("test", stream=buf, run="r1")
        log.info("hello", x=1)
        log.debug("hidden", feature="verbose-x")
        enable_feature("verbose-x")
        log.with_fields(step="a").debug("shown", feature="verbose-x")
        enable_feature("verbose-x", False)
        lines = [_json.loads(l) for l in buf.getvalue().splitlines()]
        assert lines[0]["msg"] == "hello" and lines[0]["run"] == "r1"
        assert len(lines) == 2 and lines[1]["step"] == "a"
        cl = ContractLogger("boot", log)
        cl.stage("load", detail="ok")
        assert cl.stages[0]["stage"] == "load"

    def test_engine_warns_on_failed_run(self, capsys):
        import io

        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.enums import Phase

        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        try:
            buf = io.StringIO()
            eng.log.stream = buf
            eng.apply_yaml(
                """
kind: EngramTemplate
metadata: {name: fail}
spec: {builtin: fail}
---
kind: Engram
metadata: {name: f}
spec: {templateRef: {name: fail}}
---
kind: Story
metadata: {name: boom}
spec:
  steps: [{name: a, ref: {name: f}, with: {succeedAfter: 99}}]
"""
            )
            run = eng.run_story("default/boom", {}, timeout=30)
            assert run.phase == Phase.FAILED
            assert '"level": "warn"' in buf.getvalue()
        finally:
            eng.stop()


class TestOTLPExport:
    """OTLP span export (reference: pkg/observability/exporter.go)."""

    def test_engine_exports_spans_to_file(self, tmp_path):
        import json

        from bobrapet_amd.engine import EngineConfig, RunEngine

        path = str(tmp_path / "spans.otlp.jsonl")
        eng = RunEngine(EngineConfig(cpu_workers=1, otlp_endpoint=path)).start()
        try:
            eng.apply_yaml(
                """
kind: EngramTemplate
metadata: {name: echo-tpl}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: e}
spec: {templateRef: {name: echo-tpl}}
---
kind: Story
metadata: {name: s}
spec:
  steps: [{name: a, ref: {name: e}, with: {v: 1}}]
"""
            )
            eng.run_story("default/s", {}, timeout=10)
        finally:
            eng.stop()
        doc = json.loads(open(path).read().splitlines()[0])
        spans = doc["resourceSpans"][0]["scopeSpans"][0]["spans"]
        names = {s["name"] for s in spans}
        assert "engram.run" in names and "dag.tick" in names
        one = spans[0]
        assert len(one["traceId"]) == 32 and len(one["spanId"]) == 16
        assert int(one["endTimeUnixNano"]) >= int(one["startTimeUnixNano"])
