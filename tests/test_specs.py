"""Spec loading + validation (reference parity: api/v1alpha1 types and the
admission webhooks internal/webhook/v1alpha1/story_webhook.go)."""
import pytest

from bobrapet_amd.enums import StepType, StoryPattern
from bobrapet_amd.specs import (
    Story,
    load_yaml,
    dump_yaml,
    validate_story,
    validate_transport,
)
from bobrapet_amd.specs.yaml_loader import SpecLoadError


STORY_YAML = """
apiVersion: bubustack.io/v1alpha1
kind: Story
metadata:
  name: order-flow
  namespace: shop
spec:
  pattern: batch
  inputsSchema:
    type: object
    properties:
      orderId: {type: string}
    required: [orderId]
  steps:
    - name: fetch-order
      ref: {name: fetcher}
      with:
        id: "{{ inputs.orderId }}"
    - name: check
      type: condition
      needs: [fetch-order]
      with:
        expression: "{{ steps.fetch_order.output.total > 0 }}"
    - name: pause
      type: sleep
      with: {duration: 5ms}
    - name: finish-up
      ref: {name: shipper}
      needs: [check, pause]
      allowFailure: true
  compensations:
    - name: undo-ship
      ref: {name: shipper}
  finally:
    - name: report
      ref: {name: reporter}
  output:
    shipped: "{{ steps.finish_up.output.ok }}"
  policy:
    concurrency: 4
    queue: orders
    priority: 5
    timeouts: {story: 5m, step: 1m}
"""


def _load_story(text=STORY_YAML) -> Story:
    (story,) = load_yaml(text)
    return story


def test_story_yaml_roundtrip_fields():
    s = _load_story()
    assert s.name == "order-flow"
    assert s.namespace == "shop"
    assert s.pattern == StoryPattern.BATCH
    assert [st.name for st in s.steps] == ["fetch-order", "check", "pause", "finish-up"]
    assert s.steps[0].ref.name == "fetcher"
    assert s.steps[0].with_ == {"id": "{{ inputs.orderId }}"}
    assert s.steps[1].type == StepType.CONDITION
    assert s.steps[1].needs == ["fetch-order"]
    assert s.steps[3].allow_failure is True
    assert s.compensations[0].name == "undo-ship"
    assert s.finally_[0].name == "report"
    assert s.policy.concurrency == 4
    assert s.policy.queue == "orders"
    assert s.policy.timeouts.story == "5m"
    assert s.inputs_schema["required"] == ["orderId"]


def test_story_alias():
    s = _load_story()
    assert s.steps[0].alias == "fetch_order"


def test_story_validates():
    res = validate_story(_load_story())
    assert res.ok, res.errors


def test_dump_yaml_roundtrip():
    s = _load_story()
    text = dump_yaml(s)
    s2 = load_yaml(text)[0]
    assert [st.name for st in s2.steps] == [st.name for st in s.steps]
    assert s2.steps[0].with_ == s.steps[0].with_
    assert s2.policy.concurrency == 4
    assert s2.finally_[0].name == "report"


def test_unknown_kind_rejected():
    with pytest.raises(SpecLoadError):
        load_yaml("kind: Widget\nmetadata: {name: x}\n")


def test_step_shape_exactly_one_of_ref_type():
    bad = """
kind: Story
metadata: {name: bad}
spec:
  steps:
    - name: both
      type: sleep
      ref: {name: x}
      with: {duration: 1s}
    - name: neither
"""
    res = validate_story(load_yaml(bad)[0])
    assert not res.ok
    assert any("exactly one of" in e for e in res.errors)
    assert sum("exactly one of" in e for e in res.errors) == 2


def test_duplicate_names_rejected_across_phases():
    bad = """
kind: Story
metadata: {name: dup}
spec:
  steps:
    - {name: a, type: sleep, with: {duration: 1s}}
  compensations:
    - {name: a, ref: {name: x}}
"""
    res = validate_story(load_yaml(bad)[0])
    assert any("duplicate step name" in e for e in res.errors)


def test_needs_unknown_step_rejected():
    bad = """
kind: Story
metadata: {name: unknown-dep}
spec:
  steps:
    - {name: a, type: sleep, needs: [ghost], with: {duration: 1s}}
"""
    res = validate_story(load_yaml(bad)[0])
    assert any("unknown step" in e for e in res.errors)


def test_needs_cycle_rejected():
    bad = """
kind: Story
metadata: {name: cyclic}
spec:
  steps:
    - {name: a, type: sleep, needs: [b], with: {duration: 1s}}
    - {name: b, type: sleep, needs: [a], with: {duration: 1s}}
"""
    res = validate_story(load_yaml(bad)[0])
    assert any("cycle" in e for e in res.errors)


def test_compensations_may_reference_main_steps():
    ok = """
kind: Story
metadata: {name: comp-ref}
spec:
  steps:
    - {name: main-step, ref: {name: worker}}
  compensations:
    - {name: undo, ref: {name: worker}, needs: [main-step]}
"""
    res = validate_story(load_yaml(ok)[0])
    assert res.ok, res.errors


def test_batch_only_primitives_rejected_in_streaming():
    bad = """
kind: Story
metadata: {name: stream}
spec:
  pattern: streaming
  steps:
    - {name: w, type: wait, with: {until: "{{ true }}"}}
    - {name: g, type: gate}
"""
    res = validate_story(load_yaml(bad)[0])
    assert sum("batch-only" in e for e in res.errors) == 2


def test_primitive_with_shape_checks():
    bad = """
kind: Story
metadata: {name: bad-with}
spec:
  steps:
    - {name: s, type: sleep}
    - {name: w, type: wait, with: {onTimeout: explode}}
    - {name: e, type: executeStory}
    - {name: p, type: parallel, with: {steps: []}}
"""
    res = validate_story(load_yaml(bad)[0])
    msgs = "\n".join(res.errors)
    assert "sleep requires with.duration" in msgs
    assert "onTimeout" in msgs
    assert "executeStory requires with.storyRef" in msgs
    assert "parallel requires with.steps" in msgs


def test_undeclared_transport_rejected():
    bad = """
kind: Story
metadata: {name: tr}
spec:
  pattern: streaming
  steps:
    - {name: s, ref: {name: worker}, transport: fast-lane}
"""
    res = validate_story(load_yaml(bad)[0])
    assert any("undeclared transport" in e for e in res.errors)


def test_max_steps_cap():
    steps = "\n".join(
        f"    - {{name: s{i}, type: sleep, with: {{duration: 1s}}}}" for i in range(101)
    )
    bad = f"kind: Story\nmetadata: {{name: big}}\nspec:\n  steps:\n{steps}\n"
    res = validate_story(load_yaml(bad)[0])
    assert any("max is 100" in e for e in res.errors)


def test_transport_validation():
    (tr,) = load_yaml(
        """
kind: Transport
apiVersion: transport.bubustack.io/v1alpha1
metadata: {name: xgmi-fast}
spec:
  driver: xgmi
  streaming:
    lanes:
      - {name: media, priority: 0}
      - {name: data, priority: 1}
    flowControl: {mode: credit, initialCredits: 16, maxCredits: 64}
    delivery: {semantics: atLeastOnce, ordering: perLane}
"""
    )
    assert validate_transport(tr).ok
    tr.driver = "nvlink"
    assert not validate_transport(tr).ok


def test_requires_paths_checked():
    bad = """
kind: Story
metadata: {name: req}
spec:
  steps:
    - {name: a, ref: {name: w}}
    - {name: b, ref: {name: w}, requires: ["steps.ghost.output.x"], needs: [a]}
    - {name: c, ref: {name: w}, requires: ["steps.a.output.x", "inputs.z"]}
"""
    res = validate_story(load_yaml(bad)[0])
    assert len([e for e in res.errors if "requires path" in e]) == 1


def test_engram_and_template_yaml():
    docs = load_yaml(
        """
apiVersion: catalog.bubustack.io/v1alpha1
kind: EngramTemplate
metadata: {name: llm-infer}
spec:
  builtin: llm-infer
  supportedModes: [job, deployment]
  configSchema:
    type: object
    properties:
      model: {type: string}
---
apiVersion: bubustack.io/v1alpha1
kind: Engram
metadata: {name: llama, namespace: prod}
spec:
  templateRef: {name: llm-infer}
  with: {model: llama-3-8b}
  mode: job
"""
    )
    tpl, eng = docs
    assert tpl.builtin == "llm-infer"
    assert eng.template_ref.name == "llm-infer"
    assert eng.with_ == {"model": "llama-3-8b"}

    from bobrapet_amd.specs import validate_engram, validate_engram_template

    assert validate_engram_template(tpl).ok
    assert validate_engram(eng, tpl).ok
    eng.with_ = {"model": 42}
    assert not validate_engram(eng, tpl).ok


class TestTemplateSafety:
    """Apply-time template validation (reference: pkg/templatesafety
    ValidateTemplateString; webhook ValidateJSONTemplates)."""

    def test_bad_template_rejected_at_apply(self):
        from bobrapet_amd.specs import load_yaml, validation

        (story,) = load_yaml(
            """
kind: Story
metadata: {name: bad-tpl}
spec:
  steps:
    - {name: a, type: sleep, with: {duration: 1s, note: "{{ steps.a.output. }}"}}
"""
        )
        res = validation.validate_story(story)
        assert any("bad template" in e for e in res.errors)

    def test_bad_output_template_rejected(self):
        from bobrapet_amd.specs import load_yaml, validation

        (story,) = load_yaml(
            """
kind: Story
metadata: {name: bad-out}
spec:
  steps:
    - {name: a, type: sleep, with: {duration: 1s}}
  output: {v: "{{ 1 + }}"}
"""
        )
        res = validation.validate_story(story)
        assert any("story output" in e and "bad template" in e for e in res.errors)

    def test_good_templates_pass(self):
        from bobrapet_amd.specs import load_yaml, validation

        (story,) = load_yaml(
            """
kind: Story
metadata: {name: ok-tpl}
spec:
  steps:
    - {name: a, type: sleep, with: {duration: 1s}}
    - name: b
      type: condition
      needs: [a]
      if: "{{ steps.a.phase == 'Succeeded' }}"
      with: {expression: "{{ inputs.x > 3 && size(inputs.name) < 10 }}"}
  output: {v: "{{ steps.b.output.result }}"}
"""
        )
        res = validation.validate_story(story)
        assert not res.errors, res.errors


class TestContextVariableDiscipline:
    """with-block context rules per pattern (reference:
    story_webhook_test.go packet/steps/now() context tests)."""

    def _validate(self, yaml_text):
        from bobrapet_amd.specs import load_yaml
        from bobrapet_amd.specs.validation import validate_story

        (story,) = load_yaml(yaml_text)
        return validate_story(story)

    def test_rejects_packet_context_in_batch(self):
        res = self._validate("""
kind: Story
metadata: {name: b1}
spec:
  steps:
    - {name: a, ref: {name: x}, with: {value: "{{ packet.id }}"}}
""")
        assert any("packet" in e for e in res.errors), res.errors

    def test_rejects_steps_context_in_streaming_with(self):
        res = self._validate("""
kind: Story
metadata: {name: s1}
spec:
  pattern: streaming
  steps:
    - {name: a, ref: {name: x}}
    - {name: b, ref: {name: x}, needs: [a], with: {value: "{{ steps.a.output }}"}}
""")
        assert any("steps" in e for e in res.errors), res.errors

    def test_rejects_now_in_streaming_with(self):
        res = self._validate("""
kind: Story
metadata: {name: s2}
spec:
  pattern: streaming
  steps:
    - {name: a, ref: {name: x}, with: {t: "{{ now() }}"}}
""")
        assert any("now()" in e for e in res.errors), res.errors

    def test_allows_steps_context_in_batch_and_runtime_packet(self):
        res = self._validate("""
kind: Story
metadata: {name: ok1}
spec:
  steps:
    - {name: a, ref: {name: x}, with: {v: 1}}
    - {name: b, ref: {name: x}, needs: [a], with: {v: "{{ steps.a.output.v }}"}}
""")
        assert not any("context" in e or "packet" in e for e in res.errors), res.errors
        res2 = self._validate("""
kind: Story
metadata: {name: ok2}
spec:
  pattern: streaming
  steps:
    - {name: a, ref: {name: x}, runtime: {item: "{{ packet.id }}"}}
""")
        assert not res2.errors, res2.errors

    def test_warns_on_unknown_step_ref_in_with(self):
        res = self._validate("""
kind: Story
metadata: {name: w1}
spec:
  steps:
    - {name: a, ref: {name: x}, with: {v: "{{ steps.nosuch.output.v }}"}}
""")
        assert any("nosuch" in w for w in res.warnings), res.warnings
        assert not res.errors


class TestImpulseValidationGuards:
    def _impulse(self, yaml_text):
        from bobrapet_amd.specs import load_yaml
        from bobrapet_amd.specs.validation import validate_impulse

        (imp,) = load_yaml(yaml_text)
        return validate_impulse(imp)

    def test_rejects_cross_namespace_story_ref(self):
        res = self._impulse("""
kind: Impulse
metadata: {name: imp1, namespace: default}
spec:
  templateRef: {name: t}
  storyRef: {name: s, namespace: other}
""")
        assert any("namespace" in e for e in res.errors), res.errors

    def test_rejects_negative_throttle(self):
        res = self._impulse("""
kind: Impulse
metadata: {name: imp2}
spec:
  templateRef: {name: t}
  storyRef: {name: s}
  throttle: {maxInFlight: -1}
""")
        assert any("maxInFlight" in e for e in res.errors), res.errors

    def test_transport_rejects_plaintext_default_security(self):
        from bobrapet_amd.specs import load_yaml
        from bobrapet_amd.specs.validation import validate_transport

        (tr,) = load_yaml("""
kind: Transport
metadata: {name: t1}
spec:
  driver: xgmi
  defaultSettings:
    env: {BUBU_TRANSPORT_SECURITY_MODE: plaintext}
""")
        res = validate_transport(tr)
        assert any("plaintext" in e for e in res.errors), res.errors
