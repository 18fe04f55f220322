"""Template/expression engine tests (role parity with the reference's
templating engine usage: ResolveWithInputs / EvaluateCondition /
ResolveTemplateString, deterministic mode, output caps, offloaded-data
policy — SURVEY.md §2.6)."""
import pytest

from bobrapet_amd.enums import OffloadedDataPolicy
from bobrapet_amd.templating import (
    EvalConfig,
    EvaluationBlocked,
    Evaluator,
    OffloadedDataUsage,
    OutputTooLarge,
    TemplateSyntaxError,
    is_template,
    parse_expression,
)
from bobrapet_amd.templating.deps import (
    extract_referenced_steps,
    referenced_steps_of_value,
    resolve_aliases,
)


SCOPE = {
    "inputs": {"orderId": "o-1", "count": 3, "flag": True, "items": [1, 2, 3]},
    "steps": {
        "fetch-order": {"output": {"total": 42.5, "tags": ["a", "b"]}, "phase": "Succeeded"},
        "check": {"output": None, "phase": "Skipped"},
    },
    "run": {"name": "r1", "namespace": "default"},
}


@pytest.fixture
def ev():
    return Evaluator()


class TestResolve:
    def test_plain_string_passthrough(self, ev):
        assert ev.resolve_string("hello", SCOPE) == "hello"

    def test_single_expression_preserves_type(self, ev):
        assert ev.resolve_string("{{ inputs.count }}", SCOPE) == 3
        assert ev.resolve_string("{{ inputs.items }}", SCOPE) == [1, 2, 3]
        assert ev.resolve_string("{{ inputs.flag }}", SCOPE) is True

    def test_interpolation_stringifies(self, ev):
        assert ev.resolve_string("id={{ inputs.orderId }}!", SCOPE) == "id=o-1!"
        assert ev.resolve_string("n={{ inputs.count + 1 }}", SCOPE) == "n=4"

    def test_dashed_step_names_via_underscore_alias(self, ev):
        assert ev.resolve_string("{{ steps.fetch_order.output.total }}", SCOPE) == 42.5

    def test_bracket_step_access(self, ev):
        assert ev.resolve_string('{{ steps["fetch-order"].output.total }}', SCOPE) == 42.5

    def test_leading_dot_go_template_style(self, ev):
        assert ev.resolve_string("{{ .inputs.count }}", SCOPE) == 3

    def test_missing_becomes_null(self, ev):
        assert ev.resolve_string("{{ steps.ghost.output.x }}", SCOPE) is None
        assert ev.resolve_string("v={{ steps.ghost.output.x }}", SCOPE) == "v="

    def test_resolve_value_recurses(self, ev):
        value = {
            "id": "{{ inputs.orderId }}",
            "nested": {"n": "{{ inputs.count }}"},
            "list": ["{{ inputs.flag }}", "static"],
        }
        out = ev.resolve_value(value, SCOPE)
        assert out == {"id": "o-1", "nested": {"n": 3}, "list": [True, "static"]}

    def test_ternary_and_functions(self, ev):
        assert ev.resolve_string("{{ inputs.count > 2 ? 'big' : 'small' }}", SCOPE) == "big"
        assert ev.resolve_string("{{ size(inputs.items) }}", SCOPE) == 3
        assert ev.resolve_string("{{ has(inputs, 'orderId') }}", SCOPE) is True
        assert ev.resolve_string("{{ default(steps.ghost.output, 'dflt') }}", SCOPE) == "dflt"
        assert ev.resolve_string("{{ upper(inputs.orderId) }}", SCOPE) == "O-1"
        assert ev.resolve_string("{{ join(inputs.items, ',') }}", SCOPE) == "1,2,3"

    def test_method_calls(self, ev):
        assert ev.resolve_string("{{ inputs.items.size() }}", SCOPE) == 3
        assert ev.resolve_string("{{ inputs.orderId.startsWith('o-') }}", SCOPE) is True

    def test_arithmetic(self, ev):
        assert ev.resolve_string("{{ (1 + 2) * 3 }}", SCOPE) == 9
        assert ev.resolve_string("{{ 7 % 3 }}", SCOPE) == 1
        assert ev.resolve_string("{{ -inputs.count }}", SCOPE) == -3

    def test_string_concat(self, ev):
        assert ev.resolve_string("{{ 'a' + 'b' }}", SCOPE) == "ab"
        assert ev.resolve_string("{{ 'n=' + inputs.count }}", SCOPE) == "n=3"

    def test_list_and_map_literals(self, ev):
        assert ev.resolve_string("{{ [1, 2, inputs.count] }}", SCOPE) == [1, 2, 3]
        assert ev.resolve_string("{{ {a: 1, 'b': inputs.flag} }}", SCOPE) == {"a": 1, "b": True}


class TestConditions:
    def test_bool_expressions(self, ev):
        assert ev.evaluate_condition("inputs.count == 3", SCOPE)
        assert ev.evaluate_condition("inputs.count >= 2 && inputs.flag", SCOPE)
        assert not ev.evaluate_condition("inputs.count < 2 || !inputs.flag", SCOPE)
        assert ev.evaluate_condition("'a' in steps.fetch_order.output.tags", SCOPE)

    def test_template_style_condition(self, ev):
        assert ev.evaluate_condition("{{ steps.fetch_order.phase == 'Succeeded' }}", SCOPE)
        assert not ev.evaluate_condition("{{ steps.check.phase == 'Succeeded' }}", SCOPE)

    def test_missing_is_false(self, ev):
        assert not ev.evaluate_condition("steps.ghost.output.x", SCOPE)
        assert not ev.evaluate_condition("{{ steps.ghost.output.x }}", SCOPE)

    def test_null_comparisons(self, ev):
        assert ev.evaluate_condition("steps.check.output == null", SCOPE)
        assert not ev.evaluate_condition("steps.check.output != null", SCOPE)


class TestGuards:
    def test_deterministic_mode_blocks_now(self):
        ev = Evaluator(EvalConfig(deterministic=True))
        with pytest.raises(EvaluationBlocked):
            ev.resolve_string("{{ now() }}", SCOPE)
        ev2 = Evaluator(EvalConfig(deterministic=False))
        assert isinstance(ev2.resolve_string("{{ now() }}", SCOPE), float)

    def test_output_cap(self):
        ev = Evaluator(EvalConfig(max_output_bytes=64))
        with pytest.raises(OutputTooLarge):
            ev.resolve_value({"big": "x" * 100}, {})

    def test_allocation_bounded_before_materializing(self):
        # range()/'*'/'+' must be bounded at evaluation time,
        # not after allocation — these would OOM before _check_size runs
        ev = Evaluator(EvalConfig(max_output_bytes=1 << 20))
        with pytest.raises(OutputTooLarge):
            ev.resolve_string("{{ range(1000000000) }}", SCOPE)
        with pytest.raises(OutputTooLarge):
            ev.resolve_string("{{ 'abc' * 1000000000 }}", SCOPE)
        with pytest.raises(OutputTooLarge):
            ev.resolve_string("{{ 100000000 * 'abc' }}", SCOPE)
        # small allocations still work
        assert ev.resolve_string("{{ range(3) }}", SCOPE) == [0, 1, 2]
        assert ev.resolve_string("{{ 'ab' * 2 }}", SCOPE) == "abab"

    def test_op_budget(self):
        ev = Evaluator(EvalConfig(max_ops=10))
        from bobrapet_amd.templating.evaluator import EvaluationBudgetExceeded

        with pytest.raises(EvaluationBudgetExceeded):
            ev.resolve_string("{{ 1+1+1+1+1+1+1+1+1+1+1+1 }}", SCOPE)

    def test_syntax_error(self, ev):
        with pytest.raises(TemplateSyntaxError):
            ev.resolve_string("{{ inputs..x }}", SCOPE)

    def test_strict_mode(self):
        ev = Evaluator(EvalConfig(strict=True))
        from bobrapet_amd.templating import TemplateError

        with pytest.raises(TemplateError):
            ev.resolve_string("{{ steps.ghost.output }}", SCOPE)


class TestOffloadedPolicy:
    SCOPE_OFF = {
        "steps": {"big": {"output": {"$storageRef": {"key": "outputs/big", "size": 4096}}}}
    }

    def test_block_policy_raises(self):
        ev = Evaluator(EvalConfig(offloaded_policy=OffloadedDataPolicy.BLOCK))
        with pytest.raises(OffloadedDataUsage):
            ev.resolve_string("{{ steps.big.output.x }}", self.SCOPE_OFF)

    def test_inject_policy_hydrates(self):
        ev = Evaluator(
            EvalConfig(offloaded_policy=OffloadedDataPolicy.INJECT),
            hydrator=lambda ref: {"x": 99},
        )
        assert ev.resolve_string("{{ steps.big.output.x }}", self.SCOPE_OFF) == 99

    def test_ignore_policy_passes_ref_through(self):
        ev = Evaluator(EvalConfig(offloaded_policy=OffloadedDataPolicy.IGNORE))
        out = ev.resolve_string("{{ steps.big.output }}", self.SCOPE_OFF)
        assert out == {"$storageRef": {"key": "outputs/big", "size": 4096}}


class TestDeps:
    def test_extract_dot_form(self):
        assert extract_referenced_steps("{{ steps.alpha.output.x }}") == {"alpha"}

    def test_extract_bracket_and_index_forms(self):
        assert extract_referenced_steps('{{ steps["beta-1"].output }}') == {"beta-1"}
        assert extract_referenced_steps('{{ (index .steps "gamma") }}') == {"gamma"}

    def test_extract_from_value(self):
        value = {"a": "{{ steps.one.output }}", "b": ["{{ steps.two.phase }}"]}
        assert referenced_steps_of_value(value) == {"one", "two"}

    def test_alias_resolution(self):
        assert resolve_aliases({"fetch_order"}, {"fetch_order": "fetch-order"}) == {
            "fetch-order"
        }

    def test_is_template(self):
        assert is_template("{{ x }}")
        assert not is_template("plain")
        assert not is_template(42)


def test_parse_expression_ast_shape():
    ast = parse_expression("a.b == 1")
    assert ast[0] == "cmp"
