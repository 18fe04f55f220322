"""Submit-time validation of spec objects.

Parity with the reference's admission webhooks
(reference: internal/webhook/v1alpha1/story_webhook.go:290-404 and
engram_webhook.go / impulse_webhook.go / transport_webhook.go): step shape
(exactly one of ref/type), name rules, unique names, needs existence +
acyclicity per phase, requires paths, size caps, batch-only primitives,
transports declared, template config-schema checks.

Returns a ValidationResult (errors + warnings) instead of raising, so the
engine can surface ValidationStatus like the reference's Story status does.
"""
from __future__ import annotations

import json
import re
import typing as _t
from dataclasses import dataclass, field

from ..enums import BATCH_ONLY_STEP_TYPES, StepType, StoryPattern
from ..templating import TemplateSyntaxError, deps as tdeps, is_template, parse_template
from . import types as T

MAX_STORY_BYTES = 1 << 20  # 1 MiB total Story cap (story_webhook.go:418-428)
MAX_BLOCK_BYTES = 256 << 10  # per-step with/output cap (story_webhook.go:430-468)

_NAME_RE = re.compile(r"^[a-z0-9]([-a-z0-9]*[a-z0-9])?$")  # DNS-1123 label
_MAX_NAME_LEN = 63


@dataclass
class ValidationResult:
    errors: _t.List[str] = field(default_factory=list)
    warnings: _t.List[str] = field(default_factory=list)

    @property
    def ok(self) -> bool:
        return not self.errors

    def error(self, msg: str) -> None:
        self.errors.append(msg)

    def warn(self, msg: str) -> None:
        self.warnings.append(msg)

    def raise_if_invalid(self) -> None:
        if self.errors:
            raise SpecValidationError(self.errors, self.warnings)


class SpecValidationError(ValueError):
    def __init__(self, errors: _t.List[str], warnings: _t.Optional[_t.List[str]] = None):
        self.errors = list(errors)
        self.warnings = list(warnings or [])
        super().__init__("; ".join(self.errors))


def _json_size(value) -> int:
    try:
        return len(json.dumps(value, separators=(",", ":"), default=str))
    except (TypeError, ValueError):
        return 0


def _valid_name(name: str) -> bool:
    return bool(name) and len(name) <= _MAX_NAME_LEN and bool(_NAME_RE.match(name))


# ---------------------------------------------------------------------------
# Story
# ---------------------------------------------------------------------------


def validate_story(story: T.Story) -> ValidationResult:
    res = ValidationResult()
    if not _valid_name(story.name):
        res.error(f"story name {story.name!r} must be a DNS-1123 label (<=63 chars)")

    if _json_size(T.to_dict(story)) > MAX_STORY_BYTES:
        res.error("story spec exceeds the 1 MiB size cap")

    if not story.steps:
        res.error("story must declare at least one step")
    if len(story.steps) > T.MAX_STEPS:
        res.error(f"story has {len(story.steps)} steps; max is {T.MAX_STEPS}")
    if len(story.compensations) > T.MAX_COMPENSATIONS:
        res.error(f"too many compensations ({len(story.compensations)} > {T.MAX_COMPENSATIONS})")
    if len(story.finally_) > T.MAX_FINALLY:
        res.error(f"too many finally steps ({len(story.finally_)} > {T.MAX_FINALLY})")

    main_names = _validate_step_list(res, story.steps, "steps", story)
    comp_names = _validate_step_list(res, story.compensations, "compensations", story)
    fin_names = _validate_step_list(res, story.finally_, "finally", story)

    # Names must be unique across all three phases.
    seen: _t.Dict[str, str] = {}
    for phase, names in (("steps", main_names), ("compensations", comp_names), ("finally", fin_names)):
        for n in names:
            if n in seen:
                res.error(f"duplicate step name {n!r} ({seen[n]} and {phase})")
            else:
                seen[n] = phase

    # needs existence + acyclicity per phase; compensations/finally may also
    # reference main-phase steps (story_webhook.go:332-360).
    _check_graph(res, story.steps, set(main_names), "steps")
    _check_graph(res, story.compensations, set(comp_names) | set(main_names), "compensations")
    _check_graph(res, story.finally_, set(fin_names) | set(main_names), "finally")

    # requires paths must reference known steps (story_webhook.go:362-365)
    known = set(seen)
    aliases = {s.alias: s.name for s in story.all_steps()}
    for s in story.all_steps():
        for path in s.requires:
            root = path.split(".")[0]
            if root in ("inputs", "run", "story"):
                continue
            head = path.split(".")
            if root == "steps" and len(head) >= 2:
                target = head[1]
                if target not in known and target not in aliases:
                    res.error(f"step {s.name!r}: requires path {path!r} references unknown step")
            elif root not in known and root not in aliases:
                res.error(f"step {s.name!r}: requires path {path!r} must start with steps./inputs.")

    # batch-only primitives rejected in streaming stories (story_webhook.go:564-576)
    if story.pattern == StoryPattern.STREAMING:
        for s in story.all_steps():
            if s.type in BATCH_ONLY_STEP_TYPES:
                res.error(
                    f"step {s.name!r}: primitive {s.type} is batch-only and not "
                    f"allowed in a streaming story"
                )
        # routing.maxDownstreams guardrail (reference:
        # TransportRoutingSettings.MaxDownstreams) — enforced at apply time
        for t in story.transports:
            streaming = t.streaming if isinstance(t.streaming, dict) else None
            cap = (
                (streaming.get("routing") or {}).get("maxDownstreams")
                if streaming
                else None
            )
            if not cap:
                continue
            downstream: _t.Dict[str, int] = {}
            for s2 in story.steps:
                for dep in s2.needs:
                    downstream[dep] = downstream.get(dep, 0) + 1
            for name2, n2 in downstream.items():
                if n2 > int(cap):
                    res.error(
                        f"step {name2!r} has {n2} downstream steps, over "
                        f"routing.maxDownstreams={cap}"
                    )

    # context-variable discipline in `with` blocks (reference:
    # story_webhook_test.go — "rejects packet context in batch with
    # blocks", "rejects steps context in streaming with blocks",
    # "rejects now() in streaming with blocks"; step `runtime` templates
    # are the per-packet surface and stay exempt)
    _BLOCK_RE = re.compile(r"\{\{(.*?)\}\}", re.S)
    for s in story.all_steps():
        if not s.with_:
            continue
        blocks = " ".join(_BLOCK_RE.findall(json.dumps(s.with_)))
        if story.pattern == StoryPattern.STREAMING:
            if re.search(r"\bsteps\.", blocks):
                res.error(
                    f"step {s.name!r}: `steps.` context is not available in "
                    f"streaming `with` blocks (packets flow, steps do not finish)"
                )
            if "now(" in blocks:
                res.error(
                    f"step {s.name!r}: now() is non-deterministic per packet "
                    f"and not allowed in streaming `with` blocks"
                )
        else:
            if re.search(r"\bpacket\.", blocks):
                res.error(
                    f"step {s.name!r}: `packet.` context only exists in "
                    f"streaming stories"
                )

    # declared transports must exist in the story's transport list
    declared = {t.name for t in story.transports}
    for s in story.all_steps():
        if s.transport and s.transport not in declared:
            res.error(f"step {s.name!r} references undeclared transport {s.transport!r}")

    # output template refs reachability warning (story_webhook.go:306-312)
    if story.output is not None:
        for ref in tdeps.extract_referenced_steps(json.dumps(story.output, default=str)):
            if ref not in known and ref not in aliases:
                res.warn(f"story output references unknown step {ref!r}")

    # per-step template refs to unknown steps (reference: "warns on unknown
    # step references in template expressions"); known refs become implicit
    # deps at runtime, unknown ones can never resolve
    for s in story.all_steps():
        for blob in (s.with_, s.if_, getattr(s, "requires", None)):
            if blob is None:
                continue
            for ref in tdeps.extract_referenced_steps(json.dumps(blob, default=str)):
                if ref not in known and ref not in aliases:
                    res.warn(
                        f"step {s.name!r} references unknown step {ref!r} in a "
                        f"template expression"
                    )

    # schemas must be JSON-schema-shaped mappings (story_webhook.go:321-330)
    for label, schema in (("inputsSchema", story.inputs_schema), ("outputsSchema", story.outputs_schema)):
        if schema is not None and not isinstance(schema, dict):
            res.error(f"{label} must be a JSON Schema object")

    # template safety: every template string must parse at apply time
    # (reference: pkg/templatesafety ValidateTemplateString + webhook-side
    # ValidateJSONTemplates — fail at apply, not mid-run)
    def _walk_templates(where: str, value) -> None:
        if isinstance(value, str):
            if is_template(value):
                try:
                    parse_template(value)
                except TemplateSyntaxError as exc:
                    res.error(f"{where}: bad template: {exc}")
        elif isinstance(value, dict):
            for k, v in value.items():
                _walk_templates(f"{where}.{k}", v)
        elif isinstance(value, list):
            for i, v in enumerate(value):
                _walk_templates(f"{where}[{i}]", v)

    for s in story.all_steps():
        _walk_templates(f"step {s.name!r} with", s.with_)
        _walk_templates(f"step {s.name!r} runtime", s.runtime)
        if s.if_:
            _walk_templates(f"step {s.name!r} if", s.if_)
        if s.idempotency_key_template:
            _walk_templates(f"step {s.name!r} idempotencyKey", s.idempotency_key_template)
    _walk_templates("story output", story.output)

    return res


def _validate_step_list(
    res: ValidationResult, steps: _t.List[T.Step], phase: str, story: T.Story
) -> _t.List[str]:
    names = []
    for s in steps:
        if not _valid_name(s.name):
            res.error(f"{phase}: step name {s.name!r} must be a DNS-1123 label")
        names.append(s.name)
        has_ref = s.ref is not None and bool(s.ref.name)
        has_type = s.type is not None
        if has_ref == has_type:
            res.error(
                f"step {s.name!r}: exactly one of 'ref' (engram) or 'type' "
                f"(primitive) must be set"
            )
        if s.with_ is not None and _json_size(s.with_) > MAX_BLOCK_BYTES:
            res.error(f"step {s.name!r}: 'with' block exceeds {MAX_BLOCK_BYTES} bytes")
        if has_type:
            _validate_primitive_with(res, s, story)
    return names


def _validate_primitive_with(res: ValidationResult, step: T.Step,
                             story: _t.Optional[T.Story] = None) -> None:
    """Shape checks for primitive `with` schemas (reference: dag.go:1549-1668,
    step_executor.go:1084-1107)."""
    w = step.with_ if isinstance(step.with_, dict) else {}
    st = step.type
    if st == StepType.SLEEP:
        if "duration" not in w:
            res.error(f"step {step.name!r}: sleep requires with.duration")
    elif st == StepType.WAIT:
        if "until" not in w:
            res.error(f"step {step.name!r}: wait requires with.until (template)")
        if w.get("onTimeout") not in (None, "fail", "skip"):
            res.error(f"step {step.name!r}: wait onTimeout must be fail|skip")
    elif st == StepType.GATE:
        if w.get("onTimeout") not in (None, "fail", "skip"):
            res.error(f"step {step.name!r}: gate onTimeout must be fail|skip")
    elif st == StepType.STOP:
        if w.get("phase") not in (None, "Succeeded", "Failed", "Finished") and w.get(
            "mode"
        ) not in (None, "success", "failure", "cancel"):
            res.error(f"step {step.name!r}: stop phase/mode invalid")
    elif st == StepType.EXECUTE_STORY:
        target = w.get("storyRef") or w.get("story")
        if not target:
            res.error(f"step {step.name!r}: executeStory requires with.storyRef")
        else:
            # self-reference and cross-namespace rejection at apply time
            # (reference: story_webhook_test.go "rejects executeStory steps
            # that reference the same story" / "...another namespace")
            t_ns = (w.get("namespace") or story.namespace) if story else None
            t_name = target
            if isinstance(target, dict):
                t_ns = target.get("namespace") or t_ns
                t_name = target.get("name")
            if story is not None and t_name == story.name and t_ns == story.namespace:
                res.error(
                    f"step {step.name!r}: executeStory must not reference its "
                    f"own story (unbounded recursion)"
                )
            if story is not None and t_ns != story.namespace:
                res.error(
                    f"step {step.name!r}: executeStory may not reference a "
                    f"story in another namespace ({t_ns!r})"
                )
    elif st == StepType.PARALLEL:
        branches = w.get("steps")
        if not isinstance(branches, list) or not branches:
            res.error(f"step {step.name!r}: parallel requires with.steps (non-empty list)")
        else:
            seen = set()
            for i, b in enumerate(branches):
                if not isinstance(b, dict) or not b.get("name"):
                    res.error(f"step {step.name!r}: parallel branch {i} must have a name")
                    continue
                if b["name"] in seen:
                    res.error(f"step {step.name!r}: duplicate parallel branch {b['name']!r}")
                seen.add(b["name"])
    elif st == StepType.CONDITION:
        if not (w.get("expression") or w.get("if") or step.if_ or step.runtime):
            res.error(f"step {step.name!r}: condition requires with.expression")


def _check_graph(
    res: ValidationResult, steps: _t.List[T.Step], known: _t.Set[str], phase: str
) -> None:
    """needs existence + cycle detection (story_webhook.go:332-360).

    Only explicit `needs` edges participate in the webhook-level cycle check;
    template-implied deps are validated at run time like the reference does
    (dag.go:3076)."""
    adj: _t.Dict[str, _t.List[str]] = {}
    for s in steps:
        for dep in s.needs:
            if dep not in known:
                res.error(f"{phase}: step {s.name!r} needs unknown step {dep!r}")
            if dep == s.name:
                res.error(f"{phase}: step {s.name!r} needs itself")
            adj.setdefault(s.name, []).append(dep)

    # Kahn's algorithm over this phase's steps only.
    local = {s.name for s in steps}
    indeg = {n: 0 for n in local}
    for name, ds in adj.items():
        indeg[name] = sum(1 for d in ds if d in local)
    queue = sorted(n for n, d in indeg.items() if d == 0)
    dependents: _t.Dict[str, _t.List[str]] = {}
    for name, ds in adj.items():
        for d in ds:
            if d in local:
                dependents.setdefault(d, []).append(name)
    visited = 0
    while queue:
        cur = queue.pop()
        visited += 1
        for nxt in dependents.get(cur, []):
            indeg[nxt] -= 1
            if indeg[nxt] == 0:
                queue.append(nxt)
    if visited < len(local):
        cyc = sorted(n for n, d in indeg.items() if d > 0)
        res.error(f"{phase}: dependency cycle among {cyc}")


# ---------------------------------------------------------------------------
# Engram / Impulse / Transport / templates
# ---------------------------------------------------------------------------


def validate_engram(
    engram: T.Engram, template: _t.Optional[T.EngramTemplate] = None
) -> ValidationResult:
    res = ValidationResult()
    if not _valid_name(engram.name):
        res.error(f"engram name {engram.name!r} must be a DNS-1123 label")
    if engram.template_ref is None or not engram.template_ref.name:
        res.error("engram.templateRef is required")
    if template is not None:
        if engram.mode and template.supported_modes and engram.mode not in template.supported_modes:
            res.error(
                f"engram mode {engram.mode} not in template supportedModes "
                f"{[str(m) for m in template.supported_modes]}"
            )
        if template.config_schema is not None:
            from ..utils.jsonschema import validate_instance

            for err in validate_instance(engram.with_ or {}, template.config_schema):
                res.error(f"engram.with: {err}")
        if template.secret_schema is not None:
            from ..utils.jsonschema import validate_instance

            # secret VALUES are env/file indirections; the schema governs
            # which keys must exist (reference: engram_webhook secretSchema)
            for err in validate_instance(dict(engram.secrets or {}), template.secret_schema):
                res.error(f"engram.secrets: {err}")
    return res


def validate_impulse(
    impulse: T.Impulse, template: _t.Optional[T.ImpulseTemplate] = None
) -> ValidationResult:
    res = ValidationResult()
    if not _valid_name(impulse.name):
        res.error(f"impulse name {impulse.name!r} must be a DNS-1123 label")
    if impulse.template_ref is None or not impulse.template_ref.name:
        res.error("impulse.templateRef is required")
    if impulse.story_ref is None or not impulse.story_ref.name:
        res.error("impulse.storyRef is required")
    # cross-namespace story references rejected (impulse_webhook_test.go)
    if (impulse.story_ref is not None and impulse.story_ref.namespace
            and impulse.story_ref.namespace != impulse.namespace):
        res.error(
            f"impulse.storyRef may not reference a story in another "
            f"namespace ({impulse.story_ref.namespace!r})"
        )
    # throttle sanity (impulse_webhook_test.go: negative maxInFlight
    # rejected; zero retry delays are fine elsewhere)
    th = impulse.throttle
    if th is not None:
        if th.max_in_flight is not None and th.max_in_flight < 0:
            res.error("impulse.throttle.maxInFlight must be >= 0")
        if th.rate_per_second is not None and th.rate_per_second < 0:
            res.error("impulse.throttle.ratePerSecond must be >= 0")
        if th.burst is not None and th.burst < 0:
            res.error("impulse.throttle.burst must be >= 0")
    if template is not None and template.config_schema is not None:
        from ..utils.jsonschema import validate_instance

        for err in validate_instance(impulse.with_ or {}, template.config_schema):
            res.error(f"impulse.with: {err}")
    return res


def validate_transport(transport: T.Transport) -> ValidationResult:
    res = ValidationResult()
    if not _valid_name(transport.name):
        res.error(f"transport name {transport.name!r} must be a DNS-1123 label")
    if transport.driver not in T.KNOWN_TRANSPORT_DRIVERS:
        res.error(
            f"unknown transport driver {transport.driver!r} "
            f"(known: {list(T.KNOWN_TRANSPORT_DRIVERS)})"
        )
    # plaintext default security mode rejected (reference:
    # transport_webhook_test.go RejectsPlaintextDefaultSecurityMode — the
    # default wiring must never downgrade the wire to plaintext; an
    # explicit per-binding override remains possible)
    ds = transport.extra.get("defaultSettings")
    if isinstance(ds, dict):
        env = ds.get("env")
        if isinstance(env, dict) and str(
            env.get("BUBU_TRANSPORT_SECURITY_MODE", "")
        ).lower() == "plaintext":
            res.error(
                "transport defaultSettings must not set "
                "BUBU_TRANSPORT_SECURITY_MODE=plaintext"
            )
    s = transport.streaming
    if s is not None:
        lane_names = [l.name for l in s.lanes]
        if len(lane_names) != len(set(lane_names)):
            res.error("transport streaming lanes must have unique names")
        fc = s.flow_control
        if fc is not None and fc.mode not in (None, "credit", "none"):
            res.error(f"flowControl.mode must be credit|none, got {fc.mode!r}")
        if fc is not None and fc.mode == "credit":
            if (fc.initial_credits or 0) <= 0:
                res.error("flowControl.initialCredits must be > 0 in credit mode")
            if fc.max_credits is not None and fc.initial_credits is not None:
                if fc.max_credits < fc.initial_credits:
                    res.error("flowControl.maxCredits must be >= initialCredits")
        d = s.delivery
        if d is not None and d.semantics not in (None, "atMostOnce", "atLeastOnce", "exactlyOnce"):
            res.error(f"delivery.semantics invalid: {d.semantics!r}")
        if d is not None and d.ordering not in (None, "none", "perLane", "total"):
            res.error(f"delivery.ordering invalid: {d.ordering!r}")
        bp = s.backpressure
        if bp is not None and bp.policy not in (None, "block", "dropOldest", "dropNewest"):
            res.error(f"backpressure.policy invalid: {bp.policy!r}")
    return res


def validate_engram_template(tpl: T.EngramTemplate) -> ValidationResult:
    res = ValidationResult()
    if not _valid_name(tpl.name):
        res.error(f"engram template name {tpl.name!r} must be a DNS-1123 label")
    if not tpl.implementation and not tpl.command:
        res.error(
            "engram template must name an implementation (builtin, image, or command)"
        )
    for label, schema in (
        ("configSchema", tpl.config_schema),
        ("inputSchema", tpl.input_schema),
        ("outputSchema", tpl.output_schema),
        ("secretSchema", tpl.secret_schema),
    ):
        if schema is not None and not isinstance(schema, dict):
            res.error(f"{label} must be a JSON Schema object")
    return res


def validate_impulse_template(tpl: T.ImpulseTemplate) -> ValidationResult:
    res = ValidationResult()
    if not _valid_name(tpl.name):
        res.error(f"impulse template name {tpl.name!r} must be a DNS-1123 label")
    if not tpl.implementation:
        res.error("impulse template must name an implementation (builtin or image)")
    return res
