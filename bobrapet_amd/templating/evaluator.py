"""Template evaluator with deterministic mode, output caps and offloaded-data
policy.

Role parity with the external templating engine the reference wires in
(reference: cmd/main.go:585-597 — Config{EvaluationTimeout, MaxOutputBytes,
Deterministic}; errors ErrEvaluationBlocked / ErrOffloadedDataUsage), plus
the in-controller helpers ResolveWithInputs / EvaluateCondition /
ResolveTemplateString the controllers call.
"""
from __future__ import annotations

import hashlib
import json
import math
import time
import typing as _t
from dataclasses import dataclass, field

from ..enums import OffloadedDataPolicy
from .parser import (
    CompiledTemplate,
    TemplateSyntaxError,
    is_template,
    parse_expression,
    parse_template,
)

STORAGE_REF_KEY = "$storageRef"


class TemplateError(ValueError):
    pass


class EvaluationBlocked(TemplateError):
    """Raised when a non-deterministic construct is used in deterministic mode."""


class OffloadedDataUsage(TemplateError):
    """Raised when a template touches `$storageRef` data under policy=block
    (reference: templating ErrOffloadedDataUsage; dag.go offloaded 3-way policy)."""

    def __init__(self, path: str = ""):
        super().__init__(f"template references offloaded data at {path or '<value>'}")
        self.path = path


class OutputTooLarge(TemplateError):
    pass


class EvaluationBudgetExceeded(TemplateError):
    pass


class MissingValue:
    """Sentinel for absent scope members; falsy, propagates through access."""

    _instance: _t.Optional["MissingValue"] = None

    def __new__(cls):
        if cls._instance is None:
            cls._instance = super().__new__(cls)
        return cls._instance

    def __bool__(self) -> bool:
        return False

    def __repr__(self) -> str:
        return "<missing>"


MISSING = MissingValue()


@dataclass
class EvalConfig:
    """Evaluator knobs (reference: internal/config/controller_config.go:137-144)."""

    deterministic: bool = True
    max_output_bytes: int = 1 << 20
    max_ops: int = 200_000  # deterministic stand-in for the eval timeout
    offloaded_policy: OffloadedDataPolicy = OffloadedDataPolicy.INJECT
    strict: bool = False  # raise on missing scope members instead of null


@dataclass
class Evaluator:
    config: EvalConfig = field(default_factory=EvalConfig)
    hydrator: _t.Optional[_t.Callable[[dict], _t.Any]] = None  # $storageRef → value
    _cache: _t.Dict[str, CompiledTemplate] = field(default_factory=dict)
    # optional MetricsRegistry (reference: bobrapet_cel_evaluation_total /
    # _duration_seconds / _cache_hits_total) — set by the engine
    metrics: _t.Optional[_t.Any] = None

    # -- public API ---------------------------------------------------------

    def resolve_string(self, text: str, scope: _t.Mapping[str, _t.Any]):
        """Resolve a template string; a pure `{{ expr }}` keeps its type."""
        if not is_template(text):
            return text
        tpl = self._compiled(text)
        ctx = _Ctx(self, scope)
        if tpl.single:
            value = _unwrap_missing(_eval(tpl.parts[0][1], ctx))
        else:
            out = []
            for kind, part in tpl.parts:
                if kind == "lit":
                    out.append(part)
                else:
                    out.append(_stringify(_unwrap_missing(_eval(part, ctx))))
            value = "".join(out)
        self._check_size(value)
        return value

    def resolve_value(self, value, scope: _t.Mapping[str, _t.Any]):
        """Recursively resolve templates inside a JSON-like value
        (the `with` block resolver — reference: step_executor.go:961-1079)."""
        out = self._resolve_value_inner(value, scope)
        self._check_size(out)
        return out

    def _resolve_value_inner(self, value, scope):
        if isinstance(value, str):
            if is_template(value):
                tpl = self._compiled(value)
                ctx = _Ctx(self, scope)
                if tpl.single:
                    return _unwrap_missing(_eval(tpl.parts[0][1], ctx))
                parts = []
                for kind, part in tpl.parts:
                    parts.append(
                        part if kind == "lit" else _stringify(_unwrap_missing(_eval(part, ctx)))
                    )
                return "".join(parts)
            return value
        if isinstance(value, dict):
            return {k: self._resolve_value_inner(v, scope) for k, v in value.items()}
        if isinstance(value, list):
            return [self._resolve_value_inner(v, scope) for v in value]
        return value

    def evaluate_condition(self, expr_or_template: str, scope: _t.Mapping[str, _t.Any]) -> bool:
        """Evaluate an `if`/`until`/postExecution condition to a bool
        (truthiness rules: null/0/""/[]/{}/missing are false)."""
        src = expr_or_template.strip()
        if is_template(src):
            value = self.resolve_string(src, scope)
            if isinstance(value, str):
                s = value.strip().lower()
                if s in ("true", "1", "yes"):
                    return True
                if s in ("false", "0", "no", "", "null", "none"):
                    return False
                return True
            return _truthy(value)
        ast = parse_expression(src)
        return _truthy(_eval(ast, _Ctx(self, scope)))

    # -- internals ----------------------------------------------------------

    def _compiled(self, text: str) -> CompiledTemplate:
        tpl = self._cache.get(text)
        if tpl is None:
            tpl = parse_template(text)
            if len(self._cache) < 4096:
                self._cache[text] = tpl
        elif self.metrics is not None:
            self.metrics.inc("template_cache_hits_total")
        if self.metrics is not None:
            self.metrics.inc("template_evaluations_total")
        return tpl

    def _check_size(self, value) -> None:
        cap = self.config.max_output_bytes
        if cap <= 0:
            return
        try:
            size = len(json.dumps(value, separators=(",", ":"), default=_cheap_default))
        except (TypeError, ValueError):
            return
        if size > cap:
            raise OutputTooLarge(f"template output {size} bytes exceeds cap {cap}")


def _cheap_default(value):
    """Opaque objects (tensors) serialize as a short tag — never stringify
    payload data on the control plane."""
    t = type(value)
    if t.__module__ == "torch" and t.__name__ in ("Tensor", "Parameter"):
        return f"<tensor {tuple(value.shape)}>"
    return str(value)[:256]


def _unwrap_missing(value):
    return None if value is MISSING else value


class _Ctx:
    __slots__ = ("ev", "scope", "ops")

    def __init__(self, ev: Evaluator, scope: _t.Mapping[str, _t.Any]):
        self.ev = ev
        self.scope = scope
        self.ops = 0

    def tick(self) -> None:
        self.ops += 1
        if self.ops > self.ev.config.max_ops:
            raise EvaluationBudgetExceeded(
                f"expression exceeded the {self.ev.config.max_ops}-op budget"
            )


def _truthy(v) -> bool:
    if v is MISSING or v is None:
        return False
    if isinstance(v, (list, dict, str)):
        return len(v) > 0
    return bool(v)


def _stringify(v) -> str:
    if v is None:
        return ""
    t = type(v)
    if t.__module__ == "torch" and t.__name__ in ("Tensor", "Parameter"):
        return f"<tensor {tuple(v.shape)}>"
    if isinstance(v, bool):
        return "true" if v else "false"
    if isinstance(v, float) and v.is_integer():
        return str(int(v))
    if isinstance(v, (dict, list)):
        return json.dumps(v, separators=(",", ":"), default=str)
    return str(v)


def _maybe_hydrate(ctx: _Ctx, value, path: str):
    """Apply the offloaded-data policy when touching a `$storageRef` value."""
    if isinstance(value, dict) and STORAGE_REF_KEY in value:
        policy = ctx.ev.config.offloaded_policy
        if policy == OffloadedDataPolicy.BLOCK:
            raise OffloadedDataUsage(path)
        if policy == OffloadedDataPolicy.INJECT and ctx.ev.hydrator is not None:
            return ctx.ev.hydrator(value)
    return value


def _get(ctx: _Ctx, obj, key, path: str):
    ctx.tick()
    if obj is MISSING or obj is None:
        if ctx.ev.config.strict:
            raise TemplateError(f"missing value at {path!r}")
        return MISSING
    if isinstance(obj, dict):
        if key in obj:
            return _maybe_hydrate(ctx, obj[key], path)
        # alias tolerance: step names use '-', template identifiers '_'
        if isinstance(key, str) and "_" in key:
            alt = key.replace("_", "-")
            if alt in obj:
                return _maybe_hydrate(ctx, obj[alt], path)
        if ctx.ev.config.strict:
            raise TemplateError(f"missing key {key!r} at {path!r}")
        return MISSING
    if isinstance(obj, (list, tuple, str)):
        try:
            return obj[int(key)]
        except (ValueError, TypeError, IndexError):
            if ctx.ev.config.strict:
                raise TemplateError(f"bad index {key!r} at {path!r}") from None
            return MISSING
    attr = getattr(obj, str(key), MISSING)
    if attr is MISSING and ctx.ev.config.strict:
        raise TemplateError(f"missing attribute {key!r} at {path!r}")
    return attr


def _eval(node, ctx: _Ctx):
    ctx.tick()
    op = node[0]
    if op == "const":
        return node[1]
    if op == "var":
        name = node[1]
        if name in ctx.scope:
            return _maybe_hydrate(ctx, ctx.scope[name], name)
        if ctx.ev.config.strict:
            raise TemplateError(f"unknown variable {name!r}")
        return MISSING
    if op == "get":
        obj = _eval(node[1], ctx)
        return _get(ctx, obj, node[2], _describe(node))
    if op == "index":
        obj = _eval(node[1], ctx)
        key = _eval(node[2], ctx)
        if key is MISSING:
            return MISSING
        return _get(ctx, obj, key, _describe(node))
    if op == "and":
        left = _eval(node[1], ctx)
        if not _truthy(left):
            return False
        return _truthy(_eval(node[2], ctx))
    if op == "or":
        left = _eval(node[1], ctx)
        if _truthy(left):
            return True
        return _truthy(_eval(node[2], ctx))
    if op == "not":
        return not _truthy(_eval(node[1], ctx))
    if op == "cmp":
        return _compare(node[1], _eval(node[2], ctx), _eval(node[3], ctx))
    if op == "bin":
        return _binop(
            node[1],
            _eval(node[2], ctx),
            _eval(node[3], ctx),
            cap=ctx.ev.config.max_output_bytes,
        )
    if op == "neg":
        v = _eval(node[1], ctx)
        if v is MISSING or v is None:
            return MISSING
        return -v
    if op == "cond":
        return _eval(node[2] if _truthy(_eval(node[1], ctx)) else node[3], ctx)
    if op == "list":
        return [_unwrap_missing(_eval(item, ctx)) for item in node[1]]
    if op == "map":
        return {
            _eval(k, ctx): _unwrap_missing(_eval(v, ctx)) for k, v in node[1]
        }
    if op == "call":
        return _call(ctx, node[1], [_eval(a, ctx) for a in node[2]])
    if op == "method":
        obj = _eval(node[1], ctx)
        args = [_eval(a, ctx) for a in node[3]]
        return _method(ctx, obj, node[2], args)
    raise TemplateError(f"unknown AST node {op!r}")


def _describe(node) -> str:
    if node[0] == "var":
        return node[1]
    if node[0] == "get":
        return f"{_describe(node[1])}.{node[2]}"
    if node[0] == "index":
        return f"{_describe(node[1])}[...]"
    return "<expr>"


def _compare(op: str, a, b):
    if a is MISSING:
        a = None
    if b is MISSING:
        b = None
    if op == "==":
        return a == b
    if op == "!=":
        return a != b
    if op == "in":
        if b is None:
            return False
        try:
            return a in b
        except TypeError:
            return False
    if a is None or b is None:
        return False
    try:
        if op == "<":
            return a < b
        if op == "<=":
            return a <= b
        if op == ">":
            return a > b
        if op == ">=":
            return a >= b
    except TypeError:
        return False
    raise TemplateError(f"unknown comparison {op!r}")


def _estimated_len(v) -> int:
    """Cheap element/char count for allocation bounding (no serialization)."""
    if isinstance(v, (str, bytes, list, dict)):
        return len(v)
    return 1


def _guard_alloc(n: int, cap: _t.Optional[int]) -> None:
    """Bound result size BEFORE allocating: the op-count budget caps AST
    steps, not bytes — a single `{{ 'x' * 10**9 }}` or huge concat could
    otherwise OOM the shared engine process before _check_size runs."""
    if cap is not None and cap > 0 and n > cap:
        raise OutputTooLarge(f"operation result ~{n} elements exceeds cap {cap}")


def _binop(op: str, a, b, cap: _t.Optional[int] = None):
    if a is MISSING:
        a = None
    if b is MISSING:
        b = None
    if op == "+":
        if isinstance(a, str) or isinstance(b, str):
            _guard_alloc(_estimated_len(a) + _estimated_len(b), cap)
            return _stringify(a) + _stringify(b)
        if isinstance(a, list) and isinstance(b, list):
            _guard_alloc(len(a) + len(b), cap)
            return a + b
        if a is None or b is None:
            return a if b is None else b
        return a + b
    if a is None or b is None:
        return None
    if op == "-":
        return a - b
    if op == "*":
        if isinstance(a, (str, list)) and isinstance(b, (int, float)):
            _guard_alloc(int(_estimated_len(a) * max(b, 0)), cap)
        elif isinstance(b, (str, list)) and isinstance(a, (int, float)):
            _guard_alloc(int(_estimated_len(b) * max(a, 0)), cap)
        return a * b
    if op == "/":
        if b == 0:
            raise TemplateError("division by zero")
        out = a / b
        return out
    if op == "%":
        if b == 0:
            raise TemplateError("modulo by zero")
        return a % b
    raise TemplateError(f"unknown operator {op!r}")


def _call(ctx: _Ctx, name: str, args: _t.List):
    args = [None if a is MISSING else a for a in args]
    fn = _FUNCTIONS.get(name)
    if fn is None:
        raise TemplateError(f"unknown function {name!r}")
    if name in _NONDETERMINISTIC and ctx.ev.config.deterministic:
        raise EvaluationBlocked(f"function {name!r} is blocked in deterministic mode")
    if name == "range":
        # bound BEFORE materializing: list(range(10**9)) would OOM the
        # shared engine process before _check_size ever sees the result
        r = range(*[int(x) for x in args])
        _guard_alloc(len(r), ctx.ev.config.max_output_bytes)
        return list(r)
    return fn(*args)


def _method(ctx: _Ctx, obj, name: str, args: _t.List):
    args = [None if a is MISSING else a for a in args]
    if obj is MISSING:
        obj = None
    fn = _METHODS.get(name)
    if fn is None:
        raise TemplateError(f"unknown method .{name}()")
    return fn(obj, *args)


def _size(v) -> int:
    if v is None:
        return 0
    if isinstance(v, (str, list, dict, tuple)):
        return len(v)
    raise TemplateError(f"size() of {type(v).__name__}")


def _has(v, key=None) -> bool:
    if key is None:
        return v is not None
    if isinstance(v, dict):
        return key in v
    return False


def _coalesce(*args):
    for a in args:
        if a is not None:
            return a
    return None


_FUNCTIONS: _t.Dict[str, _t.Callable] = {
    "size": _size,
    "len": _size,
    "has": _has,
    "string": _stringify,
    "int": lambda v: int(float(v)) if v is not None else None,
    "float": lambda v: float(v) if v is not None else None,
    "bool": _truthy,
    "abs": lambda v: abs(v),
    "min": lambda *a: min(a[0] if len(a) == 1 and isinstance(a[0], list) else a),
    "max": lambda *a: max(a[0] if len(a) == 1 and isinstance(a[0], list) else a),
    "floor": math.floor,
    "ceil": math.ceil,
    "round": round,
    "coalesce": _coalesce,
    "default": lambda v, d: d if v is None else v,
    "contains": lambda a, b: b in a if a is not None else False,
    "startsWith": lambda a, b: str(a).startswith(str(b)) if a is not None else False,
    "endsWith": lambda a, b: str(a).endswith(str(b)) if a is not None else False,
    "lower": lambda v: str(v).lower() if v is not None else None,
    "upper": lambda v: str(v).upper() if v is not None else None,
    "trim": lambda v: str(v).strip() if v is not None else None,
    "split": lambda v, sep: str(v).split(sep) if v is not None else [],
    "join": lambda items, sep="": sep.join(_stringify(i) for i in (items or [])),
    "keys": lambda v: sorted(v.keys()) if isinstance(v, dict) else [],
    "values": lambda v: [v[k] for k in sorted(v)] if isinstance(v, dict) else [],
    "range": lambda *a: list(range(*[int(x) for x in a])),
    "sha256": lambda v: hashlib.sha256(
        (v if isinstance(v, str) else json.dumps(v, sort_keys=True, default=str)).encode()
    ).hexdigest(),
    "toJson": lambda v: json.dumps(v, separators=(",", ":"), sort_keys=True, default=str),
    "fromJson": lambda v: json.loads(v) if v else None,
    # non-deterministic (blocked in deterministic mode):
    "now": lambda: time.time(),
    "uuid": lambda: __import__("uuid").uuid4().hex,
}

_NONDETERMINISTIC = {"now", "uuid"}

_METHODS: _t.Dict[str, _t.Callable] = {
    "size": lambda obj: _size(obj),
    "contains": lambda obj, item: item in obj if obj is not None else False,
    "startsWith": lambda obj, p: str(obj).startswith(str(p)) if obj is not None else False,
    "endsWith": lambda obj, p: str(obj).endswith(str(p)) if obj is not None else False,
    "lower": lambda obj: str(obj).lower() if obj is not None else None,
    "upper": lambda obj: str(obj).upper() if obj is not None else None,
    "trim": lambda obj: str(obj).strip() if obj is not None else None,
    "split": lambda obj, sep: str(obj).split(sep) if obj is not None else [],
    "join": lambda obj, sep="": sep.join(_stringify(i) for i in (obj or [])),
    "keys": lambda obj: sorted(obj.keys()) if isinstance(obj, dict) else [],
    "get": lambda obj, key, default=None: obj.get(key, default) if isinstance(obj, dict) else default,
}
