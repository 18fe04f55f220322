"""Implicit-dependency extraction from template text.

Parity with the reference's template-implied dependency scan
(reference: internal/controller/runs/dag.go:3024-3074 — regex over
`steps.NAME.`, `steps["NAME"]` and `(index .steps "NAME")` forms, with
underscore-alias resolution back to dashed step names; and
offloaded_refs.go:128 extractReferencedSteps).
"""
from __future__ import annotations

import json
import typing as _t

import re

_STEP_REF_RE = re.compile(
    r"""steps\.([A-Za-z0-9_\-]+)
      | steps\s*\[\s*['"]([A-Za-z0-9_\-]+)['"]\s*\]
      | \(\s*index\s+\.?steps\s+['"]([A-Za-z0-9_\-]+)['"]\s*\)
    """,
    re.X,
)


def extract_referenced_steps(text: _t.Optional[str]) -> _t.Set[str]:
    """Names of steps referenced by template text (raw names, pre-alias)."""
    if not text:
        return set()
    out: _t.Set[str] = set()
    for m in _STEP_REF_RE.finditer(text):
        out.add(next(g for g in m.groups() if g))
    return out


def referenced_steps_of_value(value) -> _t.Set[str]:
    """Scan a JSON-like value (a `with` block) for step references."""
    if value is None:
        return set()
    if isinstance(value, str):
        return extract_referenced_steps(value)
    try:
        text = json.dumps(value, default=str)
    except (TypeError, ValueError):
        return set()
    return extract_referenced_steps(text)


def resolve_aliases(names: _t.Set[str], alias_to_real: _t.Mapping[str, str]) -> _t.Set[str]:
    """Map underscore aliases back to real (dashed) step names
    (reference: dag.go:3223-3268 normalizeStepIdentifier aliasing)."""
    return {alias_to_real.get(n, n) for n in names}
