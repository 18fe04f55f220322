"""YAML loader for CRD-style spec documents.

Accepts the reference's ``apiVersion/kind/metadata/spec`` YAML surface
verbatim (reference: config/crd/bases/*.yaml, config/samples/) and produces
the spec dataclasses of :mod:`bobrapet_amd.specs.types`.  Also accepts bare
spec dicts for programmatic construction.
"""
from __future__ import annotations

import typing as _t

import yaml

from . import types as T


class SpecLoadError(ValueError):
    pass


_KIND_MAP: _t.Dict[str, type] = {
    "Story": T.Story,
    "Engram": T.Engram,
    "Impulse": T.Impulse,
    "EngramTemplate": T.EngramTemplate,
    "ImpulseTemplate": T.ImpulseTemplate,
    "Transport": T.Transport,
    "ReferenceGrant": T.ReferenceGrant,
}

_API_GROUPS = (
    "bubustack.io",
    "runs.bubustack.io",
    "catalog.bubustack.io",
    "transport.bubustack.io",
    "policy.bubustack.io",
    "bobrapet.amd",  # native group alias
)


def load_yaml(text: str) -> _t.List[object]:
    """Load one or more YAML documents into spec objects."""
    out = []
    for doc in yaml.safe_load_all(text):
        if doc is None:
            continue
        out.append(load_document(doc))
    return out


def load_path(path: str) -> _t.List[object]:
    with open(path, "r", encoding="utf-8") as fh:
        return load_yaml(fh.read())


def load_document(doc: dict) -> object:
    if not isinstance(doc, dict):
        raise SpecLoadError(f"expected a mapping document, got {type(doc).__name__}")
    kind = doc.get("kind")
    if kind is None:
        raise SpecLoadError("document has no 'kind'")
    cls = _KIND_MAP.get(kind)
    if cls is None:
        raise SpecLoadError(f"unknown kind {kind!r} (known: {sorted(_KIND_MAP)})")
    api_version = doc.get("apiVersion", "")
    if api_version and not any(api_version.startswith(g) for g in _API_GROUPS):
        raise SpecLoadError(f"unknown apiVersion {api_version!r} for kind {kind}")

    meta = doc.get("metadata") or {}
    spec = dict(doc.get("spec") or {})

    # Story-specific key remaps ("if"/"with"/"finally" are Python keywords).
    if cls is T.Story:
        obj = _load_story(spec)
    elif cls is T.Engram:
        if "with" in spec:
            spec["with_"] = spec.pop("with")
        obj = T.from_dict(cls, spec)
    elif cls is T.Impulse:
        if "with" in spec:
            spec["with_"] = spec.pop("with")
        obj = T.from_dict(cls, spec)
    elif cls is T.ReferenceGrant:
        if "from" in spec:
            spec["from_"] = spec.pop("from")
        obj = T.from_dict(cls, spec)
    else:
        obj = T.from_dict(cls, spec)

    name = meta.get("name")
    if name:
        obj.name = name
    if hasattr(obj, "namespace"):
        obj.namespace = meta.get("namespace") or getattr(obj, "namespace", None) or "default"
    if hasattr(obj, "labels") and meta.get("labels"):
        obj.labels = dict(meta["labels"])
    if hasattr(obj, "annotations") and meta.get("annotations"):
        obj.annotations = dict(meta["annotations"])
    if hasattr(obj, "generation") and meta.get("generation"):
        obj.generation = int(meta["generation"])
    if not getattr(obj, "name", None):
        raise SpecLoadError(f"{kind}: metadata.name is required")
    return obj


def _load_story(spec: dict) -> T.Story:
    spec = dict(spec)
    steps = [T._step_from_dict(s) for s in spec.pop("steps", []) or []]
    compensations = [T._step_from_dict(s) for s in spec.pop("compensations", []) or []]
    finally_ = [T._step_from_dict(s) for s in spec.pop("finally", []) or []]
    policy = spec.pop("policy", None)
    story = T.from_dict(T.Story, spec)
    story.steps = steps
    story.compensations = compensations
    story.finally_ = finally_
    if policy is not None:
        story.policy = T._story_policy_from_dict(policy)
    return story


def dump_yaml(obj, kind: _t.Optional[str] = None) -> str:
    """Serialize a spec object back to a CRD-style YAML document."""
    kind = kind or type(obj).__name__
    group = {
        "Story": "bubustack.io/v1alpha1",
        "Engram": "bubustack.io/v1alpha1",
        "Impulse": "bubustack.io/v1alpha1",
        "EngramTemplate": "catalog.bubustack.io/v1alpha1",
        "ImpulseTemplate": "catalog.bubustack.io/v1alpha1",
        "Transport": "transport.bubustack.io/v1alpha1",
        "ReferenceGrant": "policy.bubustack.io/v1alpha1",
    }.get(kind, "bobrapet.amd/v1")
    spec = T.to_dict(obj)
    meta = {"name": spec.pop("name", None)}
    ns = spec.pop("namespace", None)
    if ns:
        meta["namespace"] = ns
    for k in ("labels", "annotations"):
        v = spec.pop(k, None)
        if v:
            meta[k] = v
    spec.pop("generation", None)
    # Python-keyword remaps back to the YAML surface
    for py_key, yaml_key in (("if_", "if"), ("with_", "with"), ("finally_", "finally"), ("from_", "from")):
        if py_key in spec:
            spec[yaml_key] = spec.pop(py_key)
    doc = {"apiVersion": group, "kind": kind, "metadata": meta, "spec": spec}
    return yaml.safe_dump(doc, sort_keys=False)
