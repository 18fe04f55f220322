"""In-process run-state store + resource registry + event bus.

This replaces the reference's coordination bus — kube-apiserver CRs,
watches and field indexes (reference: internal/setup/indexing.go:51-163,
SURVEY.md §5.8a) — with an in-memory store guarded by one lock, watch
callbacks delivered through the engine's event queue, and the same
terminal-phase-wins merge discipline on writes (dag.go:780-792).
"""
from __future__ import annotations

import threading
import typing as _t
from collections import defaultdict

from ..specs import types as T
from .records import StepRun, StoryRun


class NotFound(KeyError):
    pass


class Conflict(RuntimeError):
    pass


class ResourceRegistry:
    """Definition objects: Stories, Engrams, Impulses, templates, transports,
    reference grants.  The engine validates on apply (like admission webhooks)
    and keeps usage counters (reference: story_controller.go:119-246)."""

    def __init__(self):
        self._lock = threading.RLock()
        self.stories: _t.Dict[str, T.Story] = {}
        self.engrams: _t.Dict[str, T.Engram] = {}
        self.impulses: _t.Dict[str, T.Impulse] = {}
        self.engram_templates: _t.Dict[str, T.EngramTemplate] = {}
        self.impulse_templates: _t.Dict[str, T.ImpulseTemplate] = {}
        self.transports: _t.Dict[str, T.Transport] = {}
        self.reference_grants: _t.Dict[str, T.ReferenceGrant] = {}
        self.mutations = 0  # bumped on every apply; cheap cache-invalidation tag

    def apply(self, obj) -> None:
        with self._lock:
            self.mutations += 1
            if isinstance(obj, T.Story):
                obj.generation = self.stories.get(obj.key, obj).generation + (
                    1 if obj.key in self.stories else 0
                )
                self.stories[obj.key] = obj
            elif isinstance(obj, T.Engram):
                obj.generation = self.engrams.get(obj.key, obj).generation + (
                    1 if obj.key in self.engrams else 0
                )
                self.engrams[obj.key] = obj
            elif isinstance(obj, T.Impulse):
                self.impulses[obj.key] = obj
            elif isinstance(obj, T.EngramTemplate):
                obj.generation = self.engram_templates.get(obj.key, obj).generation + (
                    1 if obj.key in self.engram_templates else 0
                )
                self.engram_templates[obj.key] = obj
            elif isinstance(obj, T.ImpulseTemplate):
                self.impulse_templates[obj.key] = obj
            elif isinstance(obj, T.Transport):
                self.transports[obj.key] = obj
            elif isinstance(obj, T.ReferenceGrant):
                self.reference_grants[obj.key] = obj
            else:
                raise TypeError(f"unknown resource type {type(obj).__name__}")

    def story(self, name: str, namespace: str = "default") -> T.Story:
        with self._lock:
            obj = self.stories.get(f"{namespace}/{name}")
        if obj is None:
            raise NotFound(f"story {namespace}/{name}")
        return obj

    def engram(self, name: str, namespace: str = "default") -> T.Engram:
        with self._lock:
            obj = self.engrams.get(f"{namespace}/{name}")
        if obj is None:
            raise NotFound(f"engram {namespace}/{name}")
        return obj

    def engram_template(self, name: str) -> T.EngramTemplate:
        with self._lock:
            obj = self.engram_templates.get(name)
        if obj is None:
            raise NotFound(f"engram template {name}")
        return obj

    def impulse_template(self, name: str) -> T.ImpulseTemplate:
        with self._lock:
            obj = self.impulse_templates.get(name)
        if obj is None:
            raise NotFound(f"impulse template {name}")
        return obj

    def transport(self, name: str) -> T.Transport:
        with self._lock:
            obj = self.transports.get(name)
        if obj is None:
            raise NotFound(f"transport {name}")
        return obj

    def try_engram(self, name: str, namespace: str = "default") -> _t.Optional[T.Engram]:
        with self._lock:
            return self.engrams.get(f"{namespace}/{name}")

    def story_usage_count(self, story_key: str) -> int:
        """Impulses referencing a story (reference: story_controller.go:119-246)."""
        with self._lock:
            count = 0
            for imp in self.impulses.values():
                if imp.story_ref is None:
                    continue
                ns = imp.story_ref.resolve_namespace(imp.namespace)
                if f"{ns}/{imp.story_ref.name}" == story_key:
                    count += 1
            return count

    def engram_usage_count(self, engram_key: str) -> int:
        """Stories referencing an engram (reference: engram_controller.go:323-352)."""
        with self._lock:
            count = 0
            for story in self.stories.values():
                for step in story.all_steps():
                    if step.ref is None:
                        continue
                    ns = step.ref.resolve_namespace(story.namespace)
                    if f"{ns}/{step.ref.name}" == engram_key:
                        count += 1
                        break
            return count

    def transport_status(self, name: str, engine=None) -> dict:
        """Aggregate live binding state for one Transport (reference:
        pkg/transport/capabilities_aggregation.go:47-172)."""
        transport = self.transport(name)
        bindings = []
        if engine is not None:
            for stream in list(getattr(engine, "_streams", {}).values()):
                for b in getattr(stream, "bindings", {}).values():
                    if b.transport in (name, ""):
                        bindings.append(b)
        codecs = sorted({c for b in bindings for c in b.codecs}) or list(transport.codecs)
        return {
            "driver": transport.driver,
            "availableCodecs": codecs,
            "bindings": {
                "total": len(bindings),
                "ready": sum(1 for b in bindings if b.phase == "Ready"),
                "pending": sum(1 for b in bindings if b.phase == "Pending"),
            },
        }

    def story_status(self, story_key: str) -> dict:
        """Definition status (reference: story_controller.go ValidationStatus
        + usageCount)."""
        from ..specs import validation as V

        with self._lock:
            story = self.stories.get(story_key)
        if story is None:
            raise NotFound(f"story {story_key}")
        res = V.validate_story(story)
        # soft admission: warn about engram refs that do not resolve yet —
        # across MAIN, COMPENSATION, FINALLY and parallel branches
        # (reference: TestValidateEngramReferencesIncludesCompensation /
        # ...IncludesFinallyExecuteStory — the scan must not stop at main)
        warnings = list(res.warnings)
        def _scan(steps):
            for st in steps:
                if st.ref is not None and st.ref.name:
                    ns = st.ref.namespace or story.namespace
                    with self._lock:
                        known = f"{ns}/{st.ref.name}" in self.engrams
                    if not known:
                        warnings.append(
                            f"step {st.name!r} references unknown engram "
                            f"{ns}/{st.ref.name}"
                        )
                w = st.with_ if isinstance(st.with_, dict) else {}
                branches = w.get("steps")
                if isinstance(branches, list):
                    from ..specs import types as _T
                    _scan([_T._step_from_dict(dict(b)) for b in branches
                           if isinstance(b, dict)])
        _scan(story.all_steps())
        return {
            "validationStatus": "valid" if res.ok else "invalid",
            "validationErrors": res.errors,
            "validationWarnings": warnings,
            "usageCount": self.story_usage_count(story_key),
            "stepsTotal": len(story.steps),
            "generation": story.generation,
        }

    def engram_status(self, engram_key: str) -> dict:
        with self._lock:
            engram = self.engrams.get(engram_key)
        if engram is None:
            raise NotFound(f"engram {engram_key}")
        return {
            "usageCount": self.engram_usage_count(engram_key),
            "generation": engram.generation,
        }

    def allows_cross_namespace(
        self, from_kind: str, from_ns: str, to_kind: str, to_ns: str, to_name: str = ""
    ) -> bool:
        """ReferenceGrant check (reference: pkg/refs/reference_grant.go:26-88):
        same-namespace always allowed; cross-namespace requires a grant in the
        TARGET namespace covering (from_kind, from_ns) → (to_kind, to_name)."""
        if from_ns == to_ns:
            return True
        with self._lock:
            for grant in self.reference_grants.values():
                if grant.namespace != to_ns:
                    continue
                from_ok = any(
                    p.kind in (from_kind, "*") and (p.namespace in (from_ns, None, "*"))
                    for p in grant.from_
                )
                to_ok = any(
                    p.kind in (to_kind, "*") and (p.name in (to_name, None, "", "*"))
                    for p in grant.to
                )
                if from_ok and to_ok:
                    return True
        return False


class RunStore:
    """StoryRun + StepRun records with indexes and watch events.

    Indexes mirror the reference's field indexes (indexing.go): StepRuns by
    story-run, by phase; StoryRuns by story."""

    def __init__(self):
        self._lock = threading.RLock()
        self.story_runs: _t.Dict[str, StoryRun] = {}
        self.step_runs: _t.Dict[str, StepRun] = {}
        self._steps_by_run: _t.Dict[str, _t.Set[str]] = defaultdict(set)
        self._runs_by_story: _t.Dict[str, _t.Set[str]] = defaultdict(set)

    # -- StoryRun ----------------------------------------------------------

    def create_story_run(self, run: StoryRun) -> StoryRun:
        with self._lock:
            if run.key in self.story_runs:
                raise Conflict(f"story run {run.key} already exists")
            run.resource_version = 1
            self.story_runs[run.key] = run
            self._runs_by_story[f"{run.story_namespace}/{run.story_name}"].add(run.key)
            return run

    def get_story_run(self, key: str) -> StoryRun:
        with self._lock:
            run = self.story_runs.get(key)
        if run is None:
            raise NotFound(f"story run {key}")
        return run

    def try_get_story_run(self, key: str) -> _t.Optional[StoryRun]:
        with self._lock:
            return self.story_runs.get(key)

    def delete_story_run(self, key: str) -> None:
        with self._lock:
            run = self.story_runs.pop(key, None)
            if run is not None:
                self._runs_by_story[f"{run.story_namespace}/{run.story_name}"].discard(key)

    def runs_of_story(self, story_key: str) -> _t.List[StoryRun]:
        with self._lock:
            return [self.story_runs[k] for k in self._runs_by_story.get(story_key, ()) if k in self.story_runs]

    def all_runs(self) -> _t.List[StoryRun]:
        with self._lock:
            return list(self.story_runs.values())

    # -- StepRun -----------------------------------------------------------

    def create_or_get_step_run(self, sr: StepRun) -> _t.Tuple[StepRun, bool]:
        """Create-or-adopt idempotency via deterministic naming
        (reference: kubeutil.ComposeName + create-or-adopt everywhere)."""
        with self._lock:
            existing = self.step_runs.get(sr.key)
            if existing is not None:
                return existing, False
            sr.resource_version = 1
            self.step_runs[sr.key] = sr
            run_key = f"{sr.namespace}/{sr.spec.story_run}"
            self._steps_by_run[run_key].add(sr.key)
            return sr, True

    def get_step_run(self, key: str) -> StepRun:
        with self._lock:
            sr = self.step_runs.get(key)
        if sr is None:
            raise NotFound(f"step run {key}")
        return sr

    def try_get_step_run(self, key: str) -> _t.Optional[StepRun]:
        with self._lock:
            return self.step_runs.get(key)

    def step_runs_of(self, run_key: str) -> _t.List[StepRun]:
        with self._lock:
            return [
                self.step_runs[k]
                for k in self._steps_by_run.get(run_key, ())
                if k in self.step_runs
            ]

    def delete_step_run(self, key: str) -> None:
        with self._lock:
            sr = self.step_runs.pop(key, None)
            if sr is not None:
                self._steps_by_run[f"{sr.namespace}/{sr.spec.story_run}"].discard(key)

    def delete_steps_of(self, run_key: str) -> int:
        with self._lock:
            keys = list(self._steps_by_run.get(run_key, ()))
            for k in keys:
                self.step_runs.pop(k, None)
            self._steps_by_run.pop(run_key, None)
            return len(keys)

    def counts(self) -> _t.Tuple[int, int]:
        with self._lock:
            return len(self.story_runs), len(self.step_runs)
