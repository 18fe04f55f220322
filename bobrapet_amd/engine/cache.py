"""Step output cache (reference: steprun_controller.go:3106-3477
tryCacheHit/maybeWriteCache — key = hash(resolved inputs) + salt, modes
read/write/readWrite, TTL; CachePolicy shared_types.go:249-277).
The cache is the resume-skip mechanism for re-runs."""
from __future__ import annotations

import hashlib
import threading
import time
import typing as _t

from ..specs import types as T
from .config import ResolvedExecutionConfig
from .records import canonical_json


class StepCache:
    def __init__(self, evaluator=None):
        self._data: _t.Dict[str, _t.Tuple[_t.Any, float, _t.Optional[float]]] = {}
        self._lock = threading.Lock()
        self.evaluator = evaluator

    def cache_key(
        self,
        step: T.Step,
        cfg: ResolvedExecutionConfig,
        resolved_input,
        scope: _t.Optional[dict] = None,
    ) -> str:
        if cfg.cache_key_template and self.evaluator is not None and scope is not None:
            base = str(self.evaluator.resolve_string(cfg.cache_key_template, scope))
        else:
            base = canonical_json(resolved_input)
        ref = step.ref.name if step.ref else str(step.type)
        payload = f"{ref}|{cfg.cache_salt}|{base}"
        return hashlib.sha256(payload.encode()).hexdigest()

    def lookup(
        self,
        step: T.Step,
        cfg: ResolvedExecutionConfig,
        resolved_input,
        scope: _t.Optional[dict] = None,
    ):
        key = self.cache_key(step, cfg, resolved_input, scope)
        with self._lock:
            item = self._data.get(key)
            if item is None:
                return None
            value, ts, ttl = item
            if ttl is not None and (time.time() - ts) > ttl:
                del self._data[key]
                return None
            return value

    def write(
        self,
        step: T.Step,
        cfg: ResolvedExecutionConfig,
        resolved_input,
        output,
        scope: _t.Optional[dict] = None,
    ) -> None:
        key = self.cache_key(step, cfg, resolved_input, scope)
        with self._lock:
            self._data[key] = (output, time.time(), cfg.cache_ttl_seconds)

    def clear(self) -> None:
        with self._lock:
            self._data.clear()

    def __len__(self) -> int:
        with self._lock:
            return len(self._data)
