"""Structured, feature-gated logging.

Role parity with the reference's logging kit (reference: pkg/logging/
structured.go + features.go — entity-scoped wrappers with feature-gated
verbosity; the "contract logger" stage logging of bootstrap flows).
"""
from __future__ import annotations

import json
import logging
import sys
import time
import typing as _t

_FEATURES: _t.Dict[str, bool] = {}


def enable_feature(name: str, on: bool = True) -> None:
    _FEATURES[name] = on


def feature_enabled(name: str) -> bool:
    return _FEATURES.get(name, False)


class StructuredLogger:
    """Entity-scoped logger emitting JSON lines; verbose records gated by
    feature flags."""

    def __init__(self, component: str, stream=None, **bound):
        self.component = component
        self.bound = bound
        self.stream = stream or sys.stderr
        self._std = logging.getLogger(f"bobrapet_amd.{component}")

    def with_fields(self, **fields) -> "StructuredLogger":
        merged = dict(self.bound)
        merged.update(fields)
        return StructuredLogger(self.component, self.stream, **merged)

    def _emit(self, level: str, msg: str, fields: dict) -> None:
        record = {
            "ts": round(time.time(), 6),
            "level": level,
            "component": self.component,
            "msg": msg,
        }
        record.update(self.bound)
        record.update(fields)
        print(json.dumps(record, default=str), file=self.stream, flush=True)

    def info(self, msg: str, **fields) -> None:
        self._emit("info", msg, fields)

    def warn(self, msg: str, **fields) -> None:
        self._emit("warn", msg, fields)

    def error(self, msg: str, **fields) -> None:
        self._emit("error", msg, fields)

    def debug(self, msg: str, feature: str = "debug", **fields) -> None:
        if feature_enabled(feature):
            self._emit("debug", msg, fields)


class ContractLogger:
    """Stage logging for multi-stage flows (reference: bootstrap contract
    logger): collects (stage, status, detail) transitions."""

    def __init__(self, flow: str, logger: _t.Optional[StructuredLogger] = None):
        self.flow = flow
        self.logger = logger
        self.stages: _t.List[dict] = []

    def stage(self, name: str, status: str = "ok", **detail) -> None:
        entry = {"flow": self.flow, "stage": name, "status": status, **detail}
        self.stages.append(entry)
        if self.logger is not None:
            self.logger.info(f"{self.flow}:{name}", status=status, **detail)
