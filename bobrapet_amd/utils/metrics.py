"""Prometheus-style metrics registry.

Series parity with the reference's `bobrapet_*` metrics
(reference: pkg/metrics/controller_metrics.go:44-290 — storyruns_total,
storyrun_duration_seconds, stepruns_total, steprun_duration_seconds,
steprun_retries_total, steprun_cache_lookups_total, dag_iteration_steps,
controller_reconcile_*), renamed under the `bobrapet_amd_` prefix and
exportable in the Prometheus text format.
"""
from __future__ import annotations

import threading
import typing as _t
from collections import defaultdict

PREFIX = "bobrapet_amd_"

_BUCKETS = (
    0.0001, 0.00025, 0.0005, 0.001, 0.0025, 0.005, 0.01, 0.025, 0.05,
    0.1, 0.25, 0.5, 1.0, 2.5, 5.0, 10.0, 30.0, 60.0, 300.0,
)


def _label_key(labels: dict) -> _t.Tuple:
    return tuple(sorted(labels.items()))


class MetricsRegistry:
    def __init__(self):
        self._lock = threading.Lock()
        self._counters: _t.Dict[str, _t.Dict[_t.Tuple, float]] = defaultdict(dict)
        self._gauges: _t.Dict[str, _t.Dict[_t.Tuple, float]] = defaultdict(dict)
        self._histograms: _t.Dict[str, _t.Dict[_t.Tuple, _t.List]] = defaultdict(dict)

    def inc(self, name: str, value: float = 1.0, **labels) -> None:
        key = _label_key(labels)
        with self._lock:
            series = self._counters[name]
            series[key] = series.get(key, 0.0) + value

    def set_gauge(self, name: str, value: float, **labels) -> None:
        with self._lock:
            self._gauges[name][_label_key(labels)] = value

    def observe(self, name: str, value: float, **labels) -> None:
        key = _label_key(labels)
        with self._lock:
            series = self._histograms[name]
            if key not in series:
                series[key] = [[0] * (len(_BUCKETS) + 1), 0.0, 0]  # buckets, sum, count
            buckets, _s, _c = series[key]
            for i, ub in enumerate(_BUCKETS):
                if value <= ub:
                    buckets[i] += 1
                    break
            else:
                buckets[-1] += 1
            series[key][1] += value
            series[key][2] += 1

    def counter_value(self, name: str, **labels) -> float:
        with self._lock:
            return self._counters.get(name, {}).get(_label_key(labels), 0.0)

    def gauge_value(self, name: str, **labels) -> _t.Optional[float]:
        with self._lock:
            return self._gauges.get(name, {}).get(_label_key(labels))

    def histogram_stats(self, name: str, **labels) -> _t.Optional[dict]:
        with self._lock:
            item = self._histograms.get(name, {}).get(_label_key(labels))
            if item is None:
                return None
            buckets, total, count = item
            return {"sum": total, "count": count, "mean": total / count if count else 0.0}

    def export_text(self) -> str:
        """Prometheus text exposition format."""
        out = []
        with self._lock:
            for name, series in sorted(self._counters.items()):
                out.append(f"# TYPE {PREFIX}{name} counter")
                for key, val in sorted(series.items()):
                    out.append(f"{PREFIX}{name}{_fmt_labels(key)} {val}")
            for name, series in sorted(self._gauges.items()):
                out.append(f"# TYPE {PREFIX}{name} gauge")
                for key, val in sorted(series.items()):
                    out.append(f"{PREFIX}{name}{_fmt_labels(key)} {val}")
            for name, series in sorted(self._histograms.items()):
                out.append(f"# TYPE {PREFIX}{name} histogram")
                for key, (buckets, total, count) in sorted(series.items()):
                    cum = 0
                    for i, ub in enumerate(_BUCKETS):
                        cum += buckets[i]
                        out.append(
                            f"{PREFIX}{name}_bucket{_fmt_labels(key, le=ub)} {cum}"
                        )
                    cum += buckets[-1]
                    out.append(f'{PREFIX}{name}_bucket{_fmt_labels(key, le="+Inf")} {cum}')
                    out.append(f"{PREFIX}{name}_sum{_fmt_labels(key)} {total}")
                    out.append(f"{PREFIX}{name}_count{_fmt_labels(key)} {count}")
        return "\n".join(out) + "\n"


def _fmt_labels(key: _t.Tuple, **extra) -> str:
    items = list(key) + list(extra.items())
    if not items:
        return ""
    body = ",".join(f'{k}="{v}"' for k, v in items)
    return "{" + body + "}"


GLOBAL = MetricsRegistry()
