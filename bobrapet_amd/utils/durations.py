"""Go-style duration string parsing ("300ms", "1.5h", "2m30s")."""
from __future__ import annotations

import re
import typing as _t

_UNIT_SECONDS = {
    "ns": 1e-9,
    "us": 1e-6,
    "µs": 1e-6,
    "ms": 1e-3,
    "s": 1.0,
    "m": 60.0,
    "h": 3600.0,
    "d": 86400.0,
}

_PART = re.compile(r"(\d+(?:\.\d+)?)(ns|us|µs|ms|s|m|h|d)")


class DurationError(ValueError):
    pass


def parse_duration(text: _t.Union[str, int, float, None]) -> _t.Optional[float]:
    """Parse a Go-style duration to seconds; numbers pass through as seconds.

    Returns None for None; raises DurationError on malformed strings."""
    if text is None:
        return None
    if isinstance(text, (int, float)) and not isinstance(text, bool):
        return float(text)
    s = str(text).strip()
    if not s:
        raise DurationError("empty duration")
    neg = s.startswith("-")
    if neg or s.startswith("+"):
        s = s[1:]
    if s in ("0", "0.0"):
        return 0.0
    pos = 0
    total = 0.0
    for m in _PART.finditer(s):
        if m.start() != pos:
            raise DurationError(f"malformed duration {text!r}")
        total += float(m.group(1)) * _UNIT_SECONDS[m.group(2)]
        pos = m.end()
    if pos != len(s):
        # bare number: treat as seconds
        try:
            return -float(s) if neg else float(s)
        except ValueError:
            raise DurationError(f"malformed duration {text!r}") from None
    return -total if neg else total


def format_duration(seconds: float) -> str:
    if seconds == 0:
        return "0s"
    neg = seconds < 0
    seconds = abs(seconds)
    parts = []
    for unit, mult in (("h", 3600.0), ("m", 60.0)):
        if seconds >= mult:
            n = int(seconds // mult)
            parts.append(f"{n}{unit}")
            seconds -= n * mult
    if seconds:
        if seconds >= 1 or not parts:
            parts.append(f"{seconds:g}s")
    return ("-" if neg else "") + "".join(parts)
