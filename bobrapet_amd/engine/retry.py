"""Retry delay computation (reference:
internal/controller/runs/steprun_controller.go:2165-2343
scheduleRetryIfNeeded / computeRetryDelay: exponential|linear|constant
backoff, percent jitter, max-delay clamp; ExitClass gating lives in
enums.classify_exit_code / ExitClass)."""
from __future__ import annotations

import random
import typing as _t

from ..enums import BackoffStrategy, ExitClass
from .config import ResolvedExecutionConfig

RATE_LIMITED_MIN_DELAY = 5.0  # rateLimited failures back off harder


def compute_retry_delay(
    cfg: ResolvedExecutionConfig,
    attempt: int,
    exit_class: ExitClass = ExitClass.RETRY,
    rng: _t.Optional[random.Random] = None,
) -> float:
    """Delay before retry number ``attempt`` (1-based)."""
    base = max(cfg.retry_delay, 0.0)
    if cfg.backoff == BackoffStrategy.EXPONENTIAL:
        delay = base * (2 ** max(attempt - 1, 0))
    elif cfg.backoff == BackoffStrategy.LINEAR:
        delay = base * attempt
    else:
        delay = base
    if exit_class == ExitClass.RATE_LIMITED:
        delay = max(delay, RATE_LIMITED_MIN_DELAY)
    delay = min(delay, cfg.retry_max_delay) if cfg.retry_max_delay else delay
    jitter_pct = max(0, min(cfg.retry_jitter_pct, 100))
    if jitter_pct:
        rng = rng or random
        delay *= 1.0 + rng.uniform(-jitter_pct / 100.0, jitter_pct / 100.0)
    return max(delay, 0.0)


def should_retry(
    cfg: ResolvedExecutionConfig, exit_class: ExitClass, retries_used: int
) -> bool:
    """UNKNOWN retries without consuming budget (enums.go:302-305)."""
    if not exit_class.is_retryable:
        return False
    if not exit_class.consumes_retry_budget:
        return True
    return retries_used < cfg.max_retries
