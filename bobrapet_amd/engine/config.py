"""Engine configuration + hierarchical execution-config resolution.

Role parity with the reference's operator config and resolver
(reference: internal/config/operator.go:107-1740 ~80 keys,
controller_config.go:80-167, resolver.go:257-321 — the 5-layer merge
operator defaults → EngramTemplate recommendations → Engram overrides →
Story policy → Step overrides).
"""
from __future__ import annotations

import typing as _t
from dataclasses import dataclass, field

from ..enums import BackoffStrategy, CacheMode, OffloadedDataPolicy
from ..specs import types as T
from ..utils.durations import parse_duration


@dataclass
class QueueConfig:
    """Scheduling queue (reference: controller_config.go:527-544)."""

    concurrency: int = 0  # 0 = unlimited
    default_priority: int = 0
    priority_aging_seconds: float = 60.0


@dataclass
class EngineConfig:
    """Operator-level configuration with hot-reloadable semantics
    (update() swaps values atomically under the engine lock)."""

    # payload / storage
    max_inline_size: int = 8 << 10
    max_storyrun_input_bytes: int = 5 << 10  # then offload (storyrun_controller.go:71-74)
    max_output_bytes: int = 1 << 20  # story/step output cap → Degraded past it
    storage_retention_seconds: float = 3600.0
    storage_gc_interval: float = 120.0
    storage_gc_max_scan: int = 2000
    storage_gc_max_delete: int = 200

    # templating (reference: controller_config.go:137-144)
    template_deterministic: bool = True
    template_max_ops: int = 200_000
    offloaded_data_policy: OffloadedDataPolicy = OffloadedDataPolicy.INJECT

    # defaults for execution
    default_step_timeout: _t.Optional[float] = 600.0
    default_story_timeout: _t.Optional[float] = None
    default_graceful_shutdown: float = 30.0
    default_max_retries: int = 0
    default_retry_delay: float = 1.0
    default_retry_max_delay: float = 60.0
    default_retry_jitter_pct: int = 10
    default_backoff: BackoffStrategy = BackoffStrategy.EXPONENTIAL
    max_retries_cap: int = 10  # hard cap (shared_types.go:400-405)
    max_recursion_depth: int = 8

    # scheduling (reference: controller_config.go:527-544)
    global_concurrency: int = 0  # 0 = unlimited concurrent StoryRuns
    queues: _t.Dict[str, QueueConfig] = field(default_factory=dict)
    default_queue: str = "default"

    # durability: periodic state snapshots (the reference's state lives in
    # etcd and is always durable; here checkpointing is opt-in)
    checkpoint_path: _t.Optional[str] = None
    checkpoint_interval_seconds: float = 30.0

    # telemetry export (reference: pkg/observability/exporter.go — OTLP
    # exporter; here a file path or an OTLP/HTTP endpoint)
    otlp_endpoint: _t.Optional[str] = None
    otlp_flush_interval_seconds: float = 10.0

    # retention (reference: handleTerminalStoryRun defaults 2044/2056)
    child_ttl_seconds: float = 3600.0
    storyrun_retention_seconds: float = 86400.0

    # workers
    workers_per_device: int = 2
    cpu_workers: int = 4

    # dag engine
    min_poll_interval: float = 0.001  # wait-step floor; event-driven otherwise
    default_wait_poll_interval: float = 0.05

    def queue(self, name: str) -> QueueConfig:
        return self.queues.get(name, QueueConfig())

    def update(self, **kwargs) -> None:
        """Hot-reload semantics (reference: operator.go:356-383)."""
        for k, v in kwargs.items():
            if not hasattr(self, k):
                raise KeyError(f"unknown config key {k!r}")
            if k == "queues" and isinstance(v, dict):
                v = {
                    name: q if isinstance(q, QueueConfig) else QueueConfig(**q)
                    for name, q in v.items()
                }
            setattr(self, k, v)

    def load_file(self, path: str) -> None:
        """Apply a YAML/JSON config file (the ConfigMap role)."""
        import yaml as _yaml

        with open(path, "r", encoding="utf-8") as fh:
            data = _yaml.safe_load(fh) or {}
        self.update(**data)

    def watch_file(self, path: str, interval: float = 2.0):
        """Hot-reload the file when its mtime changes (reference: the
        operator-config reconciler watching its ConfigMap).  Returns a
        stop() callable."""
        import os
        import threading

        stop = threading.Event()
        state = {"mtime": None}

        def loop():
            while not stop.wait(interval):
                try:
                    m = os.path.getmtime(path)
                except OSError:
                    continue
                if state["mtime"] is None:
                    state["mtime"] = m
                    continue
                if m != state["mtime"]:
                    state["mtime"] = m
                    try:
                        self.load_file(path)
                    except Exception:
                        pass  # bad config keeps the previous values

        t = threading.Thread(target=loop, daemon=True, name="config-watch")
        t.start()
        return stop.set


@dataclass
class ResolvedExecutionConfig:
    """Per-step effective execution config (reference: resolver.go
    ResolvedExecutionConfig)."""

    timeout_seconds: _t.Optional[float] = None
    max_retries: int = 0
    retry_delay: float = 1.0
    retry_max_delay: float = 60.0
    retry_jitter_pct: int = 10
    backoff: BackoffStrategy = BackoffStrategy.EXPONENTIAL
    cache_enabled: bool = False
    cache_mode: CacheMode = CacheMode.READ_WRITE
    cache_key_template: _t.Optional[str] = None
    cache_salt: str = ""
    cache_ttl_seconds: _t.Optional[float] = None
    max_inline_size: int = 8 << 10
    placement_gpu: _t.Optional[int] = None
    placement_gpus: _t.Optional[_t.List[int]] = None
    mode: str = "job"
    override_layers: _t.List[str] = field(default_factory=list)  # attribution


class ExecutionConfigResolver:
    """5-layer hierarchical merge (reference: resolver.go:257-321)."""

    def __init__(self, config: EngineConfig):
        self.config = config

    def resolve(
        self,
        step: _t.Optional[T.Step] = None,
        story: _t.Optional[T.Story] = None,
        engram: _t.Optional[T.Engram] = None,
        template: _t.Optional[T.EngramTemplate] = None,
    ) -> ResolvedExecutionConfig:
        cfg = self.config
        out = ResolvedExecutionConfig(
            timeout_seconds=cfg.default_step_timeout,
            max_retries=cfg.default_max_retries,
            retry_delay=cfg.default_retry_delay,
            retry_max_delay=cfg.default_retry_max_delay,
            retry_jitter_pct=cfg.default_retry_jitter_pct,
            backoff=cfg.default_backoff,
            max_inline_size=cfg.max_inline_size,
        )
        out.override_layers.append("operator")

        # layer 2: template recommendations
        if template is not None and template.execution_policy is not None:
            pol = template.execution_policy
            self._apply(out, timeout=pol.timeout, retry=pol.retry, cache=pol.cache)
            out.override_layers.append(f"template:{template.name}")
        if template is not None and template.supported_modes:
            out.mode = str(template.supported_modes[0])

        # layer 3: engram overrides
        if engram is not None:
            if engram.mode is not None:
                out.mode = str(engram.mode)
            if engram.execution is not None:
                e = engram.execution
                self._apply(
                    out,
                    timeout=e.timeout,
                    retry=e.retry,
                    cache=e.cache,
                    placement=e.placement,
                    max_inline=e.max_inline_size,
                )
                out.override_layers.append(f"engram:{engram.name}")

        # layer 4: story policy
        if story is not None and story.policy is not None:
            pol = story.policy
            if pol.timeouts is not None and pol.timeouts.step:
                out.timeout_seconds = parse_duration(pol.timeouts.step)
            if pol.retries is not None and pol.retries.step_retry_policy is not None:
                self._apply(out, retry=pol.retries.step_retry_policy)
            if pol.execution is not None:
                e = pol.execution
                self._apply(
                    out,
                    timeout=e.timeout,
                    retry=e.retry,
                    cache=e.cache,
                    placement=e.placement,
                )
            out.override_layers.append(f"story:{story.name}")

        # layer 5: step overrides
        if step is not None and step.execution is not None:
            e = step.execution
            self._apply(
                out,
                timeout=e.timeout,
                retry=e.retry,
                cache=e.cache,
                placement=e.placement,
                max_inline=e.max_inline_size,
            )
            out.override_layers.append(f"step:{step.name}")

        # side-effect steps default to 0 retries unless idempotency-keyed
        # (reference: story_types.go:195-199)
        if step is not None and step.side_effects and step.idempotency_key_template is None:
            out.max_retries = 0

        out.max_retries = min(out.max_retries, cfg.max_retries_cap)
        return out

    @staticmethod
    def _apply(
        out: ResolvedExecutionConfig,
        timeout=None,
        retry: _t.Optional[T.RetryPolicy] = None,
        cache: _t.Optional[T.CachePolicy] = None,
        placement: _t.Optional[T.PlacementPolicy] = None,
        max_inline: _t.Optional[int] = None,
    ) -> None:
        if timeout is not None:
            out.timeout_seconds = parse_duration(timeout)
        if retry is not None:
            if retry.max_retries is not None:
                out.max_retries = retry.max_retries
            if retry.delay is not None:
                out.retry_delay = parse_duration(retry.delay) or 0.0
            if retry.max_delay is not None:
                out.retry_max_delay = parse_duration(retry.max_delay) or 0.0
            if retry.jitter is not None:
                out.retry_jitter_pct = retry.jitter
            if retry.backoff is not None:
                out.backoff = retry.backoff
        if cache is not None:
            if cache.enabled is not None:
                out.cache_enabled = cache.enabled
            if cache.mode is not None:
                out.cache_mode = CacheMode(cache.mode)
            if cache.key is not None:
                out.cache_key_template = cache.key
            if cache.salt is not None:
                out.cache_salt = cache.salt
            if cache.ttl_seconds is not None:
                out.cache_ttl_seconds = float(cache.ttl_seconds)
        if placement is not None:
            if placement.gpu is not None:
                out.placement_gpu = placement.gpu
            if placement.gpus is not None:
                out.placement_gpus = list(placement.gpus)
        if max_inline is not None:
            out.max_inline_size = max_inline
