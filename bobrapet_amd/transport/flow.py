"""Credit-based flow control rings for streaming edges.

Keeps the reference's streaming-settings vocabulary as behavior
(reference: api/transport/v1alpha1/transport_settings_types.go:225-300 —
credit flow control, backpressure buffers + policy block/dropOldest/
dropNewest, delivery semantics counters, per-lane priority).
"""
from __future__ import annotations

import threading
import typing as _t
from collections import deque
from dataclasses import dataclass, field

from ..specs.types import TransportStreamingSettings, default_streaming_settings


class RingClosed(Exception):
    pass


SENTINEL = object()  # end-of-stream marker


@dataclass
class RingStats:
    pushed: int = 0
    popped: int = 0
    dropped_oldest: int = 0
    dropped_newest: int = 0
    blocked_waits: int = 0


class CreditRing:
    """Bounded packet ring with credit-based flow control.

    The consumer grants credits (initially `initial_credits`); the producer
    blocks (or drops, per backpressure policy) when credits are exhausted.
    Credits replenish as packets are consumed."""

    def __init__(
        self,
        name: str = "",
        settings: _t.Optional[TransportStreamingSettings] = None,
        lane: str = "data",
    ):
        self.name = name
        self.lane = lane
        s = settings or default_streaming_settings()
        fc = s.flow_control
        bp = s.backpressure
        self._credit_mode = fc is not None and fc.mode == "credit"
        self._credits = float(fc.initial_credits or 32) if self._credit_mode else float("inf")
        self._max_credits = float(fc.max_credits or 256) if self._credit_mode else float("inf")
        self._capacity = (bp.buffer_packets if bp and bp.buffer_packets else 256)
        self._policy = (bp.policy if bp and bp.policy else "block")
        self._q: deque = deque()
        self._lock = threading.Lock()
        self._not_empty = threading.Condition(self._lock)
        self._can_push = threading.Condition(self._lock)
        self._closed = False
        self.stats = RingStats()

    def push(self, packet, timeout: _t.Optional[float] = None) -> bool:
        """Returns False when the packet was dropped (dropNewest) or the ring
        closed; True when enqueued."""
        with self._lock:
            while True:
                if self._closed:
                    return False
                room = len(self._q) < self._capacity and self._credits >= 1
                if room:
                    break
                if self._policy == "dropNewest":
                    self.stats.dropped_newest += 1
                    return False
                if self._policy == "dropOldest" and self._q:
                    self._q.popleft()
                    self.stats.dropped_oldest += 1
                    continue
                self.stats.blocked_waits += 1
                if not self._can_push.wait(timeout=timeout):
                    return False
            if self._credit_mode:
                self._credits -= 1
            self._q.append(packet)
            self.stats.pushed += 1
            self._not_empty.notify()
            return True

    def pop(self, timeout: _t.Optional[float] = None):
        """Blocks for the next packet; returns SENTINEL at end-of-stream,
        raises RingClosed if closed without drain."""
        with self._lock:
            while not self._q:
                if self._closed:
                    return SENTINEL
                if not self._not_empty.wait(timeout=timeout):
                    raise TimeoutError(f"ring {self.name}: pop timed out")
            packet = self._q.popleft()
            self.stats.popped += 1
            if self._credit_mode:
                self._credits = min(self._credits + 1, self._max_credits)
            self._can_push.notify()
            return packet

    def close(self) -> None:
        with self._lock:
            self._closed = True
            self._not_empty.notify_all()
            self._can_push.notify_all()

    @property
    def depth(self) -> int:
        with self._lock:
            return len(self._q)

    @property
    def credits(self) -> float:
        with self._lock:
            return self._credits
