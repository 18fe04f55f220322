"""EffectClaim ledger: lease-based exactly-once side-effect reservations.

Role parity with the reference's EffectClaim CRD + reconciler
(reference: api/runs/v1alpha1/effectclaim_types.go:25-155 — holderIdentity,
leaseDurationSeconds, acquire/renew times, phases Reserved/Completed/
Released/Abandoned with stale-holder takeover;
effectclaim_controller.go:57-181).
"""
from __future__ import annotations

import threading
import time
import typing as _t
from dataclasses import dataclass, field

from ..enums import EffectClaimPhase

DEFAULT_LEASE_SECONDS = 60.0


@dataclass
class EffectClaim:
    key: str  # idempotency key (namespaced by run/step by the caller)
    holder: str
    phase: EffectClaimPhase = EffectClaimPhase.RESERVED
    lease_duration: float = DEFAULT_LEASE_SECONDS
    acquired_at: float = field(default_factory=time.time)
    renewed_at: float = field(default_factory=time.time)
    takeovers: int = 0
    description: str = ""

    @property
    def expired(self) -> bool:
        return (
            self.phase == EffectClaimPhase.RESERVED
            and (time.time() - self.renewed_at) > self.lease_duration
        )


class EffectLedger:
    """Append-only claim table; exactly-once decisions for side effects."""

    def __init__(self):
        self._claims: _t.Dict[str, EffectClaim] = {}
        self._lock = threading.Lock()

    def acquire(
        self,
        key: str,
        holder: str,
        lease_duration: float = DEFAULT_LEASE_SECONDS,
        description: str = "",
    ) -> _t.Tuple[_t.Optional[EffectClaim], bool]:
        """Try to reserve. Returns (claim, fresh):
        fresh=True  → the holder owns the effect and must perform it;
        fresh=False → already Completed (skip the side effect) or held by a
                      live other holder (back off)."""
        with self._lock:
            cur = self._claims.get(key)
            if cur is None:
                claim = EffectClaim(key=key, holder=holder, lease_duration=lease_duration, description=description)
                self._claims[key] = claim
                return claim, True
            if cur.phase == EffectClaimPhase.COMPLETED:
                return cur, False
            if cur.holder == holder and cur.phase == EffectClaimPhase.RESERVED:
                cur.renewed_at = time.time()
                return cur, True
            if cur.expired or cur.phase in (EffectClaimPhase.RELEASED, EffectClaimPhase.ABANDONED):
                # stale-holder takeover (effectclaim_types.go:61-84)
                cur.holder = holder
                cur.phase = EffectClaimPhase.RESERVED
                cur.acquired_at = time.time()
                cur.renewed_at = cur.acquired_at
                cur.takeovers += 1
                cur.lease_duration = lease_duration
                return cur, True
            return cur, False

    def renew(self, key: str, holder: str) -> bool:
        with self._lock:
            cur = self._claims.get(key)
            if cur is None or cur.holder != holder or cur.phase != EffectClaimPhase.RESERVED:
                return False
            cur.renewed_at = time.time()
            return True

    def complete(self, key: str, holder: str) -> bool:
        with self._lock:
            cur = self._claims.get(key)
            if cur is None or cur.holder != holder:
                return False
            cur.phase = EffectClaimPhase.COMPLETED
            return True

    def release(self, key: str, holder: str) -> bool:
        """Give the effect back without completing (retry may re-run it)."""
        with self._lock:
            cur = self._claims.get(key)
            if cur is None or cur.holder != holder or cur.phase == EffectClaimPhase.COMPLETED:
                return False
            cur.phase = EffectClaimPhase.RELEASED
            return True

    def abandon_stale(self) -> int:
        with self._lock:
            n = 0
            for claim in self._claims.values():
                if claim.expired:
                    claim.phase = EffectClaimPhase.ABANDONED
                    n += 1
            return n

    def prune_prefix(self, prefix: str) -> int:
        """Drop claims under a key prefix (claims are run-scoped:
        "run/step/key").  Called when the owning StoryRun is deleted by
        retention cleanup — the reference's orphan-claim GC (EffectClaim
        owner references cascade when the StepRun disappears)."""
        with self._lock:
            doomed = [k for k in self._claims if k.startswith(prefix)]
            for k in doomed:
                del self._claims[k]
            return len(doomed)

    def get(self, key: str) -> _t.Optional[EffectClaim]:
        with self._lock:
            return self._claims.get(key)

    def __len__(self) -> int:
        with self._lock:
            return len(self._claims)
