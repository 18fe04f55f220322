"""StoryTrigger admission: durable, deduplicating trigger → StoryRun path.

Role parity with the reference's StoryTrigger controller
(reference: internal/controller/runs/storytrigger_controller.go:70-535 —
identity validation (submissionID + optional key + inputHash), cross-ns
policy, dedupe by deterministic StoryRun name DeriveStoryRunName, decision
Created/Reused/Rejected) plus the Impulse-side throttle/delivery policy
(reference: impulse_controller.go:1477-1533).
"""
from __future__ import annotations

import threading
import time
import typing as _t
from dataclasses import dataclass, field

from ..enums import TriggerDecision
from ..specs import types as T
from .records import StoryRun, derive_story_run_name, input_hash, monotonic_now

if _t.TYPE_CHECKING:
    from .engine import RunEngine


@dataclass
class StoryTrigger:
    """A durable trigger-admission request (reference:
    storytrigger_types.go:27-155)."""

    submission_id: str
    story_name: str
    story_namespace: str = "default"
    namespace: str = "default"
    key: _t.Optional[str] = None  # user dedupe key
    inputs: _t.Any = None
    impulse: _t.Optional[str] = None
    # status -----------------------------------------------------------
    decision: TriggerDecision = TriggerDecision.PENDING
    story_run_ref: _t.Optional[str] = None
    input_hash: str = ""
    message: str = ""
    created_at: float = field(default_factory=monotonic_now)

    @property
    def dedupe_token(self) -> str:
        """Deterministic identity: key if given, else submissionID+inputHash."""
        if self.key:
            return f"key:{self.key}"
        return f"sub:{self.submission_id}:{self.input_hash}"


class RateLimiter:
    """Token bucket (reference TriggerThrottlePolicy ratePerSecond/burst)."""

    def __init__(self, rate: float, burst: int):
        self.rate = rate
        self.burst = max(burst, 1)
        self._tokens = float(self.burst)
        self._last = time.monotonic()
        self._lock = threading.Lock()

    def allow(self) -> bool:
        with self._lock:
            now = time.monotonic()
            self._tokens = min(self.burst, self._tokens + (now - self._last) * self.rate)
            self._last = now
            if self._tokens >= 1.0:
                self._tokens -= 1.0
                return True
            return False


class TriggerAdmission:
    def __init__(self, engine: "RunEngine"):
        self.engine = engine
        self.triggers: _t.Dict[str, StoryTrigger] = {}
        self._lock = threading.Lock()
        self._limiters: _t.Dict[str, RateLimiter] = {}

    def submit(
        self,
        trigger: StoryTrigger,
        throttle: _t.Optional[T.TriggerThrottlePolicy] = None,
    ) -> StoryTrigger:
        """Admit one trigger; idempotent per identity."""
        eng = self.engine
        if not trigger.submission_id:
            trigger.decision = TriggerDecision.REJECTED
            trigger.message = "submissionID is required"
            return trigger
        trigger.input_hash = input_hash(trigger.inputs)

        # cross-namespace policy (reference: validateStoryRefAccess 157-193)
        if trigger.story_namespace != trigger.namespace:
            if not eng.registry.allows_cross_namespace(
                "StoryTrigger", trigger.namespace, "Story", trigger.story_namespace, trigger.story_name
            ):
                trigger.decision = TriggerDecision.REJECTED
                trigger.message = "cross-namespace story reference not granted"
                return trigger

        try:
            story = eng.registry.story(trigger.story_name, trigger.story_namespace)
        except KeyError:
            trigger.decision = TriggerDecision.REJECTED
            trigger.message = f"story {trigger.story_namespace}/{trigger.story_name} not found"
            return trigger

        # throttle (reference: impulse trigger delivery/throttle policy)
        if throttle is not None:
            if throttle.max_in_flight:
                in_flight = sum(
                    1
                    for r in eng.store.runs_of_story(story.key)
                    if not r.is_terminal
                )
                if in_flight >= throttle.max_in_flight:
                    trigger.decision = TriggerDecision.REJECTED
                    trigger.message = f"maxInFlight {throttle.max_in_flight} reached"
                    eng.metrics.inc("impulse_throttled_triggers_total", reason="maxInFlight")
                    return trigger
            if throttle.rate_per_second:
                limiter = self._limiters.setdefault(
                    story.key,
                    RateLimiter(throttle.rate_per_second, throttle.burst or throttle.rate_per_second),
                )
                if not limiter.allow():
                    trigger.decision = TriggerDecision.REJECTED
                    trigger.message = "rate limited"
                    return trigger

        # dedupe via deterministic run naming
        run_name = derive_story_run_name(
            trigger.namespace, trigger.story_name, trigger.dedupe_token
        )
        run_key = f"{trigger.namespace}/{run_name}"
        with self._lock:
            existing_run = eng.store.try_get_story_run(run_key)
            if existing_run is not None:
                if self._run_matches(existing_run, trigger):
                    trigger.decision = TriggerDecision.REUSED
                    trigger.story_run_ref = run_key
                else:
                    trigger.decision = TriggerDecision.REJECTED
                    trigger.message = (
                        "identity collision: an existing run with this trigger "
                        "identity has different inputs"
                    )
                self.triggers[trigger.dedupe_token] = trigger
                return trigger

            run = eng.submit_run(
                story=story,
                inputs=trigger.inputs,
                name=run_name,
                namespace=trigger.namespace,
                trigger_token=trigger.dedupe_token,
                labels={"impulse": trigger.impulse} if trigger.impulse else None,
            )
            trigger.decision = TriggerDecision.CREATED
            trigger.story_run_ref = run.key
            self.triggers[trigger.dedupe_token] = trigger
            eng.metrics.inc("storytriggers_total", decision=str(trigger.decision))
            return trigger

    def _run_matches(self, run: StoryRun, trigger: StoryTrigger) -> bool:
        """(reference: storyRunMatchesTrigger 331-406): trigger-token match
        plus input-hash immutability and impulse provenance."""
        # same identity resubmitted with DIFFERENT inputs is a conflict —
        # compare against the recorded trigger (raw-input hash; the run's
        # own hash may include schema defaults)
        prior = self.triggers.get(trigger.dedupe_token)
        if prior is not None and prior.input_hash != trigger.input_hash:
            return False
        # impulse provenance must agree (reference: "does not reuse a
        # StoryRun when impulse provenance differs")
        if trigger.impulse and run.labels.get("impulse") not in (None, trigger.impulse):
            return False
        if trigger.dedupe_token in run.trigger_tokens:
            return True
        return run.input_hash == trigger.input_hash
