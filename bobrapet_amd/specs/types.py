"""Spec types: the declarative API surface of bobrapet_amd.

Field-for-field vocabulary parity with the reference CRDs
(reference: api/v1alpha1/story_types.go:40-436, shared_types.go:31-563,
engram_types.go:36-158, impulse_types.go:39-156,
api/catalog/v1alpha1/*.go, api/transport/v1alpha1/*.go), so Story/Engram/
Impulse YAML written for the reference loads unchanged.  These are plain
dataclasses — no apiserver: validation happens at submit time
(specs/validation.py) instead of in admission webhooks.
"""
from __future__ import annotations

import dataclasses
import typing as _t
from dataclasses import dataclass, field

from ..enums import (
    BackoffStrategy,
    StepType,
    StoryPattern,
    TransportMode,
    WorkloadMode,
)

JSON = _t.Union[None, bool, int, float, str, _t.List["JSON"], _t.Dict[str, "JSON"]]


# ---------------------------------------------------------------------------
# generic from_dict machinery
# ---------------------------------------------------------------------------

_MISSING = object()


def _convert(value, typ):
    """Convert a plain JSON value to the annotated type (best effort)."""
    if value is None:
        return None
    origin = _t.get_origin(typ)
    if origin is _t.Union:
        args = [a for a in _t.get_args(typ) if a is not type(None)]
        if len(args) == 1:
            return _convert(value, args[0])
        return value
    if origin in (list, _t.List):
        (item_t,) = _t.get_args(typ) or (None,)
        return [_convert(v, item_t) for v in value]
    if origin in (dict, _t.Dict):
        args = _t.get_args(typ)
        val_t = args[1] if len(args) == 2 else None
        return {k: _convert(v, val_t) for k, v in value.items()}
    if isinstance(typ, type):
        if dataclasses.is_dataclass(typ):
            return from_dict(typ, value)
        if issubclass(typ, (StepType, StoryPattern, WorkloadMode, BackoffStrategy, TransportMode)):
            return typ(value)
        try:
            import enum as _enum

            if issubclass(typ, _enum.Enum):
                return typ(value)
        except TypeError:
            pass
    return value


def from_dict(cls, data: _t.Optional[dict]):
    """Build a spec dataclass from a plain dict, mapping camelCase keys.

    Unknown keys are collected into ``extra`` (if the dataclass has one) so
    forward-compatible YAML round-trips instead of erroring.
    """
    if data is None:
        return None
    if not isinstance(data, dict):
        raise TypeError(f"{cls.__name__}: expected a mapping, got {type(data).__name__}")
    hints = _t.get_type_hints(cls)
    fields = {f.name: f for f in dataclasses.fields(cls)}
    kwargs = {}
    extra = {}
    for key, value in data.items():
        name = _camel_to_snake(key)
        if name in fields:
            kwargs[name] = _convert(value, hints.get(name))
        else:
            extra[key] = value
    if extra and "extra" in fields:
        kwargs["extra"] = extra  # in-constructor so __post_init__ sees it
    obj = cls(**kwargs)
    if extra and "extra" not in fields and hasattr(obj, "extra"):
        obj.extra = extra
    return obj


def _camel_to_snake(name: str) -> str:
    out = []
    for i, ch in enumerate(name):
        if ch.isupper() and i and (not name[i - 1].isupper()):
            out.append("_")
        out.append(ch.lower())
    return "".join(out)


def _snake_to_camel(name: str) -> str:
    parts = name.split("_")
    return parts[0] + "".join(p.title() for p in parts[1:])


def to_dict(obj) -> JSON:
    """Serialize a spec dataclass back to camelCase JSON (drops None/empty)."""
    if dataclasses.is_dataclass(obj) and not isinstance(obj, type):
        out = {}
        for f in dataclasses.fields(obj):
            if f.name == "extra":
                continue
            v = to_dict(getattr(obj, f.name))
            if v is None or v == [] or v == {}:
                continue
            out[_snake_to_camel(f.name)] = v
        extra = getattr(obj, "extra", None)
        if extra:
            out.update(extra)
        return out
    if isinstance(obj, dict):
        return {k: to_dict(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [to_dict(v) for v in obj]
    import enum as _enum

    if isinstance(obj, _enum.Enum):
        return obj.value
    return obj


# ---------------------------------------------------------------------------
# shared policy types (reference: api/v1alpha1/shared_types.go)
# ---------------------------------------------------------------------------


@dataclass
class RetryPolicy:
    """Retry policy for a step (reference: shared_types.go:400-429).

    max_retries caps at 10; delay strings are Go-style durations ("1s")."""

    max_retries: _t.Optional[int] = None
    delay: _t.Optional[str] = None
    max_delay: _t.Optional[str] = None
    jitter: _t.Optional[int] = None  # percent 0-100
    backoff: _t.Optional[BackoffStrategy] = None
    extra: dict = field(default_factory=dict)


@dataclass
class CachePolicy:
    """Step output cache (reference: shared_types.go:249-277)."""

    enabled: _t.Optional[bool] = None
    key: _t.Optional[str] = None  # template override for the cache key
    salt: _t.Optional[str] = None
    mode: _t.Optional[str] = None  # read | write | readWrite
    ttl_seconds: _t.Optional[int] = None
    extra: dict = field(default_factory=dict)


@dataclass
class FileStorageProvider:
    path: _t.Optional[str] = None
    extra: dict = field(default_factory=dict)


@dataclass
class HBMStorageProvider:
    """MI355X-native storage tier: payload blobs live in device HBM
    (288 GB/GPU), spilling to pinned host memory then disk.  Replaces the
    reference's S3 provider role (shared_types.go:498-534)."""

    device: _t.Optional[int] = None
    capacity_bytes: _t.Optional[int] = None
    spill_dir: _t.Optional[str] = None
    extra: dict = field(default_factory=dict)


@dataclass
class StoragePolicy:
    """Where oversized payloads offload (reference: shared_types.go:496-557)."""

    file: _t.Optional[FileStorageProvider] = None
    hbm: _t.Optional[HBMStorageProvider] = None
    timeout_seconds: _t.Optional[int] = None
    extra: dict = field(default_factory=dict)


@dataclass
class ResourceRequests:
    cpu: _t.Optional[str] = None
    memory: _t.Optional[str] = None
    # MI355X-native resource vocabulary: device memory + CU share per step
    hbm: _t.Optional[str] = None
    compute_units: _t.Optional[int] = None
    extra: dict = field(default_factory=dict)


@dataclass
class WorkloadResources:
    requests: _t.Optional[ResourceRequests] = None
    limits: _t.Optional[ResourceRequests] = None
    extra: dict = field(default_factory=dict)


@dataclass
class PlacementPolicy:
    """Placement constraints.  The reference targets K8s nodes
    (shared_types.go:356-366); here placement is (gpu, stream)."""

    gpu: _t.Optional[int] = None  # pin to a device index
    gpus: _t.Optional[_t.List[int]] = None  # allowed devices
    node_selector: _t.Optional[_t.Dict[str, str]] = None
    extra: dict = field(default_factory=dict)


@dataclass
class JobPolicy:
    ttl_seconds_after_finished: _t.Optional[int] = None
    backoff_limit: _t.Optional[int] = None
    active_deadline_seconds: _t.Optional[int] = None
    parallelism: _t.Optional[int] = None
    completions: _t.Optional[int] = None
    extra: dict = field(default_factory=dict)


@dataclass
class ExecutionOverrides:
    """Per-step execution overrides (reference: shared_types.go:92-147)."""

    timeout: _t.Optional[str] = None
    retry: _t.Optional[RetryPolicy] = None
    debug: _t.Optional[bool] = None
    placement: _t.Optional[PlacementPolicy] = None
    max_inline_size: _t.Optional[int] = None
    storage: _t.Optional[StoragePolicy] = None
    cache: _t.Optional[CachePolicy] = None
    resources: _t.Optional[WorkloadResources] = None
    job: _t.Optional[JobPolicy] = None
    extra: dict = field(default_factory=dict)


@dataclass
class ExecutionPolicy:
    """Story/template-level execution policy (reference: shared_types.go:171-218)."""

    resources: _t.Optional[WorkloadResources] = None
    placement: _t.Optional[PlacementPolicy] = None
    job: _t.Optional[JobPolicy] = None
    retry: _t.Optional[RetryPolicy] = None
    timeout: _t.Optional[str] = None
    max_recursion_depth: _t.Optional[int] = None
    storage: _t.Optional[StoragePolicy] = None
    cache: _t.Optional[CachePolicy] = None
    extra: dict = field(default_factory=dict)


@dataclass
class TriggerDedupePolicy:
    mode: _t.Optional[str] = None  # submission | key | inputHash
    key_template: _t.Optional[str] = None
    extra: dict = field(default_factory=dict)


@dataclass
class TriggerRetryPolicy:
    max_attempts: _t.Optional[int] = None
    base_delay: _t.Optional[str] = None
    max_delay: _t.Optional[str] = None
    backoff: _t.Optional[BackoffStrategy] = None
    extra: dict = field(default_factory=dict)


@dataclass
class TriggerDeliveryPolicy:
    dedupe: _t.Optional[TriggerDedupePolicy] = None
    retry: _t.Optional[TriggerRetryPolicy] = None
    extra: dict = field(default_factory=dict)


@dataclass
class TriggerThrottlePolicy:
    max_in_flight: _t.Optional[int] = None
    rate_per_second: _t.Optional[int] = None
    burst: _t.Optional[int] = None
    extra: dict = field(default_factory=dict)


# ---------------------------------------------------------------------------
# references (reference: pkg/refs/refs.go)
# ---------------------------------------------------------------------------


@dataclass(frozen=True)
class ObjectRef:
    """Namespaced reference to another object."""

    name: str = ""
    namespace: _t.Optional[str] = None

    def resolve_namespace(self, default: str) -> str:
        return self.namespace or default

    def key(self, default_ns: str) -> str:
        return f"{self.resolve_namespace(default_ns)}/{self.name}"


# ---------------------------------------------------------------------------
# Story (reference: api/v1alpha1/story_types.go)
# ---------------------------------------------------------------------------

MAX_STEPS = 100
MAX_COMPENSATIONS = 50
MAX_FINALLY = 50


@dataclass
class PostExecutionCheck:
    """Condition template evaluated against a step's output after success
    (reference: story_types.go:286-298)."""

    condition: str = ""
    failure_message: _t.Optional[str] = None
    extra: dict = field(default_factory=dict)


@dataclass
class Step:
    """One step of a Story (reference: story_types.go:156-284).

    Exactly one of ``ref`` (engram step) or ``type`` (primitive) is set."""

    name: str = ""
    id: _t.Optional[str] = None  # template alias (underscored) for the name
    needs: _t.List[str] = field(default_factory=list)
    type: _t.Optional[StepType] = None
    if_: _t.Optional[str] = None  # YAML key "if"
    allow_failure: _t.Optional[bool] = None
    side_effects: _t.Optional[bool] = None
    requires: _t.List[str] = field(default_factory=list)  # dot-paths that must be non-nil
    idempotency_key_template: _t.Optional[str] = None
    ref: _t.Optional[ObjectRef] = None
    with_: _t.Optional[JSON] = None  # YAML key "with"
    runtime: _t.Optional[JSON] = None  # per-packet templates (streaming hub routing)
    transport: _t.Optional[str] = None
    secrets: _t.Dict[str, str] = field(default_factory=dict)
    execution: _t.Optional[ExecutionOverrides] = None
    post_execution: _t.Optional[PostExecutionCheck] = None
    extra: dict = field(default_factory=dict)

    @property
    def is_primitive(self) -> bool:
        return self.type is not None

    @property
    def alias(self) -> str:
        """Template alias: explicit id, else the name with '-' → '_'."""
        return self.id or self.name.replace("-", "_")


def _step_from_dict(data: dict) -> Step:
    data = dict(data)
    if "if" in data:
        data["if_"] = data.pop("if")
    if "with" in data:
        data["with_"] = data.pop("with")
    return from_dict(Step, data)


@dataclass
class StoryRetries:
    step_retry_policy: _t.Optional[RetryPolicy] = None
    continue_on_step_failure: _t.Optional[bool] = None
    extra: dict = field(default_factory=dict)


@dataclass
class StoryTimeouts:
    story: _t.Optional[str] = None
    step: _t.Optional[str] = None
    graceful_shutdown_timeout: _t.Optional[str] = None
    extra: dict = field(default_factory=dict)


@dataclass
class StoryPolicy:
    """Story-level policy (reference: story_types.go:300-331)."""

    timeouts: _t.Optional[StoryTimeouts] = None
    with_: _t.Optional[JSON] = None  # defaults merged under run inputs
    retries: _t.Optional[StoryRetries] = None
    concurrency: _t.Optional[int] = None
    queue: _t.Optional[str] = None
    priority: _t.Optional[int] = None
    storage: _t.Optional[StoragePolicy] = None
    execution: _t.Optional[ExecutionPolicy] = None
    extra: dict = field(default_factory=dict)


def _story_policy_from_dict(data: dict) -> StoryPolicy:
    data = dict(data)
    if "with" in data:
        data["with_"] = data.pop("with")
    return from_dict(StoryPolicy, data)


@dataclass
class StoryTransport:
    """A transport declared for use by a Story's streaming steps
    (reference: story_types.go:404-421)."""

    name: str = ""
    transport_ref: str = ""
    description: _t.Optional[str] = None
    streaming: _t.Optional[JSON] = None  # TransportStreamingSettings overrides
    settings: _t.Optional[JSON] = None
    extra: dict = field(default_factory=dict)


@dataclass
class Story:
    """Workflow definition (reference: story_types.go:40-153)."""

    name: str = ""
    namespace: str = "default"
    version: _t.Optional[str] = None
    pattern: StoryPattern = StoryPattern.BATCH
    inputs_schema: _t.Optional[JSON] = None
    outputs_schema: _t.Optional[JSON] = None
    output: _t.Optional[JSON] = None  # output template over {inputs, steps}
    steps: _t.List[Step] = field(default_factory=list)
    compensations: _t.List[Step] = field(default_factory=list)
    finally_: _t.List[Step] = field(default_factory=list)
    policy: _t.Optional[StoryPolicy] = None
    transports: _t.List[StoryTransport] = field(default_factory=list)
    labels: _t.Dict[str, str] = field(default_factory=dict)
    annotations: _t.Dict[str, str] = field(default_factory=dict)
    generation: int = 1
    extra: dict = field(default_factory=dict)

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.name}"

    def step(self, name: str) -> _t.Optional[Step]:
        for s in self.steps:
            if s.name == name:
                return s
        return None

    def all_steps(self) -> _t.List[Step]:
        return list(self.steps) + list(self.compensations) + list(self.finally_)


# ---------------------------------------------------------------------------
# Engram / Impulse (reference: engram_types.go, impulse_types.go)
# ---------------------------------------------------------------------------


@dataclass
class Engram:
    """Configured worker instance: templateRef + with + mode + overrides
    (reference: engram_types.go:36-158)."""

    name: str = ""
    namespace: str = "default"
    template_ref: _t.Optional[ObjectRef] = None
    with_: _t.Optional[JSON] = None
    secrets: _t.Dict[str, str] = field(default_factory=dict)
    mode: _t.Optional[WorkloadMode] = None
    execution: _t.Optional[ExecutionOverrides] = None
    generation: int = 1
    extra: dict = field(default_factory=dict)

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.name}"


@dataclass
class ImpulseMapping:
    """Maps trigger payload into StoryRun inputs (template JSON)."""

    inputs: _t.Optional[JSON] = None
    extra: dict = field(default_factory=dict)


@dataclass
class Impulse:
    """Always-on trigger instance (reference: impulse_types.go:39-156)."""

    name: str = ""
    namespace: str = "default"
    template_ref: _t.Optional[ObjectRef] = None
    story_ref: _t.Optional[ObjectRef] = None
    with_: _t.Optional[JSON] = None
    mapping: _t.Optional[ImpulseMapping] = None
    delivery: _t.Optional[TriggerDeliveryPolicy] = None
    throttle: _t.Optional[TriggerThrottlePolicy] = None
    mode: _t.Optional[WorkloadMode] = None
    generation: int = 1
    extra: dict = field(default_factory=dict)

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.name}"


# ---------------------------------------------------------------------------
# Catalog templates (reference: api/catalog/v1alpha1/)
# ---------------------------------------------------------------------------


@dataclass
class TemplateExecutionPolicy:
    """Recommended execution policy carried by a template
    (reference: catalog shared_types.go:76-107)."""

    timeout: _t.Optional[str] = None
    retry: _t.Optional[RetryPolicy] = None
    resources: _t.Optional[WorkloadResources] = None
    cache: _t.Optional[CachePolicy] = None
    extra: dict = field(default_factory=dict)


@dataclass
class SecretDefinition:
    name: str = ""
    required: _t.Optional[bool] = None
    description: _t.Optional[str] = None
    mount_type: _t.Optional[str] = None  # env | file | both
    extra: dict = field(default_factory=dict)


@dataclass
class EngramTemplate:
    """Reusable worker package: implementation + schemas + recommended policy
    (reference: catalog/engramtemplate_types.go:47-122).

    ``image`` in the reference names a container; here ``builtin`` names a
    registered in-process engram implementation (engrams/registry.py)."""

    name: str = ""
    version: _t.Optional[str] = None
    builtin: _t.Optional[str] = None  # registered engram implementation name
    image: _t.Optional[str] = None  # accepted for YAML parity; resolved via registry
    command: _t.List[str] = field(default_factory=list)  # external-process engram
    supported_modes: _t.List[WorkloadMode] = field(default_factory=list)
    config_schema: _t.Optional[JSON] = None
    secret_schema: _t.Optional[JSON] = None
    input_schema: _t.Optional[JSON] = None
    output_schema: _t.Optional[JSON] = None
    declared_output_keys: _t.List[str] = field(default_factory=list)
    execution_policy: _t.Optional[TemplateExecutionPolicy] = None
    secrets: _t.List[SecretDefinition] = field(default_factory=list)
    generation: int = 1
    extra: dict = field(default_factory=dict)

    @property
    def key(self) -> str:  # cluster-scoped in the reference
        return self.name

    @property
    def implementation(self) -> _t.Optional[str]:
        return self.builtin or self.image


@dataclass
class ImpulseTemplate:
    """Reusable trigger package (reference: catalog/impulsetemplate_types.go)."""

    name: str = ""
    version: _t.Optional[str] = None
    builtin: _t.Optional[str] = None
    image: _t.Optional[str] = None
    supported_modes: _t.List[WorkloadMode] = field(default_factory=list)
    config_schema: _t.Optional[JSON] = None
    secret_schema: _t.Optional[JSON] = None
    delivery: _t.Optional[TriggerDeliveryPolicy] = None
    generation: int = 1
    extra: dict = field(default_factory=dict)

    @property
    def key(self) -> str:
        return self.name

    @property
    def implementation(self) -> _t.Optional[str]:
        return self.builtin or self.image


# ---------------------------------------------------------------------------
# Transport (reference: api/transport/v1alpha1/)
# ---------------------------------------------------------------------------


@dataclass
class TransportLane:
    """One streaming lane: media/data/control (reference:
    transport_settings_types.go:133-161)."""

    name: str = ""
    priority: _t.Optional[int] = None
    max_bytes_per_second: _t.Optional[int] = None
    extra: dict = field(default_factory=dict)


@dataclass
class TransportFlowControl:
    """Credit-based flow control (reference: transport_settings_types.go:225-246)."""

    mode: _t.Optional[str] = None  # credit | none
    initial_credits: _t.Optional[int] = None
    max_credits: _t.Optional[int] = None
    low_watermark: _t.Optional[int] = None
    extra: dict = field(default_factory=dict)


@dataclass
class TransportDelivery:
    """Delivery semantics (reference: transport_settings_types.go:287-300)."""

    semantics: _t.Optional[str] = None  # atMostOnce | atLeastOnce | exactlyOnce
    ordering: _t.Optional[str] = None  # none | perLane | total
    extra: dict = field(default_factory=dict)


@dataclass
class TransportBackpressure:
    buffer_packets: _t.Optional[int] = None
    buffer_bytes: _t.Optional[int] = None
    policy: _t.Optional[str] = None  # block | dropOldest | dropNewest
    extra: dict = field(default_factory=dict)


@dataclass
class TransportReplay:
    """Replay storage/retention (reference:
    transport_settings_types.go:301-315, modes none|memory|durable)."""

    mode: str = "none"
    retention_seconds: _t.Optional[int] = None
    checkpoint_interval: _t.Optional[str] = None
    extra: dict = field(default_factory=dict)


@dataclass
class TransportRecording:
    """Stream recording (reference: transport_settings_types.go:508-529):
    metadata-only or full payloads, sampling, redaction."""

    mode: str = ""  # "" (off) | metadata | full
    sample_rate: _t.Optional[int] = None  # 0..100
    retention_seconds: _t.Optional[int] = None
    redact_fields: _t.List[str] = field(default_factory=list)
    extra: dict = field(default_factory=dict)


@dataclass
class TransportRoutingRule:
    """Conditional downstream routing (reference:
    transport_settings_types.go:353-370): `when` evaluated per packet,
    action allow|deny over the targeted downstream steps."""

    name: str = ""
    when: _t.Optional[str] = None
    action: str = "allow"
    steps: _t.List[str] = field(default_factory=list)  # empty = all downstreams
    extra: dict = field(default_factory=dict)

    def __post_init__(self):
        # accept the reference's nested target: {steps: [...]} shape
        if not self.steps and isinstance(self.extra.get("target"), dict):
            self.steps = list(self.extra["target"].get("steps") or [])


@dataclass
class TransportRouting:
    """Routing topology + per-packet rules (reference:
    transport_settings_types.go:372-390)."""

    mode: str = "auto"  # auto | hub | p2p
    fan_out: str = "parallel"
    max_downstreams: _t.Optional[int] = None
    rules: _t.List[TransportRoutingRule] = field(default_factory=list)
    extra: dict = field(default_factory=dict)


@dataclass
class TransportPartitioning:
    """Partition assignment for stream envelopes (reference:
    transport_settings_types.go:391-419, modes none|preserve|hash)."""

    mode: str = "none"
    key: _t.Optional[str] = None  # dot path into the packet, e.g. "meta.user"
    partitions: _t.Optional[int] = None
    sticky: bool = True
    extra: dict = field(default_factory=dict)


@dataclass
class TransportLifecycle:
    """Drain/cutover behavior for live streaming steps (reference:
    transport_settings_types.go:431-447)."""

    strategy: str = "drain"  # drain | recreate
    drain_timeout_seconds: _t.Optional[int] = None
    max_in_flight: _t.Optional[int] = None
    extra: dict = field(default_factory=dict)


@dataclass
class TransportFanIn:
    """Fan-in join behavior when multiple upstreams feed a step
    (reference: transport_settings_types.go:174-192, modes all|any|quorum)."""

    mode: str = "all"
    quorum: _t.Optional[int] = None
    timeout_seconds: _t.Optional[int] = None
    max_entries: _t.Optional[int] = None
    extra: dict = field(default_factory=dict)


@dataclass
class TransportStreamingSettings:
    """Streaming policy vocabulary (reference:
    transport_settings_types.go:68-110): lanes, flow control, backpressure,
    delivery/ordering semantics, fan-in joins."""

    lanes: _t.List[TransportLane] = field(default_factory=list)
    flow_control: _t.Optional[TransportFlowControl] = None
    delivery: _t.Optional[TransportDelivery] = None
    backpressure: _t.Optional[TransportBackpressure] = None
    fan_in: _t.Optional[TransportFanIn] = None
    replay: _t.Optional[TransportReplay] = None
    recording: _t.Optional[TransportRecording] = None
    lifecycle: _t.Optional[TransportLifecycle] = None
    partitioning: _t.Optional[TransportPartitioning] = None
    routing: _t.Optional[TransportRouting] = None
    extra: dict = field(default_factory=dict)


def default_streaming_settings() -> TransportStreamingSettings:
    """Defaults incl. the 3 standard lanes (reference: pkg/transport/settings.go:149-181)."""
    return TransportStreamingSettings(
        lanes=[
            TransportLane(name="media", priority=0),
            TransportLane(name="data", priority=1),
            TransportLane(name="control", priority=2),
        ],
        flow_control=TransportFlowControl(
            mode="credit", initial_credits=32, max_credits=256, low_watermark=8
        ),
        delivery=TransportDelivery(semantics="atLeastOnce", ordering="perLane"),
        backpressure=TransportBackpressure(
            buffer_packets=256, buffer_bytes=64 << 20, policy="block"
        ),
    )


@dataclass
class Transport:
    """Streaming-provider description (reference: transport_types.go:11-49).

    Drivers here are in-process: ``xgmi`` (RCCL p2p over xGMI links),
    ``hostring`` (pinned host-memory rings), ``inproc`` (same-device queues)."""

    name: str = ""
    driver: str = "xgmi"
    codecs: _t.List[str] = field(default_factory=lambda: ["tensor", "json"])
    streaming: _t.Optional[TransportStreamingSettings] = None
    generation: int = 1
    extra: dict = field(default_factory=dict)

    @property
    def key(self) -> str:
        return self.name


KNOWN_TRANSPORT_DRIVERS = ("xgmi", "hostring", "inproc")


# ---------------------------------------------------------------------------
# ReferenceGrant (reference: api/policy/v1alpha1/referencegrant_types.go)
# ---------------------------------------------------------------------------


@dataclass
class ReferenceGrantPeer:
    kind: str = ""
    namespace: _t.Optional[str] = None
    name: _t.Optional[str] = None
    extra: dict = field(default_factory=dict)


@dataclass
class ReferenceGrant:
    """Cross-namespace reference permission (Gateway-API style;
    reference: referencegrant_types.go:29-136)."""

    name: str = ""
    namespace: str = "default"
    from_: _t.List[ReferenceGrantPeer] = field(default_factory=list)
    to: _t.List[ReferenceGrantPeer] = field(default_factory=list)
    extra: dict = field(default_factory=dict)

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.name}"
