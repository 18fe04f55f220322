"""Streaming (PerStoryRun) execution: persistent engram pipelines.

Role replacement for the reference's realtime path (reference:
steprun_controller.go:2527-4494 — per-run Deployment+Service+
TransportBinding with connector sidecars and gRPC hub/P2P routing,
SURVEY.md §3.5): a streaming Story materializes as in-process pipeline
stages, one persistent worker per engram step pinned to its own
(device, HIP stream), connected by credit-flow rings (transport/flow.py)
per the analyzed topology (transport/topology.py).  Hub-routed steps
evaluate their per-packet `runtime` templates in the engine's evaluator.

hipGraph capture (BASELINE config #4): a stage whose engram implements
``tensor_compute`` and whose step sets ``with.capture: true`` replays its
per-packet GPU work as a captured hipGraph (torch.cuda.CUDAGraph is
hipGraph on ROCm) after a warmup packet — fixed launch latency per packet.

The full TransportStreamingSettings vocabulary is ENFORCED, not just
parsed: fan-in all/any/quorum (accumulate-to-quorum joins), replay
memory/durable (``StreamingRun.replay``), recording metadata/full with
sampling + redaction, lifecycle drain/cutover (``StreamingRun.upgrade``
with binding-generation bumps), hash/preserve partitioning (per-partition
rings + parallel stage lanes, per-partition ordering), and per-packet
routing rules (allow/deny over targeted downstreams).
"""
from __future__ import annotations

import threading
import time
import typing as _t
from dataclasses import dataclass, field

from ..engrams import registry as engram_registry
from ..engrams.base import EngramContext, EngramFailure, EngramResult
from ..enums import Phase
from ..specs import types as T
from ..transport import flow, topology
from .records import StoryRun, StructuredError, monotonic_now

if _t.TYPE_CHECKING:
    from .engine import RunEngine


@dataclass
class TransportBinding:
    """Per-(storyRun, step) negotiated connector binding (reference:
    api/transport/v1alpha1/transportbinding_types.go:108-199): here the
    negotiation is in-process — driver/codec selection + ring endpoints."""

    name: str
    story_run: str
    step: str
    transport: str = ""
    driver: str = "inproc"
    codecs: _t.List[str] = field(default_factory=lambda: ["tensor", "json"])
    endpoint: str = ""
    upstream: _t.List[str] = field(default_factory=list)
    downstream: _t.List[str] = field(default_factory=list)
    phase: str = "Ready"
    generation: int = 0  # bumped on live cutover (reference: connector gen)
    heartbeat: float = field(default_factory=monotonic_now)
    # per-lane negotiated codec (reference: DeriveNegotiatedCapabilities —
    # defaults to the FIRST offered codec per modality, preserves an
    # already-negotiated value across re-derivation)
    negotiated: _t.Dict[str, str] = field(default_factory=dict)


def derive_negotiated(binding: "TransportBinding",
                      lanes: _t.Sequence[str]) -> None:
    """Default each lane's negotiated codec to the binding's first
    offered codec; existing negotiations are PRESERVED (reference:
    pkg/transport/capabilities.go DeriveNegotiatedCapabilities)."""
    first = binding.codecs[0] if binding.codecs else "json"
    for lane in lanes or ("data",):
        binding.negotiated.setdefault(lane, first)


@dataclass
class StageStats:
    packets_in: int = 0
    packets_out: int = 0
    errors: int = 0
    graph_replays: int = 0
    busy_seconds: float = 0.0


class _GraphGate:
    """Capture/replay exclusion for partition lanes on one device: hipGraph
    CAPTURE must run quiesced (no sibling lane mid-replay — stream capture
    racing concurrent GPU work wedges intermittently on ROCm), while
    replays proceed concurrently with each other."""

    def __init__(self):
        self._cv = threading.Condition()
        self._replays = 0
        self._capturing = False

    def begin_replay(self):
        with self._cv:
            while self._capturing:
                self._cv.wait()
            self._replays += 1

    def end_replay(self):
        with self._cv:
            self._replays -= 1
            self._cv.notify_all()

    def begin_capture(self):
        with self._cv:
            while self._capturing:
                self._cv.wait()
            self._capturing = True
            while self._replays > 0:
                self._cv.wait()

    def end_capture(self):
        with self._cv:
            self._capturing = False
            self._cv.notify_all()


_GRAPH_GATE = _GraphGate()


class _Stage(threading.Thread):
    def __init__(
        self,
        sr: "StreamingRun",
        step: T.Step,
        in_rings: _t.List[flow.CreditRing],
        out_edges: _t.List[_t.Tuple[_t.Tuple[str, str], _t.List[flow.CreditRing]]],
        device: _t.Optional[int],
        lane: int = 0,
    ):
        super().__init__(name=f"stage-{sr.run.name}-{step.name}.{lane}", daemon=True)
        self.sr = sr
        self.step = step
        self.in_rings = in_rings
        self.out_edges = out_edges  # [(edge_key, [ring per partition])]
        self.out_rings = [r for _, rings in out_edges for r in rings]  # leaf test
        self.lane = lane
        self.device = device
        self.stats = StageStats()
        self.stream = None
        self._graph = None
        self._graph_in = None
        self._graph_out = None
        self.error: _t.Optional[str] = None

    # ------------------------------------------------------------------

    def run(self) -> None:
        try:
            if self.device is not None:
                import torch

                torch.cuda.set_device(self.device)
                self.stream = torch.cuda.Stream(device=self.device)
            self._loop()
        except Exception as exc:  # stage crash fails the pipeline
            self.error = f"{type(exc).__name__}: {exc}"
            self.sr.on_stage_error(self.step.name, self.error, fatal=True)
        finally:
            # downstream rings close only when EVERY lane of this step is
            # done (a shared ring closed early would cut sibling lanes off)
            for edge_key, _rings in self.out_edges:
                self.sr._producer_done(edge_key)

    def _loop(self) -> None:
        import contextlib

        while True:
            packet = self._pop_fan_in()
            if packet is flow.SENTINEL:
                return
            if packet is None:  # dropped join round (quorum/timeout)
                continue
            if self.sr.canceled:
                return
            t0 = time.monotonic()
            ctx = contextlib.nullcontext()
            if self.stream is not None:
                import torch

                ctx = torch.cuda.stream(self.stream)
            with ctx:
                out = self._process(packet)
            if self.stream is not None:
                self.stream.synchronize()
            self.stats.busy_seconds += time.monotonic() - t0
            self.stats.packets_in += 1
            if out is not None:
                # per-partition ordering: a packet's partition id routes it
                # to that partition's ring (reference: ordering per_partition)
                if isinstance(out, dict) and isinstance(packet, dict) and "$partition" in packet:
                    out.setdefault("$partition", packet["$partition"])
                for (src, dst), rings in self.out_edges:
                    if not self.sr.route_allows(dst, out):
                        continue  # denied by a transport routing rule
                    if len(rings) == 1:
                        rings[0].push(out)
                    else:
                        part = out.get("$partition", 0) if isinstance(out, dict) else 0
                        rings[int(part) % len(rings)].push(out)
                self.stats.packets_out += 1
                self.sr.on_packet(self.step.name, leaf=not self.out_rings, packet=out)

    # ------------------------------------------------------------------

    def _pop_fan_in(self):
        """Join packets from the upstream rings per the transport fan-in
        settings (reference: TransportFanInSettings
        transport_settings_types.go:174-192 — modes all|any|quorum).

        Joins are arrival-round based (one packet per ring per round), not
        envelope-ID based: with `timeoutSeconds` a straggler's packet stays
        queued and joins the NEXT round — at-least-once, perLane-ordered
        semantics over in-process rings.  Returns the merged packet,
        ``None`` for a dropped round, or SENTINEL at end-of-stream."""
        rings = self.in_rings
        if len(rings) == 1:
            return rings[0].pop()
        fi = self.sr.settings.fan_in
        mode = fi.mode if fi is not None else "all"
        timeout = fi.timeout_seconds if fi is not None else None

        if mode == "any":
            # first packet from any live upstream wins; round-robin poll
            live = [r for r in rings if r not in getattr(self, "_done_rings", set())]
            if not hasattr(self, "_done_rings"):
                self._done_rings = set()
                live = list(rings)
            while True:
                progressed = False
                for ring in list(live):
                    try:
                        pkt = ring.pop(timeout=0.005)
                    except TimeoutError:
                        continue
                    if pkt is flow.SENTINEL:
                        self._done_rings.add(ring)
                        live.remove(ring)
                        progressed = True
                        continue
                    return pkt
                if not live:
                    return flow.SENTINEL
                if not progressed:
                    time.sleep(0)  # yield between poll sweeps

        # all / quorum: accumulate one packet per ring per round, emitting
        # as soon as `need` arrived — a slow ring never gates a met quorum
        # (stragglers' packets stay queued and join the NEXT round)
        closed = getattr(self, "_closed_rings", None)
        if closed is None:
            closed = self._closed_rings = set()
        need = len([r for r in rings if r not in closed])
        if mode == "quorum":
            need = max(1, min(int(fi.quorum or len(rings)), need))
        deadline = (time.monotonic() + timeout) if timeout else None
        packets, arrived_from = [], []
        pending = {i: r for i, r in enumerate(rings) if r not in closed}
        while True:
            for i in list(pending):
                ring = pending[i]
                try:
                    pkt = ring.pop(timeout=0.005)
                except TimeoutError:
                    continue
                if pkt is flow.SENTINEL:
                    closed.add(ring)
                    del pending[i]
                    continue
                packets.append(pkt)
                arrived_from.append(i)
                del pending[i]
            live = len(rings) - len(closed)
            if live < need:
                need = max(1, min(need, live)) if mode == "quorum" else live
            if need <= 0 or (live <= 0 and not packets):
                return flow.SENTINEL if not packets else self._fan_in_merge(rings, packets, arrived_from)
            if len(packets) >= need:
                return self._fan_in_merge(rings, packets, arrived_from)
            if deadline is not None and time.monotonic() >= deadline:
                if packets and len(packets) >= need:
                    return self._fan_in_merge(rings, packets, arrived_from)
                self.sr.engine.metrics.inc("stream_fanin_dropped_total")
                return None
            if self.sr.canceled:
                return flow.SENTINEL

    def _fan_in_merge(self, rings, packets, arrived_from):
        merged = {"fanIn": packets}
        if len(packets) < len(rings):
            merged["arrivedFrom"] = arrived_from
        return merged

    def _process(self, packet):
        eng = self.sr.engine
        step = self.step
        # hub routing: per-packet runtime templates transform the payload
        # (reference: BUBU_TEMPLATE_CONTEXT hub evaluation)
        if step.name in self.sr.topo.hub_steps and isinstance(step.runtime, dict):
            route = step.runtime.get("route")
            if route is not None:
                scope = {"packet": packet, "inputs": self.sr.run.inputs}
                if not eng.evaluator.evaluate_condition(str(route), scope):
                    return None  # dropped by routing rule
            transform = step.runtime.get("transform")
            if transform is not None:
                scope = {"packet": packet, "inputs": self.sr.run.inputs}
                packet = eng.evaluator.resolve_value(transform, scope)

        impl = self._impl()
        if impl is None:  # pure hub/transform step without an engram
            return packet

        captured = self._maybe_graph(impl, packet)
        if captured is not None:
            self.stats.graph_replays += 1
            return captured

        ctx = self._ctx(packet)
        try:
            result = impl.run(ctx)
        except EngramFailure as exc:
            self.stats.errors += 1
            self.sr.on_stage_error(step.name, str(exc))
            return None
        if isinstance(result, EngramResult):
            return result.output
        return result

    def _impl(self):
        if self.step.ref is None:
            return None
        cached = getattr(self, "_impl_cache", None)
        if cached is not None:
            return cached
        eng = self.sr.engine
        engram = eng.registry.try_engram(
            self.step.ref.name, self.step.ref.resolve_namespace(self.sr.story.namespace)
        )
        if engram is None:
            raise RuntimeError(f"engram {self.step.ref.name} not found")
        tpl = (
            eng.registry.engram_template(engram.template_ref.name)
            if engram.template_ref is not None
            else None
        )
        name = tpl.implementation if tpl is not None else self.step.ref.name
        self._engram_cfg = engram.with_
        self._impl_cache = engram_registry.resolve(name)
        return self._impl_cache

    def invalidate_impl(self) -> None:
        """Next packet re-resolves the engram + config from the registry
        (live cutover — reference: connector generation bumping
        steprun_controller.go:2693-2760)."""
        self._impl_cache = None

    def _ctx(self, packet) -> EngramContext:
        return EngramContext(
            story_name=self.sr.story.name,
            story_run=self.sr.run.name,
            step_name=self.step.name,
            namespace=self.sr.run.namespace,
            input=packet,
            config=getattr(self, "_engram_cfg", None),
            runtime=self.step.runtime,
            execution_mode="deployment",
            device=self.device,
            stream=self.stream,
            storage=self.sr.engine.storage,
            cancel_check=lambda: self.sr.canceled,
        )

    # ------------------------------------------------------------------

    def _maybe_graph(self, impl, packet):
        """hipGraph capture path: engram.tensor_compute over a fixed-shape
        tensor packet, captured once then replayed per packet."""
        want = isinstance(self.step.with_, dict) and self.step.with_.get("capture")
        fn = getattr(impl, "tensor_compute", None)
        if not want or fn is None or self.device is None:
            return None
        import torch

        tensor = packet.get("tensor") if isinstance(packet, dict) else None
        if not torch.is_tensor(tensor):
            return None
        tensor = tensor.to(f"cuda:{self.device}", non_blocking=False)
        if self._graph is False:  # capture failed earlier: permanent eager
            return None
        if self._graph is None:
            # one capture at a time: concurrent partition lanes each own a
            # graph, but global-mode stream capture poisons OTHER threads'
            # in-flight GPU work (thread_local relaxes the check; the lock
            # keeps captures from overlapping at all)
            _GRAPH_GATE.begin_capture()
            try:
                ctx = self._ctx(packet)
                # warmup (allocations settle), then capture
                fn(ctx, tensor)
                torch.cuda.synchronize(self.device)
                self._graph_in = tensor.clone()
                g = torch.cuda.CUDAGraph()
                capture_stream = torch.cuda.Stream(device=self.device)
                with torch.cuda.stream(capture_stream):
                    with torch.cuda.graph(
                        g, stream=capture_stream, capture_error_mode="thread_local"
                    ):
                        self._graph_out = fn(ctx, self._graph_in)
                self._graph = g
            except Exception as exc:
                # never fail the pipeline over a capture problem: degrade
                # THIS lane to eager permanently
                self._graph = False
                self.sr.engine.metrics.inc("stream_capture_fallbacks_total")
                import warnings

                warnings.warn(f"hipGraph capture fell back to eager: {exc}")
                return None
            finally:
                _GRAPH_GATE.end_capture()
        _GRAPH_GATE.begin_replay()
        try:
            self._graph_in.copy_(tensor)
            self._graph.replay()
            if self.stream is not None:
                torch.cuda.synchronize(self.device)
            out = dict(packet) if isinstance(packet, dict) else {}
            out["tensor"] = self._graph_out.clone()
        finally:
            _GRAPH_GATE.end_replay()
        return out


class StreamingRun:
    """A live streaming StoryRun: pipeline of stages + ingress."""

    def __init__(self, engine: "RunEngine", run: StoryRun, story: T.Story):
        self.engine = engine
        self.run = run
        self.story = story
        self.topo = topology.analyze(story)
        self.canceled = False
        self.degraded: _t.Optional[str] = None  # fatal stage crash message
        self._lock = threading.Lock()
        self._finish_lock = threading.Lock()
        self._finish_result: _t.Optional[StoryRun] = None
        self._leaf_packets = 0
        self._last_outputs: _t.List = []
        self._recorded: _t.List = []     # recording entries (bounded)
        self._replay_log: _t.List = []   # ingress packets for replay
        self._rec_seq = 0
        self.settings = self._settings(story)

        # partitioning (reference: TransportPartitioningSettings hash mode):
        # an edge into a fan-in-free stage splits into P rings, one per
        # partition, and the stage runs P parallel lanes — per-partition
        # order preserved, cross-partition work parallel on worker threads
        part = self.settings.partitioning
        self._nparts = (
            max(1, int(part.partitions or 4))
            if part is not None and part.mode == "hash"
            else 1
        )
        import itertools as _it

        self._part_rr = _it.count()  # thread-safe round-robin (no key case)

        # rings: per edge, 1 ring (ordered lane) or P partition rings
        self._edge_rings: _t.Dict[_t.Tuple[str, str], _t.List[flow.CreditRing]] = {}
        self._edge_producers: _t.Dict[_t.Tuple[str, str], int] = {}
        self.ingress: _t.List[_t.List[flow.CreditRing]] = []
        upstream_count = {name: len(self.topo.upstream_of(name)) for name in self.topo.stages}
        for e in self.topo.edges:
            nrings = self._nparts if upstream_count.get(e.dst, 1) == 1 else 1
            rings = [
                flow.CreditRing(
                    name=f"{run.name}:{e.src or '@'}->{e.dst}#{p_}",
                    settings=self.settings,
                    lane=e.lane,
                )
                for p_ in range(nrings)
            ]
            self._edge_rings[(e.src, e.dst)] = rings
            if e.src == "":
                self.ingress.append(rings)

        self.stages: _t.List[_Stage] = []
        by_name = {s.name: s for s in story.steps}
        n_dev = engine.workers.device_count
        for i, name in enumerate(self.topo.stages):
            step = by_name[name]
            up = self.topo.upstream_of(name)
            down = self.topo.downstream_of(name)
            out_edges = [((e.src, e.dst), self._edge_rings[(e.src, e.dst)]) for e in down]
            device = None
            if n_dev > 0 and step.ref is not None:
                device = i % n_dev  # stage-per-device pipeline placement
            lanes = len(self._edge_rings[(up[0].src, up[0].dst)]) if len(up) == 1 else 1
            for lane in range(lanes):
                if lanes > 1:
                    in_rings = [self._edge_rings[(up[0].src, up[0].dst)][lane]]
                else:
                    in_rings = [self._edge_rings[(e.src, e.dst)][0] for e in up]
                self.stages.append(_Stage(self, step, in_rings, out_edges, device, lane))
            for (ekey, _r) in out_edges:
                self._edge_producers[ekey] = self._edge_producers.get(ekey, 0) + lanes
            st = run.step_state(name)
            st.phase = Phase.RUNNING
            st.started_at = monotonic_now()
        # negotiated bindings (reference: ensureRunTransportBinding
        # steprun_controller.go:3701; driver from the story's transport)
        driver = "inproc"
        transport_name = ""
        if story.transports:
            transport_name = story.transports[0].transport_ref or story.transports[0].name
            try:
                driver = engine.registry.transport(transport_name).driver
            except KeyError:
                pass
        self.bindings: _t.Dict[str, TransportBinding] = {}
        for name in self.topo.stages:
            self.bindings[name] = TransportBinding(
                name=f"{run.name}-{name}",
                story_run=run.name,
                step=name,
                transport=transport_name,
                driver=driver,
                endpoint=f"ring://{run.name}/{name}",
                upstream=[e.src for e in self.topo.upstream_of(name) if e.src],
                downstream=[e.dst for e in self.topo.downstream_of(name)],
            )
        lane_names = [l.name for l in (self.settings.lanes or [])] or ["data"]
        for b in self.bindings.values():
            derive_negotiated(b, lane_names)
        engine.metrics.set_gauge("transport_bindings_total", len(self.bindings))
        engine.metrics.set_gauge("transport_bindings_ready", len(self.bindings))
        run.phase = Phase.RUNNING
        run.started_at = run.started_at or monotonic_now()
        for s in self.stages:
            s.start()

    @staticmethod
    def _settings(story: T.Story) -> T.TransportStreamingSettings:
        if story.transports and story.transports[0].streaming:
            from ..specs.types import from_dict

            return from_dict(T.TransportStreamingSettings, story.transports[0].streaming)
        return T.default_streaming_settings()

    # ------------------------------------------------------------------

    def push(self, packet, timeout: _t.Optional[float] = 30.0) -> bool:
        rp = self.settings.replay
        if rp is not None and rp.mode in ("memory", "durable") and len(self._replay_log) < 4096:
            # durable mode additionally survives payload offload: markers,
            # not live tensors, are retained
            self._replay_log.append(
                self.engine.storage.dehydrate_document(packet)
                if rp.mode == "durable"
                else packet
            )
        packet = self._stamp_partition(packet)
        ok = True
        for rings in self.ingress:
            if len(rings) == 1:
                ok = rings[0].push(packet, timeout=timeout) and ok
            else:
                part = packet.get("$partition", 0) if isinstance(packet, dict) else 0
                ok = rings[int(part) % len(rings)].push(packet, timeout=timeout) and ok
        return ok

    def route_allows(self, dst: str, packet) -> bool:
        """Per-packet transport routing rules (reference:
        TransportRoutingRule transport_settings_types.go:353-370): first
        matching rule targeting `dst` decides; no match → allow."""
        routing = self.settings.routing
        if routing is None or not routing.rules:
            return True
        scope = {"packet": packet, "inputs": self.run.inputs}
        for rule in routing.rules:
            if rule.steps and dst not in rule.steps:
                continue
            if rule.when:
                try:
                    if not self.engine.evaluator.evaluate_condition(str(rule.when), scope):
                        continue
                except Exception:
                    continue  # unevaluable rule never matches
            if rule.action == "deny":
                self.engine.metrics.inc("stream_route_denied_total", rule=rule.name or "?")
                return False
            return True
        return True

    def _stamp_partition(self, packet):
        """Assign a partition id (reference: TransportPartitioningSettings):
        hash of the configured key path (round-robin without a key),
        preserved when sticky and already present."""
        part = self.settings.partitioning
        if part is None or part.mode in (None, "", "none") or not isinstance(packet, dict):
            return packet
        if part.sticky and "$partition" in packet:
            return packet
        if part.mode == "preserve":
            packet.setdefault("$partition", 0)
            return packet
        n = self._nparts
        if part.key:
            from ..storage.manager import extract_path

            try:
                key_val = extract_path(packet, part.key)
            except Exception:
                key_val = None
            pid = (hash(str(key_val)) & 0x7FFFFFFF) % n
        else:
            pid = next(self._part_rr) % n
        packet = dict(packet)
        packet["$partition"] = pid
        return packet

    def _producer_done(self, edge_key) -> None:
        with self._lock:
            left = self._edge_producers.get(edge_key, 1) - 1
            self._edge_producers[edge_key] = left
        if left <= 0:
            for ring in self._edge_rings.get(edge_key, []):
                ring.close()

    def replay(self, last: _t.Optional[int] = None, timeout: _t.Optional[float] = 30.0) -> int:
        """Re-push recorded ingress packets (reference: TransportReplaySettings
        memory/durable replay).  Returns how many packets were re-delivered."""
        rp = self.settings.replay
        if rp is None or rp.mode not in ("memory", "durable"):
            raise ValueError("replay requires streaming.replay.mode memory|durable")
        src = self._replay_log[-last:] if last else list(self._replay_log)
        n = 0
        for pkt in src:
            if rp.mode == "durable":
                pkt = self.engine.storage.hydrate(pkt)
            pkt = self._stamp_partition(pkt)
            ok = True
            for rings in self.ingress:
                if len(rings) == 1:
                    ok = rings[0].push(pkt, timeout=timeout) and ok
                else:
                    part = pkt.get("$partition", 0) if isinstance(pkt, dict) else 0
                    ok = rings[int(part) % len(rings)].push(pkt, timeout=timeout) and ok
            n += 1 if ok else 0
        return n

    def finish(self, timeout: float = 60.0) -> StoryRun:
        """Close the ingress, drain and finalize the run.  Idempotent (a
        topology termination may have finalized concurrently).  A failed
        pipeline with compensations/finally hands the run to the batch DAG
        machinery, which launches those steps and finalizes with the
        proper phase (Compensated etc.)."""
        with self._finish_lock:
            if self._finish_result is not None:
                return self._finish_result
            self._finish_result = self._finish_inner(timeout)
            return self._finish_result

    def _finish_inner(self, timeout: float) -> StoryRun:
        for rings in self.ingress:
            for ring in rings:
                ring.close()
        deadline = time.monotonic() + timeout
        for s in self.stages:
            s.join(timeout=max(deadline - time.monotonic(), 0.1))
        now = monotonic_now()
        failed = False
        agg: _t.Dict[str, dict] = {}
        errs: _t.Dict[str, str] = {}
        for s in self.stages:  # sum stats across a step's partition lanes
            a = agg.setdefault(
                s.step.name,
                {"packetsIn": 0, "packetsOut": 0, "errors": 0, "graphReplays": 0,
                 "busySeconds": 0.0, "lanes": 0},
            )
            a["packetsIn"] += s.stats.packets_in
            a["packetsOut"] += s.stats.packets_out
            a["errors"] += s.stats.errors
            a["graphReplays"] += s.stats.graph_replays
            a["busySeconds"] = round(a["busySeconds"] + s.stats.busy_seconds, 6)
            a["lanes"] += 1
            if s.error:
                errs[s.step.name] = s.error
        for name, a in agg.items():
            st = self.run.step_state(name)
            st.output = a
            st.finished_at = now
            if name in errs:
                st.phase = Phase.FAILED
                st.error = StructuredError(message=errs[name])
                failed = True
            else:
                st.phase = Phase.SUCCEEDED
        if failed and (self.story.compensations or self.story.finally_):
            # topology terminated with compensations declared: the batch
            # DAG takes over (main steps are terminal; it launches the
            # compensation/finally steps and finalizes the phase)
            self.run.phase = Phase.RUNNING
            self.run.output = {"packets": self._leaf_packets, "stages": len(agg)}
            self.engine._streams.pop(self.run.key, None)
            self.engine._post(("tick", self.run.key))
            return self.engine.wait(self.run, timeout=timeout)
        self.run.phase = Phase.FAILED if failed else Phase.FINISHED
        self.run.output = {
            "packets": self._leaf_packets,
            "stages": len(agg),
        }
        if self._recorded:
            # recordings ride the normal $storageRef offload path
            self.run.output["recording"] = self.engine.storage.dehydrate_document(
                {"entries": self._recorded}
            )
            self.run.output["recordedPackets"] = len(self._recorded)
        self.run.finished_at = now
        self.engine.on_run_terminal(self.run)
        self.engine._streams.pop(self.run.key, None)
        return self.run

    def cancel(self) -> None:
        self.canceled = True
        for rings in self._edge_rings.values():
            for ring in rings:
                ring.close()

    def upgrade(self, step: _t.Optional[str] = None, timeout: _t.Optional[float] = None) -> int:
        """Live cutover of streaming stage(s) to the CURRENTLY-applied
        Engram definition (reference: TransportLifecycleSettings drain/
        cutover + connector generation bumping).  strategy=drain waits for
        the stage's input rings to empty (bounded by drainTimeoutSeconds /
        `timeout`); strategy=recreate cuts over immediately.  Returns the
        number of stages cut over; binding generations are bumped."""
        lc = self.settings.lifecycle
        strategy = lc.strategy if lc is not None else "drain"
        if timeout is None:
            timeout = float(lc.drain_timeout_seconds) if lc and lc.drain_timeout_seconds else 5.0
        targets = [s for s in self.stages if step is None or s.step.name == step]
        if step is not None and not targets:
            raise KeyError(f"no streaming stage named {step!r}")
        n = 0
        for stage in targets:
            if strategy == "drain":
                deadline = time.monotonic() + timeout
                while any(r.depth > 0 for r in stage.in_rings):
                    if time.monotonic() >= deadline:
                        break
                    time.sleep(0.002)
            stage.invalidate_impl()
            binding = self.bindings.get(stage.step.name)
            if binding is not None:
                binding.generation = getattr(binding, "generation", 0) + 1
            self.engine.metrics.inc("stream_stage_upgrades_total")
            n += 1
        return n

    def _record(self, stage: str, packet) -> None:
        rec = self.settings.recording
        if rec is None or not rec.mode or len(self._recorded) >= 8192:
            return
        self._rec_seq += 1
        rate = rec.sample_rate if rec.sample_rate is not None else 100
        if rate <= 0 or (self._rec_seq - 1) % max(1, 100 // max(rate, 1)) != 0:
            return
        entry = {"seq": self._rec_seq, "stage": stage, "ts": monotonic_now()}
        if rec.mode == "full":
            body = _strip_tensors(packet)
            for path in rec.redact_fields:
                body = _redact(body, path.split("."))
            entry["packet"] = body
        self._recorded.append(entry)

    def on_packet(self, stage: str, leaf: bool, packet) -> None:
        rec = self.settings.recording
        if leaf or (rec is not None and rec.mode):
            # only leaf counting / recording needs the run lock — keep the
            # hot per-packet path lock-free for parallel partition lanes
            with self._lock:
                self._record(stage, packet)
                if leaf:
                    self._leaf_packets += 1
                    if len(self._last_outputs) < 8:
                        self._last_outputs.append(_strip_tensors(packet))
        self.engine.metrics.inc("stream_packets_total", stage=stage)

    def on_stage_error(self, stage: str, message: str, fatal: bool = False) -> None:
        self.engine.metrics.inc("stream_stage_errors_total", stage=stage)
        if not fatal:
            return
        # a stage crash terminates the topology (reference: the DAG
        # reconciler's Degraded(TopologyTerminated) handling — main steps
        # fail and the compensation phase is entered); close the ingress
        # and finalize from a separate thread (the dying stage's own
        # thread cannot join itself)
        with self._lock:
            if self.degraded is not None:
                return
            self.degraded = f"{stage}: {message}"
        for rings in self.ingress:
            for ring in rings:
                ring.close()
        threading.Thread(
            target=lambda: self.finish(timeout=10.0), daemon=True,
            name=f"stream-terminate-{self.run.name}",
        ).start()

    @property
    def leaf_packets(self) -> int:
        with self._lock:
            return self._leaf_packets


def _redact(value, path):
    """Replace the value at a dot path with "<redacted>" (recording
    redactFields — reference: transport_settings_types.go:526)."""
    if not path or not isinstance(value, dict) or path[0] not in value:
        return value
    out = dict(value)
    if len(path) == 1:
        out[path[0]] = "<redacted>"
    else:
        out[path[0]] = _redact(out[path[0]], path[1:])
    return out


def _strip_tensors(value):
    import torch

    if torch.is_tensor(value):
        return {"tensor": {"shape": list(value.shape), "dtype": str(value.dtype)}}
    if isinstance(value, dict):
        return {k: _strip_tensors(v) for k, v in value.items()}
    if isinstance(value, list):
        return [_strip_tensors(v) for v in value]
    return value
