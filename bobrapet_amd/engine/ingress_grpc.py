"""gRPC ingress: the network data plane of the engine.

Role parity with the reference's gRPC connectors (reference:
pkg/transport/transportutil.go:9-16 — the tractatus envelope protocol
between engram pods/hub — and the Impulse workload's network ingress,
internal/controller/impulse_controller.go:451-598).  BASELINE config #4
names "gRPC Impulse → 3-stage Engram pipeline": packets enter HERE over
gRPC and flow into the streaming runtime's partition lanes.

Wire format (no protoc / generated code — a length-prefixed JSON header +
raw payload, the MI355X-native analogue of the tractatus envelope):

    request bytes  = u32le header_len | header JSON | payload bytes
    response bytes = JSON

Service ``/bobrapet.Ingress/``:
  Trigger      unary   — impulse event or direct StoryTrigger admission
  PushPacket   unary   — one streaming packet (header.tensor describes
                         the payload bytes: shape + dtype)
  PushStream   client-streaming — the hot path: a stream of packet
                         frames into one streaming StoryRun
  StreamStats  unary   — {session} → leaf-packet counters
  FinishStream unary   — drain + finish the streaming run
"""
from __future__ import annotations

import json
import struct
import threading
import typing as _t

if _t.TYPE_CHECKING:  # pragma: no cover
    from .engine import RunEngine

_SERVICE = "bobrapet.Ingress"


def pack_frame(header: dict, payload: bytes = b"") -> bytes:
    h = json.dumps(header, separators=(",", ":")).encode()
    return struct.pack("<I", len(h)) + h + payload


def unpack_frame(data: bytes) -> _t.Tuple[dict, bytes]:
    (hlen,) = struct.unpack_from("<I", data, 0)
    header = json.loads(data[4 : 4 + hlen].decode())
    return header, data[4 + hlen :]


def pack_batch(frames: _t.Sequence[bytes]) -> bytes:
    """Bundle N packet frames into ONE gRPC message (transport batching:
    amortizes the per-message gRPC + Python envelope, ~40 us/message).
    Wire: a normal frame whose header is {"sizes": [len, ...]} and whose
    payload is the frames concatenated; the server fans them back out."""
    return pack_frame({"sizes": [len(f) for f in frames]}, b"".join(frames))


def _tensor_from(header: dict, payload: bytes):
    spec = header.get("tensor")
    if not spec:
        return None
    import numpy as np
    import torch

    dtype = spec.get("dtype", "int32")
    arr = np.frombuffer(payload, dtype=np.dtype(dtype)).reshape(spec["shape"])
    t = torch.from_numpy(arr.copy())
    if torch.cuda.is_available():
        t = t.cuda(non_blocking=True)
    return t


class _Ingress:
    def __init__(self, engine: "RunEngine"):
        self.engine = engine
        self._streams: _t.Dict[str, _t.Any] = {}
        self._lock = threading.Lock()

    # ------------------------------------------------------------------

    def _stream_for(self, header: dict):
        session = header.get("session") or header.get("stream")
        with self._lock:
            s = self._streams.get(session)
            if s is None:
                story = header.get("stream") or session
                s = self.engine.submit_stream(story)
                self._streams[session] = s
        return s

    def _push_one(self, header: dict, payload: bytes) -> None:
        stream = self._stream_for(header)
        packet = dict(header.get("meta") or {})
        packet.setdefault("seq", header.get("seq", 0))
        t = _tensor_from(header, payload)
        if t is not None:
            packet["tensor"] = t
        stream.push(packet)

    # ---- RPC handlers (bytes in / bytes out) --------------------------

    def trigger(self, data: bytes, ctx) -> bytes:
        header, _ = unpack_frame(data)
        if "impulse" in header:
            from .impulses import ManualImpulse

            live = self.engine.impulses.live.get(header["impulse"])
            if live is None:
                return json.dumps({"error": f"impulse {header['impulse']} not running"}).encode()
            payload = header.get("payload") or {}
            handler = live.handler
            if isinstance(handler, ManualImpulse):
                result = handler.emit(payload)
            else:
                result = self.engine.impulses._on_event(live, payload)
            out = {
                "decision": str(getattr(result, "decision", "")),
                "storyRun": getattr(result, "story_run_ref", None),
                "message": getattr(result, "message", ""),
            }
            return json.dumps(out).encode()
        # direct StoryTrigger admission (reference: StoryTrigger CR path)
        from .triggers import StoryTrigger

        ns, _, name = str(header.get("story", "")).rpartition("/")
        trig = StoryTrigger(
            submission_id=header.get("submissionID", ""),
            story_name=name,
            story_namespace=ns or "default",
            namespace=ns or "default",
            key=header.get("dedupKey"),
            inputs=header.get("inputs") or {},
        )
        result = self.engine.triggers.submit(trig)
        return json.dumps(
            {
                "decision": str(result.decision),
                "storyRun": result.story_run_ref,
                "message": result.message,
            }
        ).encode()

    def _push_frame(self, data: bytes) -> int:
        header, payload = unpack_frame(data)
        sizes = header.get("sizes")
        if sizes is None:
            self._push_one(header, payload)
            return 1
        # validate the whole batch BEFORE applying any frame — a malformed
        # sizes vector must not partially apply
        if (not isinstance(sizes, list)
                or any(not isinstance(sz, int) or sz < 4 for sz in sizes)
                or sum(sizes) != len(payload)):
            raise ValueError("batched frame: sizes do not tile the payload")
        off = 0
        frames = []
        for sz in sizes:
            frames.append(unpack_frame(payload[off : off + sz]))
            off += sz
        for h, p in frames:  # batched message: fan the frames back out
            self._push_one(h, p)
        return len(sizes)

    def push_packet(self, data: bytes, ctx) -> bytes:
        self._push_frame(data)
        return b'{"ok":true}'

    def push_stream(self, request_iter, ctx) -> bytes:
        n = 0
        for data in request_iter:
            n += self._push_frame(data)
        return json.dumps({"pushed": n}).encode()

    def stream_stats(self, data: bytes, ctx) -> bytes:
        header, _ = unpack_frame(data)
        s = self._streams.get(header.get("session") or header.get("stream"))
        if s is None:
            return b'{"leafPackets":0}'
        return json.dumps({"leafPackets": s.leaf_packets}).encode()

    def finish_stream(self, data: bytes, ctx) -> bytes:
        header, _ = unpack_frame(data)
        key = header.get("session") or header.get("stream")
        s = self._streams.pop(key, None)
        if s is None:
            return json.dumps({"error": "unknown stream session"}).encode()
        run = s.finish(timeout=float(header.get("timeout", 60.0)))
        return json.dumps({"phase": str(run.phase)}).encode()


def serve_grpc(engine: "RunEngine", port: int = 0, workers: int = 8):
    """Start the gRPC ingress; returns (server, bound_port)."""
    import grpc
    from concurrent import futures

    ingress = _Ingress(engine)
    ident = lambda b: b  # noqa: E731 — bytes-in/bytes-out codec

    handlers = {
        "Trigger": grpc.unary_unary_rpc_method_handler(
            ingress.trigger, request_deserializer=ident, response_serializer=ident
        ),
        "PushPacket": grpc.unary_unary_rpc_method_handler(
            ingress.push_packet, request_deserializer=ident, response_serializer=ident
        ),
        "PushStream": grpc.stream_unary_rpc_method_handler(
            ingress.push_stream, request_deserializer=ident, response_serializer=ident
        ),
        "StreamStats": grpc.unary_unary_rpc_method_handler(
            ingress.stream_stats, request_deserializer=ident, response_serializer=ident
        ),
        "FinishStream": grpc.unary_unary_rpc_method_handler(
            ingress.finish_stream, request_deserializer=ident, response_serializer=ident
        ),
    }
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=workers))
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(_SERVICE, handlers),)
    )
    bound = server.add_insecure_port(f"127.0.0.1:{port}")
    server.start()
    return server, bound


class IngressClient:
    """Minimal client for the bytes-framed ingress service."""

    def __init__(self, target: str):
        import grpc

        self.channel = grpc.insecure_channel(target)
        ident = lambda b: b  # noqa: E731
        self._trigger = self.channel.unary_unary(
            f"/{_SERVICE}/Trigger", request_serializer=ident, response_deserializer=ident
        )
        self._push = self.channel.unary_unary(
            f"/{_SERVICE}/PushPacket", request_serializer=ident, response_deserializer=ident
        )
        self._push_stream = self.channel.stream_unary(
            f"/{_SERVICE}/PushStream", request_serializer=ident, response_deserializer=ident
        )
        self._stats = self.channel.unary_unary(
            f"/{_SERVICE}/StreamStats", request_serializer=ident, response_deserializer=ident
        )
        self._finish = self.channel.unary_unary(
            f"/{_SERVICE}/FinishStream", request_serializer=ident, response_deserializer=ident
        )

    def trigger(self, **header) -> dict:
        return json.loads(self._trigger(pack_frame(header)))

    def push_packet(self, header: dict, payload: bytes = b"") -> dict:
        return json.loads(self._push(pack_frame(header, payload)))

    def push_stream(self, frames: _t.Iterable[bytes]) -> dict:
        return json.loads(self._push_stream(iter(frames)))

    def stream_stats(self, session: str) -> dict:
        return json.loads(self._stats(pack_frame({"session": session})))

    def finish_stream(self, session: str, timeout: float = 60.0) -> dict:
        return json.loads(
            self._finish(pack_frame({"session": session, "timeout": timeout}))
        )

    def close(self):
        self.channel.close()
