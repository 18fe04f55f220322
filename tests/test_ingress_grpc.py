"""gRPC ingress (BASELINE config #4's network data plane): packets and
triggers enter over a real gRPC loopback connection — the role of the
reference's tractatus gRPC envelope + Impulse connectors
(pkg/transport/transportutil.go:9-16)."""
import numpy as np
import pytest

pytest.importorskip("grpc")

from bobrapet_amd.engine import EngineConfig, RunEngine
from bobrapet_amd.engine.ingress_grpc import IngressClient, pack_frame, serve_grpc

RESOURCES = """
kind: EngramTemplate
metadata: {name: echo-tpl}
spec: {builtin: echo}
---
kind: Engram
metadata: {name: st-a}
spec: {templateRef: {name: echo-tpl}}
---
kind: Engram
metadata: {name: st-b}
spec: {templateRef: {name: echo-tpl}}
---
kind: Engram
metadata: {name: st-c}
spec: {templateRef: {name: echo-tpl}}
---
kind: Story
metadata: {name: pipe3}
spec:
  pattern: streaming
  steps:
    - {name: s1, ref: {name: st-a}}
    - {name: s2, ref: {name: st-b}, needs: [s1]}
    - {name: s3, ref: {name: st-c}, needs: [s2]}
---
kind: Story
metadata: {name: trig-target}
spec:
  steps:
    - {name: one, ref: {name: st-a}, with: {v: "{{ inputs.x }}"}}
  output: {v: "{{ steps.one.output.v }}"}
"""


@pytest.fixture
def served():
    eng = RunEngine(EngineConfig(cpu_workers=2)).start()
    eng.apply_yaml(RESOURCES)
    server, port = serve_grpc(eng, port=0)
    client = IngressClient(f"127.0.0.1:{port}")
    yield eng, client
    client.close()
    server.stop(0)
    eng.stop()


def test_stream_packets_through_3_stages(served):
    eng, client = served
    n = 40
    frames = [
        pack_frame({"stream": "default/pipe3", "session": "s0", "seq": i,
                    "meta": {"items": [{}], "seq": i}})
        for i in range(n)
    ]
    out = client.push_stream(frames)
    assert out["pushed"] == n
    import time
    deadline = time.time() + 15
    while client.stream_stats("s0")["leafPackets"] < n and time.time() < deadline:
        time.sleep(0.01)
    assert client.stream_stats("s0")["leafPackets"] == n
    fin = client.finish_stream("s0")
    assert fin["phase"] == "Finished", fin


def test_stream_batched_frames(served):
    # transport batching: several packet frames bundled into one gRPC
    # message via pack_batch; the server fans them back out in order
    from bobrapet_amd.engine.ingress_grpc import pack_batch

    eng, client = served
    n = 30
    frames = [
        pack_frame({"stream": "default/pipe3", "session": "sb", "seq": i,
                    "meta": {"items": [{}], "seq": i}})
        for i in range(n)
    ]
    msgs = [pack_batch(frames[i : i + 8]) for i in range(0, n, 8)]
    out = client.push_stream(msgs)
    assert out["pushed"] == n
    import time
    deadline = time.time() + 15
    while client.stream_stats("sb")["leafPackets"] < n and time.time() < deadline:
        time.sleep(0.01)
    assert client.stream_stats("sb")["leafPackets"] == n
    fin = client.finish_stream("sb")
    assert fin["phase"] == "Finished", fin


def test_packet_with_tensor_payload(served):
    eng, client = served
    arr = np.arange(12, dtype=np.int32).reshape(3, 4)
    client.push_packet(
        {"stream": "default/pipe3", "session": "s1", "seq": 0,
         "meta": {"items": [{}]}, "tensor": {"shape": [3, 4], "dtype": "int32"}},
        arr.tobytes(),
    )
    import time
    deadline = time.time() + 15
    while client.stream_stats("s1")["leafPackets"] < 1 and time.time() < deadline:
        time.sleep(0.01)
    assert client.finish_stream("s1")["phase"] == "Finished"


def test_trigger_admission_and_dedupe(served):
    eng, client = served
    r1 = client.trigger(story="default/trig-target", inputs={"x": 7}, submissionID="sub-1")
    assert r1["decision"] == "Created", r1
    run = eng.wait(r1["storyRun"], timeout=10)
    assert run.output == {"v": 7}
    # same identity → Reused, not a second run
    r2 = client.trigger(story="default/trig-target", inputs={"x": 7}, submissionID="sub-1")
    assert r2["decision"] == "Reused", r2
    assert r2["storyRun"] == r1["storyRun"]


def test_trigger_via_manual_impulse(served):
    eng, client = served
    eng.apply_yaml("""
kind: ImpulseTemplate
metadata: {name: manual-tpl}
spec: {builtin: manual}
---
kind: Impulse
metadata: {name: poke}
spec:
  templateRef: {name: manual-tpl}
  storyRef: {name: trig-target}
  mapping: {x: "{{ event.x }}"}
""")
    eng.impulses.start("default/poke")
    r = client.trigger(impulse="default/poke", payload={"x": 3})
    assert r["storyRun"], r
    run = eng.wait(r["storyRun"], timeout=10)
    assert run.output == {"v": 3}
