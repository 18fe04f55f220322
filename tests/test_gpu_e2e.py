"""GPU end-to-end tests: streaming pipeline with hipGraph capture
(BASELINE config #4) and the nested executeStory + gate/wait story with
multi-GB HBM payload offload (config #5)."""
import time

import pytest
import torch

from bobrapet_amd.engine import EngineConfig, RunEngine
from bobrapet_amd.enums import Phase

pytestmark = pytest.mark.gpu


@pytest.fixture()
def eng():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    engine = RunEngine(EngineConfig(cpu_workers=2, workers_per_device=2)).start()
    yield engine
    engine.stop()


class TestStreamingGraphCapture:
    def test_three_stage_pipeline_with_capture(self, eng):
        eng.apply_yaml(
            """
kind: EngramTemplate
metadata: {name: embed}
spec: {builtin: embed}
---
kind: Engram
metadata: {name: embedder}
spec:
  templateRef: {name: embed}
  with: {dim: 2048, vocab: 8000}
---
kind: EngramTemplate
metadata: {name: transform-tpl}
spec: {builtin: transform}
---
kind: Engram
metadata: {name: transformer}
spec: {templateRef: {name: transform-tpl}}
---
kind: Story
metadata: {name: stream-gpu}
spec:
  pattern: streaming
  steps:
    - name: featurize
      ref: {name: embedder}
      with: {capture: true}
    - name: tag
      ref: {name: transformer}
      needs: [featurize]
      runtime: {map: {ok: "{{ true }}"}}
"""
        )
        stream = eng.submit_stream("default/stream-gpu")
        ids = torch.randint(0, 8000, (8, 64), device="cuda", dtype=torch.int32)
        n = 12
        for i in range(n):
            stream.push({"tensor": ids.clone(), "seq": i, "items": [{"v": i}]})
        run = stream.finish(timeout=60)
        assert run.phase == Phase.FINISHED, {
            k: (str(v.phase), v.error) for k, v in run.step_states.items()
        }
        st = run.step_states["featurize"].output
        assert st["packetsIn"] == n
        # first packet captures, the rest replay the hipGraph
        assert st["graphReplays"] >= n - 1, st


class TestBigPayloadNestedStory:
    def test_nested_gate_wait_with_hbm_offload(self, eng):
        """Config #5 shape: nested executeStory + gate/wait; a multi-GB
        tensor payload crosses the step edge as a $storageRef (HBM
        resident, never serialized)."""
        from bobrapet_amd.engrams.base import Engram, EngramContext, EngramResult
        from bobrapet_amd.engrams.registry import register

        class BigProducer(Engram):
            name = "big-producer"
            wants_gpu = True

            def run(self, ctx: EngramContext) -> EngramResult:
                gb = float((ctx.input or {}).get("gb", 1.0))
                n = int(gb * (1 << 30) // 2)  # bf16 elements
                t = torch.ones(n, dtype=torch.bfloat16, device=f"cuda:{ctx.device}")
                return EngramResult(
                    output={"blob": ctx.storage.offload_tensor(t), "gb": gb}
                )

        class BigConsumer(Engram):
            name = "big-consumer"
            wants_gpu = True

            def run(self, ctx: EngramContext) -> EngramResult:
                blob = (ctx.input or {}).get("blob")
                assert torch.is_tensor(blob), type(blob)
                s = float(blob[:1000].float().sum().item())
                return EngramResult(output={"checksum": s, "numel": blob.numel()})

        register("big-producer", BigProducer)
        register("big-consumer", BigConsumer)
        eng.apply_yaml(
            """
kind: EngramTemplate
metadata: {name: big-producer}
spec: {builtin: big-producer}
---
kind: EngramTemplate
metadata: {name: big-consumer}
spec: {builtin: big-consumer}
---
kind: Engram
metadata: {name: producer}
spec: {templateRef: {name: big-producer}}
---
kind: Engram
metadata: {name: consumer}
spec: {templateRef: {name: big-consumer}}
---
kind: Story
metadata: {name: inner-big}
spec:
  steps:
    - {name: make, ref: {name: producer}, with: {gb: "{{ inputs.gb }}"}}
    - name: ready
      type: wait
      with: {until: "{{ steps.make.phase == 'Succeeded' }}", pollInterval: 5ms, timeout: 60s}
    - name: use
      ref: {name: consumer}
      needs: [ready]
      with: {blob: "{{ steps.make.output.blob }}"}
  output: {checksum: "{{ steps.use.output.checksum }}", numel: "{{ steps.use.output.numel }}"}
---
kind: Story
metadata: {name: outer-big}
spec:
  steps:
    - {name: approval, type: gate}
    - name: sub
      type: executeStory
      needs: [approval]
      with: {storyRef: inner-big, with: {gb: 4.0}}
  output: {result: "{{ steps.sub.output.output }}"}
"""
        )
        run = eng.submit_run("default/outer-big", {})
        time.sleep(0.1)
        eng.approve_gate(run.key, "approval")
        run = eng.wait(run.key, timeout=180)
        assert run.phase == Phase.SUCCEEDED, (run.error, {
            k: (str(v.phase), str(v.error)) for k, v in run.step_states.items()
        })
        assert run.output["result"]["checksum"] == 1000.0
        assert run.output["result"]["numel"] == int(4.0 * (1 << 30) // 2)


@pytest.mark.gpu
class TestServedModelReuse:
    """A process-cached model must serve SEQUENTIAL stories correctly:
    re-prefill re-syncs the graphed decode position, the KV allocation is
    reused (captured graph pointers stay live), and batch changes get their
    own graphs."""

    def test_sequential_generations_match_fresh_model(self):
        import torch

        from bobrapet_amd.models.llama import LlamaModel

        served = LlamaModel("llama-tiny", device="cuda")
        torch.manual_seed(5)
        p1 = torch.randint(0, served.cfg.vocab_size, (2, 24), device="cuda")
        p2 = torch.randint(0, served.cfg.vocab_size, (2, 16), device="cuda")
        served.generate(p1, 4)          # story 1 (captures graphs)
        got = served.generate(p2, 4)    # story 2 reuses cache + graphs
        fresh = LlamaModel("llama-tiny", device="cuda")
        want = fresh.generate(p2, 4, use_graph=False)
        assert torch.equal(got, want), (got.tolist(), want.tolist())

    def test_batch_change_recaptures(self):
        import torch

        from bobrapet_amd.models.llama import LlamaModel

        m = LlamaModel("llama-tiny", device="cuda")
        a = torch.randint(0, m.cfg.vocab_size, (1, 8), device="cuda")
        b = torch.randint(0, m.cfg.vocab_size, (3, 8), device="cuda")
        m.generate(a, 2)
        out = m.generate(b, 2)  # different batch: second graph, fresh cache
        assert out.shape == (3, 2)


class TestNativeLane:
    """The GIL-free built-in engram lane (csrc/hip/native_engrams.cpp):
    the parallel8 story's embed branches + local join run entirely in
    C++/HIP off the core's loop thread — no Python launcher bodies."""

    def test_parallel8_story_on_native_lane(self):
        from bobrapet_amd import _hipops
        from bobrapet_amd.runtime.native import NativeRunner

        resources = """
kind: EngramTemplate
metadata: {name: embed}
spec: {builtin: embed}
---
kind: Engram
metadata: {name: embedder}
spec:
  templateRef: {name: embed}
  with: {dim: 1024, vocab: 4096, batch: 8, seqLen: 32}
---
kind: EngramTemplate
metadata: {name: allgather-join}
spec: {builtin: allgather-join}
---
kind: Engram
metadata: {name: joiner}
spec: {templateRef: {name: allgather-join}}
---
kind: Story
metadata: {name: lane8}
spec:
  steps:
    - name: fanout
      type: parallel
      with:
        steps:
""" + "\n".join(
            f"          - {{name: b{i}, ref: {{name: embedder}}, with: {{seed: {i}}}}}"
            for i in range(8)
        ) + """
    - name: join
      ref: {name: joiner}
      needs: [fanout]
      with:
        branches: "{{ steps.fanout.output.branches }}"
  output:
    rows: "{{ steps.join.output.worldRows }}"
"""
        eng = RunEngine(EngineConfig(cpu_workers=2, workers_per_device=4)).start()
        try:
            eng.apply_yaml(resources)
            runner = NativeRunner.from_run_engine(eng)
            assert runner.native_kinds.get("embed") == 1, "lane not registered"
            story = eng.registry.story("lane8", "default")
            runner.compile(story)
            for _ in range(3):
                st = runner.run_story(story, {}, timeout=30.0, gc=False)
                assert st["phase"] == "Succeeded", st
                assert st["output"]["rows"] == 8 * 8, st["output"]
            assert _hipops.native_registry_size() > 0, "lane never held payloads"
            # joined rows are L2-normalized embed rows: hydrate + check norms
            joined_ref = st["steps"]["join"]["output"]["joined"]
            t = eng.storage.hydrate(joined_ref)
            norms = t.float().norm(dim=-1)
            assert abs(norms.mean().item() - 1.0) < 0.05
            # gc releases the lane-held payloads of reclaimed runs
            before = _hipops.native_registry_size()
            rid = runner.submit(story, {})
            runner.wait(rid, 30.0)
            runner.engine.gc_run(rid)
            time.sleep(0.1)
            assert _hipops.native_registry_size() <= before + 1
            runner.stop()
        finally:
            eng.stop()


class TestRcclWorldOne:
    """RCCL actually exercised on hardware: a real
    nccl-backend (RCCL on ROCm) process group at world_size=1 runs the
    collective code paths the multi-rank bench uses — all_gather join,
    ring-attention pass-through — so an 8-GPU driver run hits code that
    has executed under RCCL, not only under gloo."""

    def test_collectives_under_rccl(self):
        import os

        import torch.distributed as dist

        from bobrapet_amd.parallel import collectives

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29617")
        dist.init_process_group("nccl", rank=0, world_size=1)
        try:
            t = torch.randn(64, 128, device="cuda", dtype=torch.bfloat16)
            g = collectives.all_gather_tensor(t, None)
            assert g.shape[0] == 1 and torch.equal(g[0], t)
            r = t.clone()
            dist.all_reduce(r)
            assert torch.equal(r, t)
            # ring attention degenerates to local attention at world 1
            from bobrapet_amd.parallel.ring_attention import ring_attention
            from bobrapet_amd import ops

            q = torch.randn(1, 256, 4, 128, device="cuda", dtype=torch.bfloat16)
            k = torch.randn(1, 256, 2, 128, device="cuda", dtype=torch.bfloat16)
            v = torch.randn(1, 256, 2, 128, device="cuda", dtype=torch.bfloat16)
            out = ring_attention(q, k, v, causal=True)
            ref = ops.attn_prefill(q, k, v, causal=True)
            assert (out.float() - ref.float()).abs().max().item() < 1e-3
        finally:
            dist.destroy_process_group()
