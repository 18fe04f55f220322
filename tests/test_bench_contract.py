"""Bench contract tests: the driver runs bench.py here (CPU) and on GPU
boxes at N=1..8; these tests rehearse the exact multi-rank path (gloo,
world 2) including the parallel story + all-gather join lockstep."""
import json
import multiprocessing as mp
import os
import socket
import subprocess
import sys

import pytest


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_default_single_rank():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "5", "--warmup", "1"],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=180,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = json.loads(out.stdout.strip().splitlines()[-1])
    for key in (
        "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
        "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
    ):
        assert key in line, key
    assert line["n_gpus"] == 1
    assert line["steps"] == 5
    assert line["value"] > 0


def _rank_main(rank: int, world: int, port: int, q) -> None:
    os.environ.update(
        {
            "RANK": str(rank),
            "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank),
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
        }
    )
    try:
        sys.path.insert(0, REPO)
        from bobrapet_amd.engine import EngineConfig, RunEngine
        from bobrapet_amd.enums import Phase
        from bobrapet_amd.parallel import group

        group.init_distributed(backend="gloo")
        eng = RunEngine(EngineConfig(cpu_workers=2)).start()
        # the bench's parallel story shape, shrunk for CPU: 4 embed branches
        # + the cross-rank all-gather join (the lockstep collective)
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
  with: {dim: 64, vocab: 100, batch: 2, seqLen: 4}
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
metadata: {name: par}
spec:
  steps:
    - name: fanout
      type: parallel
      with:
        steps:
          - {name: b0, ref: {name: embedder}, with: {seed: 0}}
          - {name: b1, ref: {name: embedder}, with: {seed: 1}}
          - {name: b2, ref: {name: embedder}, with: {seed: 2}}
          - {name: b3, ref: {name: embedder}, with: {seed: 3}}
    - name: join
      ref: {name: joiner}
      needs: [fanout]
      with:
        branches: "{{ steps.fanout.output.branches }}"
  output:
    rows: "{{ steps.join.output.worldRows }}"
"""
        )
        rows = []
        for i in range(3):  # sequential lockstep runs, like the bench loop
            run = eng.run_story("default/par", {"i": i}, timeout=120)
            assert run.phase == Phase.SUCCEEDED, (run.error, {
                k: str(v.phase) for k, v in run.step_states.items()
            })
            rows.append(run.output["rows"])
        agg = group.sum_over_ranks(float(rows[-1]), device="cpu")
        q.put((rank, {"rows": rows, "agg": agg}))
        eng.stop()
        group.teardown()
    except Exception as exc:
        import traceback

        q.put((rank, {"error": f"{exc}\n{traceback.format_exc()}"}))


@pytest.mark.timeout(240)
def test_parallel_join_two_ranks_lockstep():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    world = 2
    procs = [ctx.Process(target=_rank_main, args=(r, world, port, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, payload = q.get(timeout=220)
        results[rank] = payload
    for p in procs:
        p.join(timeout=30)
    for rank, payload in results.items():
        assert "error" not in payload, f"rank {rank}: {payload.get('error')}"
        # 4 branches x batch 2 x 2 ranks = 16 joined rows on every rank
        assert payload["rows"] == [16, 16, 16], payload
    assert results[0]["agg"] == results[1]["agg"]


@pytest.mark.timeout(300)
def test_bench_parallel8_pipelined_two_ranks():
    """Rehearse the driver's multi-rank bench launch exactly: torchrun world=2
    (gloo on CPU), the REAL parallel8 story (no cpu fallback), pipelined
    in-flight slots + comm-slot communicators for the all-gather join."""
    port = _free_port()
    env = dict(os.environ)
    env.update({"BOBRA_BENCH_NO_CPU_FALLBACK": "1", "BOBRA_BENCH_INFLIGHT": "4"})
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", str(port),
            "bench.py", "--gpus", "2", "--steps", "6", "--warmup", "1",
            "--config", "parallel8",
        ],
        cwd=REPO,
        env=env,
        capture_output=True,
        text=True,
        timeout=280,
    )
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-2500:])
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert lines, out.stdout[-1500:]
    line = json.loads(lines[-1])
    assert line["n_gpus"] == 2
    assert line["value"] > 0
    assert line["config"]["branches"] == 8
