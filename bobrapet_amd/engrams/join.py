"""allgather-join engram: the RCCL join for `parallel` fan-outs.

BASELINE.json config #3's join: branch outputs (embedding tensors resident
in HBM via `$storageRef`) are concatenated locally and all-gathered across
ranks over RCCL/xGMI when a process group is live.

input: {branches: {name: branchOutput}}  (the parallel step's output), or
       {refs: [storageRef, ...]}
output: {rows, dim, worldRows, latencyMs, joined: $storageRef}
"""
from __future__ import annotations

import time

import torch

from ..parallel import collectives, group
from .base import Engram, EngramContext, EngramFailure, EngramResult
from .registry import register_class


@register_class
class AllGatherJoinEngram(Engram):
    name = "allgather-join"
    wants_gpu = False  # works on CPU too (gloo)

    def run(self, ctx: EngramContext) -> EngramResult:
        inp = ctx.input if isinstance(ctx.input, dict) else {}
        tensors = []
        refs = []
        if "branches" in inp and isinstance(inp["branches"], dict):
            for name in sorted(inp["branches"]):
                out = inp["branches"][name]
                if isinstance(out, dict):
                    emb = out.get("embeddings")
                    if emb is None:
                        emb = out.get("logits")
                    if emb is not None:
                        refs.append(emb)
        elif "refs" in inp:
            refs = list(inp["refs"])
        if not refs:
            raise EngramFailure("allgather-join: no branch tensors found", exit_code=2)
        for ref in refs:
            t = ctx.storage.hydrate(ref) if ctx.storage is not None else ref
            if not torch.is_tensor(t):
                raise EngramFailure("allgather-join: branch output is not a tensor", exit_code=2)
            tensors.append(t)

        t0 = time.monotonic()
        local = torch.cat([t.reshape(-1, t.shape[-1]) for t in tensors], dim=0)
        pg = group.comm_slot(inp.get("commSlot"))
        gathered = collectives.all_gather_tensor(local, pg)  # [world, rows, dim]
        joined = gathered.reshape(-1, local.shape[-1])
        if joined.is_cuda:
            # slot-stream sync only: a device-wide sync would stall the
            # sibling branch streams running concurrently on this GPU
            torch.cuda.current_stream(joined.device).synchronize()
        latency_ms = (time.monotonic() - t0) * 1000.0

        out = {
            "rows": int(local.shape[0]),
            "dim": int(local.shape[-1]),
            "worldRows": int(joined.shape[0]),
            "world": group.world_size(),
            "latencyMs": latency_ms,
        }
        if ctx.storage is not None:
            out["joined"] = ctx.storage.offload_tensor(joined)
        return EngramResult(output=out)
