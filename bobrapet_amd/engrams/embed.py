"""embed engram: batched embedding on MI355X.

BASELINE.json config #3: 8-branch `parallel` fan-out Story, one embed
engram per branch, RCCL all-gather join on 8×MI355X.  The hot op is the
hand-written fused gather + mean-pool + L2-normalize kernel
(csrc/hip/elementwise.hip launch_embed_pool).

config: {dim: 4096, vocab: 32000, seed}
input:  {ids: [[int]]} explicit, or {batch, seqLen, seed} synthetic
output: {batch, dim, latencyMs, embeddings: $storageRef (tensor)}
"""
from __future__ import annotations

import time
import typing as _t

import torch

from .base import Engram, EngramContext, EngramFailure, EngramResult
from .registry import register_class

_TABLE_CACHE: _t.Dict[tuple, torch.Tensor] = {}


def _table(vocab: int, dim: int, seed: int, device) -> torch.Tensor:
    key = (vocab, dim, seed, str(device))
    t = _TABLE_CACHE.get(key)
    if t is None:
        gen = torch.Generator(device=device)
        gen.manual_seed(seed)
        t = torch.empty(vocab, dim, device=device, dtype=torch.bfloat16).normal_(
            0.0, 0.05, generator=gen
        )
        _TABLE_CACHE[key] = t
    return t


@register_class
class EmbedEngram(Engram):
    name = "embed"
    wants_gpu = True

    def tensor_compute(self, ctx: EngramContext, ids: "torch.Tensor") -> "torch.Tensor":
        """Pure tensor path for hipGraph capture (streaming stages with
        with.capture: true replay this as a captured graph per packet)."""
        from .. import ops

        cfg = dict(ctx.config or {})
        dim = int(cfg.get("dim", 4096))
        vocab = int(cfg.get("vocab", 32000))
        table = _table(vocab, dim, int(cfg.get("seed", 7)), ids.device)
        return ops.embed_pool(table, ids.to(torch.int32))

    def run(self, ctx: EngramContext) -> EngramResult:
        from .. import ops

        cfg = dict(ctx.config or {})
        inp = dict(ctx.input or {}) if isinstance(ctx.input, dict) else {}
        dim = int(inp.get("dim", cfg.get("dim", 4096)))
        vocab = int(inp.get("vocab", cfg.get("vocab", 32000)))
        seed = int(cfg.get("seed", 7))

        device = f"cuda:{ctx.device}" if ctx.device is not None else "cpu"
        table = _table(vocab, dim, seed, device)

        ids = inp.get("ids")
        if ids is not None:
            ids_t = torch.tensor(ids, dtype=torch.int32, device=device)
        else:
            batch = int(inp.get("batch", cfg.get("batch", 32)))
            seq = int(inp.get("seqLen", cfg.get("seqLen", 128)))
            sd = int(inp.get("seed", 0))
            key = ("ids", vocab, batch, seq, sd, str(device))
            ids_t = _TABLE_CACHE.get(key)
            if ids_t is None:
                gen = torch.Generator(device="cpu").manual_seed(sd)
                ids_t = torch.randint(0, vocab, (batch, seq), generator=gen).to(
                    device=device, dtype=torch.int32
                )
                if len(_TABLE_CACHE) < 256:
                    _TABLE_CACHE[key] = ids_t
        if ids_t.ndim != 2:
            raise EngramFailure("embed: ids must be [batch][seq]", exit_code=2)

        t0 = time.monotonic()
        emb = ops.embed_pool(table, ids_t)
        if table.is_cuda:
            torch.cuda.current_stream(table.device).synchronize()
        latency_ms = (time.monotonic() - t0) * 1000.0

        out = {
            "batch": int(ids_t.shape[0]),
            "dim": dim,
            "latencyMs": latency_ms,
        }
        if ctx.storage is not None:
            out["embeddings"] = ctx.storage.offload_tensor(emb)
        else:
            out["sample"] = emb[0, :4].float().tolist()
        return EngramResult(output=out)
