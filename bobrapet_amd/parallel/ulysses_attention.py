"""Ulysses attention: head-parallel sequence-parallel attention.

The complement of ring attention: instead of rotating K/V
blocks, an all-to-all re-shards the activation from sequence-sharded
[B, S_local, H, D] to head-sharded [B, S_global, H/world, D]; each rank
then runs ORDINARY full-sequence attention over its head group (exact
softmax, no cross-block merging), and a second all-to-all restores the
sequence sharding.  Two all-to-alls move the same bytes as one ring
rotation, but as a single dense exchange — better when S_local is small
(latency-bound rings) or when exact single-pass softmax is preferred.

Constraints: Hq % world == 0 and Hkv % world == 0 (Llama-3-8B's Hkv=8
supports up to 8-way; GQA groups stay intact per rank).

Backend notes: NCCL/RCCL runs true all_to_all over xGMI; gloo (CPU tests)
lacks all_to_all, so the exchange falls back to all_gather + local slice —
identical semantics, more traffic, test-only.
"""
from __future__ import annotations

import typing as _t

import torch
import torch.distributed as dist

from .. import ops


def _exchange_s_to_h(t: torch.Tensor, world: int, pg) -> torch.Tensor:
    """[B, S_loc, H, D] sequence-sharded → [B, S_glob, H/world, D] head-sharded."""
    rank = dist.get_rank(pg)
    hg = t.shape[2] // world
    send = [t[:, :, g * hg : (g + 1) * hg].contiguous() for g in range(world)]
    if dist.get_backend(pg) == "gloo":
        mine = []
        for g, chunk in enumerate(send):  # all_gather my-head-group rows from all ranks
            gathered = [torch.empty_like(chunk) for _ in range(world)]
            dist.all_gather(gathered, chunk, group=pg)
            if g == rank:
                mine = gathered
        return torch.cat(mine, dim=1)
    recv = [torch.empty_like(send[0]) for _ in range(world)]
    dist.all_to_all(recv, send, group=pg)
    return torch.cat(recv, dim=1)


def _exchange_h_to_s(t: torch.Tensor, world: int, pg) -> torch.Tensor:
    """[B, S_glob, H/world, D] head-sharded → [B, S_loc, H, D] sequence-sharded."""
    rank = dist.get_rank(pg)
    s_loc = t.shape[1] // world
    send = [t[:, j * s_loc : (j + 1) * s_loc].contiguous() for j in range(world)]
    if dist.get_backend(pg) == "gloo":
        mine = []
        for j, chunk in enumerate(send):
            gathered = [torch.empty_like(chunk) for _ in range(world)]
            dist.all_gather(gathered, chunk, group=pg)
            if j == rank:
                mine = gathered
        return torch.cat(mine, dim=2)
    recv = [torch.empty_like(send[0]) for _ in range(world)]
    dist.all_to_all(recv, send, group=pg)
    return torch.cat(recv, dim=2)


def ulysses_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    scale: _t.Optional[float] = None,
    causal: bool = True,
    pg=None,
) -> torch.Tensor:
    """Every rank passes its LOCAL BSHD shard (rank-major global sequence)
    and receives its local output shard [B, S_local, Hq, D]."""
    if not dist.is_initialized() or dist.get_world_size(pg) == 1:
        return ops.attn_prefill(q, k, v, scale, causal)
    world = dist.get_world_size(pg)
    if q.shape[2] % world or k.shape[2] % world:
        raise ValueError(
            f"ulysses: Hq={q.shape[2]} and Hkv={k.shape[2]} must divide world={world}"
        )
    qh = _exchange_s_to_h(q, world, pg)
    kh = _exchange_s_to_h(k, world, pg)
    vh = _exchange_s_to_h(v, world, pg)
    out = ops.attn_prefill(qh, kh, vh, scale, causal)
    return _exchange_h_to_s(out, world, pg)
