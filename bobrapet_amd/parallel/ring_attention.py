"""Ring attention over xGMI: sequence-parallel prefill across GPUs.

Closes SURVEY.md §5.7's long-context answer for contexts that exceed one
GPU: the sequence is partitioned rank-major (rank r holds rows
[r*S_loc, (r+1)*S_loc)), K/V blocks rotate around the ring by RCCL
send/recv (xGMI is point-to-point — a ring pass uses exactly one link per
GPU per step, the topology-native pattern), and each rank merges per-block
softmax partials online.

The merge surface is `ops.attn_prefill_stats`: the CDNA4 prefill kernel
exports each row's running (m, l) in its exp2 domain (attention.hip
epilogue), and the fp32 CPU reference produces the SAME domain, so this
module is implementation-uniform (gloo/CPU tests, RCCL/GPU production).

Causality with rank-major blocks is block-triangular: block j < r is fully
visible (non-causal kernel call), j == r is locally causal, j > r is
skipped entirely — rank 0 does 1 compute step, rank w-1 does w, matching
the LPT observation that deep rows carry the work.
"""
from __future__ import annotations

import typing as _t

import torch
import torch.distributed as dist

from .. import ops


def _merge(acc, o, stats):
    """Online merge of a new block's (normalized O, (m,l)) into the
    accumulator; all in the stats' exp2 domain."""
    if acc is None:
        return [o.float(), stats[..., 0].clone(), stats[..., 1].clone()]
    o_a, m_a, l_a = acc
    m_b = stats[..., 0]
    l_b = stats[..., 1]
    m = torch.maximum(m_a, m_b)
    wa = l_a * torch.exp2(m_a - m)
    wb = l_b * torch.exp2(m_b - m)
    # weights are [B,H,S] — broadcast over the BSHD output's D
    wa_ = wa.permute(0, 2, 1).unsqueeze(-1)
    wb_ = wb.permute(0, 2, 1).unsqueeze(-1)
    o_new = (o_a * wa_ + o.float() * wb_) / (wa_ + wb_)
    return [o_new, m, wa + wb]


def _ring_pass(k: torch.Tensor, v: torch.Tensor, pg) -> _t.Tuple[torch.Tensor, torch.Tensor]:
    """Pass (k, v) to rank+1, receive from rank-1 (one xGMI link each way)."""
    rank = dist.get_rank(pg)
    world = dist.get_world_size(pg)
    nxt = (rank + 1) % world
    prv = (rank - 1) % world
    rk = torch.empty_like(k)
    rv = torch.empty_like(v)
    # batch_isend_irecv groups the four P2P ops so RCCL schedules the
    # send+recv pairs together: plain isend-before-irecv on every rank is
    # a classic NCCL/RCCL P2P deadlock pattern (and NCCL ignores tags —
    # correctness must come from posting order, which batching guarantees).
    reqs = dist.batch_isend_irecv(
        [
            dist.P2POp(dist.isend, k, peer=nxt, group=pg),
            dist.P2POp(dist.irecv, rk, peer=prv, group=pg),
            dist.P2POp(dist.isend, v, peer=nxt, group=pg),
            dist.P2POp(dist.irecv, rv, peer=prv, group=pg),
        ]
    )
    for r in reqs:
        r.wait()
    return rk, rv


def ring_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    scale: _t.Optional[float] = None,
    causal: bool = True,
    pg=None,
) -> torch.Tensor:
    """Sequence-parallel attention: every rank passes its LOCAL BSHD blocks
    (q/k/v [B, S_local, H, D]; global sequence = rank-major concatenation)
    and receives its local output block [B, S_local, Hq, D]."""
    if not dist.is_initialized() or dist.get_world_size(pg) == 1:
        return ops.attn_prefill(q, k, v, scale, causal)
    rank = dist.get_rank(pg)
    world = dist.get_world_size(pg)
    cur_k = k.contiguous()
    cur_v = v.contiguous()
    acc = None
    for step in range(world):
        j = (rank - step) % world  # origin rank of the current K/V block
        if not (causal and j > rank):
            o, stats = ops.attn_prefill_stats(q, cur_k, cur_v, scale, causal and j == rank)
            acc = _merge(acc, o, stats)
        if step < world - 1:
            cur_k, cur_v = _ring_pass(cur_k, cur_v, pg)
    assert acc is not None  # j == rank always computes
    return acc[0].to(q.dtype)
