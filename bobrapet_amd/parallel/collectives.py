"""Step-edge collectives over xGMI.

The MI355X replacement for the reference's payload-passing mechanisms
(SURVEY.md §2.6): step outputs cross GPUs by RCCL send/recv; `parallel`
joins aggregate by one-shot all-gather (xGMI is 7 point-to-point links per
GPU — for ≤8 ranks a direct all-gather drives all links concurrently,
unlike a per-link-bound ring — SURVEY.md §5.8).
"""
from __future__ import annotations

import typing as _t

import torch
import torch.distributed as dist


def all_gather_tensor(t: torch.Tensor, pg=None) -> torch.Tensor:
    """Gather a same-shape tensor from every rank → stacked [world, ...].

    all_gather_into_tensor requires the output's FIRST dim to be
    world * input_first_dim (flat concat) — allocate flat, then view.
    `pg` selects a comm-slot communicator (group.comm_slot) so concurrent
    stories can all-gather without cross-rank order hazards."""
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return t.unsqueeze(0)
    world = dist.get_world_size()
    src = t.contiguous()
    flat = torch.empty(
        (world * src.shape[0],) + tuple(src.shape[1:]), dtype=src.dtype, device=src.device
    )
    dist.all_gather_into_tensor(flat, src, group=pg)
    return flat.view((world,) + tuple(src.shape))


def all_gather_object(obj) -> _t.List:
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return [obj]
    out = [None] * dist.get_world_size()
    dist.all_gather_object(out, obj)
    return out


def send_tensor(t: torch.Tensor, dst: int, tag: int = 0) -> None:
    dist.send(t.contiguous(), dst=dst, tag=tag)


def recv_tensor(shape, dtype, src: int, device=None, tag: int = 0) -> torch.Tensor:
    t = torch.empty(shape, dtype=dtype, device=device)
    dist.recv(t, src=src, tag=tag)
    return t


def broadcast_tensor(t: torch.Tensor, src: int = 0) -> torch.Tensor:
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.broadcast(t, src=src)
    return t


def all_reduce_sum(t: torch.Tensor) -> torch.Tensor:
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t
