"""Process-group management: one process per GPU over RCCL/xGMI.

Role replacement for the reference's cross-pod coordination (SURVEY.md
§5.8): torch.distributed with backend "nccl" (RCCL on ROCm) on GPU boxes,
"gloo" on CPU (multi-process CPU tests).  Rendezvous always on 127.0.0.1.
"""
from __future__ import annotations

import datetime
import os
import typing as _t

import torch
import torch.distributed as dist


def env_rank() -> int:
    return int(os.environ.get("RANK", "0"))


def env_world_size() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def env_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))


def init_distributed(backend: _t.Optional[str] = None, timeout_s: float = 300.0) -> bool:
    """Initialize torch.distributed from the env; returns True when a
    multi-rank group is live. Safe to call with WORLD_SIZE unset/1."""
    if dist.is_initialized():
        return dist.get_world_size() > 1
    world = env_world_size()
    if world <= 1:
        return False
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29517")
    if backend == "nccl":
        torch.cuda.set_device(env_local_rank())
    dist.init_process_group(
        backend=backend,
        rank=env_rank(),
        world_size=world,
        timeout=datetime.timedelta(seconds=timeout_s),
    )
    return True


_SLOT_GROUPS: _t.List = []


def ensure_comm_slots(n: int) -> None:
    """Create n extra all-rank communicators (collective call — every rank
    must call with the same n, in the same order).  Slot communicators make
    CONCURRENT in-flight stories safe: a story pinned to slot s only issues
    collectives on communicator s, per-slot submission is sequential, so
    the cross-rank operation order matches per communicator even when W
    stories overlap (xGMI has no global stream order to rely on)."""
    if not dist.is_initialized() or dist.get_world_size() <= 1:
        return
    ranks = list(range(dist.get_world_size()))
    while len(_SLOT_GROUPS) < n:
        _SLOT_GROUPS.append(dist.new_group(ranks=ranks))


def comm_slot(i: _t.Optional[int]):
    """The process group for in-flight slot i (None → default group)."""
    if i is None or not _SLOT_GROUPS:
        return None
    return _SLOT_GROUPS[int(i) % len(_SLOT_GROUPS)]


def teardown() -> None:
    global _SLOT_GROUPS
    _SLOT_GROUPS = []
    if dist.is_initialized():
        dist.destroy_process_group()


def rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def barrier() -> None:
    if dist.is_initialized():
        if dist.get_backend() == "nccl":
            dist.barrier(device_ids=[torch.cuda.current_device()])
        else:
            dist.barrier()


def max_over_ranks(value: float, device=None) -> float:
    """MAX of a scalar over all ranks (bench contract)."""
    if not dist.is_initialized():
        return value
    backend = dist.get_backend()
    dev = device if device is not None else (
        torch.cuda.current_device() if backend == "nccl" else "cpu"
    )
    t = torch.tensor([value], dtype=torch.float64, device=dev)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def sum_over_ranks(value: float, device=None) -> float:
    if not dist.is_initialized():
        return value
    backend = dist.get_backend()
    dev = device if device is not None else (
        torch.cuda.current_device() if backend == "nccl" else "cpu"
    )
    t = torch.tensor([value], dtype=torch.float64, device=dev)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return float(t.item())
