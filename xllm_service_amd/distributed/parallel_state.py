"""Process-group state for tensor parallelism.

One process per GPU; torch.distributed backend "nccl" IS RCCL on ROCm, riding
xGMI between the 8 MI355X GPUs of a node. CPU tests use gloo.
"""
from __future__ import annotations

import os
from datetime import timedelta

import torch
import torch.distributed as dist

_TP_GROUP = None
_TP_RANK = 0
_TP_SIZE = 1


def init_distributed(backend: str | None = None, timeout_s: int = 120) -> None:
    """Initialise the default process group from torchrun env vars."""
    if dist.is_initialized():
        return
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    dist.init_process_group(backend=backend,
                            timeout=timedelta(seconds=timeout_s))
    if backend == "nccl":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))


def shutdown() -> None:
    """Barrier + destroy the default group. Spawned worker processes must
    call this before exiting: gloo's background threads abort the process
    ("terminate called without an active exception") if torn down by
    process exit instead."""
    global _TP_GROUP, _TP_RANK, _TP_SIZE
    if dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()
    _TP_GROUP, _TP_RANK, _TP_SIZE = None, 0, 1


def init_tensor_parallel(tp_size: int = 1, group=None) -> None:
    """Declare the TP group for model layers created afterwards."""
    global _TP_GROUP, _TP_RANK, _TP_SIZE
    if tp_size <= 1:
        _TP_GROUP, _TP_RANK, _TP_SIZE = None, 0, 1
        return
    assert dist.is_initialized(), "init_distributed() first"
    _TP_GROUP = group if group is not None else dist.group.WORLD
    _TP_RANK = dist.get_rank(_TP_GROUP)
    _TP_SIZE = tp_size
    assert dist.get_world_size(_TP_GROUP) == tp_size


def tp_size() -> int:
    return _TP_SIZE


def tp_rank() -> int:
    return _TP_RANK


def tp_group():
    return _TP_GROUP


def tp_all_reduce(x: torch.Tensor) -> torch.Tensor:
    if _TP_SIZE > 1:
        dist.all_reduce(x, group=_TP_GROUP)
    return x


def tp_all_gather(x: torch.Tensor, dim: int = -1) -> torch.Tensor:
    if _TP_SIZE == 1:
        return x
    parts = [torch.empty_like(x) for _ in range(_TP_SIZE)]
    dist.all_gather(parts, x, group=_TP_GROUP)
    return torch.cat(parts, dim=dim)
