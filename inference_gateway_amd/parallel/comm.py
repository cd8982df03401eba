"""Distributed communication over RCCL/xGMI.

MI355X-first design: one process per GPU,
torch.distributed with the "nccl" backend (RCCL on ROCm) for TP all-reduce
over the node's fully-connected xGMI mesh; "gloo" for CPU tests. All
collectives go through this module so the engine code never touches
process-group plumbing directly.

TP membership is EXPLICIT (init_tp), not inferred from torch.distributed
being initialized: a torchrun launch may be data-parallel replicas (the
weak-scaling bench) where each rank runs an unsharded model and the process
group is only used for barriers.
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist

_TP_SIZE = 1
_TP_RANK = 0


def init_distributed(backend: Optional[str] = None) -> None:
    """Initialize the process group from torchrun env vars if present."""
    if dist.is_initialized():
        return
    if "RANK" not in os.environ or int(os.environ.get("WORLD_SIZE", "1")) <= 1:
        return
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend)
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))


def init_tp(size: Optional[int] = None) -> None:
    """Declare that the whole process group is one tensor-parallel group."""
    global _TP_SIZE, _TP_RANK
    if size is None:
        size = dist.get_world_size() if dist.is_initialized() else 1
    if size > 1 and (not dist.is_initialized()
                     or dist.get_world_size() != size):
        raise RuntimeError(
            f"init_tp({size}) requires an initialized process group of that size")
    _TP_SIZE = size
    _TP_RANK = dist.get_rank() if size > 1 else 0


def tp_world_size() -> int:
    return _TP_SIZE


def tp_rank() -> int:
    return _TP_RANK


def tp_all_reduce(t: torch.Tensor) -> torch.Tensor:
    """In-place sum all-reduce across the TP group (no-op when TP=1)."""
    if _TP_SIZE > 1:
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


def tp_broadcast(t: torch.Tensor, src: int = 0) -> torch.Tensor:
    if _TP_SIZE > 1:
        dist.broadcast(t, src=src)
    return t


def world_barrier() -> None:
    """Barrier over the WHOLE process group (TP or DP alike)."""
    if dist.is_initialized():
        dist.barrier()


def world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def world_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0
