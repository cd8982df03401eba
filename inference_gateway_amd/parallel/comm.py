"""Distributed communication over RCCL/xGMI.

MI355X-first design: one process per GPU,
torch.distributed with the "nccl" backend (RCCL on ROCm) for TP
collectives over the node's fully-connected xGMI mesh; "gloo" for CPU
tests.  All collectives go through this module so the engine code never
touches process-group plumbing directly.

TP membership is EXPLICIT (init_tp), not inferred from torch.distributed
being initialized: a torchrun launch may be data-parallel replicas (the
weak-scaling bench) where each rank runs an unsharded model and the
process group is only used for barriers.

r2 additions:
  * one-shot all-reduce (parallel/oneshot.py) for decode-size bf16
    tensors: direct xGMI peer reads, one sync round, graph-capturable
    custom kernel — RCCL's ring is per-link-bound on the point-to-point
    topology and RCCL-in-graph capture is version-fragile;
  * a dedicated comm stream + chunked overlap for LARGE (prefill-size)
    row-parallel all-reduces: the all-reduce of chunk i overlaps the
    GEMM of chunk i+1 (tp_all_reduce_overlapped);
  * warm_up_collectives(): RCCL lazily builds communicators on first
    use — a first collective inside hipGraph capture would fail, so the
    engine warms the comm up eagerly before any capture.
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist

_TP_SIZE = 1
_TP_RANK = 0
_ONESHOT = None            # OneShotAllReduce or None (ring fallback)
_COMM_STREAM: Optional[torch.cuda.Stream] = None
# decode-size threshold for the one-shot path: beyond this the parallel
# per-link reads stop beating the pipelined ring (and the staging buffer
# is sized for it)
_ONESHOT_MAX_BYTES = 16 << 20


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
    global _TP_SIZE, _TP_RANK, _ONESHOT
    if size is None:
        size = dist.get_world_size() if dist.is_initialized() else 1
    if size > 1 and (not dist.is_initialized()
                     or dist.get_world_size() != size):
        raise RuntimeError(
            f"init_tp({size}) requires an initialized process group of that size")
    _TP_SIZE = size
    _TP_RANK = dist.get_rank() if size > 1 else 0
    if size > 1 and torch.cuda.is_available():
        from . import oneshot

        _ONESHOT = oneshot.try_init(
            _TP_RANK, size, torch.device("cuda",
                                         torch.cuda.current_device()),
            max_bytes=_ONESHOT_MAX_BYTES)
        warm_up_collectives()


def tp_world_size() -> int:
    return _TP_SIZE


def tp_rank() -> int:
    return _TP_RANK


def comm_stream() -> Optional[torch.cuda.Stream]:
    global _COMM_STREAM
    if _COMM_STREAM is None and torch.cuda.is_available():
        _COMM_STREAM = torch.cuda.Stream()
    return _COMM_STREAM


def oneshot_active() -> bool:
    return _ONESHOT is not None


def warm_up_collectives() -> None:
    """Run one eager all-reduce so RCCL builds its communicators BEFORE
    any hipGraph capture (a communicator bootstrap inside capture
    fails); also touches the one-shot path so its first (allocation-
    side-effect-free) launch is out of the way."""
    if _TP_SIZE <= 1 or not dist.is_initialized():
        return
    dev = ("cuda" if torch.cuda.is_available() else "cpu")
    t = torch.ones(8, dtype=torch.bfloat16 if dev == "cuda"
                   else torch.float32, device=dev)
    dist.all_reduce(t)
    if _ONESHOT is not None:
        _ONESHOT.all_reduce_(
            torch.ones(8, dtype=torch.bfloat16, device=dev))
        torch.cuda.synchronize()
        _ONESHOT.check_error()


def tp_all_reduce(t: torch.Tensor) -> torch.Tensor:
    """In-place sum all-reduce across the TP group (no-op when TP=1).
    Decode-size bf16 tensors take the one-shot xGMI kernel when it
    initialized; everything else goes through RCCL."""
    if _TP_SIZE > 1:
        if _ONESHOT is not None and _ONESHOT.usable(t):
            _ONESHOT.all_reduce_(t)
        else:
            dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


# rows above which the chunked comm-stream overlap pays: the all-reduce
# of chunk 0 hides under the GEMM of chunk 1 (prefill activations)
_OVERLAP_MIN_ROWS = 2048
_OVERLAP_CHUNKS = 2


def row_parallel_matmul_allreduce(x: torch.Tensor,
                                  weight: torch.Tensor) -> torch.Tensor:
    """Y = X @ W^T followed by the TP sum all-reduce, with comm/GEMM
    overlap for large row counts: X is split along rows into chunks, the
    all-reduce of chunk i (on the comm stream) overlaps the GEMM of
    chunk i+1 (compute stream).  Falls back to the plain sequential form
    when TP=1, on CPU/gloo, for small tensors, or under graph capture
    (stream forks inside capture are legal but the gain is decode-size
    -irrelevant there; the one-shot kernel owns decode)."""
    import torch.nn.functional as F

    if (_TP_SIZE <= 1 or not torch.cuda.is_available()
            or x.shape[0] < _OVERLAP_MIN_ROWS
            or torch.cuda.is_current_stream_capturing()):
        return tp_all_reduce(F.linear(x, weight))
    cs = comm_stream()
    cur = torch.cuda.current_stream()
    n = _OVERLAP_CHUNKS
    rows = x.shape[0]
    step = (rows + n - 1) // n
    outs = []
    events = []
    for i in range(n):
        xi = x[i * step: min(rows, (i + 1) * step)]
        if xi.shape[0] == 0:
            continue
        yi = F.linear(xi, weight)
        # hand yi to the comm stream: all-reduce overlaps the NEXT
        # chunk's GEMM on the compute stream
        cs.wait_stream(cur)
        with torch.cuda.stream(cs):
            if _ONESHOT is not None and _ONESHOT.usable(yi):
                _ONESHOT.all_reduce_(yi)
            else:
                dist.all_reduce(yi, op=dist.ReduceOp.SUM)
            ev = torch.cuda.Event()
            ev.record(cs)
        # yi's lifetime: record_stream so the allocator does not reuse
        # it while the comm stream still reads it
        yi.record_stream(cs)
        outs.append(yi)
        events.append(ev)
    for ev in events:
        cur.wait_event(ev)
    return torch.cat(outs, dim=0) if len(outs) > 1 else outs[0]


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
