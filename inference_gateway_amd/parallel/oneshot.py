"""One-shot all-reduce over xGMI peer-to-peer (decode-size tensors).

The 8-GPU MI355X node is fully connected (7 xGMI links/GPU at ~153 GB/s,
SURVEY.md §5), so for the ≤-few-MB all-reduces of decode-time tensor
parallelism every rank can read all peers' contributions DIRECTLY in one
synchronization round instead of ring-stepping over a single link.  The
device protocol (double-buffered shared staging, per-block epoch flags,
system-scope release/acquire, bounded spins) lives in
ops/csrc/allreduce_oneshot.hip; this module owns the host plumbing:
buffer allocation, hipIpc handle exchange over the process group, and
the dispatch decision.

Requires HSA_ENABLE_IPC_MODE_LEGACY=0 (dmabuf IPC — exported by the
environment); initialization failures fall back silently to
dist.all_reduce (comm.py), so the ring path always works.  The kernel is
hipGraph-capturable (plain launches, device-side epochs), which is what
lets the TP decode graph replay without RCCL nodes.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


class OneShotAllReduce:
    MAX_RANKS = 8

    def __init__(self, rank: int, world: int, device,
                 max_bytes: int = 32 << 20):
        from .. import ops

        assert 1 < world <= self.MAX_RANKS
        k = ops._kernels
        self.ops = k
        self.rank = rank
        self.world = world
        self.device = device
        self.max_elems = max_bytes // 2          # bf16
        # double-buffered data (parity halves) + fine-grained signal page
        self.data_ptr = k.ar_alloc(2 * max_bytes, False)
        self.sig_ptr = k.ar_alloc(k.ar_sig_bytes(), True)
        handles = [None] * world
        dist.all_gather_object(
            handles, (k.ar_get_handle(self.data_ptr),
                      k.ar_get_handle(self.sig_ptr)))
        self.data_ptrs: List[int] = []
        self.sig_ptrs: List[int] = []
        self._opened: List[int] = []
        for r, (dh, sh) in enumerate(handles):
            if r == rank:
                self.data_ptrs.append(self.data_ptr)
                self.sig_ptrs.append(self.sig_ptr)
            else:
                dp = k.ar_open_handle(dh)
                sp = k.ar_open_handle(sh)
                self.data_ptrs.append(dp)
                self.sig_ptrs.append(sp)
                self._opened.extend((dp, sp))

    def usable(self, t: torch.Tensor) -> bool:
        return (t.is_cuda and t.dtype == torch.bfloat16
                and t.is_contiguous() and t.numel() <= self.max_elems)

    def all_reduce_(self, t: torch.Tensor) -> torch.Tensor:
        """In-place sum all-reduce on the current stream."""
        self.ops.oneshot_allreduce(t, self.rank, self.world,
                                   self.data_ptrs, self.sig_ptrs,
                                   self.max_elems)
        return t

    def check_error(self) -> None:
        if self.ops.ar_check_error(self.sig_ptr):
            raise RuntimeError(
                "one-shot all-reduce spin timeout (a peer rank stalled "
                "or died); outputs since the timeout are invalid")

    def close(self) -> None:
        for p in self._opened:
            try:
                self.ops.ar_close_handle(p)
            except Exception:
                pass
        self._opened = []
        for p in (self.data_ptr, self.sig_ptr):
            try:
                self.ops.ar_free(p)
            except Exception:
                pass


def try_init(rank: int, world: int, device) -> Optional[OneShotAllReduce]:
    """Best-effort: any failure (no IPC support, no kernels, driver
    refuses dmabuf) returns None and the caller keeps dist.all_reduce.

    DEFAULT OFF (IGW_ONESHOT_AR=1 opts in): single-GPU loopback
    validation (tests/test_ops_gpu.py, tools/debug_oneshot.py) found
    rare but reproducible flag-visibility stalls between concurrent
    kernels on one device — a peer's committed system-scope flag store
    stays invisible to acquire-polls for whole spin windows while a
    host-side dump sees it (cross-XCD cache behavior not covered by the
    documented fence semantics).  The protocol is bounded-spin safe
    (poisons + raises, never hangs), but until it is validated on a
    real 8-GPU node — where each rank is its own process/GPU and the
    fabric path differs — the RCCL ring stays the default."""
    import os

    if os.environ.get("IGW_ONESHOT_AR", "0") != "1":
        return None
    if world < 2 or world > OneShotAllReduce.MAX_RANKS:
        return None
    if not torch.cuda.is_available():
        return None
    try:
        return OneShotAllReduce(rank, world, device)
    except Exception:
        return None
