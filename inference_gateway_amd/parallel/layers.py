"""Tensor-parallel linear layers for the MI355X engine.

Megatron-style sharding adapted to the 8-GPU xGMI node:
column-parallel layers keep their shard's activations local; row-parallel
layers end in one RCCL all-reduce. GEMMs go through F.linear (hipBLASLt on
ROCm); the fused hot ops around them are the hand-written gfx950 kernels.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .comm import row_parallel_matmul_allreduce, tp_all_reduce


def _init_weight(shape, device, dtype, seed: int) -> torch.Tensor:
    """Deterministic random init (synthetic weights; no network for real
    checkpoints). Seed is derived from the param identity so every rank
    materializes the same full tensor before slicing its shard — which makes
    TP=N bitwise-consistent with TP=1 up to reduction order (tested on CPU
    with gloo, world_size=2). Generated on the target device: materializing
    70B params through a CPU generator would take minutes."""
    dev = torch.device(device)
    g = torch.Generator(device=dev).manual_seed(seed & 0x7FFFFFFF)
    w = torch.randn(shape, generator=g, device=dev, dtype=torch.float32) * 0.02
    return w.to(dtype)


class ColumnParallelLinear(nn.Module):
    """Y = X W^T with W sharded along the output dim."""

    def __init__(self, in_features: int, out_features: int, tp_rank: int,
                 tp_size: int, device, dtype, seed: int):
        super().__init__()
        assert out_features % tp_size == 0
        self.shard = out_features // tp_size
        full = _init_weight((out_features, in_features), device, dtype, seed)
        w = full[tp_rank * self.shard:(tp_rank + 1) * self.shard].contiguous()
        self.weight = nn.Parameter(w, requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.linear(x, self.weight)


class RowParallelLinear(nn.Module):
    """Y = X W^T with W sharded along the input dim; all-reduce on output."""

    def __init__(self, in_features: int, out_features: int, tp_rank: int,
                 tp_size: int, device, dtype, seed: int):
        super().__init__()
        assert in_features % tp_size == 0
        self.shard = in_features // tp_size
        full = _init_weight((out_features, in_features), device, dtype, seed)
        w = full[:, tp_rank * self.shard:(tp_rank + 1) * self.shard].contiguous()
        self.weight = nn.Parameter(w, requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # prefill-size inputs chunk the GEMM and overlap each chunk's
        # all-reduce (comm stream) with the next chunk's GEMM; decode
        # sizes go straight through (one-shot xGMI kernel when active)
        return row_parallel_matmul_allreduce(x, self.weight)


class ReplicatedLinear(nn.Module):
    def __init__(self, in_features: int, out_features: int, device, dtype,
                 seed: int):
        super().__init__()
        w = _init_weight((out_features, in_features), device, dtype, seed)
        self.weight = nn.Parameter(w, requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.linear(x, self.weight)
