"""Mixtral-8x7B MoE decoder for the MI355X engine (BASELINE config 5;
net-new engine scope, SURVEY.md §2.3 expert parallelism).

Same attention stack as Llama (hand-written gfx950 kernels); the MLP is a
top-2 router + 8 SwiGLU experts. Expert parallelism strategy on the 8-GPU
xGMI node: experts are TP-sharded like dense MLPs (each rank holds a slice
of every expert), so dispatch stays local and the existing row-parallel
all-reduce covers the combine — the right choice for the fully-connected
xGMI topology at Mixtral's scale (8 experts / 8 GPUs would leave EP
all-to-all bound by single-link bandwidth with no GEMM-shape benefit at
decode batch sizes). Token routing/gather is done with index_select /
index_add (hipBLASLt batched GEMMs per expert); a fused grouped-GEMM HIP
kernel is the planned next optimization.
"""
from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...parallel.comm import tp_rank, tp_world_size
from ...parallel.layers import (ColumnParallelLinear, ReplicatedLinear,
                                RowParallelLinear, _init_weight)
from ..config import ModelConfig
from .batch import ForwardBatch
from .llama import LlamaAttention, _seed


class MixtralMoE(nn.Module):
    def __init__(self, cfg: ModelConfig, layer: int, device, dtype,
                 expert_parallel: bool = False):
        super().__init__()
        tp, rank = tp_world_size(), tp_rank()
        h, inter = cfg.hidden_size, cfg.intermediate_size
        self.num_experts = cfg.num_experts
        self.top_k = cfg.num_experts_per_tok
        self.expert_parallel = expert_parallel and tp > 1
        self.router = ReplicatedLinear(h, cfg.num_experts, device, dtype,
                                       _seed(f"l{layer}.router"))
        if self.expert_parallel:
            # EP: experts PARTITIONED across ranks, each expert full-width;
            # dispatch/combine go over all-to-all (xGMI point-to-point)
            assert cfg.num_experts % tp == 0
            self.ep_size = tp
            self.ep_local = cfg.num_experts // tp
            eids = range(rank * self.ep_local, (rank + 1) * self.ep_local)
            shard_tp, shard_rank = 1, 0
        else:
            # default: experts TP-SHARDED like dense MLPs (every rank holds
            # a slice of every expert) — the right fit for the
            # fully-connected 8-GPU xGMI node at Mixtral's scale
            self.ep_size = 1
            self.ep_local = cfg.num_experts
            eids = range(cfg.num_experts)
            shard_tp, shard_rank = tp, rank
        assert inter % shard_tp == 0
        # per-expert fused [gate|up] and down weights, stacked:
        # gate_up: [E_local, 2*shard, h], down: [E_local, h, shard]
        gus, downs = [], []
        for e in eids:
            g = ColumnParallelLinear(h, inter, shard_rank, shard_tp, device,
                                     dtype, _seed(f"l{layer}.e{e}.gate"))
            u = ColumnParallelLinear(h, inter, shard_rank, shard_tp, device,
                                     dtype, _seed(f"l{layer}.e{e}.up"))
            d = RowParallelLinear(inter, h, shard_rank, shard_tp, device,
                                  dtype, _seed(f"l{layer}.e{e}.down"))
            gus.append(torch.cat([g.weight.data, u.weight.data], 0))
            downs.append(d.weight.data)
        self.gate_up = nn.Parameter(torch.stack(gus).contiguous(),
                                    requires_grad=False)
        self.down = nn.Parameter(torch.stack(downs).contiguous(),
                                 requires_grad=False)
        # K-major copies for the batched-GEMM path: hipBLASLt/Tensile on
        # ROCm 7.2 has a layout-dependent OOB read with a TRANSPOSED-VIEW
        # B at [8,512,4096]x[8,4096,28672] (memory faults, reproduced in
        # isolation — gpurun_out/iso3); contiguous-B runs clean
        self.gate_up_t = nn.Parameter(
            self.gate_up.data.transpose(1, 2).contiguous(),
            requires_grad=False)
        self.down_t = nn.Parameter(
            self.down.data.transpose(1, 2).contiguous(),
            requires_grad=False)

    def forward(self, x: torch.Tensor,
                is_prefill: bool = False) -> torch.Tensor:
        from ...parallel.comm import tp_all_reduce

        if self.expert_parallel:
            if is_prefill:
                # dynamic splits: exact traffic for big token counts
                return self._forward_expert_parallel(x)
            # decode: static-capacity, sync-free (graph-capturable)
            return self._forward_expert_parallel_padded(x)
        if is_prefill and x.shape[0] > 1024:
            out = self._forward_sorted(x)
        else:
            out = self._forward_padded(x)
        return tp_all_reduce(out)

    def _forward_expert_parallel(self, x: torch.Tensor) -> torch.Tensor:
        """EP dispatch/combine over torch.distributed all-to-all (RCCL on
        xGMI; gloo in the CPU tests).  Activations enter replicated (the
        attention block ends in an all-reduce), so each rank routes its
        own sequence shard, all-to-alls the routed token copies to the
        expert-owning ranks, computes its local full-width experts, and
        all-to-alls the results back; an all-gather re-replicates."""
        import torch.distributed as dist

        W, r = self.ep_size, tp_rank()
        T, h = x.shape
        K, El = self.top_k, self.ep_local
        logits = self.router(x).float()
        weights, topk = torch.topk(logits, K, dim=-1)
        weights = torch.softmax(weights, dim=-1).to(x.dtype)
        # this rank's sequence shard
        t0, t1 = (T * r) // W, (T * (r + 1)) // W
        flat_e = topk[t0:t1].reshape(-1)                     # [n]
        flat_t = (torch.arange(t0, t1, device=x.device)
                  .unsqueeze(1).expand(t1 - t0, K).reshape(-1))
        flat_w = weights[t0:t1].reshape(-1)
        dest = flat_e // El                                  # owning rank
        order = torch.argsort(dest, stable=True)
        send_e = flat_e[order]
        send_x = x.index_select(0, flat_t[order])
        out_splits = torch.bincount(dest, minlength=W)
        in_splits = torch.zeros_like(out_splits)
        dist.all_to_all_single(in_splits, out_splits)
        n_in = int(in_splits.sum())
        recv_x = x.new_empty((n_in, h))
        recv_e = send_e.new_empty(n_in)
        dist.all_to_all_single(recv_x, send_x,
                               in_splits.tolist(), out_splits.tolist())
        dist.all_to_all_single(recv_e, send_e,
                               in_splits.tolist(), out_splits.tolist())
        # local experts on the received rows
        ys = torch.zeros_like(recv_x)
        for le in range(El):
            rows = (recv_e == r * El + le).nonzero(as_tuple=True)[0]
            if rows.numel() == 0:
                continue
            xe = recv_x.index_select(0, rows)
            gu = F.linear(xe, self.gate_up[le])
            ys.index_copy_(0, rows,
                           F.linear(ops.silu_mul(gu), self.down[le]))
        # combine: send results back along the reverse routes
        back = torch.empty_like(send_x)
        dist.all_to_all_single(back, ys,
                               out_splits.tolist(), in_splits.tolist())
        out_local = x.new_zeros((t1 - t0, h))
        out_local.index_add_(0, flat_t[order] - t0,
                             back * flat_w[order].unsqueeze(1))
        # re-replicate the sequence across ranks (pad to the max shard —
        # gloo/RCCL all-gather needs equal sizes)
        max_tl = max((T * (d + 1)) // W - (T * d) // W for d in range(W))
        padded = x.new_zeros((max_tl, h))
        padded[: t1 - t0] = out_local
        chunks = [x.new_empty((max_tl, h)) for _ in range(W)]
        dist.all_gather(chunks, padded)
        out = torch.cat(
            [chunks[d][: (T * (d + 1)) // W - (T * d) // W]
             for d in range(W)], 0)
        return out

    def _forward_expert_parallel_padded(self, x: torch.Tensor
                                        ) -> torch.Tensor:
        """Sync-free EP for DECODE (the dynamic-split
        EP path host-syncs on .tolist() and cannot be graph-captured).
        Every (src, dst) pair exchanges a STATIC capacity of
        cap = ceil(Tl)*K rows — the worst case, so routing is EXACT
        (zero token drops) — which makes both all-to-alls equal-split
        (no host-known split lists), every shape static, and the whole
        layer hipGraph-capturable.  Decode-size tensors keep the padding
        traffic trivial (T=256, K=2, W=8: 4 MB/layer send buffer); the
        dynamic path remains for prefill where padding would be W-fold
        HBM traffic.  Numerics: identical to the dynamic EP path
        (tested rank-for-rank on gloo world=2)."""
        import torch.distributed as dist

        W, r = self.ep_size, tp_rank()
        T, h = x.shape
        K, El = self.top_k, self.ep_local
        dev = x.device
        logits = self.router(x).float()
        weights, topk = torch.topk(logits, K, dim=-1)
        weights = torch.softmax(weights, dim=-1).to(x.dtype)
        # this rank's sequence shard (same split as the dynamic path)
        t0, t1 = (T * r) // W, (T * (r + 1)) // W
        tl_max = max((T * (d + 1)) // W - (T * d) // W for d in range(W))
        cap = tl_max * K                   # per-pair slot budget (exact)
        n = (t1 - t0) * K
        flat_e = topk[t0:t1].reshape(-1)                     # [n]
        flat_t = (torch.arange(t0, t1, device=dev)
                  .unsqueeze(1).expand(t1 - t0, K).reshape(-1))
        flat_w = weights[t0:t1].reshape(-1)
        dest = flat_e // El                                  # [n]
        # slot within the destination pair: stable rank among same-dest
        onehot = torch.nn.functional.one_hot(dest, W)
        rank_in = (torch.cumsum(onehot, 0) - onehot).mul(onehot).sum(-1)
        slot = dest * cap + rank_in                          # [n] < W*cap
        sx = x.new_zeros((W * cap, h))
        sx.index_copy_(0, slot, x.index_select(0, flat_t))
        se = torch.full((W * cap,), -1, dtype=torch.int64, device=dev)
        se.index_copy_(0, slot, flat_e)
        # equal-split all-to-all: cap rows per pair, no split lists
        rx = torch.empty_like(sx)
        re = torch.empty_like(se)
        dist.all_to_all_single(rx, sx)
        dist.all_to_all_single(re, se)
        # local experts over the received padded rows (same padded-bmm
        # shape as the TP decode path; invalid rows route to a scrap row)
        local_e = re - r * El
        valid = (local_e >= 0) & (local_e < El)
        le_safe = local_e.clamp(0, El - 1)
        oh = torch.nn.functional.one_hot(le_safe, El) * valid.unsqueeze(1)
        rk = (torch.cumsum(oh, 0) - oh).mul(oh).sum(-1)
        C2 = W * cap
        # invalid (padding) rows get DISTINCT scrap slots past the real
        # region: index_copy_ with duplicate indices is undefined
        scrap = El * C2 + torch.arange(C2, device=dev)
        dst2 = torch.where(valid, le_safe * C2 + rk, scrap)
        buf = x.new_zeros((El * C2 + C2, h))
        buf.index_copy_(0, dst2, rx)
        gu = torch.bmm(buf[: El * C2].view(El, C2, h), self.gate_up_t)
        act = ops.silu_mul(gu.view(El * C2, -1))
        y = torch.bmm(act.view(El, C2, -1), self.down_t).view(El * C2, h)
        ys = torch.cat([y, y.new_zeros((C2, h))], 0).index_select(0, dst2)
        # combine back along the same static routes
        back = torch.empty_like(ys)
        dist.all_to_all_single(back, ys)
        out_local = x.new_zeros((tl_max, h))
        out_local.index_add_(
            0, flat_t - t0,
            back.index_select(0, slot) * flat_w.unsqueeze(1))
        # re-replicate (equal shards of tl_max rows)
        chunks = [x.new_empty((tl_max, h)) for _ in range(W)]
        dist.all_gather(chunks, out_local)
        return torch.cat(
            [chunks[d][: (T * (d + 1)) // W - (T * d) // W]
             for d in range(W)], 0)

    def _forward_padded(self, x: torch.Tensor) -> torch.Tensor:
        """Sync-free dispatch for decode: tokens are scattered into a
        per-expert padded buffer [E, C=T, h] purely with device ops (no
        .tolist()/argsort round trips), so the whole MoE layer is
        hipGraph-capturable. The padding rows cost batched-GEMM FLOPs but
        no extra HBM traffic — decode MoE is bound by streaming all 8
        experts' weights (~1.4 GB/layer), so the padded rows are free."""
        T, h = x.shape
        E, K = self.num_experts, self.top_k
        C = (T + 63) // 64 * 64    # grouped-GEMM row-tile granularity
        router_logits = self.router(x).float()                # [T, E]
        # fused routing kernel: top-k + softmax + atomic-counter dispatch
        # ranks in one launch (the torch glue — topk/one-hot/cumsum — was
        # ~5.6 ms/step at batch 256; profiles/r01)
        w32, dst32, _, counts = ops.moe_route(router_logits, K, C)
        weights = w32.to(x.dtype)                             # [T, K]
        dst = dst32.to(torch.int64)
        flat_token = (
            torch.arange(T, device=x.device)
            .unsqueeze(1)
            .expand(T, K)
            .reshape(-1)
        )
        import os

        if (x.is_cuda and ops.HAS_KERNELS
                and os.environ.get("IGW_MOE_GROUPED", "0") == "1"):
            # grouped MFMA GEMMs compute only the real rows per expert.
            # Opt-in while the kernel is below hipBLASLt's batched GEMM on
            # these shapes (tools/bench_moe_gemm.py tracks the gap; the
            # small-N down projection needs split-K to fill the chip)
            buf = x.new_empty((E * C, h))
            buf.index_copy_(0, dst, x[flat_token])
            gu = ops.moe_grouped_gemm(buf, self.gate_up, counts, C)
            act = ops.silu_mul_grouped(gu, counts, C)
            y = ops.moe_grouped_gemm(act, self.down, counts, C)
        elif x.is_cuda and ops.HAS_KERNELS:
            # padded batched GEMM (hipBLASLt), with the fused routing
            # kernel and padding-skipping SwiGLU
            buf = x.new_zeros((E * C, h))
            buf.index_copy_(0, dst, x[flat_token])
            gu = torch.bmm(buf.view(E, C, h), self.gate_up_t)  # [E, C, 2s]
            act = ops.silu_mul_grouped(gu.view(E * C, -1), counts, C)
            y = torch.bmm(act.view(E, C, -1),
                          self.down_t).view(E * C, h)
        else:
            buf = x.new_zeros((E * C, h))
            buf.index_copy_(0, dst, x[flat_token])
            gu = torch.bmm(buf.view(E, C, h), self.gate_up_t)  # [E, C, 2s]
            act = ops.silu_mul(gu.view(E * C, -1))
            y = torch.bmm(act.view(E, C, -1),
                          self.down_t).view(E * C, h)
        ys = y.index_select(0, dst)                           # [T*K, h]
        out = torch.zeros_like(x)
        out.index_add_(0, flat_token, ys * weights.reshape(-1, 1))
        return out

    def _forward_sorted(self, x: torch.Tensor) -> torch.Tensor:
        """Per-expert contiguous GEMMs for prefill-size batches, where the
        padded path's E/K-fold FLOP overhead would be compute-bound."""
        T, h = x.shape
        router_logits = self.router(x).float()            # [T, E]
        weights, topk_idx = torch.topk(router_logits, self.top_k, dim=-1)
        weights = torch.softmax(weights, dim=-1).to(x.dtype)  # [T, K]
        out = torch.zeros_like(x)
        flat_expert = topk_idx.reshape(-1)                # [T*K]
        flat_token = (
            torch.arange(T, device=x.device)
            .unsqueeze(1)
            .expand(T, self.top_k)
            .reshape(-1)
        )
        # sort tokens by expert so each expert runs one contiguous GEMM
        order = torch.argsort(flat_expert, stable=True)
        sorted_expert = flat_expert[order]
        sorted_token = flat_token[order]
        counts = torch.bincount(sorted_expert, minlength=self.num_experts)
        counts_list = counts.tolist()
        start = 0
        for e, cnt in enumerate(counts_list):
            if cnt == 0:
                continue
            toks = sorted_token[start:start + cnt]
            xe = x.index_select(0, toks)
            gu = F.linear(xe, self.gate_up[e])
            ye = F.linear(ops.silu_mul(gu), self.down[e])
            w = weights.reshape(-1)[order[start:start + cnt]].unsqueeze(1)
            out.index_add_(0, toks, ye * w)
            start += cnt
        return out


class MixtralLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, layer: int, device, dtype):
        super().__init__()
        self.input_norm_w = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=dtype, device=device),
            requires_grad=False)
        self.post_norm_w = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=dtype, device=device),
            requires_grad=False)
        self.attn = LlamaAttention(cfg, layer, device, dtype)
        self.moe = MixtralMoE(cfg, layer, device, dtype)
        self.eps = cfg.norm_eps

    def forward(self, x, residual, fb, kv_cache, cos_sin):
        if residual is None:
            residual = x.clone()
            x = ops.rmsnorm(x, self.input_norm_w, self.eps)
        else:
            x, residual = ops.fused_add_rmsnorm(
                x, residual, self.input_norm_w, self.eps)
        x = self.attn(x, fb, kv_cache, cos_sin)
        x, residual = ops.fused_add_rmsnorm(x, residual, self.post_norm_w,
                                            self.eps)
        x = self.moe(x, is_prefill=fb.is_prefill)
        return x, residual


class MixtralForCausalLM(nn.Module):
    def __init__(self, cfg: ModelConfig, device, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Parameter(
            _init_weight((cfg.vocab_size, cfg.hidden_size), device, dtype,
                         _seed("embed")),
            requires_grad=False)
        self.layers = nn.ModuleList(
            MixtralLayer(cfg, i, device, dtype) for i in range(cfg.num_layers))
        self.final_norm_w = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=dtype, device=device),
            requires_grad=False)
        self.lm_head = ReplicatedLinear(cfg.hidden_size, cfg.vocab_size,
                                        device, dtype, _seed("lm_head"))
        self.cos_sin = ops.build_cos_sin_cache(
            cfg.head_dim, cfg.max_position, cfg.rope_base, device=device,
            rope_scaling=cfg.rope_scaling)

    def forward(self, fb: ForwardBatch,
                kv_caches: List[Tuple[torch.Tensor, torch.Tensor]]
                ) -> torch.Tensor:
        x = F.embedding(fb.token_ids, self.embed)
        residual = None
        for i, layer in enumerate(self.layers):
            x, residual = layer(x, residual, fb, kv_caches[i], self.cos_sin)
        x, _ = ops.fused_add_rmsnorm(x, residual, self.final_norm_w,
                                     self.cfg.norm_eps)
        return x

    def compute_logits(self, hidden_last: torch.Tensor) -> torch.Tensor:
        return self.lm_head(hidden_last)
