"""Llama-family decoder (Llama-3 8B / 70B architectures) for the MI355X
engine.

Compute path (SURVEY.md §7 stage 3): GEMMs via F.linear (hipBLASLt on ROCm,
TP-sharded through parallel.layers); everything between them is a
hand-written gfx950 kernel — fused add+RMSNorm, NeoX RoPE, paged-cache
append, MFMA flash prefill / paged decode attention, fused SwiGLU. On CPU
the ops package transparently runs the fp32 references so the same module
is testable without a GPU.
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


def _seed(name: str) -> int:
    # deterministic ACROSS PROCESSES: python's hash() is salted per process,
    # which would give each TP rank different weights (caught by the gloo
    # world_size=2 CPU test)
    import zlib

    return zlib.crc32(name.encode()) & 0x7FFFFFFF


class LlamaAttention(nn.Module):
    def __init__(self, cfg: ModelConfig, layer: int, device, dtype):
        super().__init__()
        tp, rank = tp_world_size(), tp_rank()
        assert cfg.num_heads % tp == 0 and cfg.num_kv_heads % tp == 0, (
            f"TP={tp} must divide heads {cfg.num_heads}/{cfg.num_kv_heads}"
        )
        self.nq = cfg.num_heads // tp
        self.nkv = cfg.num_kv_heads // tp
        self.hd = cfg.head_dim
        self.scale = cfg.head_dim**-0.5
        h = cfg.hidden_size
        # fused qkv: column-parallel over heads (q shard + k shard + v shard)
        q_proj = ColumnParallelLinear(
            h, cfg.num_heads * self.hd, rank, tp, device, dtype,
            _seed(f"l{layer}.q"))
        k_proj = ColumnParallelLinear(
            h, cfg.num_kv_heads * self.hd, rank, tp, device, dtype,
            _seed(f"l{layer}.k"))
        v_proj = ColumnParallelLinear(
            h, cfg.num_kv_heads * self.hd, rank, tp, device, dtype,
            _seed(f"l{layer}.v"))
        # one fused qkv GEMM per layer (decode is launch/skinny-M bound)
        self.qkv = nn.Parameter(
            torch.cat([q_proj.weight.data, k_proj.weight.data,
                       v_proj.weight.data], 0).contiguous(),
            requires_grad=False)
        del q_proj, k_proj, v_proj
        if cfg.qkv_bias:
            # Qwen2 family: bias on qkv only, sharded with the head dims
            # (zeros under synthetic init; real values come from the
            # checkpoint loader)
            self.qkv_b = nn.Parameter(
                torch.zeros((self.nq + 2 * self.nkv) * self.hd,
                            dtype=dtype, device=device),
                requires_grad=False)
        else:
            self.qkv_b = None
        self.o_proj = RowParallelLinear(
            cfg.num_heads * self.hd, h, rank, tp, device, dtype,
            _seed(f"l{layer}.o"))

    def forward(self, x: torch.Tensor, fb: ForwardBatch,
                kv_cache: Tuple[torch.Tensor, torch.Tensor],
                cos_sin: torch.Tensor) -> torch.Tensor:
        T = x.shape[0]
        qkv = F.linear(x, self.qkv, self.qkv_b)
        # strided views into the fused QKV output — the HIP kernels take a
        # row stride, so there are no .contiguous() copies on the hot path
        q, k, v = qkv.split(
            [self.nq * self.hd, self.nkv * self.hd, self.nkv * self.hd], dim=-1)
        ops.rotary_embedding(fb.positions, q, k, self.hd, cos_sin)
        kc, vc = kv_cache
        k3 = k.view(T, self.nkv, self.hd)
        v3 = v.view(T, self.nkv, self.hd)
        ops.append_kv_cache(k3, v3, kc, vc, fb.slot_mapping)
        if fb.is_prefill:
            P = fb.num_prefill_tokens if fb.is_mixed else T
            qv = q.view(T, self.nq, self.hd)
            if fb.prefix_lens is not None:
                # prefix-cache hit in the batch: q rows are suffixes;
                # cached prefix + the just-appended suffix K/V are read
                # back from the paged cache (ops/csrc/flash_prefill.hip,
                # flash_prefill_paged_kernel)
                o = ops.flash_prefill_paged(
                    qv[:P], kc, vc, fb.block_table,
                    fb.prefix_lens, fb.cu_seqlens, fb.seq_lens, self.scale)
            else:
                o = ops.flash_prefill(
                    qv[:P], k3[:P], v3[:P], fb.cu_seqlens,
                    fb.seq_lens, self.scale)
            if fb.is_mixed:
                # mixed round: rows [P, T) are one decode token per
                # running sequence (engine._build_forward_batch)
                od = ops.paged_attn_decode(
                    qv[P:].contiguous(), kc, vc, fb.decode_block_table,
                    fb.decode_ctx_lens, self.scale)
                o = torch.cat([o.view(P, self.nq, self.hd),
                               od.view(T - P, self.nq, self.hd)], 0)
        else:
            o = ops.paged_attn_decode(
                q.view(T, self.nq, self.hd), kc, vc, fb.block_table,
                fb.ctx_lens, self.scale)
        return self.o_proj(o.view(T, self.nq * self.hd))


class LlamaMLP(nn.Module):
    def __init__(self, cfg: ModelConfig, layer: int, device, dtype):
        super().__init__()
        tp, rank = tp_world_size(), tp_rank()
        h, inter = cfg.hidden_size, cfg.intermediate_size
        # gate and up fused in one column-parallel GEMM; the HIP silu_mul
        # kernel expects [gate | up] halves per shard
        self.gate_proj = ColumnParallelLinear(
            h, inter, rank, tp, device, dtype, _seed(f"l{layer}.gate"))
        self.up_proj = ColumnParallelLinear(
            h, inter, rank, tp, device, dtype, _seed(f"l{layer}.up"))
        self.down_proj = RowParallelLinear(
            inter, h, rank, tp, device, dtype, _seed(f"l{layer}.down"))
        # fuse gate+up into a single weight for one GEMM
        w = torch.cat([self.gate_proj.weight.data, self.up_proj.weight.data], 0)
        self.gate_up = nn.Parameter(w.contiguous(), requires_grad=False)
        del self.gate_proj, self.up_proj

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        gu = F.linear(x, self.gate_up)
        return self.down_proj(ops.silu_mul(gu))


class LlamaLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, layer: int, device, dtype):
        super().__init__()
        self.input_norm_w = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=dtype, device=device),
            requires_grad=False)
        self.post_norm_w = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=dtype, device=device),
            requires_grad=False)
        self.attn = LlamaAttention(cfg, layer, device, dtype)
        self.mlp = LlamaMLP(cfg, layer, device, dtype)
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
        x = self.mlp(x)
        return x, residual


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: ModelConfig, device, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        self.device = device
        self.embed = nn.Parameter(
            _init_weight((cfg.vocab_size, cfg.hidden_size), device, dtype,
                         _seed("embed")),
            requires_grad=False)
        self.layers = nn.ModuleList(
            LlamaLayer(cfg, i, device, dtype) for i in range(cfg.num_layers))
        self.final_norm_w = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=dtype, device=device),
            requires_grad=False)
        if cfg.tie_embeddings:
            self.lm_head = None
        else:
            self.lm_head = ReplicatedLinear(
                cfg.hidden_size, cfg.vocab_size, device, dtype,
                _seed("lm_head"))
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
        """hidden_last: [B, hidden] -> bf16 logits [B, vocab] (the
        sampler's bf16 radix kernel consumes them directly)."""
        if self.lm_head is None:
            return F.linear(hidden_last, self.embed)
        return self.lm_head(hidden_last)
