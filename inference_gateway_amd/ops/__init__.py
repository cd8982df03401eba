"""Op dispatch for the MI355X engine.

On a GPU box the hand-written gfx950 HIP kernels (_kernels extension) are
REQUIRED — if a CUDA/HIP device is visible and the extension is missing,
every op raises instead of silently falling back to eager PyTorch (the
round-end harness records which .so files the GPU processes load). On CPU
the fp32 torch references (torch_ref) run so the whole engine and gateway
are testable without a GPU.
"""
from __future__ import annotations

import torch

from . import torch_ref

try:
    from . import _kernels  # compiled by setup.py build_ext --inplace

    HAS_KERNELS = True
except ImportError:  # pragma: no cover - absent only before first build
    _kernels = None
    HAS_KERNELS = False


def _require_kernels() -> None:
    if not HAS_KERNELS:
        raise RuntimeError(
            "inference_gateway_amd.ops._kernels is not built but a GPU tensor "
            "was passed. Build it in-tree with: "
            "PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace"
        )


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if x.is_cuda:
        _require_kernels()
        out = torch.empty_like(x)
        _kernels.rmsnorm(out, x, weight, eps)
        return out
    return torch_ref.rmsnorm_ref(x, weight, eps)


def fused_add_rmsnorm(x, residual, weight, eps: float):
    """In-place: residual += x; x = rmsnorm(residual) * w. Returns (x, residual)."""
    if x.is_cuda:
        _require_kernels()
        _kernels.fused_add_rmsnorm(x, residual, weight, eps)
        return x, residual
    nx, nres = torch_ref.fused_add_rmsnorm_ref(x, residual, weight, eps)
    x.copy_(nx)
    residual.copy_(nres)
    return x, residual


def rotary_embedding(positions, q, k, head_dim: int, cos_sin_cache):
    """In-place NeoX RoPE on q [T, NQ*HD] and k [T, NKV*HD]."""
    if q.is_cuda:
        _require_kernels()
        _kernels.rotary_embedding(positions, q, k, head_dim, cos_sin_cache)
        return q, k
    nq, nk = torch_ref.rope_ref(positions, q, k, head_dim, cos_sin_cache)
    q.copy_(nq)
    k.copy_(nk)
    return q, k


def append_kv_cache(k, v, kcache, vcache, slot_mapping):
    if k.is_cuda:
        _require_kernels()
        _kernels.append_kv_cache(k, v, kcache, vcache, slot_mapping)
        return
    torch_ref.append_kv_cache_ref(k, v, kcache, vcache, slot_mapping)


def paged_attn_decode(q, kcache, vcache, block_table, ctx_lens, scale: float):
    if q.is_cuda:
        _require_kernels()
        out = torch.empty_like(q)
        _kernels.paged_attn_decode(
            out, q, kcache, vcache, block_table, ctx_lens, scale
        )
        return out
    return torch_ref.paged_attn_decode_ref(
        q, kcache, vcache, block_table, ctx_lens, scale
    )


def flash_prefill(q, k, v, cu_seqlens, seq_lens, scale: float):
    """Varlen causal prefill attention. seq_lens: python list of ints."""
    if q.is_cuda:
        _require_kernels()
        out = torch.empty_like(q)
        tile_seq, tile_off = torch_ref.make_prefill_tiles(
            seq_lens, tile_m=64, device=q.device
        )
        _kernels.flash_prefill(
            out, q, k, v, cu_seqlens, tile_seq, tile_off, scale
        )
        return out
    return torch_ref.flash_prefill_ref(q, k, v, cu_seqlens, scale)


def flash_prefill_paged(q, kcache, vcache, block_table, prefix_lens,
                        cu_seqlens, seq_lens, scale: float):
    """Paged-EXTEND prefill (chunked-prefill / prefix-cache enabler):
    q rows are a per-seq CHUNK whose own K/V plus `prefix_lens` cached
    tokens before it are all in the paged cache; each q row attends over
    cache positions [0, prefix + row].  seq_lens: python ints (q rows
    per sequence)."""
    if q.is_cuda:
        _require_kernels()
        out = torch.empty_like(q)
        tile_seq, tile_off = torch_ref.make_prefill_tiles(
            seq_lens, tile_m=64, device=q.device
        )
        _kernels.flash_prefill_paged(
            out, q, kcache, vcache, block_table, prefix_lens, cu_seqlens,
            tile_seq, tile_off, scale
        )
        return out
    return torch_ref.flash_prefill_paged_ref(
        q, kcache, vcache, block_table, prefix_lens, cu_seqlens, scale
    )


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    if gate_up.is_cuda:
        _require_kernels()
        inter = gate_up.shape[-1] // 2
        out = torch.empty(
            (*gate_up.shape[:-1], inter),
            dtype=gate_up.dtype,
            device=gate_up.device,
        )
        _kernels.silu_mul(out, gate_up)
        return out
    return torch_ref.silu_mul_ref(gate_up)


def sample(logits, temperature, top_k, top_p, uniforms):
    """logits [B, V] fp32 -> sampled token ids [B] int32."""
    if logits.is_cuda:
        _require_kernels()
        out = torch.empty(
            logits.shape[0], dtype=torch.int32, device=logits.device
        )
        _kernels.sample(out, logits, temperature, top_k, top_p, uniforms)
        return out
    return torch_ref.sample_ref(logits, temperature, top_k, top_p, uniforms)


build_cos_sin_cache = torch_ref.build_cos_sin_cache
make_prefill_tiles = torch_ref.make_prefill_tiles


def moe_route(router_logits: torch.Tensor, top_k: int, capacity: int):
    """Fused MoE routing: top-k + softmax + padded-dispatch row indices.

    router_logits: [T, E] fp32. Returns (weights [T,K] fp32,
    dst [T*K] int32, topk_idx [T,K] int32, counts [E] int32)."""
    if router_logits.is_cuda:
        _require_kernels()
        T, E = router_logits.shape
        weights = torch.empty((T, top_k), dtype=torch.float32,
                              device=router_logits.device)
        dst = torch.empty(T * top_k, dtype=torch.int32,
                          device=router_logits.device)
        topk_idx = torch.empty((T, top_k), dtype=torch.int32,
                               device=router_logits.device)
        counts = torch.empty(E, dtype=torch.int32,
                             device=router_logits.device)
        _kernels.moe_route(weights, dst, topk_idx, counts, router_logits,
                           capacity)
        return weights, dst, topk_idx, counts
    return torch_ref.moe_route_ref(router_logits, top_k, capacity)


def moe_grouped_gemm(a: torch.Tensor, w: torch.Tensor,
                     counts: torch.Tensor, capacity: int) -> torch.Tensor:
    """Per-expert GEMM over the padded dispatch buffer.

    a: [E*C, K] bf16 (rows >= counts[e] in each expert segment are
    ignored), w: [E, N, K] bf16 -> out [E*C, N] bf16 (only real rows
    written)."""
    if a.is_cuda:
        _require_kernels()
        E, N, _K = w.shape
        out = torch.empty((a.shape[0], N), dtype=a.dtype, device=a.device)
        _kernels.moe_grouped_gemm(out, a, w, counts, capacity)
        return out
    return torch_ref.moe_grouped_gemm_ref(a, w, counts, capacity)


def silu_mul_grouped(gate_up: torch.Tensor, counts: torch.Tensor,
                     capacity: int) -> torch.Tensor:
    """SwiGLU over only the real rows of the padded dispatch buffer."""
    if gate_up.is_cuda:
        _require_kernels()
        inter = gate_up.shape[-1] // 2
        out = torch.empty((*gate_up.shape[:-1], inter), dtype=gate_up.dtype,
                          device=gate_up.device)
        _kernels.silu_mul_grouped(out, gate_up, counts, capacity)
        return out
    return torch_ref.silu_mul_grouped_ref(gate_up, counts, capacity)


def fused_gemm_tn(x: torch.Tensor, w: torch.Tensor, swiglu: bool = False,
                  nt_weights: bool = False) -> torch.Tensor:
    """TN bf16 GEMM out = x @ w.T with an optional fused SwiGLU
    epilogue (w = [gate; up] rows, out = silu(g)*u).  Decode
    weight-streaming shapes (M%128==0, N%64==0, K%32==0); callers fall
    back to the library GEMM otherwise (ops/csrc/fused_gemm.hip)."""
    if x.is_cuda:
        _require_kernels()
        N = w.shape[0] // 2 if swiglu else w.shape[0]
        out = torch.empty((x.shape[0], N), dtype=x.dtype, device=x.device)
        _kernels.fused_gemm_tn(out, x, w, swiglu, nt_weights)
        return out
    y = x.float() @ w.float().t()
    if swiglu:
        N = w.shape[0] // 2
        g, u = y[:, :N], y[:, N:]
        y = torch.nn.functional.silu(g) * u
    return y.to(x.dtype)


def fused_gemm_tn_supported(M: int, N_rows: int, K: int,
                            swiglu: bool) -> bool:
    N = N_rows // 2 if swiglu else N_rows
    return M % 128 == 0 and N % 64 == 0 and K % 32 == 0
