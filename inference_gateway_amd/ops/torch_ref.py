"""Plain-PyTorch fp32 reference implementations of every HIP kernel.

These are the *numerics oracle* for the gfx950 kernels (the
rebuild adds a GPU-kernel test tier comparing hand-written kernels against
a PyTorch fp32 reference) and the CPU execution path for the engine so the
full request flow is testable without a GPU. On a GPU box the engine uses
the _kernels extension and refuses to fall back silently.
"""
from __future__ import annotations

import torch


def rmsnorm_ref(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv * weight.float()).to(x.dtype)


def fused_add_rmsnorm_ref(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
) -> tuple[torch.Tensor, torch.Tensor]:
    """Returns (normed, new_residual); matches the in-place HIP op."""
    new_residual = (x.float() + residual.float()).to(x.dtype)
    return rmsnorm_ref(new_residual, weight, eps), new_residual


def rope_ref(
    positions: torch.Tensor,
    q: torch.Tensor,
    k: torch.Tensor,
    head_dim: int,
    cos_sin_cache: torch.Tensor,
) -> tuple[torch.Tensor, torch.Tensor]:
    """NeoX-style rotate-half RoPE. q: [T, NQ*HD], k: [T, NKV*HD]."""
    half = head_dim // 2
    cs = cos_sin_cache[positions.long()].float()  # [T, HD]
    cos, sin = cs[:, :half], cs[:, half:]

    def rot(x: torch.Tensor) -> torch.Tensor:
        T = x.shape[0]
        xs = x.float().view(T, -1, head_dim)
        x1, x2 = xs[..., :half], xs[..., half:]
        c = cos.unsqueeze(1)
        s = sin.unsqueeze(1)
        out = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
        return out.view(x.shape).to(x.dtype)

    return rot(q), rot(k)


def build_cos_sin_cache(
    head_dim: int, max_pos: int, base: float = 10000.0, device="cpu",
    rope_scaling: dict = None,
) -> torch.Tensor:
    """[max_pos, HD] fp32 — cos in [0, HD/2), sin in [HD/2, HD).

    rope_scaling: HF `rope_scaling` dict.  Supported: the Llama-3.1
    "llama3" scheme (wavelength-banded frequency rescale: short
    wavelengths untouched, long wavelengths divided by `factor`, a
    smooth ramp between — the published long-context extension) and
    "linear" (all frequencies divided by `factor`).  None/unknown-less
    configs keep plain RoPE; an unsupported rope_type raises (silent
    wrong positions would corrupt every long-context generation)."""
    half = head_dim // 2
    inv_freq = 1.0 / (
        base ** (torch.arange(0, half, dtype=torch.float64, device=device) / half)
    )
    if rope_scaling:
        rtype = (rope_scaling.get("rope_type")
                 or rope_scaling.get("type") or "llama3")
        factor = float(rope_scaling.get("factor", 8.0))
        if rtype == "llama3":
            lo = float(rope_scaling.get("low_freq_factor", 1.0))
            hi = float(rope_scaling.get("high_freq_factor", 4.0))
            orig = float(rope_scaling.get(
                "original_max_position_embeddings", 8192))
            import math

            wavelen = 2 * math.pi / inv_freq
            low_wl = orig / lo            # longest untouched band edge
            high_wl = orig / hi           # shortest scaled band edge
            smooth = (orig / wavelen - lo) / (hi - lo)
            scaled = torch.where(
                wavelen > low_wl, inv_freq / factor, inv_freq)
            ramp = ((1 - smooth) * inv_freq / factor
                    + smooth * inv_freq)
            inv_freq = torch.where(
                (wavelen <= low_wl) & (wavelen >= high_wl), ramp, scaled)
        elif rtype == "linear":
            inv_freq = inv_freq / factor
        else:
            raise ValueError(
                f"unsupported rope_scaling type {rtype!r} (supported: "
                "llama3, linear)")
    t = torch.arange(max_pos, dtype=torch.float64, device=device)
    freqs = torch.outer(t, inv_freq)
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).float()


def append_kv_cache_ref(k, v, kcache, vcache, slot_mapping):
    """kcache/vcache: [NB, NKV, BS, HD]; k/v: [T, NKV, HD]."""
    bs = kcache.shape[2]
    for t in range(k.shape[0]):
        slot = int(slot_mapping[t])
        if slot < 0:
            continue
        blk, off = slot // bs, slot % bs
        kcache[blk, :, off, :] = k[t]
        vcache[blk, :, off, :] = v[t]


def paged_attn_decode_ref(q, kcache, vcache, block_table, ctx_lens, scale):
    """q: [B, NQ, HD] -> out [B, NQ, HD]; fp32 math."""
    B, NQ, HD = q.shape
    NKV, BS = kcache.shape[1], kcache.shape[2]
    G = NQ // NKV
    out = torch.empty_like(q)
    for b in range(B):
        ctx = int(ctx_lens[b])
        nblk = (ctx + BS - 1) // BS
        blocks = block_table[b, :nblk].long()
        # gather [ctx, NKV, HD]
        keys = kcache[blocks].permute(0, 2, 1, 3).reshape(-1, NKV, HD)[:ctx]
        vals = vcache[blocks].permute(0, 2, 1, 3).reshape(-1, NKV, HD)[:ctx]
        for h in range(NQ):
            kvh = h // G
            s = (keys[:, kvh].float() @ q[b, h].float()) * scale
            p = torch.softmax(s, dim=-1)
            out[b, h] = (p @ vals[:, kvh].float()).to(q.dtype)
    return out


def flash_prefill_ref(q, k, v, cu_seqlens, scale):
    """Varlen causal attention. q: [T, NQ, HD], k/v: [T, NKV, HD]."""
    T, NQ, HD = q.shape
    NKV = k.shape[1]
    G = NQ // NKV
    out = torch.empty_like(q)
    for s in range(cu_seqlens.shape[0] - 1):
        a, b = int(cu_seqlens[s]), int(cu_seqlens[s + 1])
        L = b - a
        mask = torch.tril(torch.ones(L, L, dtype=torch.bool, device=q.device))
        for h in range(NQ):
            kvh = h // G
            sc = (q[a:b, h].float() @ k[a:b, kvh].float().T) * scale
            sc = sc.masked_fill(~mask, float("-inf"))
            p = torch.softmax(sc, dim=-1)
            out[a:b, h] = (p @ v[a:b, kvh].float()).to(q.dtype)
    return out


def flash_prefill_paged_ref(q, kcache, vcache, block_table, prefix_lens,
                            cu_seqlens, scale):
    """Reference for the paged-extend prefill: gather each sequence's
    full K/V (prefix + chunk) from the cache via its block table, then
    run offset-causal attention for the chunk's q rows."""
    Tq, NQ, HD = q.shape
    NKV = kcache.shape[1]
    BS = kcache.shape[2]
    G = NQ // NKV
    out = torch.empty_like(q)
    for s in range(cu_seqlens.shape[0] - 1):
        a, b = int(cu_seqlens[s]), int(cu_seqlens[s + 1])
        Lq = b - a
        P = int(prefix_lens[s])
        KVL = P + Lq
        nblk = (KVL + BS - 1) // BS
        blocks = block_table[s, :nblk].long()
        # [nblk*BS, NKV, HD] gathered then trimmed
        keys = kcache[blocks].permute(0, 2, 1, 3).reshape(
            nblk * BS, NKV, HD)[:KVL]
        vals = vcache[blocks].permute(0, 2, 1, 3).reshape(
            nblk * BS, NKV, HD)[:KVL]
        qpos = torch.arange(P, KVL, device=q.device)
        kpos = torch.arange(KVL, device=q.device)
        mask = kpos.unsqueeze(0) <= qpos.unsqueeze(1)     # [Lq, KVL]
        for h in range(NQ):
            kvh = h // G
            sc = (q[a:b, h].float() @ keys[:, kvh].float().T) * scale
            sc = sc.masked_fill(~mask, float("-inf"))
            p = torch.softmax(sc, dim=-1)
            out[a:b, h] = (p @ vals[:, kvh].float()).to(q.dtype)
    return out


def make_prefill_tiles(seq_lens, tile_m: int = 64, device="cpu"):
    """Host-side tile map for flash_prefill: (tile_seq, tile_off) int32."""
    tseq, toff = [], []
    for i, L in enumerate(seq_lens):
        for off in range(0, int(L), tile_m):
            tseq.append(i)
            toff.append(off)
    return (
        torch.tensor(tseq, dtype=torch.int32, device=device),
        torch.tensor(toff, dtype=torch.int32, device=device),
    )


def silu_mul_ref(gate_up: torch.Tensor) -> torch.Tensor:
    inter = gate_up.shape[-1] // 2
    g = gate_up[..., :inter].float()
    u = gate_up[..., inter:].float()
    return (torch.nn.functional.silu(g) * u).to(gate_up.dtype)


def _f32_key(x: torch.Tensor) -> torch.Tensor:
    u = x.float().view(torch.int32).to(torch.int64) & 0xFFFFFFFF
    neg = (u & 0x80000000) != 0
    return torch.where(neg, u ^ 0xFFFFFFFF, u | 0x80000000)


def sample_ref(logits, temperature, top_k, top_p, uniforms):
    """Reference with the kernel's exact semantics (see sampling.hip):

    value-threshold top-k (keeps ties), value-threshold nucleus (smallest
    top set with mass >= p), inverse CDF over kept tokens in index order.
    """
    B, V = logits.shape
    out = torch.empty(B, dtype=torch.int32)
    for i in range(B):
        row = logits[i].float()
        temp = float(temperature[i])
        if temp <= 0.0:
            out[i] = int(torch.argmax(row))
            continue
        # thresholds live in the UNSCALED value domain (temperature > 0
        # preserves order; the bf16 kernel walks unscaled bf16 keys), the
        # exp-weights use the temperature
        inv_t = 1.0 / temp
        m = float(row.max())
        probs = torch.exp((row - m) * inv_t)
        Z = float(probs.sum())
        sorted_vals, _ = torch.sort(row, descending=True)
        k = int(top_k[i])
        tau_k = float("-inf")
        if 0 < k < V:
            tau_k = float(sorted_vals[k - 1])
        p = float(top_p[i])
        tau_p = float("-inf")
        if p < 1.0:
            sp = torch.exp((sorted_vals - m) * inv_t)
            cum = torch.cumsum(sp, 0)
            idx = int(torch.searchsorted(cum, p * Z))
            idx = min(idx, V - 1)
            tau_p = float(sorted_vals[idx])
        tau = max(tau_k, tau_p)
        keep = row >= tau
        masked = torch.where(keep, probs, torch.zeros(()))
        zf = float(masked.sum())
        target = float(uniforms[i]) * zf
        cdf = torch.cumsum(masked, 0)
        chosen = int(torch.searchsorted(cdf, target, right=True))
        chosen = min(chosen, V - 1)
        while masked[chosen] == 0 and chosen > 0:
            chosen -= 1
        out[i] = chosen
    return out


def moe_route_ref(logits: torch.Tensor, top_k: int, capacity: int):
    """Reference for the fused MoE routing kernel (moe_route.hip).

    Returns (weights [T,K] fp32, dst [T*K] int32, topk_idx [T,K] int32).
    dst = expert * capacity + rank; rank order within an expert is NOT
    part of the contract (the kernel uses atomic counters) — tests check
    validity, not order.
    """
    T, E = logits.shape
    vals, idx = torch.topk(logits.float(), top_k, dim=-1)
    weights = torch.softmax(vals, dim=-1)
    flat = idx.reshape(-1)
    onehot = torch.nn.functional.one_hot(flat, E)
    rank = (torch.cumsum(onehot, 0) - onehot).mul(onehot).sum(-1)
    dst = (flat * capacity + rank).to(torch.int32)
    counts = torch.bincount(flat, minlength=E).to(torch.int32)
    return weights, dst, idx.to(torch.int32), counts


def moe_grouped_gemm_ref(a, w, counts, capacity):
    """Per-expert GEMM reference: only rows < counts[e] are computed;
    padded rows of the output stay zero."""
    E, N, K = w.shape
    out = torch.zeros((a.shape[0], N), dtype=a.dtype)
    for e in range(E):
        c = int(counts[e])
        if c == 0:
            continue
        rows = a[e * capacity:e * capacity + c].float()
        out[e * capacity:e * capacity + c] = (
            rows @ w[e].float().T).to(a.dtype)
    return out


def silu_mul_grouped_ref(gate_up, counts, capacity):
    inter = gate_up.shape[-1] // 2
    out = torch.zeros((*gate_up.shape[:-1], inter), dtype=gate_up.dtype)
    E = counts.shape[0]
    for e in range(E):
        c = int(counts[e])
        if c:
            seg = gate_up[e * capacity:e * capacity + c]
            out[e * capacity:e * capacity + c] = silu_mul_ref(seg)
    return out
