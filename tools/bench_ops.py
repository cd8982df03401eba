#!/usr/bin/env python3
"""Kernel-level microbenchmarks with roofline context (run on the GPU
box; output is committed to profiles/ as evidence)."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch  # noqa: E402

from inference_gateway_amd import ops  # noqa: E402

PEAK_TBS = 6.3  # measured-achievable HBM BW (MI355X_MICROARCH)


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def row(name, us, bytes_moved, extra=""):
    tbs = bytes_moved / us / 1e6
    print(f"| {name} | {us:8.1f} | {tbs:6.2f} | {100*tbs/PEAK_TBS:5.1f}% "
          f"| {extra} |")


def main():
    torch.manual_seed(0)
    dev = "cuda"
    print("| kernel | us | TB/s | % of 6.3 | notes |")
    print("|---|---|---|---|---|")

    # paged decode attention: B=512, ctx 512, GQA 32/8, hd 128
    B, ctx, nkv, G, hd, bs = 512, 512, 8, 4, 128, 16
    nb = ctx // bs
    kc = torch.randn(B * nb + 1, nkv, bs, hd, dtype=torch.bfloat16,
                     device=dev)
    vc = torch.randn_like(kc)
    q = torch.randn(B, nkv * G, hd, dtype=torch.bfloat16, device=dev)
    bt = torch.arange(1, B * nb + 1, dtype=torch.int32,
                      device=dev).view(B, nb)
    ctxl = torch.full((B,), ctx, dtype=torch.int32, device=dev)
    us = timeit(lambda: ops.paged_attn_decode(q, kc, vc, bt, ctxl,
                                              hd ** -0.5))
    row("paged_attn_decode B512 ctx512", us,
        B * ctx * nkv * hd * 2 * 2, "KV stream")

    # sampler: B=512, V=128256 bf16
    V = 128256
    logits = (torch.randn(B, V, device=dev) * 3).to(torch.bfloat16)
    temp = torch.full((B,), 0.8, device=dev)
    tk = torch.full((B,), 50, dtype=torch.int32, device=dev)
    tp = torch.full((B,), 0.95, device=dev)
    u = torch.rand(B, device=dev).clamp(0.01, 0.99)
    us = timeit(lambda: ops.sample(logits, temp, tk, tp, u))
    row("sample topk50+topp B512 V128k", us, 4 * B * V * 2, "4 passes")

    # fused_add_rmsnorm: [512, 4096]
    x = torch.randn(512, 4096, dtype=torch.bfloat16, device=dev)
    res = torch.randn_like(x)
    w = torch.randn(4096, dtype=torch.bfloat16, device=dev)
    us = timeit(lambda: ops.fused_add_rmsnorm(x, res, w, 1e-5))
    row("fused_add_rmsnorm 512x4096", us, 4 * 512 * 4096 * 2,
        "r/w x + residual")

    # silu_mul: [512, 28672]
    gu = torch.randn(512, 28672, dtype=torch.bfloat16, device=dev)
    us = timeit(lambda: ops.silu_mul(gu))
    row("silu_mul 512x2x14336", us, 512 * (28672 + 14336) * 2, "")

    # rope + append (strided views)
    qkv = torch.randn(512, 6144, dtype=torch.bfloat16, device=dev)
    qv, kv_, vv = qkv.split([4096, 1024, 1024], dim=-1)
    pos = torch.arange(512, dtype=torch.int32, device=dev)
    cs = ops.build_cos_sin_cache(128, 8192, 500000.0, device=dev)
    us = timeit(lambda: ops.rotary_embedding(pos, qv, kv_, 128, cs))
    row("rope q4096+k1024 T512", us, 2 * 512 * 5120 * 2, "in-place")

    print()
    print("llama-3-8b decode GEMMs for reference (hipBLASLt, weights-only"
          " bytes):")
    for name, N, K in (("qkv", 6144, 4096), ("o", 4096, 4096),
                       ("gate_up", 28672, 4096), ("down", 4096, 14336)):
        wt = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.02
        xx = torch.randn(512, K, dtype=torch.bfloat16, device=dev)
        us = timeit(lambda: torch.nn.functional.linear(xx, wt))
        row(f"F.linear {name} [512x{K}]x[{N}]", us, N * K * 2, "")


if __name__ == "__main__":
    main()
