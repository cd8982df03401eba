#!/usr/bin/env python3
"""Isolate the sampler kernel's phase costs on the GPU box."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch  # noqa: E402

from inference_gateway_amd import ops  # noqa: E402


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    torch.manual_seed(0)
    B, V = 512, 128256
    logits = (torch.randn(B, V, device="cuda") * 3).to(torch.bfloat16)
    u = torch.rand(B, device="cuda").clamp(0.01, 0.99)
    ones = torch.ones(B, device="cuda")
    for name, temp, tk, tp in (
        ("greedy (2 passes: max+argmax)", torch.zeros(B, device="cuda"),
         torch.zeros(B, dtype=torch.int32, device="cuda"), ones),
        ("pure-temp (2 passes: max+cdf)", ones * 0.8,
         torch.zeros(B, dtype=torch.int32, device="cuda"), ones),
        ("top-k only", ones * 0.8,
         torch.full((B,), 50, dtype=torch.int32, device="cuda"), ones),
        ("top-p only", ones * 0.8,
         torch.zeros(B, dtype=torch.int32, device="cuda"), ones * 0.95),
        ("top-k + top-p (bench config)", ones * 0.8,
         torch.full((B,), 50, dtype=torch.int32, device="cuda"),
         ones * 0.95),
    ):
        us = timeit(lambda: ops.sample(logits, temp, tk, tp, u))
        print(f"{name:34s} {us:8.1f} us")


if __name__ == "__main__":
    main()
