#!/usr/bin/env python3
"""Microbenchmark: grouped MoE GEMM kernel vs torch padded bmm on the
Mixtral decode shapes. Run on the GPU box."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch  # noqa: E402

from inference_gateway_amd import ops  # noqa: E402


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    torch.manual_seed(0)
    E, K = 8, 4096
    for T, cnt in ((256, 64), (512, 128)):
        C = T
        for name, N in (("gate_up", 28672), ("down", 4096)):
            Kk = K if name == "gate_up" else 14336
            a = torch.randn(E * C, Kk, dtype=torch.bfloat16,
                            device="cuda")
            w = torch.randn(E, N, Kk, dtype=torch.bfloat16,
                            device="cuda") * 0.02
            counts = torch.full((E,), cnt, dtype=torch.int32, device="cuda")
            us_k = timeit(lambda: ops.moe_grouped_gemm(a, w, counts, C))
            us_bmm = timeit(lambda: torch.bmm(
                a.view(E, C, Kk), w.transpose(1, 2)))
            wbytes = E * N * Kk * 2
            useful = E * cnt * Kk * 2 + E * cnt * N * 2
            print(f"T={T} {name}: kernel {us_k:8.1f} us "
                  f"({(wbytes + useful) / us_k / 1e3:6.2f} TB/s algorithmic) "
                  f"| bmm {us_bmm:8.1f} us "
                  f"({(wbytes + E * C * (Kk + N) * 2) / us_bmm / 1e3:6.2f})")


if __name__ == "__main__":
    main()
