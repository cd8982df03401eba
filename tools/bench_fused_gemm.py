#!/usr/bin/env python3
"""A/B the fused TN decode GEMM (ops/csrc/fused_gemm.hip) against the
library path (F.linear = hipBLASLt/Tensile, + the separate silu_mul
launch for gate_up) on the flagship decode shapes.

Within-probe interleaved rounds (guide §5.4 rule 24): N_ROUNDS of
{lib, fused, fused_nt} per shape in one process, median reported.
Weight-stream TB/s = weight bytes / time (the decode GEMMs are
weight-bound; activations are noise at these shapes).
"""
from __future__ import annotations

import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402

from inference_gateway_amd import ops  # noqa: E402

SHAPES = [
    # (name, M, N_rows, K, swiglu)
    ("gate_up+silu", 1024, 28672, 4096, True),
    ("down", 1024, 4096, 14336, False),
    ("qkv", 1024, 6144, 4096, False),
    ("o_proj", 1024, 4096, 4096, False),
    ("gate_up+silu@256", 256, 28672, 4096, True),
    ("down@256", 256, 4096, 14336, False),
]
ROUNDS = 30
INNER = 4


def timed(fn):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(INNER):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / INNER


def main():
    assert torch.cuda.is_available()
    results = []
    for name, M, NR, K, swiglu in SHAPES:
        x = (torch.randn(M, K, device="cuda") * 0.3).to(torch.bfloat16)
        w = (torch.randn(NR, K, device="cuda") * 0.05).to(torch.bfloat16)
        wbytes = NR * K * 2

        def lib():
            y = F.linear(x, w)
            if swiglu:
                y = ops.silu_mul(y)
            return y

        def fused():
            return ops.fused_gemm_tn(x, w, swiglu=swiglu)

        def fused_nt():
            return ops.fused_gemm_tn(x, w, swiglu=swiglu,
                                     nt_weights=True)

        from inference_gateway_amd.ops import _kernels

        def fused2():
            N = NR // 2 if swiglu else NR
            out = torch.empty((M, N), dtype=x.dtype, device=x.device)
            _kernels.fused_gemm_tn2(out, x, w, swiglu, True)
            return out

        # correctness gate before timing
        got = fused().float()
        ref = lib().float()
        rel = (got - ref).abs().max() / ref.abs().max().clamp_min(1.0)
        assert rel < 5e-2, (name, float(rel))
        has_v2 = M % 256 == 0
        if has_v2:
            got2 = fused2().float()
            rel2 = (got2 - ref).abs().max() / ref.abs().max().clamp_min(1.0)
            assert rel2 < 5e-2, ("v2", name, float(rel2))

        variants = [lib, fused, fused_nt] + ([fused2] if has_v2 else [])
        for f in variants:
            for _ in range(3):
                f()
        t = {"lib": [], "fused": [], "fused_nt": [], "fused2": []}
        for _ in range(ROUNDS):
            t["lib"].append(timed(lib))
            t["fused"].append(timed(fused))
            t["fused_nt"].append(timed(fused_nt))
            if has_v2:
                t["fused2"].append(timed(fused2))
        if not has_v2:
            t.pop("fused2")
        med = {k: sorted(v)[len(v) // 2] for k, v in t.items()}
        row = {
            "shape": name, "M": M, "N_rows": NR, "K": K,
            "lib_us": round(med["lib"] * 1e6, 1),
            "fused_us": round(med["fused"] * 1e6, 1),
            "fused_nt_us": round(med["fused_nt"] * 1e6, 1),
            "lib_tbps": round(wbytes / med["lib"] / 1e12, 2),
            "fused_tbps": round(wbytes / med["fused"] / 1e12, 2),
            "fused_nt_tbps": round(wbytes / med["fused_nt"] / 1e12, 2),
            "fused2_us": (round(med["fused2"] * 1e6, 1)
                          if "fused2" in med else None),
            "fused2_tbps": (round(wbytes / med["fused2"] / 1e12, 2)
                            if "fused2" in med else None),
            "speedup": round(med["lib"] / min(
                [med["fused"], med["fused_nt"]]
                + ([med["fused2"]] if "fused2" in med else [])), 3),
        }
        results.append(row)
        print(json.dumps(row), flush=True)
    wins = [r for r in results if r["speedup"] > 1.02]
    print(json.dumps({"summary": "fused wins on",
                      "shapes": [r["shape"] for r in wins]}))


if __name__ == "__main__":
    main()
