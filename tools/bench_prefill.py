#!/usr/bin/env python3
"""Prefill-rate benchmark: prompt tokens/sec through the engine's
prefill path (the TTFT half of the BASELINE metric — a 1024-stream burst
at prompt 512 is ~533k prompt tokens through this path before the last
stream sees its first token).

Usage: python tools/bench_prefill.py [--model llama-3-8b] [--prompts 64]
       [--prompt-len 512] [--chunked] [--repeats 3]
Prints one JSON line with prompt_tokens_per_sec.
"""
from __future__ import annotations

import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--prompts", type=int, default=64)
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--chunked", action="store_true")
    ap.add_argument("--repeats", type=int, default=3)
    ap.add_argument("--max-batched-tokens", type=int, default=8192)
    args = ap.parse_args()

    from inference_gateway_amd.engine.config import (EngineConfig,
                                                     SamplingParams)
    from inference_gateway_amd.engine.engine import LLMEngine

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    cfg = EngineConfig.for_model(
        args.model,
        max_num_seqs=args.prompts,
        max_batched_tokens=args.max_batched_tokens,
        max_model_len=args.prompt_len + 64,
        enable_chunked_prefill=args.chunked,
        use_hip_graphs=False,        # prefill path only; no decode graphs
    )
    eng = LLMEngine(cfg, device=device)
    vocab = cfg.model.vocab_size
    gen = torch.Generator().manual_seed(7)

    def run_once():
        rids = []
        for _ in range(args.prompts):
            p = torch.randint(0, vocab, (args.prompt_len,),
                              generator=gen).tolist()
            rids.append(eng.add_request(
                p, SamplingParams(max_new_tokens=1, ignore_eos=True)))
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        while eng.scheduler.has_work:
            eng.step()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        return time.perf_counter() - t0

    run_once()                        # warmup (allocators, autotuning)
    times = [run_once() for _ in range(args.repeats)]
    tokens = args.prompts * args.prompt_len
    best = min(times)
    print(json.dumps({
        "metric": "prompt_tokens_per_sec",
        "value": round(tokens / best, 1),
        "unit": "tokens/s",
        "prompts": args.prompts,
        "prompt_len": args.prompt_len,
        "chunked": args.chunked,
        "times_s": [round(t, 3) for t in times],
        "model": args.model,
        "data": "synthetic",
    }))


if __name__ == "__main__":
    main()
