#!/usr/bin/env python3
"""Flagship serving benchmark: Llama-3-8B bf16 decode throughput
(output tokens/sec) on the MI355X-native engine — the BASELINE.json metric
("output tokens/sec ... /v1/chat/completions local, Llama-3-8B") measured at
the engine step level with synthetic data and random-init weights (no
network for checkpoints; BASELINE.md documents the HTTP/SSE overhead
separately).

Contract (driver):
  python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches this under torch.distributed.run with one rank
per GPU; ranks run independent engine replicas (weak scaling — per-GPU work
fixed), the process group is used only for barriers, and rank 0 prints ONE
JSON line with the whole-job aggregate.

A "step" is one continuous-batching decode step of the full running batch
(scheduler + forward + fused sampling kernel + token append).
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))

import torch  # noqa: E402

# GEMM backend experiment knob: IGW_BLAS=hipblaslt routes torch GEMMs
# through hipBLASLt instead of rocBLAS/Tensile (decode GEMMs are skinny-M;
# see profiles/r01).
if os.environ.get("IGW_BLAS", "") == "hipblaslt" and torch.cuda.is_available():
    torch.backends.cuda.preferred_blas_library("cublaslt")


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=32)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--batch", type=int, default=1024,
                    help="concurrent sequences per GPU")
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--model", type=str, default="llama-3-8b")
    ap.add_argument("--eager", action="store_true",
                    help="disable hipGraph capture (for TunableOp tuning)")
    ap.add_argument("--tp", type=int, default=0,
                    help="tensor-parallel degree (0 = DP weak scaling; "
                         "--tp N shards the model over N ranks via "
                         "RCCL/xGMI, requires world_size == N)")
    args = ap.parse_args()

    from inference_gateway_amd.engine.config import (EngineConfig,
                                                     SamplingParams)
    from inference_gateway_amd.engine.engine import LLMEngine
    from inference_gateway_amd.parallel import comm

    comm.init_distributed()
    nranks = comm.world_size()
    rank = comm.world_rank()
    if args.tp:
        comm.init_tp(args.tp)
    if torch.cuda.is_available():
        device = f"cuda:{int(os.environ.get('LOCAL_RANK', 0))}"
    else:
        device = "cpu"

    torch.manual_seed(1234 + rank)
    cfg = EngineConfig.for_model(
        args.model,
        max_num_seqs=args.batch,
        max_batched_tokens=max(8192, args.prompt_len * 8),
        max_model_len=args.prompt_len + args.steps + args.warmup + 64,
        use_hip_graphs=not args.eager,
        enforce_eager=args.eager,
    )
    eng = LLMEngine(cfg, device=device)

    # synthetic workload: fixed per-GPU batch of random prompts
    vocab = cfg.model.vocab_size
    gen = torch.Generator().manual_seed(42 + rank)
    for _ in range(args.batch):
        prompt = torch.randint(
            0, vocab, (args.prompt_len,), generator=gen).tolist()
        eng.add_request(
            prompt,
            SamplingParams(max_new_tokens=10**9, temperature=0.8, top_k=50,
                           top_p=0.95, ignore_eos=True),
        )

    # prefill everything + warmup decode steps (untimed)
    while eng.scheduler.waiting:
        eng.step()
    for _ in range(args.warmup):
        eng.step()

    def sync():
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        comm.world_barrier()

    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        eng.step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    comm.world_barrier()

    # MAX over ranks (the slowest rank defines the job's step time).
    # NOTE: the tensor must live on the GPU under the nccl/RCCL backend —
    # a CPU tensor raises and would kill the run after the timed region.
    if nranks > 1:
        red_dev = device if torch.cuda.is_available() else "cpu"
        t = torch.tensor([elapsed], device=red_dev)
        import torch.distributed as dist

        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0])

    # TP ranks compute ONE model together: count the batch once
    replicas = 1 if args.tp else nranks
    tokens_total = args.batch * args.steps * replicas
    value = tokens_total / elapsed
    if rank == 0:
        print(json.dumps({
            "metric": "output_tokens_per_sec",
            "value": round(value, 1),
            "unit": "tokens/s",
            "n_gpus": nranks,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": args.batch * replicas,
                "seq_len": args.prompt_len,
                "parallelism": (f"tp{args.tp}" if args.tp else
                                (f"dp{nranks}" if nranks > 1 else "tp1")),
            },
        }))


if __name__ == "__main__":
    main()
