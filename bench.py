#!/usr/bin/env python3
"""Flagship serving benchmark — the BASELINE.json metric measured at the
API surface: output tokens/sec delivered over /v1/chat/completions SSE
plus p50 TTFT, Llama-3-8B bf16, synthetic prompts, random-init weights
(no network for checkpoints).

Contract (driver):
  python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches this under torch.distributed.run with one
rank per GPU; ranks run independent gateway+engine replicas (weak
scaling — per-GPU work fixed), the process group is used only for
barriers/reduction, and rank 0 prints ONE JSON line with the whole-job
aggregate.

Default mode (serving): each rank starts the in-process gateway (native
HTTP server, gateway/hserver.py) on 127.0.0.1, drives --batch concurrent
SSE client streams from separate OS processes, waits until every stream
is in steady-state decode, then times EXACTLY K engine steps and counts
the tokens that crossed the SSE write path in that window (wire rate,
not engine rate).  TTFT is measured client-side under the initial
prompt burst.  A "step" is one continuous-batching decode step of the
full running batch.

--engine-only (also implied by --tp N) times raw engine steps without
the HTTP layer — the kernel-iteration mode, and the TP mode until
serving-over-TP lands (TP ranks step one engine SPMD; the async serving
thread is rank-local).
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import statistics
import subprocess
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))

# client workers are torch-free: at SCALE N=8 the bench spawns up to 64
# of them per node and they only push bytes over loopback sockets
if "--client-worker" not in sys.argv:
    import torch  # noqa: E402

    # GEMM backend experiment knob: IGW_BLAS=hipblaslt routes torch GEMMs
    # through hipBLASLt instead of rocBLAS/Tensile (decode GEMMs are
    # skinny-M; see profiles/r01).
    if (os.environ.get("IGW_BLAS", "") == "hipblaslt"
            and torch.cuda.is_available()):
        torch.backends.cuda.preferred_blas_library("cublaslt")
else:
    torch = None


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=32)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--batch", type=int, default=1024,
                    help="concurrent sequences (serving: SSE streams) "
                         "per GPU")
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--model", type=str, default="llama-3-8b")
    ap.add_argument("--eager", action="store_true",
                    help="disable hipGraph capture (for TunableOp tuning)")
    ap.add_argument("--tp", type=int, default=0,
                    help="tensor-parallel degree (0 = DP weak scaling; "
                         "--tp N shards the model over N ranks via "
                         "RCCL/xGMI, requires world_size == N; implies "
                         "--engine-only)")
    ap.add_argument("--engine-only", action="store_true",
                    help="time raw engine steps (no HTTP layer)")
    ap.add_argument("--client-procs", type=int, default=8)
    ap.add_argument("--port", type=int, default=18300)
    ap.add_argument("--client-worker", action="store_true",
                    help=argparse.SUPPRESS)
    ap.add_argument("--num-gpu-blocks", type=int, default=0,
                    help=argparse.SUPPRESS)  # CPU smoke tests
    return ap.parse_args()


# --------------------------------------------------------------------------
# client worker: raw-socket SSE streams against 127.0.0.1 (separate OS
# process — one Python client process saturates long before the server)
# --------------------------------------------------------------------------
async def client_worker_main(args) -> None:
    port = args.port
    ttfts = []

    async def one(i):
        content = f"req {i} " + "x " * (args.prompt_len // 2)
        body = json.dumps({
            "model": f"local/{args.model}",
            "messages": [{"role": "user", "content": content}],
            "max_tokens": args.steps + args.warmup + 512,
            "temperature": 0.8, "top_k": 50, "top_p": 0.95,
            "stream": True,
        }).encode()
        t0 = time.perf_counter()
        try:
            reader, writer = await asyncio.open_connection("127.0.0.1",
                                                           port)
        except OSError:
            return
        writer.write(
            b"POST /v1/chat/completions HTTP/1.1\r\nhost: b\r\n"
            b"content-type: application/json\r\n"
            b"content-length: %d\r\n\r\n" % len(body) + body)
        try:
            await writer.drain()
            events = 0
            ttft = None
            while True:
                try:
                    # before the first token the stream legitimately
                    # sits in the prefill queue for the whole burst ramp
                    # (minutes at batch 1024) — only time out fast once
                    # tokens are flowing, where silence means the server
                    # is gone (stop() raced the FIN)
                    data = await asyncio.wait_for(
                        reader.read(65536), 60 if ttft else 900)
                except asyncio.TimeoutError:
                    break
                if not data:
                    break
                if ttft is None:
                    events += data.count(b"data:")
                    # event 1 = role chunk (sent at stream open);
                    # event 2 = first generated token
                    if events >= 2:
                        ttft = time.perf_counter() - t0
                        ttfts.append(ttft)
                if b"[DONE]" in data:
                    break
        except (ConnectionResetError, OSError):
            pass
        finally:
            try:
                writer.close()
            except Exception:
                pass

    await asyncio.gather(*(one(i) for i in range(args.batch)))
    print(json.dumps({"ttfts": ttfts}), flush=True)


# --------------------------------------------------------------------------
# serving bench (default): gateway + engine in-process, clients external
# --------------------------------------------------------------------------
def bench_serving(args) -> None:
    from inference_gateway_amd import config as config_mod
    from inference_gateway_amd.config import LocalConfig
    from inference_gateway_amd.engine.async_engine import AsyncEngine
    from inference_gateway_amd.engine.config import EngineConfig
    from inference_gateway_amd.gateway.app import Gateway
    from inference_gateway_amd.gateway.hserver import HttpServer
    from inference_gateway_amd.logger import noop_logger
    from inference_gateway_amd.providers.local import LocalProvider
    from inference_gateway_amd.parallel import comm

    comm.init_distributed()
    nranks = comm.world_size()
    rank = comm.world_rank()
    if torch.cuda.is_available():
        device = f"cuda:{int(os.environ.get('LOCAL_RANK', 0))}"
    else:
        device = "cpu"
    port = args.port + rank

    torch.manual_seed(1234 + rank)
    max_len = args.prompt_len + 128 + args.steps + args.warmup + 512 + 64
    ecfg = EngineConfig.for_model(
        args.model,
        max_num_seqs=args.batch,
        max_batched_tokens=max(8192, args.prompt_len * 8),
        max_model_len=max_len,
        use_hip_graphs=not args.eager,
        enforce_eager=args.eager,
        num_gpu_blocks=args.num_gpu_blocks or None,
        # serving posture: capture every decode bucket at init (no lazy
        # capture stalls mid-stream) and strict prefill-priority
        # admission — the burst ramp then takes ~prefill-compute time
        # instead of 41 s of decode-starved trickle admission
        capture_at_init=True,
        scheduler_prefill_first=True,
    )
    engine = AsyncEngine(ecfg, device=device)
    lc = LocalConfig(enabled=True, model=args.model,
                     max_num_seqs=args.batch, max_model_len=max_len)
    provider = LocalProvider(lc, device=device, engine=engine)
    import httpx

    cfg = config_mod.load({})
    gw = Gateway(cfg=cfg, logger=noop_logger(),
                 http_client=httpx.AsyncClient(
                     transport=httpx.MockTransport(
                         lambda r: httpx.Response(404))),
                 local_provider=provider)

    result = asyncio.run(_serving_run(args, gw, provider, engine, port,
                                      rank, nranks, device))
    if rank == 0:
        print(json.dumps(result))


async def _serving_run(args, gw, provider, engine, port, rank, nranks,
                       device):
    from inference_gateway_amd.gateway.hserver import HttpServer

    eng = engine.engine
    srv = HttpServer(gw, "127.0.0.1", port)
    await srv.start()

    # absorb the connect burst with the engine held: the event loop
    # handles all handshakes/requests undisturbed, then the engine
    # serves the whole burst in consecutive prefill rounds
    # (prefill-first).  Without this, token delivery for early streams
    # competes with request handling and admission trickles for minutes.
    engine.hold()

    nprocs = min(args.client_procs, args.batch)
    counts = [args.batch // nprocs + (1 if i < args.batch % nprocs else 0)
              for i in range(nprocs)]
    procs = []
    for n in counts:
        procs.append(subprocess.Popen(
            [sys.executable, os.path.abspath(__file__), "--client-worker",
             "--port", str(srv.port), "--batch", str(n),
             "--prompt-len", str(args.prompt_len),
             "--model", args.model,
             "--steps", str(args.steps), "--warmup", str(args.warmup)],
            stdout=subprocess.PIPE, cwd=str(Path(__file__).parent)))

    # ---- arrival: every request registered before the engine starts ----
    t_arr0 = time.monotonic()
    while time.monotonic() < t_arr0 + 600:
        if len(eng.scheduler.waiting) >= args.batch:
            break
        await asyncio.sleep(0.05)
    else:
        raise RuntimeError(
            f"arrival timeout: {len(eng.scheduler.waiting)}/{args.batch}")
    print(f"[arrival] {args.batch} requests in "
          f"{time.monotonic() - t_arr0:.1f}s", file=sys.stderr)
    engine.release()

    # ---- ramp: wait for every stream to be admitted and prefilled ----
    t_ramp0 = time.monotonic()
    deadline = t_ramp0 + 900
    next_trace = t_ramp0
    while time.monotonic() < deadline:
        running = len(eng.scheduler.running)
        now = time.monotonic()
        if now >= next_trace:
            print(f"[ramp] t={now - t_ramp0:6.1f}s steps={eng.num_steps} "
                  f"prompt_toks={eng.num_prompt_tokens} "
                  f"gen={eng.num_generated_tokens} run={running} "
                  f"wait={len(eng.scheduler.waiting)} "
                  f"wire={provider.streamed_tokens}", file=sys.stderr)
            next_trace = now + 2.0
        if running >= args.batch and not eng.scheduler.waiting:
            break
        await asyncio.sleep(0.05)
    else:
        raise RuntimeError(
            f"ramp timeout: {len(eng.scheduler.running)}/{args.batch} "
            "streams running")
    print(f"[ramp] done in {time.monotonic() - t_ramp0:.1f}s: "
          f"steps={eng.num_steps} prompt_toks={eng.num_prompt_tokens} "
          f"gen={eng.num_generated_tokens}", file=sys.stderr)

    # ---- warmup W steps, then barrier + sync ----
    s_mark = eng.num_steps + args.warmup
    while eng.num_steps < s_mark:
        await asyncio.sleep(0.005)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    await asyncio.to_thread(_barrier)

    # ---- timed window: EXACTLY K engine steps ----
    t0 = time.perf_counter()
    s0, w0 = eng.num_steps, provider.streamed_tokens
    g0 = eng.num_generated_tokens
    run0 = len(eng.scheduler.running)
    target = s0 + args.steps
    while eng.num_steps < target:
        await asyncio.sleep(0.002)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    w1 = provider.streamed_tokens
    run1 = len(eng.scheduler.running)
    g1 = eng.num_generated_tokens
    ctxs = [s.num_tokens for s in eng.scheduler.running]
    if ctxs:
        print(f"[window] mean_ctx={sum(ctxs) / len(ctxs):.0f} "
              f"max_ctx={max(ctxs)}", file=sys.stderr)
    elapsed = t1 - t0
    wire_tokens = w1 - w0
    engine_tokens = args.batch * args.steps
    if run0 < args.batch or run1 < args.batch:
        print(f"[bench] WARNING: running set {run0}->{run1} during the "
              f"window (want {args.batch}): streams died, the window is "
              "not steady-state", file=sys.stderr)

    # ---- aggregate across ranks: MAX elapsed, SUM tokens ----
    elapsed, wire_tokens = _reduce(elapsed, wire_tokens, device)

    # ---- teardown: stop server, collect client TTFTs ----
    await srv.stop()
    engine.shutdown()
    ttfts = []
    for p in procs:
        try:
            out, _ = p.communicate(timeout=120)
            d = json.loads(out.decode().strip().splitlines()[-1])
            ttfts.extend(d["ttfts"])
        except Exception as e:
            print(f"[bench] client proc lost: {e!r}", file=sys.stderr)
            p.kill()
    ttfts = _gather_lists(ttfts)

    value = wire_tokens / elapsed
    result = {
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
        "surface": "/v1/chat/completions SSE",
        "p50_ttft_ms": (round(statistics.median(ttfts) * 1000, 1)
                        if ttfts else None),
        "p99_ttft_ms": (round(sorted(ttfts)[
            max(0, int(len(ttfts) * 0.99) - 1)] * 1000, 1)
            if ttfts else None),
        "wire_vs_engine": round(
            wire_tokens / max(1, engine_tokens * nranks), 3),
        "engine_tokens_window": g1 - g0,
        "window_running": [run0, run1],
        "config": {
            "model": args.model,
            "global_batch": args.batch * nranks,
            "seq_len": args.prompt_len,
            "parallelism": f"dp{nranks}" if nranks > 1 else "tp1",
        },
    }
    return result


def _barrier():
    import torch.distributed as dist

    if dist.is_initialized():
        if torch.cuda.is_available():
            # this runs via asyncio.to_thread: the executor thread's
            # CUDA device defaults to 0, and an NCCL barrier from the
            # wrong device wedges the whole SCALE job — bind it first
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
        dist.barrier()


def _reduce(elapsed: float, tokens: int, device):
    import torch.distributed as dist

    if not dist.is_initialized():
        return elapsed, tokens
    dev = device if torch.cuda.is_available() else "cpu"
    t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    n = torch.tensor([tokens], dtype=torch.float64, device=dev)
    dist.all_reduce(n, op=dist.ReduceOp.SUM)
    return float(t[0]), int(n[0])


def _gather_lists(local):
    import torch.distributed as dist

    if not dist.is_initialized():
        return local
    out = [None] * dist.get_world_size()
    dist.all_gather_object(out, local)
    return [x for part in out for x in part]


# --------------------------------------------------------------------------
# engine-only bench (kernel iteration + TP mode)
# --------------------------------------------------------------------------
def bench_engine(args) -> None:
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
        num_gpu_blocks=args.num_gpu_blocks or None,
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
    # NOTE: the tensor must live on the GPU under the nccl/RCCL backend.
    if nranks > 1:
        import torch.distributed as dist

        red_dev = device if torch.cuda.is_available() else "cpu"
        t = torch.tensor([elapsed], device=red_dev)
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
            "surface": "engine-step",
            "config": {
                "model": args.model,
                "global_batch": args.batch * replicas,
                "seq_len": args.prompt_len,
                "parallelism": (f"tp{args.tp}" if args.tp else
                                (f"dp{nranks}" if nranks > 1 else "tp1")),
            },
        }))


def main() -> None:
    args = parse_args()
    if args.client_worker:
        asyncio.run(client_worker_main(args))
        return
    if args.tp or args.engine_only:
        bench_engine(args)
    else:
        bench_serving(args)


if __name__ == "__main__":
    main()
