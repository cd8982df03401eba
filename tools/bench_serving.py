#!/usr/bin/env python3
"""HTTP-level serving benchmark: output tokens/sec + p50 TTFT via
POST /v1/chat/completions (SSE) — the BASELINE.json metric measured at the
wire, not the engine step loop.

Modes:
  --mode local  (default): starts the gateway with the in-process MI355X
      engine (LOCAL_MODEL, GPU box) under uvicorn on 127.0.0.1 and drives
      N concurrent streaming clients.           [BASELINE configs 2/5]
  --mode mock: gateway with a mock upstream provider (in-process ASGI, no
      GPU) — isolates the gateway/SSE overhead. [BASELINE config 1]

Prints one JSON line: {"metric": "http_output_tokens_per_sec", "value",
"ttft_p50_ms", "ttft_p99_ms", ...}.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


async def drive(base_url: str, transport, clients: int, max_tokens: int,
                prompt_chars: int, model: str, shared_prefix: int = 0):
    import httpx

    ttfts, token_counts = [], []

    async def one(client, i):
        t0 = time.perf_counter()
        ttft = None
        ntok = 0
        async with client.stream("POST", "/v1/chat/completions", json={
            "model": model,
            "messages": (
                [{"role": "system", "content": "s " * (shared_prefix // 2)}]
                if shared_prefix else []) + [
                {"role": "user",
                 "content": f"req {i} " + "x " * (prompt_chars // 2)}],
            "max_tokens": max_tokens,
            "temperature": 0.8,
            "top_k": 50,
            "top_p": 0.95,
            "stream": True,
        }) as r:
            assert r.status_code == 200, r.status_code
            async for line in r.aiter_lines():
                if not line.startswith("data:"):
                    continue
                data = line[5:].strip()
                if data == "[DONE]":
                    break
                try:
                    chunk = json.loads(data)
                except json.JSONDecodeError:
                    continue
                delta = chunk["choices"][0].get("delta", {})
                if delta.get("content"):
                    if ttft is None:
                        ttft = time.perf_counter() - t0
                    ntok += 1
                usage = chunk.get("usage")
                if usage and usage.get("completion_tokens"):
                    # authoritative: content chunks may merge several
                    # tokens (UTF-8 holdback in the byte tokenizer)
                    ntok = usage["completion_tokens"]
        ttfts.append(ttft if ttft is not None else time.perf_counter() - t0)
        token_counts.append(ntok)

    async with (await make_client(base_url, transport)) as client:
        t0 = time.perf_counter()
        await asyncio.gather(*(one(client, i) for i in range(clients)))
        wall = time.perf_counter() - t0
    return ttfts, token_counts, wall


async def make_client(base_url, transport):
    import httpx

    return httpx.AsyncClient(base_url=base_url, transport=transport,
                             timeout=300.0)


async def run_mock(args):
    import httpx

    from inference_gateway_amd import config as config_mod
    from inference_gateway_amd.gateway.app import Gateway
    from inference_gateway_amd.logger import noop_logger

    # simpler: synchronous handler
    def shandler(request: httpx.Request) -> httpx.Response:
        chunks = ['data: {"id":"c","object":"chat.completion.chunk",'
                  '"choices":[{"index":0,"delta":{"content":"t"},'
                  '"finish_reason":null}]}'] * args.max_tokens
        chunks.append("data: [DONE]")
        return httpx.Response(
            200, headers={"content-type": "text/event-stream"},
            content="\n\n".join(chunks).encode())

    cfg = config_mod.load({"GROQ_API_KEY": "tok"})
    client = httpx.AsyncClient(transport=httpx.MockTransport(shandler))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client)
    transport = httpx.ASGITransport(app=gw)
    return await drive("http://gw", transport, args.clients,
                       args.max_tokens, args.prompt_chars, "groq/mock")


async def run_local(args):
    import httpx

    # spawn the real server process on 127.0.0.1
    env = dict(os.environ,
               LOCAL_ENABLED="true",
               LOCAL_MODEL=args.model,
               LOCAL_MAX_NUM_SEQS=str(max(args.clients, 8)),
               LOCAL_MAX_MODEL_LEN="4096",
               LOCAL_PREFIX_CACHING=(
                   "true" if args.prefix_caching else "false"),
               LOCAL_CHUNKED_PREFILL=(
                   "true" if args.chunked_prefill else "false"),
               SERVER_PORT=str(args.port))
    proc = await asyncio.create_subprocess_exec(
        sys.executable, "-m", "inference_gateway_amd", env=env,
        cwd=str(Path(__file__).resolve().parent.parent))
    base = f"http://127.0.0.1:{args.port}"
    try:
        async with (await make_client(base, None)) as c:
            for _ in range(600):
                try:
                    r = await c.get("/health")
                    if r.status_code == 200:
                        break
                except Exception:
                    pass
                await asyncio.sleep(0.5)
            else:
                raise RuntimeError("server never became healthy")
        return await drive(base, None, args.clients, args.max_tokens,
                           args.prompt_chars, f"local/{args.model}",
                           shared_prefix=args.shared_prefix)
    finally:
        proc.terminate()
        await proc.wait()


async def run_local_multi(args):
    import subprocess

    env = dict(os.environ,
               LOCAL_ENABLED="true",
               LOCAL_MODEL=args.model,
               LOCAL_MAX_NUM_SEQS=str(max(args.clients, 8)),
               LOCAL_MAX_MODEL_LEN="4096",
               LOCAL_PREFIX_CACHING=(
                   "true" if args.prefix_caching else "false"),
               LOCAL_CHUNKED_PREFILL=(
                   "true" if args.chunked_prefill else "false"),
               SERVER_PORT=str(args.port))
    proc = await asyncio.create_subprocess_exec(
        sys.executable, "-m", "inference_gateway_amd", env=env,
        cwd=str(Path(__file__).resolve().parent.parent))
    base = f"http://127.0.0.1:{args.port}"
    try:
        async with (await make_client(base, None)) as c:
            for _ in range(600):
                try:
                    if (await c.get("/health")).status_code == 200:
                        break
                except Exception:
                    pass
                await asyncio.sleep(0.5)
            else:
                raise RuntimeError("server never became healthy")
        per = args.clients // args.client_procs
        workers = [await asyncio.create_subprocess_exec(
            sys.executable, __file__, "--worker", "--port", str(args.port),
            "--model", args.model, "--clients", str(per),
            "--max-tokens", str(args.max_tokens),
            "--prompt-chars", str(args.prompt_chars),
            "--shared-prefix", str(args.shared_prefix),
            stdout=subprocess.PIPE)
            for _ in range(args.client_procs)]
        t0 = time.perf_counter()
        outs = await asyncio.gather(*(w.communicate() for w in workers))
        wall = time.perf_counter() - t0
        ttfts, tokens = [], []
        for stdout, _ in outs:
            d = json.loads(stdout.decode().strip().splitlines()[-1])
            ttfts.extend(d["ttfts"])
            tokens.extend(d["tokens"])
        return ttfts, tokens, wall
    finally:
        proc.terminate()
        await proc.wait()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", choices=("local", "mock"), default="local")
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--clients", type=int, default=64)
    ap.add_argument("--max-tokens", type=int, default=128)
    ap.add_argument("--prompt-chars", type=int, default=512)
    ap.add_argument("--shared-prefix", type=int, default=0,
                    help="chars of a SHARED system prompt per request "
                         "(prefix-cache workload; pair with "
                         "--prefix-caching to measure the benefit)")
    ap.add_argument("--prefix-caching", action="store_true",
                    help="run the engine with LOCAL_PREFIX_CACHING=true")
    ap.add_argument("--chunked-prefill", action="store_true",
                    help="run the engine with LOCAL_CHUNKED_PREFILL=true")
    ap.add_argument("--port", type=int, default=18080)
    ap.add_argument("--client-procs", type=int, default=1,
                    help="drive the load from N OS processes (a single "
                         "Python client process saturates before the "
                         "server at high concurrency)")
    ap.add_argument("--worker", action="store_true", help=argparse.SUPPRESS)
    args = ap.parse_args()

    if args.worker:
        # child: drive clients against an already-running server
        async def child():
            base = f"http://127.0.0.1:{args.port}"
            return await drive(base, None, args.clients, args.max_tokens,
                               args.prompt_chars, f"local/{args.model}",
                               shared_prefix=args.shared_prefix)

        t, c, w = asyncio.run(child())
        print(json.dumps({"ttfts": t, "tokens": c, "wall": w}))
        return

    if args.mode == "local" and args.client_procs > 1:
        ttfts, token_counts, wall = asyncio.run(run_local_multi(args))
    else:
        runner = run_local if args.mode == "local" else run_mock
        ttfts, token_counts, wall = asyncio.run(runner(args))
    total_tokens = sum(token_counts)
    print(json.dumps({
        "metric": "http_output_tokens_per_sec",
        "value": round(total_tokens / wall, 1),
        "unit": "tokens/s",
        "ttft_p50_ms": round(statistics.median(ttfts) * 1000, 1),
        "ttft_p99_ms": round(
            sorted(ttfts)[max(0, int(len(ttfts) * 0.99) - 1)] * 1000, 1),
        "clients": args.clients,
        "max_tokens": args.max_tokens,
        "wall_s": round(wall, 2),
        "mode": args.mode,
        "model": args.model,
        "data": "synthetic",
    }))


if __name__ == "__main__":
    main()
