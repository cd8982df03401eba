#!/usr/bin/env python3
"""Serving-path ceiling harness (no GPU): how many tokens/sec can the
gateway's SSE pipeline deliver to the wire when the engine is infinitely
fast?

A SyntheticEngine reproduces AsyncEngine's exact delivery mechanics
(producer thread, per-token call_soon_threadsafe into per-request
asyncio queues) but generates tokens at an unbounded rate, so the
measured wire throughput IS the Python serving ceiling — the number the
real engine (30k+ tok/s on MI355X) is throttled to at high concurrency
(profiles/r02 lever 8).

Usage:
  python tools/bench_sse_ceiling.py --server hserver --clients 256
  python tools/bench_sse_ceiling.py --server uvicorn --clients 256

Orchestrates one server process + N client processes on 127.0.0.1 and
prints one JSON line with wire tokens/s and TTFT percentiles.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import statistics
import subprocess
import sys
import threading
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

TOKEN_TEXT = " the"          # decoded text per synthetic token


class _Out:
    __slots__ = ("seq_id", "token_id", "finished", "finish_reason",
                 "logprob", "top_logprobs")

    def __init__(self, token_id, finished, reason):
        self.seq_id = 0
        self.token_id = token_id
        self.finished = finished
        self.finish_reason = reason
        self.logprob = None
        self.top_logprobs = None


class SyntheticEngine:
    """AsyncEngine lookalike: same thread + queue delivery, infinite
    token rate.  Each stream gets its sampling.max_new_tokens tokens."""

    _SENTINEL = object()

    def __init__(self, real_engine, token_id: int):
        self.engine = real_engine        # for cfg access (max_model_len)
        self.token_id = token_id
        self._lock = threading.Lock()
        self._streams = {}               # rid -> [queue, loop, remaining]
        self._rid = 0
        self._stop = False
        self._wake = threading.Event()
        self._fatal = None
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def _run(self):
        batch_deliver = os.environ.get("IGW_BATCH_DELIVER", "1") == "1"
        while not self._stop:
            with self._lock:
                items = list(self._streams.items())
            if not items:
                self._wake.wait(timeout=0.05)
                self._wake.clear()
                continue
            # one "decode step": one token per live stream
            by_loop = {}
            for rid, st in items:
                q, loop, remaining = st
                fin = remaining <= 1
                out = _Out(self.token_id, fin,
                           "length" if fin else None)
                try:
                    if batch_deliver:
                        by_loop.setdefault(loop, []).append((q, out))
                    else:
                        loop.call_soon_threadsafe(self._deliver,
                                                  [(q, out)])
                except RuntimeError:
                    fin = True           # loop closed
                if fin:
                    with self._lock:
                        self._streams.pop(rid, None)
                else:
                    st[2] = remaining - 1
            for loop, entries in by_loop.items():
                try:
                    loop.call_soon_threadsafe(self._deliver, entries)
                except RuntimeError:
                    pass

    def _deliver(self, entries):
        touched = {}
        for q, out in entries:
            if isinstance(q, asyncio.Queue):
                q.put_nowait(out)
                if out.finished:
                    q.put_nowait(self._SENTINEL)
            else:
                q.feed(out)
                touched[id(q)] = q
        for sink in touched.values():
            sink.flush()

    # ---- direct-sink API (matches AsyncEngine) -------------------------
    def add_direct(self, prompt_token_ids, sampling, sink) -> int:
        loop = asyncio.get_running_loop()
        with self._lock:
            rid = self._rid = self._rid + 1
            self._streams[rid] = [sink, loop, sampling.max_new_tokens]
        self._wake.set()
        return rid

    def close_direct(self, rid: int) -> None:
        with self._lock:
            self._streams.pop(rid, None)

    async def generate(self, prompt_token_ids, sampling):
        async for outs in self.generate_batched(prompt_token_ids, sampling):
            for out in outs:
                yield out

    async def generate_batched(self, prompt_token_ids, sampling):
        loop = asyncio.get_running_loop()
        q: asyncio.Queue = asyncio.Queue()
        with self._lock:
            rid = self._rid = self._rid + 1
            self._streams[rid] = [q, loop, sampling.max_new_tokens]
        self._wake.set()
        try:
            done = False
            while not done:
                batch = [await q.get()]
                while True:
                    try:
                        batch.append(q.get_nowait())
                    except asyncio.QueueEmpty:
                        break
                outs = []
                for item in batch:
                    if item is self._SENTINEL:
                        done = True
                        break
                    outs.append(item)
                if outs:
                    yield outs
        finally:
            with self._lock:
                self._streams.pop(rid, None)

    def shutdown(self):
        self._stop = True
        self._wake.set()


def build_gateway():
    import httpx

    from inference_gateway_amd import config as config_mod
    from inference_gateway_amd.config import LocalConfig
    from inference_gateway_amd.gateway.app import Gateway
    from inference_gateway_amd.logger import noop_logger
    from inference_gateway_amd.providers.local import LocalProvider

    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=1024,
                     max_model_len=100000)
    lc.num_gpu_blocks = 64
    provider = LocalProvider(lc, device="cpu")
    real = provider.engine
    # token that decodes to TOKEN_TEXT under the byte tokenizer
    tid = provider.tokenizer.encode(TOKEN_TEXT)[-1]
    provider.engine = SyntheticEngine(real.engine, tid)
    real.shutdown()
    cfg = config_mod.load({})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(
            lambda r: __import__("httpx").Response(404)))
    return Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                   local_provider=provider)


async def serve_hserver(port: int, profile_seconds: float = 0):
    from inference_gateway_amd.gateway.hserver import HttpServer

    gw = build_gateway()
    srv = HttpServer(gw, "127.0.0.1", port)
    await srv.start()
    print(f"READY {srv.port}", flush=True)
    if profile_seconds:
        import cProfile
        import pstats

        prof = cProfile.Profile()
        prof.enable()
        await asyncio.sleep(profile_seconds)
        prof.disable()
        pstats.Stats(prof).sort_stats("cumulative").print_stats(40)
        sys.stdout.flush()
        os._exit(0)
    await asyncio.Event().wait()


def serve_uvicorn(port: int):
    import uvicorn

    gw = build_gateway()
    print(f"READY {port}", flush=True)
    uvicorn.run(gw, host="127.0.0.1", port=port, log_level="warning",
                lifespan="off")


async def client_main(port: int, clients: int, max_tokens: int):
    """Raw-socket SSE clients: count data: events, record TTFT."""
    results = []

    async def one(i):
        body = json.dumps({
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": f"req {i} hello"}],
            "max_tokens": max_tokens, "stream": True,
        }).encode()
        t0 = time.perf_counter()
        reader, writer = await asyncio.open_connection("127.0.0.1", port)
        writer.write(
            b"POST /v1/chat/completions HTTP/1.1\r\nhost: b\r\n"
            b"content-type: application/json\r\n"
            b"content-length: %d\r\n\r\n" % len(body) + body)
        await writer.drain()
        ttft = None
        tail = b""
        while True:
            data = await reader.read(65536)
            if not data:
                break
            if ttft is None and b"data:" in data:
                ttft = time.perf_counter() - t0
            tail = (tail + data)[-8192:]
            if b"[DONE]" in tail:
                break
        writer.close()
        # authoritative token count: the final usage chunk (chunks may
        # carry several merged tokens under wire-bound batching)
        import re

        m = re.search(rb'"completion_tokens":(\d+)', tail)
        results.append((ttft or 0.0, int(m.group(1)) if m else 0))

    t0 = time.perf_counter()
    await asyncio.gather(*(one(i) for i in range(clients)))
    wall = time.perf_counter() - t0
    print(json.dumps({
        "wall": wall,
        "ttfts": [r[0] for r in results],
        "events": [r[1] for r in results],
    }), flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--server", choices=("hserver", "uvicorn"),
                    default="hserver")
    ap.add_argument("--clients", type=int, default=256)
    ap.add_argument("--max-tokens", type=int, default=500)
    ap.add_argument("--client-procs", type=int, default=4)
    ap.add_argument("--port", type=int, default=18090)
    ap.add_argument("--role", choices=("main", "server", "client"),
                    default="main")
    args = ap.parse_args()

    if args.role == "server":
        if args.server == "hserver":
            asyncio.run(serve_hserver(
                args.port, float(os.environ.get("IGW_PROFILE_S", "0"))))
        else:
            serve_uvicorn(args.port)
        return
    if args.role == "client":
        asyncio.run(client_main(args.port, args.clients, args.max_tokens))
        return

    # orchestrator
    srv = subprocess.Popen(
        [sys.executable, __file__, "--role", "server",
         "--server", args.server, "--port", str(args.port)],
        stdout=subprocess.PIPE, cwd=str(Path(__file__).parent.parent),
        env=dict(os.environ))
    try:
        line = srv.stdout.readline().decode()
        assert line.startswith("READY"), line
        port = int(line.split()[1])
        time.sleep(1.0 if args.server == "hserver" else 3.0)
        per = args.clients // args.client_procs
        t0 = time.perf_counter()
        workers = [subprocess.Popen(
            [sys.executable, __file__, "--role", "client",
             "--port", str(port), "--clients", str(per),
             "--max-tokens", str(args.max_tokens)],
            stdout=subprocess.PIPE) for _ in range(args.client_procs)]
        ttfts, events, walls = [], [], []
        for w in workers:
            out, _ = w.communicate(timeout=600)
            d = json.loads(out.decode().strip().splitlines()[-1])
            ttfts.extend(d["ttfts"])
            events.extend(d["events"])
            walls.append(d["wall"])
        wall = time.perf_counter() - t0
        total_tokens = sum(events)
        print(json.dumps({
            "metric": "sse_ceiling_tokens_per_sec",
            "value": round(total_tokens / max(walls), 1),
            "server": args.server,
            "clients": args.clients,
            "client_procs": args.client_procs,
            "max_tokens": args.max_tokens,
            "wall_s": round(wall, 2),
            "ttft_p50_ms": round(
                statistics.median(ttfts) * 1000, 1),
        }), flush=True)
    finally:
        srv.terminate()
        srv.wait()


if __name__ == "__main__":
    main()
