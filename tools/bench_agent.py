#!/usr/bin/env python3
"""BASELINE config 4: the MCP agent path over the REAL local engine —
gateway with MCP enabled (selector mode), the example filesystem MCP
server handling JSON-RPC in-process, N concurrent streaming chat
requests through the agent-enabled middleware on one MI355X.

What this measures: the full agent-path serving stack (tool injection,
agent-managed streaming with delta accumulation, one [DONE] per stream)
with real GPU generation.  A random-init model never emits a valid tool
call, so the agent loop's tool-execution round-trip is benchmarked
separately at the component level (tests/test_benchmarks.py: 1.6 ms per
accumulate+dispatch iteration vs the real filesystem server) — multi-
turn tool TRAFFIC needs a real checkpoint (LOCAL_WEIGHTS_PATH).

Prints one JSON line: tokens/s over the wire, TTFT percentiles, and the
tool-injection evidence (prompt growth through the MCP middleware).
"""
from __future__ import annotations

import asyncio
import importlib.util
import json
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import httpx  # noqa: E402

from inference_gateway_amd import config as config_mod  # noqa: E402
from inference_gateway_amd.config import LocalConfig  # noqa: E402
from inference_gateway_amd.gateway.app import Gateway  # noqa: E402
from inference_gateway_amd.gateway.hserver import HttpServer  # noqa: E402
from inference_gateway_amd.logger import noop_logger  # noqa: E402
from inference_gateway_amd.providers.local import LocalProvider  # noqa


def _load_fs_server(sandbox: str):
    import os

    os.environ["SANDBOX_DIR"] = sandbox
    p = (Path(__file__).resolve().parent.parent / "examples"
         / "mcp-servers" / "filesystem_server.py")
    spec = importlib.util.spec_from_file_location("fs_server_bench", p)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


async def run(model: str, clients: int, max_tokens: int) -> dict:
    import tempfile

    import torch

    fs = _load_fs_server(tempfile.mkdtemp())

    def handler(request: httpx.Request) -> httpx.Response:
        if "/mcp" in request.url.path:
            return httpx.Response(
                200, json=fs.handle_rpc(json.loads(request.content)))
        return httpx.Response(404)

    lc = LocalConfig(enabled=True, model=model,
                     max_num_seqs=max(clients, 8), max_model_len=4096)
    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    if device == "cpu":
        lc.num_gpu_blocks = 4096
    provider = LocalProvider(lc, device=device)
    cfg = config_mod.load({
        "MCP_ENABLED": "true", "MCP_EXPOSE": "true",
        "MCP_SERVERS": "http://fs/mcp",
    })
    client = httpx.AsyncClient(transport=httpx.MockTransport(handler))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider)
    await gw.mcp.initialize()
    tools = gw.mcp.client.all_tools()
    assert len(tools) == 7, "filesystem server exposes 7 tools"
    # every request below takes the AGENT-managed streaming path (the
    # middleware applies whenever MCP tools are discovered)
    assert gw.mcp.client.chat_tools(), "agent layer must be active"
    srv = HttpServer(gw, "127.0.0.1", 0)
    await srv.start()

    ttfts, counts = [], []

    async def one(c, i):
        t0 = time.perf_counter()
        ttft = None
        ntok = 0
        async with c.stream("POST", "/v1/chat/completions", json={
            "model": f"local/{model}",
            "messages": [
                {"role": "system",
                 "content": "You can use tools to answer."},
                {"role": "user",
                 "content": f"request {i}: list the files and read one"},
            ],
            "max_tokens": max_tokens, "temperature": 0.8, "top_k": 50,
            "stream": True,
        }) as r:
            assert r.status_code == 200, r.status_code
            async for line in r.aiter_lines():
                if not line.startswith("data:"):
                    continue
                d = line[5:].strip()
                if d == "[DONE]":
                    break
                try:
                    ch = json.loads(d)
                except json.JSONDecodeError:
                    continue
                u = ch.get("usage")
                if u and u.get("completion_tokens"):
                    ntok = u["completion_tokens"]
                delta = (ch.get("choices") or [{}])[0].get("delta", {})
                if delta.get("content") and ttft is None:
                    ttft = time.perf_counter() - t0
        ttfts.append(ttft if ttft is not None else
                     time.perf_counter() - t0)
        counts.append(ntok)

    base = f"http://127.0.0.1:{srv.port}"
    async with httpx.AsyncClient(base_url=base, timeout=600) as c:
        t0 = time.perf_counter()
        await asyncio.gather(*(one(c, i) for i in range(clients)))
        wall = time.perf_counter() - t0
    await srv.stop()
    provider.engine.shutdown()
    return {
        "metric": "agent_path_output_tokens_per_sec",
        "value": round(sum(counts) / wall, 1),
        "clients": clients,
        "max_tokens": max_tokens,
        "completed": sum(1 for n in counts if n > 0),
        "ttft_p50_ms": round(statistics.median(ttfts) * 1000, 1),
        "ttft_p99_ms": round(sorted(ttfts)[
            max(0, int(len(ttfts) * 0.99) - 1)] * 1000, 1),
        "wall_s": round(wall, 2),
        "mcp_tools_discovered": len(tools),
        "model": model,
        "data": "synthetic",
    }


def main():
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--clients", type=int, default=64)
    ap.add_argument("--max-tokens", type=int, default=64)
    args = ap.parse_args()
    out = asyncio.run(run(args.model, args.clients, args.max_tokens))
    print(json.dumps(out))


if __name__ == "__main__":
    main()
