"""Soak + end-to-end tests:

* 64 concurrent SSE clients against the in-process local engine through
  the gateway (BASELINE config 5's soak bar, CPU-sized), asserting every
  stream terminates with exactly one [DONE];
* the MCP agent loop against the REAL example filesystem MCP server
  (examples/mcp-servers/filesystem_server.py) — not a canned mock — with
  a scripted provider that asks for a write then a read (BASELINE
  config 4's plumbing)."""
import asyncio
import importlib.util
import json
import sys
from pathlib import Path

import httpx
import pytest

from inference_gateway_amd import config as config_mod
from inference_gateway_amd.config import LocalConfig
from inference_gateway_amd.gateway.app import Gateway
from inference_gateway_amd.logger import noop_logger
from inference_gateway_amd.providers.local import LocalProvider

from test_gateway import gw_client

ROOT = Path(__file__).resolve().parent.parent


@pytest.mark.asyncio
async def test_soak_64_concurrent_sse_clients():
    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=64,
                     max_model_len=256)
    lc.num_gpu_blocks = 1024
    provider = LocalProvider(lc, device="cpu")
    cfg = config_mod.load({})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(404)))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider)
    try:
        async with gw_client(gw) as c:
            async def one(i: int):
                done, chunks = 0, 0
                async with c.stream("POST", "/v1/chat/completions", json={
                    "model": "local/tiny-llama",
                    "messages": [{"role": "user", "content": f"req {i}"}],
                    "max_tokens": 4, "temperature": 0.8, "stream": True,
                }) as r:
                    assert r.status_code == 200
                    async for line in r.aiter_lines():
                        if not line.startswith("data:"):
                            continue
                        if line[5:].strip() == "[DONE]":
                            done += 1
                        else:
                            chunks += 1
                return done, chunks

            results = await asyncio.gather(*(one(i) for i in range(64)))
        for done, chunks in results:
            assert done == 1, "exactly one [DONE] per stream"
            # byte-tokenizer may merge partial UTF-8 runs: role chunk
            # + >=1 content chunk + final usage chunk
            assert chunks >= 3
    finally:
        provider.engine.shutdown()


def _load_fs_server():
    spec = importlib.util.spec_from_file_location(
        "fs_server", ROOT / "examples" / "mcp-servers" /
        "filesystem_server.py")
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


@pytest.mark.asyncio
async def test_mcp_agent_against_real_filesystem_server(tmp_path,
                                                        monkeypatch):
    monkeypatch.setenv("SANDBOX_DIR", str(tmp_path))
    fs = _load_fs_server()

    # scripted provider: turn 1 asks to write, turn 2 asks to read, turn 3
    # answers with the read content
    state = {"turn": 0}

    def provider_handler(request: httpx.Request) -> httpx.Response:
        path = request.url.path
        if path.endswith("/models"):
            return httpx.Response(200, json={"object": "list", "data": [
                {"id": "m1", "object": "model"}]})
        if "/mcp" in path:
            # route MCP JSON-RPC into the real example server
            resp = fs.handle_rpc(json.loads(request.content))
            return httpx.Response(200, json=resp)
        body = json.loads(request.content)
        state["turn"] += 1
        t = state["turn"]
        if t == 1:
            msg = {"role": "assistant", "content": None, "tool_calls": [
                {"id": "c1", "type": "function", "function": {
                    "name": "mcp_write_file",
                    "arguments": json.dumps({"path": "note.txt",
                                             "content": "hello-mi355x"})}}]}
            finish = "tool_calls"
        elif t == 2:
            # the previous tool result must be in the history
            tool_msgs = [m for m in body["messages"]
                         if m.get("role") == "tool"]
            assert tool_msgs and "wrote" in tool_msgs[-1]["content"]
            msg = {"role": "assistant", "content": None, "tool_calls": [
                {"id": "c2", "type": "function", "function": {
                    "name": "mcp_read_file",
                    "arguments": json.dumps({"path": "note.txt"})}}]}
            finish = "tool_calls"
        else:
            tool_msgs = [m for m in body["messages"]
                         if m.get("role") == "tool"]
            assert tool_msgs and "hello-mi355x" in tool_msgs[-1]["content"]
            msg = {"role": "assistant",
                   "content": "the file says: hello-mi355x"}
            finish = "stop"
        return httpx.Response(200, json={
            "id": "x", "object": "chat.completion", "model": body["model"],
            "choices": [{"index": 0, "message": msg,
                         "finish_reason": finish}],
            "usage": {"prompt_tokens": 1, "completion_tokens": 1,
                      "total_tokens": 2}})

    env = {"GROQ_API_KEY": "tok", "MCP_ENABLED": "true",
           "MCP_SERVERS": "http://fs-server:8083/mcp",
           "MCP_TOOL_MODE": "direct"}
    cfg = config_mod.load(env)
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(provider_handler))
    from inference_gateway_amd.mcp.layer import MCPLayer

    mcp = MCPLayer(cfg.mcp, client, noop_logger())
    await mcp.initialize()
    assert len(mcp.client.all_tools()) == 7, "filesystem server has 7 tools"

    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 mcp_layer=mcp)
    async with gw_client(gw) as c:
        r = await c.post("/v1/chat/completions", json={
            "model": "groq/m1",
            "messages": [{"role": "user",
                          "content": "write a note then read it back"}]})
    assert r.status_code == 200
    final = r.json()["choices"][0]["message"]["content"]
    assert "hello-mi355x" in final
    # the file really exists in the sandbox
    assert (tmp_path / "note.txt").read_text() == "hello-mi355x"
    await mcp.stop()


def test_search_server_rpc_roundtrip():
    """The example search MCP server answers list/call RPCs correctly."""
    spec = importlib.util.spec_from_file_location(
        "search_server", ROOT / "examples" / "mcp-servers" /
        "search_server.py")
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)

    tools = mod.handle_rpc({"jsonrpc": "2.0", "id": 1,
                            "method": "tools/list"})["result"]["tools"]
    assert {t["name"] for t in tools} == {"search", "get_document"}

    hits = mod.handle_rpc({
        "jsonrpc": "2.0", "id": 2, "method": "tools/call",
        "params": {"name": "search",
                   "arguments": {"query": "paged kv block"}}})["result"]
    assert not hits["isError"]
    ranked = json.loads(hits["content"][0]["text"])
    assert ranked and ranked[0]["id"] == "paged-kv-cache"

    doc = mod.handle_rpc({
        "jsonrpc": "2.0", "id": 3, "method": "tools/call",
        "params": {"name": "get_document",
                   "arguments": {"id": ranked[0]["id"]}}})["result"]
    assert "block table" in doc["content"][0]["text"]

    missing = mod.handle_rpc({
        "jsonrpc": "2.0", "id": 4, "method": "tools/call",
        "params": {"name": "get_document",
                   "arguments": {"id": "nope"}}})["result"]
    assert missing["isError"]


@pytest.mark.asyncio
async def test_soak_mixed_sampling_features():
    """24 concurrent clients exercising every sampling feature at once —
    n>1 KV forks, logprobs, stop strings, seeds — against one engine:
    every stream completes with exactly one [DONE], seeded identical
    requests return identical text."""
    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=64,
                     max_model_len=256)
    lc.num_gpu_blocks = 1024
    provider = LocalProvider(lc, device="cpu")
    cfg = config_mod.load({})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(404)))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider)
    try:
        async with gw_client(gw) as c:
            async def stream_one(i):
                body = {
                    "model": "local/tiny-llama",
                    "messages": [{"role": "user", "content": f"mix {i}"}],
                    "max_tokens": 5, "temperature": 0.9, "stream": True,
                }
                if i % 3 == 0:
                    body["n"] = 2
                if i % 4 == 0:
                    body["logprobs"] = True
                    body["top_logprobs"] = 2
                if i % 5 == 0:
                    body["stop"] = ["zq!x"]
                done = 0
                async with c.stream("POST", "/v1/chat/completions",
                                    json=body) as r:
                    assert r.status_code == 200
                    async for line in r.aiter_lines():
                        if line.strip() == "data: [DONE]":
                            done += 1
                return done

            async def seeded(i):
                r = await c.post("/v1/chat/completions", json={
                    "model": "local/tiny-llama",
                    "messages": [{"role": "user", "content": "same"}],
                    "max_tokens": 6, "temperature": 1.0, "seed": 42,
                })
                return r.json()["choices"][0]["message"]["content"]

            dones = await asyncio.gather(*(stream_one(i)
                                           for i in range(16)))
            texts = await asyncio.gather(*(seeded(i) for i in range(8)))
        assert all(d == 1 for d in dones)
        assert len(set(texts)) == 1
    finally:
        provider.engine.shutdown()


@pytest.mark.asyncio
async def test_mcp_agent_over_real_local_engine(tmp_path, monkeypatch):
    """MCP layer + the REAL in-process engine: the agent loop offers
    tools to a random-init model; whatever it emits (never a valid tool
    call) the loop must terminate cleanly with a normal completion and
    exactly one [DONE] when streaming."""
    monkeypatch.setenv("SANDBOX_DIR", str(tmp_path))
    fs = _load_fs_server()

    def handler(request: httpx.Request) -> httpx.Response:
        if "/mcp" in request.url.path:
            return httpx.Response(
                200, json=fs.handle_rpc(json.loads(request.content)))
        return httpx.Response(404)

    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=4,
                     max_model_len=256)
    lc.num_gpu_blocks = 64
    provider = LocalProvider(lc, device="cpu")
    cfg = config_mod.load({
        "MCP_ENABLED": "true", "MCP_EXPOSE": "true",
        "MCP_SERVERS": "http://fs/mcp",
    })
    client = httpx.AsyncClient(transport=httpx.MockTransport(handler))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider)
    try:
        await gw.mcp.initialize()
        assert gw.mcp.client.all_tools(), "MCP tools must discover"
        async with gw_client(gw) as c:

            r = await c.post("/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "list files"}],
                "max_tokens": 6, "temperature": 0,
            })
            assert r.status_code == 200
            body = r.json()
            assert body["choices"][0]["finish_reason"] in ("stop",
                                                           "length")

            done = 0
            async with c.stream("POST", "/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "read a file"}],
                "max_tokens": 6, "temperature": 0, "stream": True,
            }) as rs:
                assert rs.status_code == 200
                async for line in rs.aiter_lines():
                    if line.strip() == "data: [DONE]":
                        done += 1
            assert done == 1
    finally:
        provider.engine.shutdown()
