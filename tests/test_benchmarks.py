"""Benchmark-tier tests with custom metrics (the analog of the reference's
`task benchmark`: ListModels@1000 reporting cpu_ms/op + peak_heap_MB
(tests/api_context_window_bench_test.go) and the selector-vs-direct
tool-injection payload-size reduction, which MUST be a reduction
(tests/mcp_selector_bench_test.go:52-73))."""
import json
import resource
import time
import tracemalloc

import httpx
import pytest

from inference_gateway_amd import config as config_mod
from inference_gateway_amd.gateway.app import Gateway
from inference_gateway_amd.logger import noop_logger

from test_gateway import gw_client


@pytest.mark.asyncio
async def test_bench_list_models_1000(capsys):
    models = [{"id": f"m-{i}", "object": "model", "created": 1,
               "owned_by": "bench"} for i in range(1000)]

    def handler(request: httpx.Request) -> httpx.Response:
        return httpx.Response(200, json={"object": "list", "data": models})

    cfg = config_mod.load({"GROQ_API_KEY": "tok"})
    client = httpx.AsyncClient(transport=httpx.MockTransport(handler))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client)

    tracemalloc.start()
    cpu0 = resource.getrusage(resource.RUSAGE_SELF)
    t0 = time.perf_counter()
    n = 20
    async with gw_client(gw) as c:
        for _ in range(n):
            r = await c.get("/v1/models?provider=groq"
                            "&include=context_window,pricing")
            assert r.status_code == 200
            assert len(r.json()["data"]) == 1000
    wall = (time.perf_counter() - t0) / n * 1000
    cpu1 = resource.getrusage(resource.RUSAGE_SELF)
    cpu_ms = ((cpu1.ru_utime - cpu0.ru_utime)
              + (cpu1.ru_stime - cpu0.ru_stime)) / n * 1000
    peak_mb = tracemalloc.get_traced_memory()[1] / 1e6
    tracemalloc.stop()
    with capsys.disabled():
        print(f"\n[bench] list_models@1000+include: {wall:.1f} ms/op "
              f"cpu {cpu_ms:.1f} ms/op peak_heap {peak_mb:.1f} MB")
    assert wall < 1000, "1000-model listing must stay under 1s/op on CPU"


def test_bench_selector_vs_direct_payload_reduction(capsys):
    """Selector mode must SHRINK the request payload at 50 tools."""
    from inference_gateway_amd.mcp.selector import selector_tool_definitions

    tools = [{
        "type": "function",
        "function": {
            "name": f"mcp_tool_{i}",
            "description": "a tool with a moderately long description "
                           "explaining exactly what it does and when " * 2,
            "parameters": {"type": "object", "properties": {
                "arg_a": {"type": "string", "description": "first arg"},
                "arg_b": {"type": "integer", "description": "second arg"},
            }, "required": ["arg_a"]},
        },
    } for i in range(50)]
    base = {"model": "m", "messages": [{"role": "user", "content": "hi"}]}
    direct_bytes = len(json.dumps({**base, "tools": tools}))
    selector_bytes = len(json.dumps(
        {**base, "tools": selector_tool_definitions()}))
    reduction = direct_bytes / selector_bytes
    with capsys.disabled():
        print(f"\n[bench] direct_bytes={direct_bytes} "
              f"selector_bytes={selector_bytes} reduction_x={reduction:.1f}")
    assert selector_bytes < direct_bytes
    assert reduction > 5.0


@pytest.mark.asyncio
async def test_bench_agent_loop_overhead(tmp_path, monkeypatch, capsys):
    """Gateway-side overhead of one MCP agent iteration against the real
    example filesystem server (BASELINE config 4's plumbing, CPU-only:
    provider latency mocked to ~0)."""
    import sys
    from pathlib import Path

    sys.path.insert(0, str(Path(__file__).resolve().parent))
    from test_soak_and_e2e import _load_fs_server

    monkeypatch.setenv("SANDBOX_DIR", str(tmp_path))
    fs = _load_fs_server()
    state = {"n": 0}

    def handler(request: httpx.Request) -> httpx.Response:
        path = request.url.path
        if "/mcp" in path:
            return httpx.Response(
                200, json=fs.handle_rpc(json.loads(request.content)))
        body = json.loads(request.content)
        state["n"] += 1
        if state["n"] % 2 == 1:
            msg = {"role": "assistant", "content": None, "tool_calls": [
                {"id": f"c{state['n']}", "type": "function", "function": {
                    "name": "mcp_list_directory",
                    "arguments": "{}"}}]}
            fin = "tool_calls"
        else:
            msg = {"role": "assistant", "content": "done"}
            fin = "stop"
        return httpx.Response(200, json={
            "id": "x", "object": "chat.completion", "model": body["model"],
            "choices": [{"index": 0, "message": msg, "finish_reason": fin}],
            "usage": {"prompt_tokens": 1, "completion_tokens": 1,
                      "total_tokens": 2}})

    from inference_gateway_amd import config as config_mod
    from inference_gateway_amd.gateway.app import Gateway
    from inference_gateway_amd.mcp.layer import MCPLayer

    cfg = config_mod.load({"GROQ_API_KEY": "t", "MCP_ENABLED": "true",
                           "MCP_SERVERS": "http://fs/mcp",
                           "MCP_TOOL_MODE": "direct"})
    client = httpx.AsyncClient(transport=httpx.MockTransport(handler))
    mcp = MCPLayer(cfg.mcp, client, noop_logger())
    await mcp.initialize()
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 mcp_layer=mcp)
    n = 50
    async with gw_client(gw) as c:
        t0 = time.perf_counter()
        for _ in range(n):
            r = await c.post("/v1/chat/completions", json={
                "model": "groq/m1",
                "messages": [{"role": "user", "content": "ls"}]})
            assert r.status_code == 200
        wall = (time.perf_counter() - t0) / n * 1000
    await mcp.stop()
    with capsys.disabled():
        print(f"\n[bench] agent loop (1 tool iteration, real MCP server): "
              f"{wall:.2f} ms/request")
    assert wall < 100
