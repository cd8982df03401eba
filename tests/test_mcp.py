"""MCP subsystem tests: fake MCP server (JSON-RPC over MockTransport) +
fake tool-calling provider; agent loop, selector mode, delta accumulation
(the analog of the reference's tests/mcp_test.go suite)."""
import asyncio
import json

import httpx
import pytest

from inference_gateway_amd import config as config_mod
from inference_gateway_amd.gateway.app import Gateway
from inference_gateway_amd.logger import noop_logger
from inference_gateway_amd.mcp.agent import accumulate_streaming_tool_calls
from inference_gateway_amd.mcp.client import (build_sse_fallback_url,
                                              is_tool_allowed)

MCP_URL = "http://mcpserver:9000/mcp"


def make_handler(provider_script, sse_framed=False):
    """provider_script: list of responses for successive chat calls; each is
    either ('tool', name, args_json) or ('text', content)."""
    state = {"chat_calls": 0, "tool_calls": []}

    def mcp_response(payload):
        body = json.dumps({"jsonrpc": "2.0", "id": payload.get("id"),
                           "result": mcp_result(payload)})
        if sse_framed:
            return httpx.Response(
                200, headers={"content-type": "text/event-stream",
                              "mcp-session-id": "sess-1"},
                content=f"event: message\ndata: {body}\n\n".encode())
        return httpx.Response(200, json=json.loads(body),
                              headers={"mcp-session-id": "sess-1"})

    def mcp_result(payload):
        method = payload.get("method")
        if method == "initialize":
            return {"protocolVersion": "2024-11-05", "capabilities": {},
                    "serverInfo": {"name": "fake", "version": "0"}}
        if method == "tools/list":
            return {"tools": [
                {"name": "read_file",
                 "description": "Read a file from disk",
                 "inputSchema": {"type": "object",
                                 "properties": {"path": {"type": "string"}},
                                 "required": ["path"]}},
                {"name": "secret_tool", "description": "hidden",
                 "inputSchema": {"type": "object", "properties": {}}},
            ]}
        if method == "tools/call":
            state["tool_calls"].append(payload["params"])
            return {"content": [{"type": "text",
                                 "text": "FILE-CONTENT-42"}]}
        return {}

    def handler(request: httpx.Request) -> httpx.Response:
        if request.url.host == "mcpserver":
            payload = json.loads(request.content)
            if payload.get("method", "").startswith("notifications/"):
                return httpx.Response(202)
            return mcp_response(payload)
        path = request.url.path
        if path.endswith("/models"):
            return httpx.Response(200, json={"object": "list", "data": []})
        if path.endswith("/chat/completions"):
            body = json.loads(request.content)
            i = state["chat_calls"]
            state["chat_calls"] += 1
            step = provider_script[min(i, len(provider_script) - 1)]
            if body.get("stream"):
                return httpx.Response(
                    200, headers={"content-type": "text/event-stream"},
                    content="\n\n".join(stream_chunks(step)).encode())
            return httpx.Response(200, json=nonstream_response(step))
        return httpx.Response(404)

    def nonstream_response(step):
        if step[0] == "tool":
            return {"id": "c", "object": "chat.completion", "choices": [{
                "index": 0, "finish_reason": "tool_calls",
                "message": {"role": "assistant", "content": None,
                            "tool_calls": [{
                                "id": "call-1", "type": "function",
                                "function": {"name": step[1],
                                             "arguments": step[2]}}]}}]}
        return {"id": "c", "object": "chat.completion", "choices": [{
            "index": 0, "finish_reason": "stop",
            "message": {"role": "assistant", "content": step[1]}}],
            "usage": {"prompt_tokens": 1, "completion_tokens": 1,
                      "total_tokens": 2}}

    def stream_chunks(step):
        if step[0] == "tool":
            # split the tool call across chunks to exercise accumulation
            args = step[2]
            half = len(args) // 2
            return [
                'data: ' + json.dumps({"choices": [{"index": 0, "delta": {
                    "tool_calls": [{"index": 0, "id": "call-1",
                                    "type": "function",
                                    "function": {"name": step[1],
                                                 "arguments": args[:half]}}]},
                    "finish_reason": None}]}),
                'data: ' + json.dumps({"choices": [{"index": 0, "delta": {
                    "tool_calls": [{"index": 0,
                                    "function": {
                                        "arguments": args[half:]}}]},
                    "finish_reason": "tool_calls"}]}),
                'data: [DONE]',
            ]
        return [
            'data: ' + json.dumps({"choices": [{"index": 0, "delta": {
                "role": "assistant", "content": step[1]},
                "finish_reason": None}]}),
            'data: ' + json.dumps({"choices": [{"index": 0, "delta": {},
                                                "finish_reason": "stop"}]}),
            'data: [DONE]',
        ]

    return handler, state


def make_gateway(script, env=None, sse_framed=False):
    handler, state = make_handler(script, sse_framed=sse_framed)
    base_env = {"GROQ_API_KEY": "k", "MCP_ENABLED": "true",
                "MCP_SERVERS": MCP_URL}
    base_env.update(env or {})
    cfg = config_mod.load(base_env)
    client = httpx.AsyncClient(transport=httpx.MockTransport(handler))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client)
    return gw, state


def gw_client(gw):
    return httpx.AsyncTransport if False else httpx.AsyncClient(
        transport=httpx.ASGITransport(app=gw), base_url="http://gw")


CHAT = {"model": "groq/m", "messages": [{"role": "user",
                                         "content": "read a.txt"}]}


@pytest.mark.asyncio
async def test_agent_nonstream_tool_loop():
    gw, state = make_gateway([
        ("tool", "mcp_read_file", '{"path": "a.txt"}'),
        ("text", "file says FILE-CONTENT-42"),
    ])
    await gw.mcp.initialize()
    async with gw_client(gw) as c:
        r = await c.post("/v1/chat/completions", json=CHAT)
        assert r.status_code == 200
        msg = r.json()["choices"][0]["message"]
        assert msg["content"] == "file says FILE-CONTENT-42"
    assert state["tool_calls"] == [{"name": "read_file",
                                    "arguments": {"path": "a.txt"}}]
    assert state["chat_calls"] == 2


@pytest.mark.asyncio
async def test_selector_mode_injects_two_meta_tools():
    gw, state = make_gateway([("text", "done")])
    await gw.mcp.initialize()
    captured = {}
    orig = gw.mcp.inject_tools

    def spy(body):
        out = orig(body)
        captured["tools"] = out["tools"]
        return out

    gw.mcp.inject_tools = spy
    async with gw_client(gw) as c:
        await c.post("/v1/chat/completions", json=CHAT)
    names = [t["function"]["name"] for t in captured["tools"]]
    assert names == ["mcp_tools_get", "mcp_tools_execute"]


@pytest.mark.asyncio
async def test_direct_mode_injects_all_tools_with_filter():
    gw, state = make_gateway(
        [("text", "done")],
        env={"MCP_TOOL_MODE": "direct", "MCP_EXCLUDE_TOOLS": "secret_tool"})
    await gw.mcp.initialize()
    tools = gw.mcp.client.chat_tools()
    names = [t["function"]["name"] for t in tools]
    assert names == ["mcp_read_file"]


@pytest.mark.asyncio
async def test_selector_execute_dispatches():
    gw, state = make_gateway([
        ("tool", "mcp_tools_execute",
         '{"name": "read_file", "arguments": {"path": "b.txt"}}'),
        ("text", "ok"),
    ])
    await gw.mcp.initialize()
    async with gw_client(gw) as c:
        r = await c.post("/v1/chat/completions", json=CHAT)
        assert r.status_code == 200
    assert state["tool_calls"][0]["arguments"] == {"path": "b.txt"}


@pytest.mark.asyncio
async def test_agent_streaming_with_tool_loop():
    gw, state = make_gateway([
        ("tool", "mcp_read_file", '{"path": "a.txt"}'),
        ("text", "answer!"),
    ])
    await gw.mcp.initialize()
    req = dict(CHAT, stream=True)
    async with gw_client(gw) as c:
        async with c.stream("POST", "/v1/chat/completions", json=req) as r:
            assert r.status_code == 200
            events = []
            async for line in r.aiter_lines():
                if line.startswith("data:"):
                    events.append(line[5:].strip())
    # exactly one [DONE], at the very end
    assert events[-1] == "[DONE]"
    assert sum(1 for e in events if e == "[DONE]") == 1
    # the tool executed and the second iteration's content came through
    assert state["tool_calls"] == [{"name": "read_file",
                                    "arguments": {"path": "a.txt"}}]
    contents = []
    for e in events[:-1]:
        chunk = json.loads(e)
        delta = chunk["choices"][0].get("delta", {})
        if delta.get("content"):
            contents.append(delta["content"])
    assert "answer!" in "".join(contents)


@pytest.mark.asyncio
async def test_sse_framed_mcp_responses():
    gw, state = make_gateway([
        ("tool", "mcp_read_file", '{"path": "a.txt"}'),
        ("text", "done"),
    ], sse_framed=True)
    await gw.mcp.initialize()
    assert gw.mcp.client.chat_tools()
    async with gw_client(gw) as c:
        r = await c.post("/v1/chat/completions", json=CHAT)
        assert r.status_code == 200
    assert state["tool_calls"]


@pytest.mark.asyncio
async def test_mcp_tools_endpoint_gated_by_expose():
    gw, _ = make_gateway([("text", "x")])
    await gw.mcp.initialize()
    async with gw_client(gw) as c:
        r = await c.get("/v1/mcp/tools")
        assert r.status_code == 403
    gw2, _ = make_gateway([("text", "x")], env={"MCP_EXPOSE": "true"})
    await gw2.mcp.initialize()
    async with gw_client(gw2) as c:
        r = await c.get("/v1/mcp/tools")
        assert r.status_code == 200
        names = [t["name"] for t in r.json()["data"]]
        assert "mcp_read_file" in names


def test_accumulate_streaming_tool_calls_drops_unnamed():
    lines = [
        'data: ' + json.dumps({"choices": [{"delta": {"tool_calls": [
            {"index": 0, "id": "a", "function": {"name": "f1",
                                                 "arguments": '{"x"'}},
            {"index": 1, "function": {"arguments": '{"never-named"'}},
        ]}}]}),
        'data: ' + json.dumps({"choices": [{"delta": {"tool_calls": [
            {"index": 0, "function": {"arguments": ': 1}'}},
        ]}}]}),
        'data: [DONE]',
    ]
    out = accumulate_streaming_tool_calls(lines)
    assert len(out) == 1
    assert out[0]["function"]["name"] == "f1"
    assert out[0]["function"]["arguments"] == '{"x": 1}'


def test_tool_filter_semantics():
    assert is_tool_allowed("mcp_Read_File", "read_file", "")
    assert not is_tool_allowed("other", "read_file", "")
    assert not is_tool_allowed("MCP_secret", "", "secret")
    assert is_tool_allowed("fine", "", "secret")
    assert is_tool_allowed("anything", "", "")


def test_sse_fallback_url():
    assert build_sse_fallback_url("http://h/mcp") == "http://h/sse"
    assert build_sse_fallback_url("http://h/x") == "http://h/x/sse"


@pytest.mark.asyncio
async def test_reconnect_recovers_dead_server():
    """Zero live servers at startup is tolerated when reconnect is on
    (init.go:64-77); the background loop picks the server up once it
    returns (init.go:330-408)."""
    state = {"up": False}

    def handler(request: httpx.Request) -> httpx.Response:
        if not state["up"]:
            return httpx.Response(503)
        body = json.loads(request.content)
        if body.get("method") == "initialize":
            return httpx.Response(200, json={
                "jsonrpc": "2.0", "id": body.get("id"),
                "result": {"protocolVersion": "2024-11-05",
                           "capabilities": {"tools": {}},
                           "serverInfo": {"name": "s", "version": "1"}}})
        if body.get("method") == "tools/list":
            return httpx.Response(200, json={
                "jsonrpc": "2.0", "id": body.get("id"),
                "result": {"tools": [{"name": "late_tool",
                                      "description": "d",
                                      "inputSchema": {"type": "object"}}]}})
        return httpx.Response(200, json={"jsonrpc": "2.0",
                                         "id": body.get("id"),
                                         "result": {}})

    cfg = config_mod.load({
        "MCP_ENABLED": "true",
        "MCP_SERVERS": "http://flaky/mcp",
        "MCP_ENABLE_RECONNECT": "true",
        "MCP_RECONNECT_INTERVAL": "50ms",
        "MCP_MAX_RETRIES": "1",
        "MCP_INITIAL_BACKOFF": "10ms",
        "MCP_RETRY_INTERVAL": "20ms",
        "MCP_POLLING_ENABLED": "false",
    })
    client = httpx.AsyncClient(transport=httpx.MockTransport(handler))
    from inference_gateway_amd.mcp.layer import MCPLayer

    mcp = MCPLayer(cfg.mcp, client, noop_logger())
    await mcp.initialize()          # all servers down: tolerated
    assert mcp.client.all_tools() == []
    state["up"] = True
    for _ in range(60):
        await asyncio.sleep(0.05)
        if mcp.client.all_tools():
            break
    names = [t["name"] for t in mcp.client.all_tools()]
    assert "late_tool" in names
    await mcp.stop()


def test_accumulate_tolerates_sparse_indices():
    """A provider that starts tool-call indices at 1 (or skips one)
    must not lose calls."""
    from inference_gateway_amd.mcp.agent import (
        accumulate_streaming_tool_calls)

    lines = [
        'data: {"choices":[{"delta":{"tool_calls":[{"index":1,'
        '"id":"b","function":{"name":"fs_read","arguments":"{}"}}]}}]}',
        'data: {"choices":[{"delta":{"tool_calls":[{"index":3,'
        '"id":"d","function":{"name":"time_now","arguments":""}}]}}]}',
    ]
    out = accumulate_streaming_tool_calls(lines)
    assert [tc["function"]["name"] for tc in out] == ["fs_read",
                                                      "time_now"]
