"""Tracing tests (analog of the reference's in-memory SpanRecorder suite,
tests/tracing_test.go): server spans per request, W3C traceparent
propagation inbound and outbound, execute_tool spans, /health filtering,
OTLP payload shape."""
import httpx
import pytest

from inference_gateway_amd import config as config_mod
from inference_gateway_amd.gateway.app import Gateway
from inference_gateway_amd.logger import noop_logger
from inference_gateway_amd.otel.tracing import (Tracer, format_traceparent,
                                                parse_traceparent)

from test_gateway import gw_client, make_upstream


def make_traced_gateway():
    handler, calls = make_upstream()
    cfg = config_mod.load({"GROQ_API_KEY": "tok",
                           "TELEMETRY_TRACING_ENABLED": "true"})
    client = httpx.AsyncClient(transport=httpx.MockTransport(handler))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client)
    return gw, calls


@pytest.mark.asyncio
async def test_server_span_and_outbound_propagation():
    gw, calls = make_traced_gateway()
    tp_in = format_traceparent("ab" * 16, "cd" * 8)
    async with gw_client(gw) as c:
        r = await c.post("/v1/chat/completions",
                         json={"model": "groq/m1",
                               "messages": [{"role": "user",
                                             "content": "x"}]},
                         headers={"traceparent": tp_in})
    assert r.status_code == 200
    spans = gw.tracer.drain()
    names = [s.name for s in spans]
    assert "POST /v1/chat/completions" in names
    server = [s for s in spans if s.kind == "server"][0]
    # inherits the caller's trace id, parents on the caller's span
    assert server.trace_id == "ab" * 16
    assert server.parent_id == "cd" * 8
    assert server.attributes["http.status_code"] == 200
    # the upstream call carried OUR span's traceparent (W3C propagation)
    sent = [req for req in calls if "chat" in req.url.path][-1]
    ctx = parse_traceparent(sent.headers["traceparent"])
    assert ctx is not None
    assert ctx[0] == "ab" * 16
    assert ctx[1] == server.span_id


@pytest.mark.asyncio
async def test_health_not_traced():
    gw, _ = make_traced_gateway()
    async with gw_client(gw) as c:
        await c.get("/health")
    assert gw.tracer.drain() == []


def test_otlp_payload_shape_and_path_fix():
    t = Tracer(endpoint="http://collector:4318")
    assert t.endpoint == "http://collector:4318/v1/traces"
    s = t.start_span("GET /v1/models", kind="server")
    s.set("k", "v")
    t.finish(s)
    payload = t._otlp_payload(t.drain())
    span = payload["resourceSpans"][0]["scopeSpans"][0]["spans"][0]
    assert span["name"] == "GET /v1/models"
    assert span["kind"] == 2
    assert span["attributes"] == [
        {"key": "k", "value": {"stringValue": "v"}}]


@pytest.mark.asyncio
async def test_execute_tool_span(tmp_path, monkeypatch):
    """The agent emits `execute_tool <name>` spans (agent.go:402-429)."""
    import json as _json

    from inference_gateway_amd.mcp.layer import MCPLayer
    from test_soak_and_e2e import _load_fs_server

    monkeypatch.setenv("SANDBOX_DIR", str(tmp_path))
    fs = _load_fs_server()

    def handler(request: httpx.Request) -> httpx.Response:
        if "/mcp" in request.url.path:
            return httpx.Response(
                200, json=fs.handle_rpc(_json.loads(request.content)))
        return httpx.Response(404)

    cfg = config_mod.load({"MCP_ENABLED": "true",
                           "MCP_SERVERS": "http://fs/mcp"})
    client = httpx.AsyncClient(transport=httpx.MockTransport(handler))
    tracer = Tracer()
    mcp = MCPLayer(cfg.mcp, client, noop_logger(), tracer=tracer)
    await mcp.initialize()
    out = await mcp.agent.execute_tools(
        [{"id": "c1", "type": "function",
          "function": {"name": "mcp_write_file",
                       "arguments": _json.dumps(
                           {"path": "t.txt", "content": "hi"})}}],
        "groq", "m1")
    assert out and "wrote" in out[0]["content"]
    names = [s.name for s in tracer.drain()]
    assert "execute_tool mcp_write_file" in names
    await mcp.stop()


@pytest.mark.asyncio
async def test_client_spans_on_outbound_calls():
    """Outbound provider calls get client spans named 'METHOD /path',
    parented on the server span (otelhttp analog)."""
    gw, _ = make_traced_gateway()
    async with gw_client(gw) as c:
        r = await c.post("/v1/chat/completions",
                         json={"model": "groq/m1",
                               "messages": [{"role": "user",
                                             "content": "x"}]})
    assert r.status_code == 200
    spans = gw.tracer.drain()
    client_spans = [s for s in spans if s.kind == "client"]
    server = [s for s in spans if s.kind == "server"][0]
    assert client_spans, [s.name for s in spans]
    cs = client_spans[-1]
    assert cs.name.startswith("POST ") and "chat/completions" in cs.name
    assert cs.trace_id == server.trace_id
    assert cs.parent_id == server.span_id
    assert cs.attributes["http.status_code"] == 200


@pytest.mark.asyncio
async def test_server_span_covers_whole_stream():
    """For SSE the server span must finish when the STREAM ends (otelgin
    semantics), not at response-start — and exactly once."""
    from inference_gateway_amd.gateway.http import Request, Response
    from inference_gateway_amd.otel.tracing import tracing_middleware

    tracer = Tracer()
    mw = tracing_middleware(tracer)

    async def gen():
        yield "data: a\n\n"
        yield "data: [DONE]\n\n"

    async def nxt(req):
        return Response.sse(gen())

    req = Request(method="POST", path="/v1/chat/completions",
                  headers={}, query={}, body=b"{}")
    resp = await mw(req, nxt)
    # the middleware returned but the stream has not run: span open
    assert [x.name for x in tracer.drain()] == []
    chunks = [c async for c in resp.stream]
    assert len(chunks) == 2
    spans = tracer.drain()
    assert [x.name for x in spans] == ["POST /v1/chat/completions"]
    assert spans[0].kind == "server"
