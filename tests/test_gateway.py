"""Black-box gateway tests: real Gateway ASGI app + httpx MockTransport
upstreams (the analog of the reference's httptest-based integration suite,
tests/api_routes_test.go)."""
import json

import httpx
import pytest

from inference_gateway_amd import config as config_mod
from inference_gateway_amd.gateway.app import Gateway
from inference_gateway_amd.logger import noop_logger


def make_upstream():
    """Mock upstream provider speaking the OpenAI wire shape."""
    calls = []

    def handler(request: httpx.Request) -> httpx.Response:
        calls.append(request)
        path = request.url.path
        if path.endswith("/models"):
            return httpx.Response(200, json={
                "object": "list",
                "data": [
                    {"id": "m-one", "object": "model", "created": 1,
                     "owned_by": "me"},
                    {"id": "m-two", "object": "model", "created": 1,
                     "owned_by": "me"},
                ],
            })
        if path.endswith("/chat/completions"):
            body = json.loads(request.content)
            if body.get("stream"):
                chunks = [
                    'data: {"id":"c1","object":"chat.completion.chunk",'
                    '"choices":[{"index":0,"delta":{"role":"assistant",'
                    '"content":"hel"},"finish_reason":null}]}',
                    'data: {"id":"c1","object":"chat.completion.chunk",'
                    '"choices":[{"index":0,"delta":{"content":"lo"},'
                    '"finish_reason":"stop"}],"usage":{"prompt_tokens":2,'
                    '"completion_tokens":2,"total_tokens":4}}',
                    'data: [DONE]',
                ]
                return httpx.Response(
                    200, headers={"content-type": "text/event-stream"},
                    content="\n\n".join(chunks).encode())
            return httpx.Response(200, json={
                "id": "c1", "object": "chat.completion",
                "model": body.get("model"),
                "choices": [{"index": 0,
                             "message": {"role": "assistant",
                                         "content": "hello"},
                             "finish_reason": "stop"}],
                "usage": {"prompt_tokens": 2, "completion_tokens": 1,
                          "total_tokens": 3},
            })
        if path.endswith("/messages"):
            return httpx.Response(200, json={"type": "message",
                                             "content": [{"type": "text",
                                                          "text": "hi"}]})
        return httpx.Response(404, json={"error": "nope"})

    return handler, calls


def make_gateway(env=None, **kw):
    handler, calls = make_upstream()
    base_env = {"GROQ_API_KEY": "groq-token",
                "ANTHROPIC_API_KEY": "anthro-key"}
    base_env.update(env or {})
    cfg = config_mod.load(base_env)
    client = httpx.AsyncClient(transport=httpx.MockTransport(handler))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client, **kw)
    return gw, calls


def gw_client(gw):
    return httpx.AsyncClient(transport=httpx.ASGITransport(app=gw),
                             base_url="http://gw")


@pytest.mark.asyncio
async def test_health():
    gw, _ = make_gateway()
    async with gw_client(gw) as c:
        r = await c.get("/health")
        assert r.status_code == 200
        assert r.json() == {"status": "healthy"}


@pytest.mark.asyncio
async def test_list_models_single_provider_prefixes_and_filters():
    gw, _ = make_gateway(env={"ALLOWED_MODELS": "groq/m-one"})
    async with gw_client(gw) as c:
        r = await c.get("/v1/models", params={"provider": "groq"})
        assert r.status_code == 200
        data = r.json()["data"]
        assert [m["id"] for m in data] == ["groq/m-one"]
        assert data[0]["served_by"] == "groq"


@pytest.mark.asyncio
async def test_list_models_fanout():
    gw, _ = make_gateway()
    async with gw_client(gw) as c:
        r = await c.get("/v1/models")
        ids = {m["id"] for m in r.json()["data"]}
        # both configured providers (groq + anthropic) are present
        assert "groq/m-one" in ids and "anthropic/m-one" in ids


@pytest.mark.asyncio
async def test_chat_completion_routing_and_auth_injection():
    gw, calls = make_gateway()
    async with gw_client(gw) as c:
        r = await c.post("/v1/chat/completions",
                         headers={"Authorization": "Bearer caller-secret"},
                         json={"model": "groq/m-one",
                               "messages": [{"role": "user",
                                             "content": "hi"}]})
        assert r.status_code == 200
        assert r.json()["choices"][0]["message"]["content"] == "hello"
    upstream = [c for c in calls if c.url.path.endswith("/chat/completions")]
    assert upstream, "no upstream call recorded"
    req = upstream[-1]
    # provider credential injected; caller Authorization NOT forwarded
    assert req.headers["authorization"] == "Bearer groq-token"
    assert req.url.host == "api.groq.com"
    # model prefix stripped before forwarding
    assert json.loads(req.content)["model"] == "m-one"


@pytest.mark.asyncio
async def test_chat_completion_unknown_provider_and_missing_key():
    gw, _ = make_gateway()
    async with gw_client(gw) as c:
        msgs = [{"role": "user", "content": "x"}]
        r = await c.post("/v1/chat/completions",
                         json={"model": "nosuch/m", "messages": msgs})
        assert r.status_code == 400
        r = await c.post("/v1/chat/completions",
                         json={"model": "openai/gpt-x", "messages": msgs})
        assert r.status_code == 401  # openai has no key configured
        r = await c.post("/v1/chat/completions",
                         json={"model": "bare-model", "messages": msgs})
        assert r.status_code == 400  # no prefix, no ?provider=


@pytest.mark.asyncio
async def test_model_deny_list():
    gw, _ = make_gateway(env={"DISALLOWED_MODELS": "m-one"})
    async with gw_client(gw) as c:
        r = await c.post("/v1/chat/completions",
                         json={"model": "groq/m-one", "messages": [
                             {"role": "user", "content": "x"}]})
        assert r.status_code == 403


@pytest.mark.asyncio
async def test_streaming_relay_sse():
    gw, _ = make_gateway()
    async with gw_client(gw) as c:
        async with c.stream("POST", "/v1/chat/completions",
                            json={"model": "groq/m-one", "stream": True,
                                  "messages": [{"role": "user",
                                                "content": "x"}]}) as r:
            assert r.status_code == 200
            assert r.headers["content-type"].startswith("text/event-stream")
            events = []
            async for line in r.aiter_lines():
                if line.startswith("data:"):
                    events.append(line[5:].strip())
    assert events[-1] == "[DONE]"
    assert sum(1 for e in events if e == "[DONE]") == 1
    first = json.loads(events[0])
    assert first["choices"][0]["delta"]["content"] == "hel"


@pytest.mark.asyncio
async def test_proxy_handler_strips_caller_auth():
    gw, calls = make_gateway()
    async with gw_client(gw) as c:
        r = await c.get("/proxy/groq/models",
                        headers={"Authorization": "Bearer caller"})
        assert r.status_code == 200
    req = calls[-1]
    assert req.headers["authorization"] == "Bearer groq-token"


@pytest.mark.asyncio
async def test_proxy_forwards_query_string():
    gw, calls = make_gateway()
    async with gw_client(gw) as c:
        r = await c.get("/proxy/groq/models", params={"limit": "5",
                                                      "after": "x"})
        assert r.status_code == 200
    hit = [x for x in calls if "models" in x.url.path][-1]
    assert hit.url.params.get("limit") == "5"
    assert hit.url.params.get("after") == "x"


@pytest.mark.asyncio
async def test_messages_passthrough_requires_anthropic():
    gw, calls = make_gateway()
    async with gw_client(gw) as c:
        r = await c.post("/v1/messages",
                         json={"model": "anthropic/claude-3", "messages": []})
        assert r.status_code == 200
        req = calls[-1]
        assert req.url.path.endswith("/messages")
        assert req.headers["x-api-key"] == "anthro-key"
        assert req.headers["anthropic-version"] == "2023-06-01"
        # wrong provider -> anthropic-shaped 400
        r = await c.post("/v1/messages",
                         json={"model": "groq/m", "messages": []})
        assert r.status_code == 400
        assert r.json()["type"] == "error"


@pytest.mark.asyncio
async def test_body_size_cap():
    gw, _ = make_gateway(env={"SERVER_MAX_REQUEST_BODY_SIZE": "100"})
    async with gw_client(gw) as c:
        r = await c.post("/v1/chat/completions", content=b"x" * 200,
                         headers={"content-type": "application/json"})
        assert r.status_code == 413


@pytest.mark.asyncio
async def test_routing_pools_round_robin(tmp_path):
    pool = tmp_path / "routing.yaml"
    pool.write_text("""
models:
  fast:
    strategy: round_robin
    deployments:
      - provider: groq
        model: m-one
      - provider: anthropic
        model: m-two
""")
    gw, calls = make_gateway(env={"ROUTING_ENABLED": "true",
                                  "ROUTING_CONFIG_PATH": str(pool)})
    async with gw_client(gw) as c:
        hosts = []
        for _ in range(4):
            r = await c.post("/v1/chat/completions",
                             json={"model": "fast",
                                   "messages": [{"role": "user",
                                                 "content": "x"}]})
            assert r.status_code == 200
            hosts.append(calls[-1].url.host)
    assert hosts == ["api.groq.com", "api.anthropic.com",
                     "api.groq.com", "api.anthropic.com"]


@pytest.mark.asyncio
async def test_images_gated_by_flag():
    gw, _ = make_gateway()
    async with gw_client(gw) as c:
        r = await c.post("/v1/images/generations", json={"prompt": "x"})
        assert r.status_code == 404


@pytest.mark.asyncio
async def test_local_models_report_context_window():
    """The local provider's own context limit survives enrichment
    (provider-supplied values are preserved, not nulled)."""
    from inference_gateway_amd.config import LocalConfig
    from inference_gateway_amd.providers.local import LocalProvider

    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=2,
                     max_model_len=256)
    lc.num_gpu_blocks = 32
    provider = LocalProvider(lc, device="cpu")
    try:
        gw, _ = make_gateway(local_provider=provider)
        async with gw_client(gw) as c:
            r = await c.get("/v1/models", params={
                "provider": "local", "include": "context_window"})
            assert r.status_code == 200
            assert r.json()["data"][0]["context_window"] == 256
    finally:
        provider.engine.shutdown()


@pytest.mark.asyncio
async def test_models_include_enrichment():
    """?include=context_window,pricing,modalities resolves provider >
    community precedence; unknown values render as explicit null."""
    gw, _ = make_gateway()
    async with gw_client(gw) as c:
        r = await c.get("/v1/models?provider=groq"
                        "&include=context_window,pricing,modalities")
    assert r.status_code == 200
    models = r.json()["data"]
    for m in models:
        assert "context_window" in m and "pricing" in m and "modalities" in m
        # mock upstream models are unknown to the community tables => null
        assert m["context_window"] is None


@pytest.mark.asyncio
async def test_models_include_community_table_hit():
    from inference_gateway_amd.providers import metadata

    assert metadata.community_context_window("openai/gpt-4o") == 128000
    assert metadata.community_pricing("openai/gpt-4o")["output"] == 1e-5
    assert metadata.model_accepts_images("openai/gpt-4o")
    assert not metadata.model_accepts_images("openai/gpt-3.5-turbo")
    assert metadata.model_accepts_images("totally/unknown-model")


@pytest.mark.asyncio
async def test_images_multipart_model_rewrite():
    """Multipart edits: provider prefix is stripped from the model field."""
    gw, calls = make_gateway(env={"OPENAI_API_KEY": "sk-x",
                                  "ENABLE_IMAGES": "true"})
    boundary = "XbOuNdArYx"
    parts = (
        f"--{boundary}\r\n"
        'Content-Disposition: form-data; name="model"\r\n\r\n'
        "openai/dall-e-2\r\n"
        f"--{boundary}\r\n"
        'Content-Disposition: form-data; name="image"; filename="a.png"\r\n'
        "Content-Type: image/png\r\n\r\n"
        "PNGBYTES\r\n"
        f"--{boundary}--\r\n"
    ).encode()
    async with gw_client(gw) as c:
        r = await c.post(
            "/v1/images/edits", content=parts,
            headers={"content-type":
                     f"multipart/form-data; boundary={boundary}"})
    # mock upstream 404s unknown paths — we only care what was SENT
    sent = [req for req in calls if "images" in req.url.path]
    assert sent, r.text
    body = sent[-1].content
    assert b"openai/dall-e-2" not in body
    assert b"dall-e-2" in body
    assert b"PNGBYTES" in body


def test_dev_truncation():
    from inference_gateway_amd.gateway.middlewares.dev_mw import (
        truncate_chat_body)

    body = {"model": "m", "messages": [
        {"role": "user", "content": "x" * 5000}]}
    out = truncate_chat_body(body)
    assert len(out["messages"][0]["content"]) < 300
    assert "5000 bytes" in out["messages"][0]["content"]


@pytest.mark.asyncio
async def test_runtime_context_window_lookups():
    """llama.cpp /props and Ollama /api/show runtime lookups win over the
    community table (reference api/context_window.go:28-137)."""
    def handler(request: httpx.Request) -> httpx.Response:
        path = request.url.path
        if path.endswith("/models"):
            return httpx.Response(200, json={"object": "list", "data": [
                {"id": "some-gguf", "object": "model"}]})
        if path.endswith("/props"):
            return httpx.Response(200, json={
                "default_generation_settings": {"n_ctx": 4096}})
        if path.endswith("/api/show"):
            return httpx.Response(200, json={
                "model_info": {"llama.context_length": 32768}})
        return httpx.Response(404)

    cfg = config_mod.load({"LLAMACPP_API_KEY": "t"})  # ollama needs no key
    client = httpx.AsyncClient(transport=httpx.MockTransport(handler))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client)
    async with gw_client(gw) as c:
        r = await c.get("/v1/models?provider=llamacpp&include=context_window")
        assert r.status_code == 200
        assert r.json()["data"][0]["context_window"] == 4096
        r = await c.get("/v1/models?provider=ollama&include=context_window")
        assert r.json()["data"][0]["context_window"] == 32768


@pytest.mark.asyncio
async def test_responses_passthrough_and_sse():
    """/v1/responses: byte passthrough (model prefix stripped) in JSON and
    SSE modes (reference routes.go:1018-1179)."""
    seen = {}

    def handler(request: httpx.Request) -> httpx.Response:
        seen["path"] = request.url.path
        seen["body"] = json.loads(request.content)
        if seen["body"].get("stream"):
            chunks = ["event: response.output_text.delta",
                      'data: {"type":"response.output_text.delta",'
                      '"delta":"hi"}',
                      "event: response.completed",
                      'data: {"type":"response.completed"}']
            return httpx.Response(
                200, headers={"content-type": "text/event-stream"},
                content="\n".join(chunks).encode())
        return httpx.Response(200, json={"id": "resp_1",
                                         "object": "response",
                                         "output": []})

    cfg = config_mod.load({"OPENAI_API_KEY": "sk"})
    client = httpx.AsyncClient(transport=httpx.MockTransport(handler))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client)
    async with gw_client(gw) as c:
        r = await c.post("/v1/responses",
                         json={"model": "openai/gpt-4o", "input": "hello"})
        assert r.status_code == 200
        assert r.json()["object"] == "response"
        assert seen["path"].endswith("/responses")
        assert seen["body"]["model"] == "gpt-4o"  # prefix stripped

        lines = []
        async with c.stream("POST", "/v1/responses", json={
                "model": "openai/gpt-4o", "input": "hello",
                "stream": True}) as resp:
            assert resp.status_code == 200
            assert "text/event-stream" in resp.headers["content-type"]
            async for line in resp.aiter_lines():
                if line:
                    lines.append(line)
        assert any("response.completed" in l for l in lines)

        # wrong provider rejected with an OpenAI-shaped error
        r = await c.post("/v1/responses",
                         json={"model": "groq/m1", "input": "x"})
        assert r.status_code == 400
        assert "error" in r.json()


@pytest.mark.asyncio
async def test_messages_passthrough_sse():
    """Anthropic /v1/messages streaming: SSE relayed verbatim."""
    def handler(request: httpx.Request) -> httpx.Response:
        body = json.loads(request.content)
        assert body["model"] == "claude-3-5-sonnet"  # prefix stripped
        if body.get("stream"):
            frames = ["event: message_start\n"
                      'data: {"type":"message_start"}',
                      "event: content_block_delta\n"
                      'data: {"type":"content_block_delta",'
                      '"delta":{"type":"text_delta","text":"hi"}}',
                      "event: message_stop\n"
                      'data: {"type":"message_stop"}']
            return httpx.Response(
                200, headers={"content-type": "text/event-stream"},
                content=("\n\n".join(frames) + "\n\n").encode())
        return httpx.Response(200, json={"type": "message"})

    cfg = config_mod.load({"ANTHROPIC_API_KEY": "k"})
    client = httpx.AsyncClient(transport=httpx.MockTransport(handler))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client)
    async with gw_client(gw) as c:
        lines = []
        async with c.stream("POST", "/v1/messages", json={
                "model": "anthropic/claude-3-5-sonnet",
                "messages": [], "stream": True}) as r:
            assert r.status_code == 200
            async for line in r.aiter_lines():
                if line:
                    lines.append(line)
    assert any("message_stop" in l for l in lines)
    assert any("text_delta" in l for l in lines)



@pytest.mark.asyncio
async def test_chat_request_shape_validation():
    """Typed wire validation (the reference's generated-types decode
    400s): wrong-typed KNOWN fields are a 400; unknown fields pass."""
    gw, _ = make_gateway()
    bads = [
        ({"model": "groq/m", "messages": "hi"},
         "messages must be"),
        ({"model": "groq/m", "messages": []}, "non-empty"),
        ({"model": "groq/m", "messages": [{"role": "robot",
                                           "content": "x"}]}, "role"),
        ({"model": "groq/m", "messages": [{"role": "user",
                                           "content": 5}]}, "content"),
        ({"model": "groq/m", "messages": [{"role": "user",
                                           "content": "x"}],
          "temperature": "hot"}, "temperature must be a number"),
        ({"model": "groq/m", "messages": [{"role": "user",
                                           "content": "x"}],
          "max_tokens": 3.5}, "max_tokens must be an integer"),
        ({"model": "groq/m", "messages": [{"role": "user",
                                           "content": "x"}],
          "stream": "yes"}, "stream must be a boolean"),
        ({"model": "groq/m", "messages": [{"role": "user",
                                           "content": "x"}],
          "stop": [1, 2]}, "stop"),
        ({"model": "groq/m", "messages": [{"role": "user",
                                           "content": "x"}],
          "stop": ["a", "b", "c", "d", "e"]}, "at most 4"),
        ({"model": "groq/m", "messages": [{"role": "user",
                                           "content": "x"}],
          "logit_bias": [1, 2]}, "logit_bias must be an object"),
        ({"model": "groq/m", "messages": [{"role": "user",
                                           "content": "x"}],
          "logit_bias": {"abc": 1}}, "token ids"),
        ({"model": "groq/m", "messages": [{"role": "user",
                                           "content": "x"}],
          "logit_bias": {"5": "big"}}, "values must be numbers"),
    ]
    async with gw_client(gw) as c:
        for body, frag in bads:
            r = await c.post("/v1/chat/completions", json=body)
            assert r.status_code == 400, (body, r.status_code)
            assert frag in str(r.json()["error"]), (body, r.json())
        # unknown fields and multimodal parts still pass through
        r = await c.post("/v1/chat/completions", json={
            "model": "groq/m", "response_format": {"type": "json"},
            "whatever": 1,
            "messages": [{"role": "user", "content": [
                {"type": "text", "text": "hi"}]}]})
        assert r.status_code == 200


@pytest.mark.asyncio
async def test_relay_sse_event_grouping():
    """Multi-line SSE frames (Anthropic event:+data:) survive the relay
    as ONE frame each — the writer adds one blank separator per yielded
    chunk, so the relay must group lines by event boundary."""
    from inference_gateway_amd.gateway.routes import _relay_sse_events

    async def upstream_lines():
        for line in ["event: message_start",
                     'data: {"a":1}',
                     "",
                     "event: delta",
                     'data: {"b":2}',
                     "",
                     "data: solo",
                     ""]:
            yield line

    evs = [e async for e in _relay_sse_events(upstream_lines())]
    assert evs == ['event: message_start\ndata: {"a":1}',
                   'event: delta\ndata: {"b":2}',
                   "data: solo"]


def test_transform_tolerates_cohere_native_shape():
    """Cohere's native /v1/models returns {"models":[{"name":...}]};
    the transformer maps it instead of returning an empty list (the
    reference's generated transformer silently drops these)."""
    from inference_gateway_amd.providers.transformers import (
        transform_list_models)

    out = transform_list_models("cohere", {
        "models": [{"name": "command-r-plus", "context_length": 128000}]})
    assert out["data"][0]["id"] == "cohere/command-r-plus"
    assert out["data"][0]["served_by"] == "cohere"
