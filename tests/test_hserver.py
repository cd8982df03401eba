"""Native asyncio HTTP server (gateway/hserver.py) over a real TCP
socket: request parsing, chunked SSE streaming, keep-alive, body cap,
malformed input, concurrent streams.  The reference's SSE liveness bar
(tests/streaming_test.go:21-90: per-line flush, streams outlive
timeouts) is covered by the incremental-arrival assertions."""
import asyncio
import json
from contextlib import asynccontextmanager

import httpx
import pytest

from inference_gateway_amd import config as config_mod
from inference_gateway_amd.config import LocalConfig
from inference_gateway_amd.gateway.app import Gateway
from inference_gateway_amd.gateway.hserver import HttpServer
from inference_gateway_amd.logger import noop_logger
from inference_gateway_amd.providers.local import LocalProvider


@pytest.fixture(scope="module")
def provider():
    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=16,
                     max_model_len=256)
    lc.num_gpu_blocks = 128
    p = LocalProvider(lc, device="cpu")
    yield p
    p.engine.shutdown()


@asynccontextmanager
async def serve(provider):
    """Start the native server on an ephemeral port inside the test's
    own event loop (no pytest-asyncio: fixtures cannot be async)."""
    cfg = config_mod.load({})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(404)))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider)
    srv = HttpServer(gw, "127.0.0.1", 0, max_body=1 << 20)
    await srv.start()
    try:
        yield f"http://127.0.0.1:{srv.port}"
    finally:
        await srv.stop()
        await client.aclose()


@pytest.mark.asyncio
async def test_health_and_keep_alive(provider):
    async with serve(provider) as base:
        async with httpx.AsyncClient(base_url=base) as c:
            r1 = await c.get("/health")
            assert r1.status_code == 200
            # same pooled connection: second request reuses it
            r2 = await c.get("/health")
            assert r2.status_code == 200
            r3 = await c.get("/v1/models", params={"provider": "local"})
            assert r3.status_code == 200
            assert r3.json()["data"][0]["id"] == "local/tiny-llama"


@pytest.mark.asyncio
async def test_chat_completion_over_socket(provider):
    async with serve(provider) as base:
        async with httpx.AsyncClient(base_url=base, timeout=60) as c:
            r = await c.post("/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "hi"}],
                "max_tokens": 5, "temperature": 0})
            assert r.status_code == 200
            assert r.json()["usage"]["completion_tokens"] == 5


@pytest.mark.asyncio
async def test_explicit_null_sampling_fields_use_defaults(provider):
    """OpenAI clients send `"temperature": null` to mean "default";
    validation lets null through and the local provider must not 500."""
    async with serve(provider) as base:
        async with httpx.AsyncClient(base_url=base, timeout=60) as c:
            r = await c.post("/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "hi"}],
                "max_tokens": 3, "temperature": None, "top_p": None,
                "top_k": None, "seed": None, "stop": None,
                "presence_penalty": None})
            assert r.status_code == 200
            assert r.json()["usage"]["completion_tokens"] == 3


@pytest.mark.asyncio
async def test_logit_bias_over_socket(provider):
    async with serve(provider) as base:
        async with httpx.AsyncClient(base_url=base, timeout=60) as c:
            r = await c.post("/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "hi"}],
                "max_tokens": 3, "temperature": 0,
                "logit_bias": {"9": 100}})
            assert r.status_code == 200
            # out-of-vocab token id -> 400, not a 500
            r = await c.post("/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "hi"}],
                "max_tokens": 3, "logit_bias": {"99999": 1}})
            assert r.status_code == 400
            assert "vocabulary" in str(r.json())


@pytest.mark.asyncio
async def test_sse_streams_incrementally(provider):
    """Chunks must arrive one event at a time (chunked framing, per-event
    flush), ending with [DONE]."""
    events = []
    async with serve(provider) as base:
        async with httpx.AsyncClient(base_url=base, timeout=60) as c:
            async with c.stream("POST", "/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "hi"}],
                "max_tokens": 6, "temperature": 0, "stream": True,
            }) as r:
                assert r.status_code == 200
                assert r.headers["content-type"].startswith(
                    "text/event-stream")
                assert r.headers.get("transfer-encoding") == "chunked"
                async for line in r.aiter_lines():
                    if line.startswith("data:"):
                        events.append(line[5:].strip())
    assert events[-1] == "[DONE]"
    body = json.loads(events[-2])
    assert body["usage"]["completion_tokens"] == 6
    content = "".join(
        json.loads(e)["choices"][0]["delta"].get("content") or ""
        for e in events[:-1])
    assert isinstance(content, str)


@pytest.mark.asyncio
async def test_concurrent_streams(provider):
    async def one(c, i):
        n = 0
        async with c.stream("POST", "/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": f"req {i}"}],
            "max_tokens": 5, "temperature": 0.8, "stream": True,
        }) as r:
            assert r.status_code == 200
            async for line in r.aiter_lines():
                if line.startswith("data:") and "[DONE]" not in line:
                    n += 1
        return n

    async with serve(provider) as base:
        async with httpx.AsyncClient(base_url=base, timeout=120) as c:
            counts = await asyncio.gather(*(one(c, i) for i in range(8)))
    assert all(n >= 2 for n in counts)


@pytest.mark.asyncio
async def test_direct_stream_equivalence_and_telemetry(provider):
    """The direct-write SSE path (DirectStream) must produce the same
    content as the non-streaming completion for the same seeded request,
    carry exact usage, and report tokens through the telemetry hooks."""
    from inference_gateway_amd.otel.metrics import Telemetry

    telemetry = Telemetry()
    cfg = config_mod.load({})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(404)))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider, telemetry=telemetry)
    srv = HttpServer(gw, "127.0.0.1", 0)
    await srv.start()
    base = f"http://127.0.0.1:{srv.port}"
    try:
        req = {"model": "local/tiny-llama",
               "messages": [{"role": "user", "content": "equiv"}],
               "max_tokens": 8, "temperature": 0.7, "seed": 42}
        async with httpx.AsyncClient(base_url=base, timeout=60) as c:
            r = await c.post("/v1/chat/completions", json=req)
            want = r.json()["choices"][0]["message"]["content"]
            events = []
            async with c.stream("POST", "/v1/chat/completions",
                                json={**req, "stream": True}) as rs:
                assert rs.status_code == 200
                async for line in rs.aiter_lines():
                    if line.startswith("data:"):
                        events.append(line[5:].strip())
        assert events[-1] == "[DONE]"
        final = json.loads(events[-2])
        assert final["choices"][0]["finish_reason"] == "length"
        assert final["usage"]["completion_tokens"] == 8
        got = "".join(
            json.loads(e)["choices"][0]["delta"].get("content") or ""
            for e in events[:-1])
        assert got == want
        expo = telemetry.exposition().decode()
        assert "gen_ai_client_token_usage" in expo
    finally:
        await srv.stop()
        await client.aclose()


@pytest.mark.asyncio
async def test_direct_stream_stop_string(provider):
    """Stop-string semantics through the direct path: the stop sequence
    never reaches the wire and finish_reason is 'stop'."""
    async with serve(provider) as base:
        async with httpx.AsyncClient(base_url=base, timeout=60) as c:
            # find deterministic output first, derive a stop from it
            r = await c.post("/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "stopteststop"}],
                "max_tokens": 12, "temperature": 0, "seed": 1})
            text = r.json()["choices"][0]["message"]["content"]
            if len(text) < 4:
                pytest.skip("output too short to derive a stop string")
            stop = text[2:4]
            events = []
            async with c.stream("POST", "/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "stopteststop"}],
                "max_tokens": 12, "temperature": 0, "seed": 1,
                "stop": [stop], "stream": True,
            }) as rs:
                async for line in rs.aiter_lines():
                    if line.startswith("data:"):
                        events.append(line[5:].strip())
            got = "".join(
                json.loads(e)["choices"][0]["delta"].get("content") or ""
                for e in events[:-1])
            fins = [json.loads(e)["choices"][0]["finish_reason"]
                    for e in events[:-1]]
            assert stop not in got
            assert got == text[:text.find(stop)]
            assert fins[-1] == "stop"


@pytest.mark.asyncio
async def test_body_cap_413(provider):
    async with serve(provider) as base:
        async with httpx.AsyncClient(base_url=base, timeout=30) as c:
            r = await c.post("/v1/chat/completions",
                             content=b"x" * (2 << 20),
                             headers={"content-type": "application/json"})
            assert r.status_code == 413


@pytest.mark.asyncio
async def test_malformed_request_400(provider):
    async with serve(provider) as base:
        host, port = base.rsplit("//", 1)[1].split(":")
        reader, writer = await asyncio.open_connection(host, int(port))
        writer.write(b"NONSENSE\r\n\r\n")
        await writer.drain()
        line = await asyncio.wait_for(reader.readline(), 10)
        assert b"400" in line
        writer.close()


@pytest.mark.asyncio
async def test_client_disconnect_aborts_engine_request(provider):
    """Closing the socket mid-stream must cancel the handler task and
    abort the engine-side sequence (no stranded running sequences)."""
    async with serve(provider) as base:
        host, port = base.rsplit("//", 1)[1].split(":")
        reader, writer = await asyncio.open_connection(host, int(port))
        body = json.dumps({
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hi"}],
            "max_tokens": 100, "temperature": 0.8, "stream": True,
        }).encode()
        writer.write(
            b"POST /v1/chat/completions HTTP/1.1\r\n"
            b"host: x\r\ncontent-type: application/json\r\n"
            b"content-length: %d\r\n\r\n" % len(body) + body)
        await writer.drain()
        # read a few bytes (stream started), then drop the connection
        await asyncio.wait_for(reader.read(512), 30)
        writer.close()
        for _ in range(100):
            if not provider.engine.engine.scheduler.has_work:
                break
            await asyncio.sleep(0.1)
        assert not provider.engine.engine.scheduler.has_work


@pytest.mark.asyncio
async def test_tls_serving(provider, tmp_path):
    """SERVER_TLS_*: the native server terminates TLS (the reference
    serves HTTPS when TLS paths are set, main.go:334-350).  Self-signed
    cert generated on the fly; client pins it as its CA."""
    import ssl
    import subprocess

    cert = tmp_path / "cert.pem"
    key = tmp_path / "key.pem"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(key), "-out", str(cert), "-days", "1",
         "-subj", "/CN=127.0.0.1",
         "-addext", "subjectAltName=IP:127.0.0.1"],
        check=True, capture_output=True)
    from inference_gateway_amd.gateway.hserver import build_ssl_context

    cfg = config_mod.load({})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(404)))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider)
    srv = HttpServer(gw, "127.0.0.1", 0,
                     ssl_context=build_ssl_context(str(cert), str(key)))
    await srv.start()
    try:
        ctx = ssl.create_default_context(cafile=str(cert))
        async with httpx.AsyncClient(
                base_url=f"https://127.0.0.1:{srv.port}",
                verify=ctx, timeout=60) as c:
            r = await c.get("/health")
            assert r.status_code == 200
            # SSE over TLS, direct path
            events = 0
            async with c.stream("POST", "/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "tls"}],
                "max_tokens": 4, "temperature": 0, "stream": True,
            }) as rs:
                assert rs.status_code == 200
                async for line in rs.aiter_lines():
                    if line.startswith("data:"):
                        events += 1
            assert events >= 3          # role + content(+) + usage + DONE
    finally:
        await srv.stop()
        await client.aclose()


@pytest.mark.asyncio
async def test_soak_64_streams_over_socket(provider):
    """Reference soak bar over the REAL socket + direct-write path: 64
    concurrent SSE streams, every stream completes with exact usage and
    exactly one [DONE]."""
    async def one(c, i):
        dones = 0
        usage = None
        async with c.stream("POST", "/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": f"soak {i}"}],
            "max_tokens": 6, "temperature": 0.9, "seed": i,
            "stream": True,
        }) as r:
            assert r.status_code == 200
            async for line in r.aiter_lines():
                if not line.startswith("data:"):
                    continue
                d = line[5:].strip()
                if d == "[DONE]":
                    dones += 1
                    continue
                u = json.loads(d).get("usage")
                usage = u or usage
        return dones, usage

    async with serve(provider) as base:
        async with httpx.AsyncClient(base_url=base, timeout=300) as c:
            results = await asyncio.gather(*(one(c, i)
                                             for i in range(64)))
    for dones, usage in results:
        assert dones == 1
        assert usage and usage["completion_tokens"] == 6


@pytest.mark.asyncio
async def test_engine_death_mid_stream_emits_error_frame():
    """ADVICE r1 (medium), end to end over the socket: an engine-thread
    crash mid-stream must surface as an in-stream error frame on the
    direct-write path — never a clean finish_reason 'stop'."""
    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=4,
                     max_model_len=256)
    lc.num_gpu_blocks = 64
    p = LocalProvider(lc, device="cpu")
    try:
        async with serve(p) as base:
            events = []

            def boom():
                raise RuntimeError("injected mid-stream fault")

            async with httpx.AsyncClient(base_url=base, timeout=60) as c:
                async with c.stream("POST", "/v1/chat/completions", json={
                    "model": "local/tiny-llama",
                    "messages": [{"role": "user", "content": "die"}],
                    "max_tokens": 200, "temperature": 0.8,
                    "stream": True,
                }) as r:
                    assert r.status_code == 200
                    n = 0
                    async for line in r.aiter_lines():
                        if not line.startswith("data:"):
                            continue
                        events.append(line[5:].strip())
                        n += 1
                        if n == 3:
                            p.engine.engine.step = boom
            errs = [e for e in events if e != "[DONE]"
                    and "error" in json.loads(e)]
            assert errs, events[-3:]
            assert "injected mid-stream fault" in errs[-1]
            # and no clean finish: none of the frames said 'stop'
            fins = [json.loads(e)["choices"][0].get("finish_reason")
                    for e in events if e != "[DONE]"
                    and "choices" in json.loads(e)]
            assert "stop" not in fins
    finally:
        p.engine.shutdown()


@pytest.mark.asyncio
async def test_direct_stream_logprobs_over_socket(provider):
    """OpenAI logprobs ride the direct-write path: per-chunk
    logprobs.content entries with top_logprobs."""
    async with serve(provider) as base:
        async with httpx.AsyncClient(base_url=base, timeout=60) as c:
            entries = []
            async with c.stream("POST", "/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "lp"}],
                "max_tokens": 5, "temperature": 0.5, "seed": 3,
                "logprobs": True, "top_logprobs": 3, "stream": True,
            }) as r:
                assert r.status_code == 200
                async for line in r.aiter_lines():
                    if not line.startswith("data:"):
                        continue
                    d = line[5:].strip()
                    if d == "[DONE]":
                        continue
                    ch = json.loads(d)["choices"][0]
                    lp = ch.get("logprobs")
                    if lp:
                        entries.extend(lp["content"])
    assert len(entries) == 5
    for e in entries:
        assert isinstance(e["logprob"], float) and e["logprob"] <= 0
        assert len(e["top_logprobs"]) == 3


@pytest.mark.asyncio
async def test_chunked_request_body(provider):
    """Transfer-Encoding: chunked request bodies decode correctly (the
    reference's Go server accepts them natively)."""
    async with serve(provider) as base:
        host, port = base.rsplit("//", 1)[1].split(":")
        body = json.dumps({
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "chunked"}],
            "max_tokens": 3, "temperature": 0}).encode()
        mid = len(body) // 2
        frames = (b"%x\r\n" % mid + body[:mid] + b"\r\n"
                  + b"%x\r\n" % (len(body) - mid) + body[mid:] + b"\r\n"
                  + b"0\r\n\r\n")
        reader, writer = await asyncio.open_connection(host, int(port))
        writer.write(
            b"POST /v1/chat/completions HTTP/1.1\r\nhost: x\r\n"
            b"content-type: application/json\r\n"
            b"transfer-encoding: chunked\r\n\r\n")
        await writer.drain()
        await asyncio.sleep(0.05)          # split across TCP segments
        writer.write(frames[:10])
        await writer.drain()
        await asyncio.sleep(0.05)
        writer.write(frames[10:])
        await writer.drain()
        data = b""
        while b"\r\n\r\n" not in data or b"completion_tokens" not in data:
            chunk = await asyncio.wait_for(reader.read(65536), 30)
            if not chunk:
                break
            data += chunk
        writer.close()
        assert b"200" in data.split(b"\r\n", 1)[0]
        assert b'"completion_tokens":3' in data.replace(b" ", b"")


@pytest.mark.asyncio
async def test_incomplete_header_times_out(provider):
    """Slowloris guard: a trickled, never-completed request head gets a
    408 and the connection is closed."""
    cfg = config_mod.load({})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(404)))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider)
    srv = HttpServer(gw, "127.0.0.1", 0, header_timeout=0.3)
    await srv.start()
    try:
        reader, writer = await asyncio.open_connection("127.0.0.1",
                                                       srv.port)
        writer.write(b"GET /health HTTP/1.1\r\nhost: x")  # never finishes
        await writer.drain()
        data = await asyncio.wait_for(reader.read(4096), 10)
        assert b"408" in data
        eof = await asyncio.wait_for(reader.read(4096), 10)
        assert eof == b""
        writer.close()
    finally:
        await srv.stop()
        await client.aclose()


@pytest.mark.asyncio
async def test_incomplete_body_times_out(provider):
    """The read deadline covers the BODY too: a complete head with a
    trickled, never-completed Content-Length body gets a 408 (slow-body
    slowloris variant)."""
    cfg = config_mod.load({})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(404)))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider)
    srv = HttpServer(gw, "127.0.0.1", 0, header_timeout=0.3)
    await srv.start()
    try:
        reader, writer = await asyncio.open_connection("127.0.0.1",
                                                       srv.port)
        writer.write(b"POST /health HTTP/1.1\r\nhost: x\r\n"
                     b"content-length: 100\r\n\r\nonly-a-few-bytes")
        await writer.drain()
        data = await asyncio.wait_for(reader.read(4096), 10)
        assert b"408" in data
        writer.close()
    finally:
        await srv.stop()
        await client.aclose()


@pytest.mark.asyncio
async def test_bad_content_length_rejected(provider):
    """Negative and conflicting-duplicate Content-Length headers are
    400-rejected (RFC 7230 request-smuggling defenses)."""
    async with serve(provider) as base:
        host, port = base[len("http://"):].split(":")
        for head in (b"POST /health HTTP/1.1\r\nhost: x\r\n"
                     b"content-length: -5\r\n\r\n",
                     b"POST /health HTTP/1.1\r\nhost: x\r\n"
                     b"content-length: 4\r\ncontent-length: 9\r\n"
                     b"\r\nabcd"):
            reader, writer = await asyncio.open_connection(host,
                                                           int(port))
            writer.write(head)
            await writer.drain()
            data = await asyncio.wait_for(reader.read(4096), 10)
            assert b"400" in data.split(b"\r\n", 1)[0]
            writer.close()


@pytest.mark.asyncio
async def test_pipelined_requests_serialize(provider):
    """Two requests written back-to-back on one connection (the second
    arriving while the first streams) must produce two well-formed
    responses in order — never interleaved writes."""
    async with serve(provider) as base:
        host, port = base.rsplit("//", 1)[1].split(":")
        reader, writer = await asyncio.open_connection(host, int(port))
        b1 = json.dumps({"model": "local/tiny-llama",
                         "messages": [{"role": "user", "content": "a"}],
                         "max_tokens": 4, "temperature": 0,
                         "stream": True}).encode()
        b2 = json.dumps({"model": "local/tiny-llama",
                         "messages": [{"role": "user", "content": "b"}],
                         "max_tokens": 2, "temperature": 0}).encode()
        writer.write(
            b"POST /v1/chat/completions HTTP/1.1\r\nhost: x\r\n"
            b"content-type: application/json\r\n"
            b"content-length: %d\r\n\r\n" % len(b1) + b1
            + b"POST /v1/chat/completions HTTP/1.1\r\nhost: x\r\n"
            b"content-type: application/json\r\n"
            b"content-length: %d\r\n\r\n" % len(b2) + b2)
        await writer.drain()
        data = b""
        deadline = asyncio.get_event_loop().time() + 60
        while data.count(b"HTTP/1.1 200") < 2:
            assert asyncio.get_event_loop().time() < deadline, data[-400:]
            chunk = await asyncio.wait_for(reader.read(65536), 30)
            if not chunk:
                break
            data += chunk
            if (b"[DONE]" in data
                    and data.count(b"HTTP/1.1 200") >= 2
                    and b'"completion_tokens":2' in data.replace(b" ", b"")):
                break
        writer.close()
        # first response: SSE with [DONE]; second: JSON completion
        assert data.count(b"HTTP/1.1 200") == 2
        first, second = data.split(b"HTTP/1.1 200", 2)[1:]
        assert b"[DONE]" in first
        assert b'"chat.completion"' in second.replace(b" ", b"")


@pytest.mark.asyncio
async def test_graceful_stop_drains_inflight(provider):
    """stop(grace=N) lets an in-flight stream finish before closing
    (reference server.Shutdown(5s) parity)."""
    cfg = config_mod.load({})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(404)))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider)
    srv = HttpServer(gw, "127.0.0.1", 0)
    await srv.start()
    base = f"http://127.0.0.1:{srv.port}"
    events = []

    async def consume():
        async with httpx.AsyncClient(base_url=base, timeout=60) as c:
            async with c.stream("POST", "/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "drain"}],
                "max_tokens": 8, "temperature": 0, "stream": True,
            }) as r:
                async for line in r.aiter_lines():
                    if line.startswith("data:"):
                        events.append(line[5:].strip())

    task = asyncio.create_task(consume())
    # let the stream start, then stop with grace
    while not events:
        await asyncio.sleep(0.01)
    await srv.stop(grace=10.0)
    await asyncio.wait_for(task, 30)
    await client.aclose()
    assert events[-1] == "[DONE]"
    assert json.loads(events[-2])["usage"]["completion_tokens"] == 8


@pytest.mark.asyncio
async def test_remote_provider_sse_relay_over_socket():
    """The generator relay path (remote providers — NOT the local
    direct-write path) over the real socket: upstream SSE chunks arrive
    framed, exactly one [DONE]."""
    def upstream(request: httpx.Request) -> httpx.Response:
        if request.url.path.endswith("/chat/completions"):
            chunks = [
                'data: {"id":"u1","object":"chat.completion.chunk",'
                '"choices":[{"index":0,"delta":{"content":"re"},'
                '"finish_reason":null}]}',
                'data: {"id":"u1","object":"chat.completion.chunk",'
                '"choices":[{"index":0,"delta":{"content":"mote"},'
                '"finish_reason":"stop"}]}',
                "data: [DONE]",
            ]
            return httpx.Response(
                200, headers={"content-type": "text/event-stream"},
                content="\n\n".join(chunks).encode())
        return httpx.Response(404)

    cfg = config_mod.load({"GROQ_API_KEY": "tok"})
    client = httpx.AsyncClient(transport=httpx.MockTransport(upstream))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client)
    srv = HttpServer(gw, "127.0.0.1", 0)
    await srv.start()
    try:
        events = []
        async with httpx.AsyncClient(
                base_url=f"http://127.0.0.1:{srv.port}",
                timeout=30) as c:
            async with c.stream("POST", "/v1/chat/completions", json={
                "model": "groq/m", "stream": True,
                "messages": [{"role": "user", "content": "x"}],
            }) as r:
                assert r.status_code == 200
                assert r.headers.get("transfer-encoding") == "chunked"
                async for line in r.aiter_lines():
                    if line.startswith("data:"):
                        events.append(line[5:].strip())
        assert events[-1] == "[DONE]"
        assert sum(1 for e in events if e == "[DONE]") == 1
        text = "".join(
            json.loads(e)["choices"][0]["delta"].get("content") or ""
            for e in events[:-1])
        assert text == "remote"
    finally:
        await srv.stop()
        await client.aclose()


@pytest.mark.asyncio
async def test_messages_passthrough_sse_over_socket():
    """Anthropic /v1/messages streaming passthrough over the native
    server: upstream event/data lines relayed verbatim, chunk-framed."""
    def upstream(request: httpx.Request) -> httpx.Response:
        if request.url.path.endswith("/messages"):
            frames = (
                "event: message_start\n"
                'data: {"type":"message_start"}\n\n'
                "event: content_block_delta\n"
                'data: {"type":"content_block_delta","delta":'
                '{"type":"text_delta","text":"hi"}}\n\n'
                "event: message_stop\n"
                'data: {"type":"message_stop"}\n\n')
            return httpx.Response(
                200, headers={"content-type": "text/event-stream"},
                content=frames.encode())
        return httpx.Response(404)

    cfg = config_mod.load({"ANTHROPIC_API_KEY": "k"})
    client = httpx.AsyncClient(transport=httpx.MockTransport(upstream))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client)
    srv = HttpServer(gw, "127.0.0.1", 0)
    await srv.start()
    try:
        lines = []
        async with httpx.AsyncClient(
                base_url=f"http://127.0.0.1:{srv.port}",
                timeout=30) as c:
            async with c.stream("POST", "/v1/messages", json={
                "model": "anthropic/claude-x", "stream": True,
                "max_tokens": 8,
                "messages": [{"role": "user", "content": "x"}],
            }) as r:
                assert r.status_code == 200
                async for line in r.aiter_lines():
                    if line.strip():
                        lines.append(line.strip())
        assert "event: message_start" in lines
        assert any("text_delta" in ln for ln in lines)
        assert "event: message_stop" in lines
        # raw framing: read the same stream as BYTES on a second
        # request — each event: line must share its frame with its
        # data: line (no blank line between them)
        host, port = "127.0.0.1", srv.port
        reader, writer = await asyncio.open_connection(host, port)
        body = json.dumps({"model": "anthropic/claude-x",
                           "stream": True, "max_tokens": 8,
                           "messages": [{"role": "user",
                                         "content": "x"}]}).encode()
        writer.write(b"POST /v1/messages HTTP/1.1\r\nhost: x\r\n"
                     b"content-type: application/json\r\n"
                     b"content-length: %d\r\n\r\n" % len(body) + body)
        await writer.drain()
        raw = b""
        while b"message_stop" not in raw:
            chunk = await asyncio.wait_for(reader.read(65536), 30)
            if not chunk:
                break
            raw += chunk
        writer.close()
        text = raw.decode(errors="replace")
        assert ("event: content_block_delta\ndata:" in text
                or "event: content_block_delta\r\ndata:" in text), text
    finally:
        await srv.stop()
        await client.aclose()


@pytest.mark.asyncio
async def test_reuse_port_balances_two_listeners():
    """SO_REUSEPORT (SERVER_WORKERS mechanics): two listeners on one
    port, kernel distributes distinct connections across both."""
    hits = {"a": 0, "b": 0}

    def app_for(tag):
        async def app(scope, receive, send):
            hits[tag] += 1
            await send({"type": "http.response.start", "status": 200,
                        "headers": [(b"content-length", b"2")]})
            await send({"type": "http.response.body", "body": b"ok"})
        return app

    s1 = HttpServer(app_for("a"), "127.0.0.1", 0, reuse_port=True)
    await s1.start()
    s2 = HttpServer(app_for("b"), "127.0.0.1", s1.port, reuse_port=True)
    await s2.start()
    try:
        for _ in range(32):
            reader, writer = await asyncio.open_connection(
                "127.0.0.1", s1.port)
            writer.write(b"GET /x HTTP/1.1\r\nhost: x\r\n"
                         b"connection: close\r\n\r\n")
            await writer.drain()
            data = await asyncio.wait_for(reader.read(4096), 10)
            assert b"200" in data
            writer.close()
    finally:
        await s1.stop()
        await s2.stop()
    assert hits["a"] + hits["b"] == 32
    assert hits["a"] > 0 and hits["b"] > 0, hits


def test_resolve_workers_clamps_with_local_engine():
    from inference_gateway_amd.server import resolve_workers

    cfg = config_mod.load({"SERVER_WORKERS": "4"})
    assert not cfg.local.enabled
    assert resolve_workers(cfg) == 4
    cfg = config_mod.load({"SERVER_WORKERS": "4", "LOCAL_ENABLED": "true",
                           "LOCAL_MODEL": "tiny-llama"})
    assert resolve_workers(cfg) == 1


@pytest.mark.asyncio
async def test_idle_keepalive_connection_reaped(provider):
    """A keep-alive connection idle past SERVER_IDLE_TIMEOUT is closed
    silently (no 408 — between-requests idling is not an error)."""
    cfg = config_mod.load({})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(404)))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider)
    srv = HttpServer(gw, "127.0.0.1", 0, idle_timeout=0.3)
    await srv.start()
    try:
        reader, writer = await asyncio.open_connection("127.0.0.1",
                                                       srv.port)
        writer.write(b"GET /health HTTP/1.1\r\nhost: x\r\n\r\n")
        await writer.drain()
        data = await asyncio.wait_for(reader.read(4096), 10)
        assert b"200" in data.split(b"\r\n", 1)[0]
        # now idle past the timeout: server closes, NOT with a 408
        rest = await asyncio.wait_for(reader.read(4096), 10)
        assert rest == b""
        writer.close()
    finally:
        await srv.stop()
        await client.aclose()


@pytest.mark.asyncio
async def test_guardrails_on_direct_sse_path(provider, tmp_path):
    """Guardrails + the direct-write fast path: pre_call still blocks
    (403, no stream), clean requests stream via DirectStream, and the
    middleware never runs post_call on a direct session."""
    (tmp_path / "policy.yaml").write_text(
        "rules:\n"
        "  - name: block-bad\n"
        "    phase: pre_call\n"
        "    action: block\n"
        "    contains: [forbidden]\n")
    cfg = config_mod.load({"GUARDRAILS_ENABLED": "true",
                           "GUARDRAILS_POLICY_DIR": str(tmp_path)})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(404)))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider)
    srv = HttpServer(gw, "127.0.0.1", 0)
    await srv.start()
    base = f"http://127.0.0.1:{srv.port}"
    try:
        async with httpx.AsyncClient(base_url=base, timeout=60) as c:
            r = await c.post("/v1/chat/completions", json={
                "model": "local/tiny-llama", "stream": True,
                "messages": [{"role": "user",
                              "content": "forbidden stuff"}]})
            assert r.status_code == 403
            assert r.json()["error"]["type"] == "guardrail_blocked"
            events = []
            async with c.stream("POST", "/v1/chat/completions", json={
                "model": "local/tiny-llama", "stream": True,
                "max_tokens": 4, "temperature": 0,
                "messages": [{"role": "user", "content": "clean"}],
            }) as r:
                assert r.status_code == 200
                async for line in r.aiter_lines():
                    if line.startswith("data:"):
                        events.append(line[5:].strip())
            assert events[-1] == "[DONE]"
            body = json.loads(events[-2])
            assert body["usage"]["completion_tokens"] == 4
    finally:
        await srv.stop()
        await client.aclose()


def test_direct_stream_holdback_equivalence_fuzz(provider):
    """Property: over random token streams, arbitrary flush batching
    and random stop strings, the CONCATENATED streamed content equals
    the non-streaming truncation exactly (UTF-8 holdback + stop-string
    holdback compose; 4000-trial randomized run was clean — this pins
    300 seeded trials)."""
    import random

    from inference_gateway_amd.engine.config import SamplingParams
    from inference_gateway_amd.providers.local import DirectStream

    tok = provider.tokenizer

    class Out:
        def __init__(self, t, fin):
            self.token_id = t
            self.finished = fin
            self.finish_reason = "length"

    def run_stream(tokens, stops, feeds):
        frames = []
        ds = DirectStream(provider, {"stop": stops} if stops else {},
                          [1, 2, 3], SamplingParams(max_new_tokens=99),
                          lambda b: frames.append(b) or True,
                          lambda: None)
        i = 0
        for n in feeds + [len(tokens)]:
            for _ in range(n):
                if i >= len(tokens):
                    break
                ds.feed(Out(tokens[i], i == len(tokens) - 1))
                i += 1
            ds.flush()
        content = ""
        for fr in frames:
            for line in fr.decode().split("\n"):
                if line.startswith("data: {"):
                    d = json.loads(line[6:])
                    content += (d["choices"][0]["delta"].get("content")
                                or "")
        return content

    def reference(tokens, stops):
        # CANONICAL per-token stop semantics (matches the non-streaming
        # _collect): grow token by token, cut at the earliest-start
        # match inside the FIRST prefix containing any stop
        for k in range(1, len(tokens) + 1):
            partial = tok.decode(list(tokens[:k]))
            best = -1
            for st_ in stops:
                j = partial.find(st_)
                if j >= 0 and (best < 0 or j < best):
                    best = j
            if best >= 0:
                return partial[:best]
        return tok.decode(list(tokens))

    rng = random.Random(7)
    for _ in range(300):
        nt = rng.randrange(1, 25)
        tokens = [rng.choice([rng.randrange(3, 259),
                              rng.randrange(3, 259),
                              rng.randrange(259, 400)])
                  for _ in range(nt)]
        stops = []
        if rng.random() < 0.7:
            full = tok.decode(list(tokens))
            for _ in range(rng.randrange(1, 3)):
                if full and rng.random() < 0.6:
                    a = rng.randrange(0, len(full))
                    b = min(len(full), a + rng.randrange(1, 4))
                    if full[a:b]:
                        stops.append(full[a:b])
                else:
                    stops.append(rng.choice(["xy", "<tok", "\u00e9"]))
        feeds = [rng.randrange(1, 5) for _ in range(10)]
        got = run_stream(tokens, stops, feeds)
        assert got == reference(tokens, stops), (tokens, stops)


@pytest.mark.asyncio
async def test_chaos_soak_disconnects_and_garbage(provider):
    """48 concurrent SSE streams with random mid-stream disconnects
    plus raw-garbage connections interleaved: no stream errors, the
    engine drains to idle, every KV block comes back."""
    import random

    async with serve(provider) as base:
        host, port = base[len("http://"):].split(":")
        port = int(port)
        rng = random.Random(11)
        stats = {"done": 0, "aborted": 0, "err": 0}

        async def stream(i):
            try:
                reader, writer = await asyncio.open_connection(host, port)
                body = json.dumps({
                    "model": "local/tiny-llama", "stream": True,
                    "max_tokens": rng.randrange(3, 12),
                    "temperature": 0.8,
                    "messages": [{"role": "user",
                                  "content": f"req {i}"}]}).encode()
                writer.write(
                    b"POST /v1/chat/completions HTTP/1.1\r\nhost: x\r\n"
                    b"content-type: application/json\r\n"
                    b"content-length: %d\r\n\r\n" % len(body) + body)
                await writer.drain()
                if rng.random() < 0.3:
                    await asyncio.sleep(rng.random() * 0.1)
                    writer.close()
                    stats["aborted"] += 1
                    return
                data = b""
                while b"data: [DONE]" not in data:
                    chunk = await asyncio.wait_for(reader.read(65536), 60)
                    if not chunk:
                        break
                    data += chunk
                assert b"200" in data.split(b"\r\n", 1)[0]
                writer.close()
                stats["done"] += 1
            except Exception:
                stats["err"] += 1
                raise

        async def garbage():
            try:
                reader, writer = await asyncio.open_connection(host, port)
                writer.write(bytes(rng.randrange(256)
                                   for _ in range(rng.randrange(1, 60))))
                await writer.drain()
                await asyncio.sleep(0.02)
                writer.close()
            except Exception:
                pass

        await asyncio.gather(*([stream(i) for i in range(48)]
                               + [garbage() for _ in range(8)]))
        assert stats["err"] == 0 and stats["done"] > 0, stats
        # engine drained; all blocks restored
        eng = provider.engine.engine
        for _ in range(200):
            if not eng.scheduler.has_work:
                break
            await asyncio.sleep(0.05)
        assert not eng.scheduler.running and not eng.scheduler.waiting


def test_worker_main_process_serves():
    """SERVER_WORKERS worker entry (_worker_main) in a real spawned
    process: it loads config from env, binds with SO_REUSEPORT and
    serves /health."""
    import multiprocessing as mp
    import os
    import socket
    import time
    import urllib.request

    # reserve a port: bind with SO_REUSEPORT so the worker can share it
    probe = socket.socket()
    probe.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEPORT, 1)
    probe.bind(("127.0.0.1", 0))
    port = probe.getsockname()[1]
    probe.close()

    env = dict(os.environ, SERVER_HOST="127.0.0.1",
               SERVER_PORT=str(port), TELEMETRY_ENABLED="false")
    ctx = mp.get_context("spawn")
    proc = ctx.Process(target=_spawn_worker_entry, args=(env,),
                       daemon=True)
    proc.start()
    try:
        deadline = time.time() + 20
        last = None
        while time.time() < deadline:
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{port}/health",
                        timeout=2) as r:
                    assert r.status == 200
                    assert b"healthy" in r.read()
                    break
            except Exception as e:
                last = e
                time.sleep(0.25)
        else:
            raise AssertionError(f"worker never served: {last}")
    finally:
        proc.terminate()
        proc.join(timeout=10)


def _spawn_worker_entry(env):
    import os

    os.environ.update(env)
    from inference_gateway_amd.server import _worker_main

    _worker_main(1)


@pytest.mark.asyncio
async def test_http10_sse_streams_raw_and_closes(provider):
    """HTTP/1.0 has no chunked transfer-encoding: an SSE response to a
    1.0 client streams raw bytes and ends by connection close."""
    async with serve(provider) as base:
        host, port = base[len("http://"):].split(":")
        reader, writer = await asyncio.open_connection(host, int(port))
        body = json.dumps({"model": "local/tiny-llama", "stream": True,
                           "max_tokens": 4, "temperature": 0,
                           "messages": [{"role": "user",
                                         "content": "x"}]}).encode()
        writer.write(b"POST /v1/chat/completions HTTP/1.0\r\nhost: x\r\n"
                     b"content-type: application/json\r\n"
                     b"content-length: %d\r\n\r\n" % len(body) + body)
        await writer.drain()
        raw = b""
        while True:
            chunk = await asyncio.wait_for(reader.read(65536), 30)
            if not chunk:
                break                     # server closed = end of body
            raw += chunk
        writer.close()
        head, _, payload = raw.partition(b"\r\n\r\n")
        assert b"200" in head.split(b"\r\n", 1)[0]
        assert b"transfer-encoding" not in head.lower()
        assert b"connection: close" in head.lower()
        # raw SSE bytes, no chunk-size hex lines
        assert payload.lstrip().startswith(b"data: ")
        assert b"data: [DONE]" in payload


@pytest.mark.asyncio
async def test_head_request_returns_no_body(provider):
    async with serve(provider) as base:
        host, port = base[len("http://"):].split(":")
        reader, writer = await asyncio.open_connection(host, int(port))
        writer.write(b"HEAD /health HTTP/1.1\r\nhost: x\r\n"
                     b"connection: close\r\n\r\n")
        await writer.drain()
        raw = b""
        while True:
            chunk = await asyncio.wait_for(reader.read(4096), 10)
            if not chunk:
                break
            raw += chunk
        writer.close()
        head, _, payload = raw.partition(b"\r\n\r\n")
        # parity: routes register GET only, so HEAD is 404 (gin does
        # the same) — but the 404 BODY must still be suppressed
        assert head.split(b"\r\n", 1)[0].endswith(b"404 Not Found")
        assert b"healthy" not in payload
        assert b"not found" not in payload.lower()


@pytest.mark.asyncio
async def test_expect_100_continue(provider):
    """Clients sending Expect: 100-continue get the interim 100 before
    the body is transmitted (curl does this for large bodies)."""
    async with serve(provider) as base:
        host, port = base[len("http://"):].split(":")
        reader, writer = await asyncio.open_connection(host, int(port))
        body = json.dumps({"model": "local/tiny-llama", "max_tokens": 2,
                           "temperature": 0,
                           "messages": [{"role": "user",
                                         "content": "x"}]}).encode()
        writer.write(b"POST /v1/chat/completions HTTP/1.1\r\nhost: x\r\n"
                     b"content-type: application/json\r\n"
                     b"expect: 100-continue\r\n"
                     b"content-length: %d\r\n\r\n" % len(body))
        await writer.drain()
        interim = await asyncio.wait_for(reader.readline(), 10)
        assert interim.startswith(b"HTTP/1.1 100")
        await asyncio.wait_for(reader.readline(), 10)   # blank line
        writer.write(body)
        await writer.drain()
        data = b""
        while b"completion_tokens" not in data:
            chunk = await asyncio.wait_for(reader.read(65536), 30)
            if not chunk:
                break
            data += chunk
        writer.close()
        assert b"200" in data.split(b"\r\n", 1)[0]
