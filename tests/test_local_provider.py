"""End-to-end: gateway -> in-process local engine provider (CPU, tiny
model) — the rebuild's core differentiator (no self-proxy
hop; engine tokens stream straight into SSE)."""
import json

import httpx
import pytest

from inference_gateway_amd import config as config_mod
from inference_gateway_amd.config import LocalConfig
from inference_gateway_amd.gateway.app import Gateway
from inference_gateway_amd.logger import noop_logger
from inference_gateway_amd.providers.local import LocalProvider

from test_gateway import gw_client


@pytest.fixture(scope="module")
def local_gateway():
    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=4,
                     max_model_len=256)
    lc.num_gpu_blocks = 64
    provider = LocalProvider(lc, device="cpu")
    cfg = config_mod.load({})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(404)))
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider)
    yield gw
    provider.engine.shutdown()


@pytest.mark.asyncio
async def test_local_in_models_list(local_gateway):
    async with gw_client(local_gateway) as c:
        r = await c.get("/v1/models", params={"provider": "local"})
        assert r.status_code == 200
        ids = [m["id"] for m in r.json()["data"]]
        assert ids == ["local/tiny-llama"]


@pytest.mark.asyncio
async def test_local_chat_completion(local_gateway):
    async with gw_client(local_gateway) as c:
        r = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hello"}],
            "max_tokens": 6,
            "temperature": 0,
        })
        assert r.status_code == 200
        body = r.json()
        assert body["object"] == "chat.completion"
        assert body["model"] == "local/tiny-llama"
        assert body["usage"]["completion_tokens"] == 6
        assert isinstance(body["choices"][0]["message"]["content"], str)


@pytest.mark.asyncio
async def test_local_chat_stream(local_gateway):
    async with gw_client(local_gateway) as c:
        async with c.stream("POST", "/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hi"}],
            "max_tokens": 5,
            "temperature": 0,
            "stream": True,
        }) as r:
            assert r.status_code == 200
            assert r.headers["content-type"].startswith("text/event-stream")
            events = []
            async for line in r.aiter_lines():
                if line.startswith("data:"):
                    events.append(line[5:].strip())
    assert events[-1] == "[DONE]"
    first = json.loads(events[0])
    assert first["choices"][0]["delta"]["role"] == "assistant"
    final = json.loads(events[-2])
    assert final["usage"]["completion_tokens"] == 5
    assert final["choices"][0]["finish_reason"] in ("stop", "length")


@pytest.mark.asyncio
async def test_local_concurrent_streams(local_gateway):
    import asyncio

    async def one(c, i):
        done = 0
        async with c.stream("POST", "/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": f"req {i}"}],
            "max_tokens": 4,
            "stream": True,
        }) as r:
            async for line in r.aiter_lines():
                if line.strip() == "data: [DONE]":
                    done += 1
        return done

    async with gw_client(local_gateway) as c:
        results = await asyncio.gather(*(one(c, i) for i in range(4)))
    assert results == [1, 1, 1, 1]


@pytest.mark.asyncio
async def test_stop_strings(local_gateway):
    """OpenAI `stop` sequences: generation ends at (and excludes) the
    stop string, in both modes."""
    # pick a stop string we can force: generate once, reuse a substring
    async with gw_client(local_gateway) as c:
        r = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hello"}],
            "max_tokens": 12, "temperature": 0})
        full = r.json()["choices"][0]["message"]["content"]
        if len(full) < 4:
            pytest.skip("output too short to pick a stop substring")
        stop = full[2:4]
        r = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hello"}],
            "max_tokens": 12, "temperature": 0, "stop": stop})
        body = r.json()
        assert body["choices"][0]["finish_reason"] == "stop"
        text = body["choices"][0]["message"]["content"]
        assert stop not in text
        assert full.startswith(text)

        # streaming: same semantics
        got = []
        async with c.stream("POST", "/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hello"}],
            "max_tokens": 12, "temperature": 0, "stop": stop,
            "stream": True,
        }) as resp:
            async for line in resp.aiter_lines():
                if not line.startswith("data:"):
                    continue
                d = line[5:].strip()
                if d == "[DONE]":
                    break
                delta = json.loads(d)["choices"][0]["delta"]
                if delta.get("content"):
                    got.append(delta["content"])
        stext = "".join(got)
        assert stop not in stext
        assert stext == text


@pytest.mark.asyncio
async def test_n_choices(local_gateway):
    async with gw_client(local_gateway) as c:
        r = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hello"}],
            "max_tokens": 5, "temperature": 0.9, "n": 3})
    body = r.json()
    assert len(body["choices"]) == 3
    assert [ch["index"] for ch in body["choices"]] == [0, 1, 2]
    assert body["usage"]["completion_tokens"] == 15


@pytest.mark.asyncio
async def test_engine_gauges_in_metrics_exposition():
    from inference_gateway_amd.otel.metrics import Telemetry

    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=4,
                     max_model_len=256)
    lc.num_gpu_blocks = 64
    provider = LocalProvider(lc, device="cpu")
    cfg = config_mod.load({"TELEMETRY_ENABLED": "true"})
    client = httpx.AsyncClient(
        transport=httpx.MockTransport(lambda r: httpx.Response(404)))
    telemetry = Telemetry()
    gw = Gateway(cfg=cfg, logger=noop_logger(), http_client=client,
                 local_provider=provider, telemetry=telemetry)
    assert gw is not None
    text = telemetry.exposition().decode()
    assert "igw_engine_kv_blocks_total 63.0" in text
    assert "igw_engine_running_seqs 0.0" in text
    assert "igw_engine_generated_tokens_total 0.0" in text
    assert "igw_engine_preemptions_total 0.0" in text
    assert "igw_engine_steps_total" in text
    assert "igw_engine_mixed_rounds_total" in text
    provider.engine.shutdown()


@pytest.mark.asyncio
async def test_local_logprobs_non_streaming(local_gateway):
    async with gw_client(local_gateway) as c:
        r = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hello"}],
            "max_tokens": 5, "temperature": 0,
            "logprobs": True, "top_logprobs": 3,
        })
        assert r.status_code == 200
        lp = r.json()["choices"][0]["logprobs"]
        assert lp is not None and len(lp["content"]) == 5
        for item in lp["content"]:
            assert isinstance(item["token"], str)
            assert item["logprob"] <= 0.0
            assert item["bytes"] == list(item["token"].encode("utf-8"))
            tops = item["top_logprobs"]
            assert len(tops) == 3
            vals = [t["logprob"] for t in tops]
            assert vals == sorted(vals, reverse=True)
            # greedy: sampled token is the top-1 alternative
            assert tops[0]["token"] == item["token"]

        # not requested -> null, like OpenAI
        r2 = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hello"}],
            "max_tokens": 2, "temperature": 0,
        })
        assert r2.json()["choices"][0]["logprobs"] is None


@pytest.mark.asyncio
async def test_local_logprobs_streaming(local_gateway):
    async with gw_client(local_gateway) as c:
        async with c.stream("POST", "/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hi"}],
            "max_tokens": 6, "temperature": 0, "stream": True,
            "logprobs": True, "top_logprobs": 2,
        }) as r:
            assert r.status_code == 200
            items = []
            async for line in r.aiter_lines():
                if not line.startswith("data: ") or line == "data: [DONE]":
                    continue
                ch = json.loads(line[6:])["choices"][0]
                if ch.get("logprobs"):
                    items.extend(ch["logprobs"]["content"])
            assert 1 <= len(items) <= 6
            for item in items:
                assert item["logprob"] <= 0.0
                assert len(item["top_logprobs"]) == 2


@pytest.mark.asyncio
async def test_local_n_greater_one_forked(local_gateway):
    """n>1 rides the engine's KV fork (one prefill); greedy branches are
    identical and identical to n=1."""
    async with gw_client(local_gateway) as c:
        r1 = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "fork me"}],
            "max_tokens": 6, "temperature": 0,
        })
        want = r1.json()["choices"][0]["message"]["content"]
        r = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "fork me"}],
            "max_tokens": 6, "temperature": 0, "n": 3,
        })
        assert r.status_code == 200
        body = r.json()
        assert [ch["index"] for ch in body["choices"]] == [0, 1, 2]
        for ch in body["choices"]:
            assert ch["message"]["content"] == want
        assert body["usage"]["completion_tokens"] == 18


@pytest.mark.asyncio
async def test_local_streaming_n_greater_one(local_gateway):
    """Streaming with n=3: interleaved per-choice chunks, one [DONE],
    group usage on the final chunk, greedy branches identical."""
    async with gw_client(local_gateway) as c:
        async with c.stream("POST", "/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "multi"}],
            "max_tokens": 5, "temperature": 0, "stream": True, "n": 3,
        }) as r:
            assert r.status_code == 200
            texts = {0: "", 1: "", 2: ""}
            finishes, dones, usage = {}, 0, None
            async for line in r.aiter_lines():
                if not line.startswith("data: "):
                    continue
                if line == "data: [DONE]":
                    dones += 1
                    continue
                body = json.loads(line[6:])
                ch = body["choices"][0]
                i = ch["index"]
                texts[i] += ch["delta"].get("content") or ""
                if ch.get("finish_reason"):
                    finishes[i] = ch["finish_reason"]
                if body.get("usage"):
                    usage = body["usage"]
            assert dones == 1
            assert set(finishes) == {0, 1, 2}
            assert all(f == "length" for f in finishes.values())
            assert texts[0] == texts[1] == texts[2] != ""
            assert usage["completion_tokens"] == 15


@pytest.mark.asyncio
async def test_local_prefix_caching_and_chunked_flags_plumbed():
    """LOCAL_PREFIX_CACHING / LOCAL_CHUNKED_PREFILL reach the engine,
    and serving with both on still answers correctly."""
    cfg = config_mod.load({"LOCAL_ENABLED": "true",
                           "LOCAL_PREFIX_CACHING": "true",
                           "LOCAL_CHUNKED_PREFILL": "true"})
    assert cfg.local.prefix_caching and cfg.local.chunked_prefill

    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=4,
                     max_model_len=256, prefix_caching=True,
                     chunked_prefill=True)
    lc.num_gpu_blocks = 64
    provider = LocalProvider(lc, device="cpu")
    try:
        eng = provider.engine.engine
        assert eng.block_manager.enable_prefix_caching
        assert eng.scheduler.chunked_prefill
        client = httpx.AsyncClient(
            transport=httpx.MockTransport(lambda r: httpx.Response(404)))
        gw = Gateway(cfg=config_mod.load({}), logger=noop_logger(),
                     http_client=client, local_provider=provider)
        async with gw_client(gw) as c:
            body = {"model": "local/tiny-llama",
                    "messages": [{"role": "user",
                                  "content": "hello " * 20}],
                    "max_tokens": 4, "temperature": 0}
            r1 = await c.post("/v1/chat/completions", json=body)
            r2 = await c.post("/v1/chat/completions", json=body)
            assert r1.status_code == r2.status_code == 200
            assert (r1.json()["choices"][0]["message"]["content"]
                    == r2.json()["choices"][0]["message"]["content"])
        assert eng.block_manager.prefix_hit_tokens > 0
    finally:
        provider.engine.shutdown()


@pytest.mark.asyncio
async def test_local_context_overflow_is_400(local_gateway):
    async with gw_client(local_gateway) as c:
        r = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "x" * 5000}],
            "max_tokens": 4,
        })
        assert r.status_code == 400
        assert "maximum context length" in str(r.json()["error"])


@pytest.mark.asyncio
async def test_local_stream_context_overflow_is_plain_400(local_gateway):
    """Overflow on a STREAMING request must be a plain 400 response,
    not an error frame inside an SSE stream."""
    async with gw_client(local_gateway) as c:
        r = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "x" * 5000}],
            "max_tokens": 4, "stream": True,
        })
        assert r.status_code == 400
        assert "text/event-stream" not in r.headers.get("content-type", "")


@pytest.mark.asyncio
async def test_local_n_over_cap_is_400(local_gateway):
    """n beyond the supported maximum is a 400, never a
    silent clamp (clients index choices by their requested n)."""
    async with gw_client(local_gateway) as c:
        for stream in (False, True):
            r = await c.post("/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "hello"}],
                "max_tokens": 4, "n": 9, "stream": stream})
            assert r.status_code == 400
            assert "n must be between" in str(r.json()["error"])


@pytest.mark.asyncio
async def test_local_prompt_plus_max_tokens_overflow_is_400(local_gateway):
    """OpenAI 400s when prompt + an explicit max_tokens
    exceeds the context window (no silent truncation); without
    max_tokens the request is accepted and truncates at the limit."""
    async with gw_client(local_gateway) as c:
        r = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hello"}],
            "max_tokens": 100000})
        assert r.status_code == 400
        assert "max_tokens" in str(r.json()["error"])
        # no explicit max_tokens: allowed, finishes with 'length'
        r = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hello"}],
            "temperature": 0})
        assert r.status_code == 200


@pytest.mark.asyncio
async def test_stream_flushes_incomplete_utf8_tail(local_gateway,
                                                   monkeypatch):
    """Regression (soak flake): tokens that never complete a UTF-8
    sequence were held back forever and silently dropped at stream end —
    the final flush must emit them (as replacement chars)."""
    provider = local_gateway.local_provider

    class _Out:
        def __init__(self, tok, fin):
            self.token_id = tok
            self.finished = fin
            self.finish_reason = "length" if fin else None
            self.logprob = None
            self.top_logprobs = None

    async def fake_generate(prompt_ids, sampling):
        # 0x80 continuation bytes (+3 special offset): every prefix of
        # this stream ends mid-UTF-8, so nothing flushes until the end
        toks = [0x80 + 3, 0x81 + 3, 0x82 + 3, 0x83 + 3]
        for i, t in enumerate(toks):
            yield [_Out(t, i == len(toks) - 1)]

    monkeypatch.setattr(provider.engine, "generate_batched",
                        fake_generate)
    content = ""
    chunks = 0
    async for line in provider.stream_chat_completions({
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "x"}],
            "max_tokens": 4}):
        if not line.startswith("data: ") or line == "data: [DONE]":
            continue
        ch = json.loads(line[6:])["choices"][0]
        content += ch["delta"].get("content") or ""
        chunks += 1
        if ch.get("finish_reason"):
            fin = ch["finish_reason"]
    assert "�" in content          # tokens surfaced, not dropped
    assert chunks >= 3                  # role + content + usage
    assert fin == "length"              # not misreported as "stop"


@pytest.mark.asyncio
async def test_stream_merges_batched_tokens_into_one_chunk(local_gateway,
                                                           monkeypatch):
    """Wire-bound batching: when the engine queue has several tokens
    ready, they are emitted as ONE content chunk (one socket write) with
    no text lost and exact usage accounting."""
    provider = local_gateway.local_provider

    class _Out:
        def __init__(self, tok, fin):
            self.token_id = tok
            self.finished = fin
            self.finish_reason = "length" if fin else None
            self.logprob = None
            self.top_logprobs = None

    # 'a','b' (+3 byte-tokenizer offset) then 'c','d','e' in one batch
    async def fake_batched(prompt_ids, sampling):
        yield [_Out(ord("a") + 3, False), _Out(ord("b") + 3, False)]
        yield [_Out(ord("c") + 3, False), _Out(ord("d") + 3, False),
               _Out(ord("e") + 3, True)]

    monkeypatch.setattr(provider.engine, "generate_batched", fake_batched)
    contents, usage = [], None
    async for line in provider.stream_chat_completions({
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "x"}],
            "max_tokens": 5}):
        if not line.startswith("data: ") or line == "data: [DONE]":
            continue
        body = json.loads(line[6:])
        d = body["choices"][0]["delta"].get("content")
        if d:
            contents.append(d)
        usage = body.get("usage") or usage
    assert contents == ["ab", "cde"]       # one chunk per ready batch
    assert usage["completion_tokens"] == 5


@pytest.mark.asyncio
async def test_local_tolerates_tools_and_extra_params(local_gateway):
    """OpenAI clients send tools/response_format/user etc. — the local
    provider must accept and ignore what it cannot honor."""
    async with gw_client(local_gateway) as c:
        r = await c.post("/v1/chat/completions", json={
            "model": "local/tiny-llama",
            "messages": [{"role": "user", "content": "hi"}],
            "max_tokens": 3,
            "tools": [{"type": "function", "function": {
                "name": "f", "parameters": {"type": "object"}}}],
            "tool_choice": "auto",
            "response_format": {"type": "text"},
            "user": "tester",
            "parallel_tool_calls": True,
        })
        assert r.status_code == 200
        assert r.json()["usage"]["completion_tokens"] == 3


@pytest.mark.asyncio
async def test_local_real_tokenizer_end_to_end(tmp_path):
    """LOCAL_TOKENIZER_PATH plumbs a real tokenizers file into the
    provider and serving works with it."""
    from tokenizers import Tokenizer
    from tokenizers.models import WordLevel
    from tokenizers.pre_tokenizers import Whitespace

    vocab = {"<s>": 0, "</s>": 1, "[UNK]": 2, "hi": 3, "there": 4}
    tok = Tokenizer(WordLevel(vocab, unk_token="[UNK]"))
    tok.pre_tokenizer = Whitespace()
    path = str(tmp_path / "tokenizer.json")
    tok.save(path)

    cfg = config_mod.load({"LOCAL_TOKENIZER_PATH": path})
    assert cfg.local.tokenizer_path == path

    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=4,
                     max_model_len=256, tokenizer_path=path)
    lc.num_gpu_blocks = 64
    provider = LocalProvider(lc, device="cpu")
    try:
        assert provider.tokenizer.bos_token_id == 0
        client = httpx.AsyncClient(
            transport=httpx.MockTransport(lambda r: httpx.Response(404)))
        gw = Gateway(cfg=config_mod.load({}), logger=noop_logger(),
                     http_client=client, local_provider=provider)
        async with gw_client(gw) as c:
            r = await c.post("/v1/chat/completions", json={
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "hi there"}],
                "max_tokens": 3, "temperature": 0,
            })
            assert r.status_code == 200
            assert isinstance(
                r.json()["choices"][0]["message"]["content"], str)
    finally:
        provider.engine.shutdown()


def test_max_queue_sheds_load_with_429():
    """LOCAL_MAX_QUEUE: requests beyond the waiting-queue cap get 429
    (load shedding), and the cap releases as the queue drains."""
    import asyncio

    from inference_gateway_amd.config import LocalConfig
    from inference_gateway_amd.providers.core import HTTPError
    from inference_gateway_amd.providers.local import LocalProvider

    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=2,
                     max_model_len=128)
    lc.num_gpu_blocks = 64
    lc.max_queue = 2
    provider = LocalProvider(lc, device="cpu")
    try:
        eng = provider.engine
        eng.hold()                     # requests queue, engine paused
        async def go():
            reqs = [asyncio.create_task(provider.chat_completions({
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": f"r{i}"}],
                "max_tokens": 2, "temperature": 0})) for i in range(2)]
            await asyncio.sleep(0.2)   # both now in waiting
            with pytest.raises(HTTPError) as ei:
                await provider.chat_completions({
                    "model": "local/tiny-llama",
                    "messages": [{"role": "user", "content": "over"}],
                    "max_tokens": 2})
            assert ei.value.status == 429
            eng.release()              # drain; the cap must release too
            out = await asyncio.gather(*reqs)
            assert all(o["usage"]["completion_tokens"] == 2 for o in out)
            ok = await provider.chat_completions({
                "model": "local/tiny-llama",
                "messages": [{"role": "user", "content": "after"}],
                "max_tokens": 2, "temperature": 0})
            assert ok["usage"]["completion_tokens"] == 2

        asyncio.run(go())
    finally:
        provider.engine.shutdown()


def test_n_gt_1_stream_equals_nonstream_with_stops():
    """Per-choice streamed content and finish_reason equal the
    non-streaming response exactly — same seed, stops on, n in {2,3}
    (cross-path consistency incl. the progressive stop scan)."""
    import asyncio
    import json
    import random

    from inference_gateway_amd.config import LocalConfig
    from inference_gateway_amd.providers.local import LocalProvider

    lc = LocalConfig(enabled=True, model="tiny-llama", max_num_seqs=8,
                     max_model_len=128)
    lc.num_gpu_blocks = 96
    provider = LocalProvider(lc, device="cpu")

    async def one(trial, rng):
        req = {"model": "local/tiny-llama", "n": rng.choice([2, 3]),
               "max_tokens": rng.randrange(3, 12), "temperature": 0.9,
               "seed": trial,
               "messages": [{"role": "user",
                             "content": f"fuzz {trial}"}]}
        if rng.random() < 0.7:
            req["stop"] = [rng.choice(["<tok", ">", "ab", "<t"])
                           for _ in range(rng.randrange(1, 3))]
        ns = {c["index"]: (c["message"]["content"], c["finish_reason"])
              for c in (await provider.chat_completions(
                  dict(req)))["choices"]}
        st = {}
        async for line in provider.stream_chat_completions(dict(req)):
            if not line.startswith("data: {"):
                continue
            ch = json.loads(line[6:])["choices"][0]
            txt, fin = st.get(ch["index"], ("", None))
            st[ch["index"]] = (txt + (ch["delta"].get("content") or ""),
                               ch["finish_reason"] or fin)
        assert st == ns, (trial, st, ns, req.get("stop"))

    async def main():
        rng = random.Random(77)
        for trial in range(20):
            await one(trial, rng)

    try:
        asyncio.run(main())
    finally:
        provider.engine.shutdown()
