"""Telemetry middleware (reference api/middlewares/telemetry.go:76-254):
records request duration + token usage + tool calls by inspecting the
response — for streams it tees the SSE events and scans the LAST few data
chunks for the usage block (telemetry.go:199-228), recording TTFT from the
first content chunk."""
from __future__ import annotations

import json
import time

from ..http import Request, Response

_TRACKED = ("/v1/chat/completions",)


def _extract_usage(payload: dict):
    u = payload.get("usage") or {}
    return int(u.get("prompt_tokens") or 0), int(
        u.get("completion_tokens") or 0)


def telemetry_middleware(telemetry):
    async def mw(req: Request, nxt) -> Response:
        if req.path not in _TRACKED:
            return await nxt(req)
        start = time.monotonic()
        resp = await nxt(req)
        provider = req.state.get("provider_id", "unknown")
        model = req.state.get("model", "unknown")

        if getattr(resp, "direct", None) is not None:
            # direct-write SSE session: it reports through hooks (there
            # is no stream generator to tee)
            sess = resp.direct
            sess.on_first_chunk = lambda: telemetry.record_ttft(
                provider, model, time.monotonic() - start)

            def _on_end(prompt_toks, completion_toks, err):
                telemetry.record_request(provider, model,
                                         time.monotonic() - start, err)
                if prompt_toks or completion_toks:
                    telemetry.record_tokens(provider, model, prompt_toks,
                                            completion_toks)

            sess.on_end = _on_end
            return resp

        if resp.stream is None:
            dur = time.monotonic() - start
            err = "" if resp.status < 400 else str(resp.status)
            telemetry.record_request(provider, model, dur, err)
            if resp.status < 400:
                try:
                    payload = json.loads(resp.body)
                    p, c = _extract_usage(payload)
                    if p or c:
                        telemetry.record_tokens(provider, model, p, c)
                    for ch in payload.get("choices") or []:
                        for tc in (ch.get("message") or {}).get(
                                "tool_calls") or []:
                            telemetry.record_tool_call(
                                provider, model, "function",
                                (tc.get("function") or {}).get("name", ""))
                except Exception:
                    pass
            return resp

        inner = resp.stream

        async def teed():
            tail = []
            first = True
            try:
                async for event in inner:
                    if first and event.startswith("data:"):
                        telemetry.record_ttft(provider, model,
                                              time.monotonic() - start)
                        first = False
                    tail.append(event)
                    if len(tail) > 4:
                        tail.pop(0)
                    yield event
            finally:
                dur = time.monotonic() - start
                telemetry.record_request(provider, model, dur, "")
                # scan the last 4 chunks for a usage block
                for event in reversed(tail):
                    if not event.startswith("data:"):
                        continue
                    data = event[5:].strip()
                    if data == "[DONE]":
                        continue
                    try:
                        payload = json.loads(data)
                    except Exception:
                        continue
                    p, c = _extract_usage(payload)
                    if p or c:
                        telemetry.record_tokens(provider, model, p, c)
                        break

        resp.stream = teed()
        return resp

    return mw
