"""Guardrails middleware (reference api/middlewares/guardrails.go:71-213):
pre_call evaluation of the request messages; for non-streaming responses a
post_call evaluation of the final body (block replaces the response,
redact rewrites it); streaming responses get pre_call only, matching the
reference."""
from __future__ import annotations

import json

from ..http import Request, Response

_GUARDED = ("/v1/chat/completions",)


def _request_text(body: dict) -> str:
    parts = []
    for m in body.get("messages") or []:
        c = m.get("content")
        if isinstance(c, str):
            parts.append(c)
        elif isinstance(c, list):
            parts.extend(p.get("text", "") for p in c
                         if isinstance(p, dict) and p.get("type") == "text")
    return "\n".join(parts)


def _rewrite_messages(body: dict, redacted: str) -> None:
    # simple parity-preserving rewrite: apply redaction per message content
    lines = redacted.split("\n")
    i = 0
    for m in body.get("messages") or []:
        if isinstance(m.get("content"), str) and i < len(lines):
            m["content"] = lines[i]
            i += 1


def guardrails_middleware(evaluator, logger):
    async def mw(req: Request, nxt) -> Response:
        if req.path not in _GUARDED:
            return await nxt(req)
        try:
            body = req.json()
        except Exception:
            return await nxt(req)
        if not isinstance(body, dict):
            return await nxt(req)
        text = _request_text(body)
        decision = await evaluator.eval("pre_call", text)
        if decision.action == "block":
            return Response.json(
                {"error": {"message": decision.message or
                           "request blocked by guardrails",
                           "type": "guardrail_blocked",
                           "rule": decision.rule}}, 403)
        if decision.action == "redact" and decision.content is not None:
            _rewrite_messages(body, decision.content)
            req.body = json.dumps(body).encode()
            req.state["parsed_body"] = body
        if decision.action == "warn":
            logger.warn("guardrail warning", "rule", decision.rule,
                        "path", req.path)

        resp = await nxt(req)

        # post_call only for non-streaming success responses; direct
        # sessions (gateway/hserver.py fast path) are streaming too —
        # their Response carries no body to evaluate.
        if (resp.stream is not None or resp.direct is not None
                or resp.status >= 400):
            return resp
        try:
            payload = json.loads(resp.body)
            content = "\n".join(
                (ch.get("message") or {}).get("content") or ""
                for ch in payload.get("choices") or [])
        except Exception:
            return resp
        decision = await evaluator.eval("post_call", content)
        if decision.action == "block":
            return Response.json(
                {"error": {"message": decision.message or
                           "response blocked by guardrails",
                           "type": "guardrail_blocked",
                           "rule": decision.rule}}, 403)
        if decision.action == "redact" and decision.content is not None:
            lines = decision.content.split("\n")
            for i, ch in enumerate(payload.get("choices") or []):
                if i < len(lines) and ch.get("message"):
                    ch["message"]["content"] = lines[i]
            return Response.json(payload, resp.status)
        return resp

    return mw
