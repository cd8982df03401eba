"""Development-mode request/response body logging.

The analog of the reference's dev proxy modifiers
(internal/proxy/proxy.go:53-217): attached only when
ENVIRONMENT=development (api/routes.go:258-269), logs request and
response bodies with smart truncation of chat message content so huge
prompts don't flood the log.
"""
from __future__ import annotations

import json

_MAX_FIELD = 200
_MAX_BODY = 4096


def _truncate(s: str, n: int = _MAX_FIELD) -> str:
    return s if len(s) <= n else s[:n] + f"... ({len(s)} bytes total)"


def truncate_chat_body(parsed) -> object:
    """Truncate message/content fields of a chat-completion-shaped body
    (reference proxy.go:84-145)."""
    if not isinstance(parsed, dict):
        return parsed
    out = dict(parsed)
    msgs = out.get("messages")
    if isinstance(msgs, list):
        tm = []
        for m in msgs:
            if isinstance(m, dict) and isinstance(m.get("content"), str):
                m = dict(m)
                m["content"] = _truncate(m["content"])
            tm.append(m)
        out["messages"] = tm
    for key in ("prompt", "input"):
        if isinstance(out.get(key), str):
            out[key] = _truncate(out[key])
    return out


def _render(body: bytes) -> str:
    if not body:
        return ""
    try:
        parsed = json.loads(body)
    except Exception:
        return _truncate(body.decode(errors="replace"), _MAX_BODY)
    return _truncate(json.dumps(truncate_chat_body(parsed)), _MAX_BODY)


def dev_logging_middleware(logger):
    async def mw(req, nxt):
        logger.debug("dev request", method=req.method, path=req.path,
                     body=_render(req.body))
        resp = await nxt(req)
        if getattr(resp, "body", None) is not None:
            logger.debug("dev response", status=resp.status,
                         body=_render(resp.body))
        else:
            logger.debug("dev response (stream)", status=resp.status)
        return resp

    return mw
