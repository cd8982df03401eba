"""Request logging middleware (reference api/middlewares/logger.go:25-61):
method/path/status/duration with sanitized headers; /health excluded."""
from __future__ import annotations

import time

from ..http import Request, Response

_SENSITIVE = {"authorization", "x-api-key", "cookie", "proxy-authorization"}


def sanitize_headers(headers: dict) -> dict:
    return {k: ("[REDACTED]" if k.lower() in _SENSITIVE else v)
            for k, v in headers.items()}


def logging_middleware(logger):
    async def mw(req: Request, nxt) -> Response:
        if req.path == "/health":
            return await nxt(req)
        start = time.monotonic()
        resp = await nxt(req)
        dur_ms = (time.monotonic() - start) * 1000
        logger.info("request", "method", req.method, "path", req.path,
                    "status", resp.status, "duration_ms", round(dur_ms, 2))
        return resp

    return mw
