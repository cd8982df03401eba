"""Minimal ASGI primitives for the gateway.

The reference is a ~10 MB Go binary; to stay in the same overhead class
 the gateway speaks raw ASGI — no framework
dispatch on the hot path. Request/Response are thin; SSE responses stream
an async iterator of pre-framed `data: ...` lines with one ASGI send per
event (per-line flush is a reference correctness bar, and uvicorn flushes
every http.response.body message)."""
from __future__ import annotations

import json
from typing import AsyncIterator, Callable, Dict, Optional, Union
from urllib.parse import parse_qs


class Request:
    __slots__ = ("method", "path", "headers", "query", "body", "state",
                 "path_params")

    def __init__(self, method: str, path: str, headers: Dict[str, str],
                 query: Dict[str, str], body: bytes):
        self.method = method
        self.path = path
        self.headers = headers          # lower-cased keys
        self.query = query
        self.body = body
        self.state: dict = {}
        self.path_params: Dict[str, str] = {}

    def json(self):
        return json.loads(self.body) if self.body else None


class Response:
    __slots__ = ("status", "headers", "body", "stream", "direct")

    def __init__(self, status: int = 200,
                 body: Union[bytes, str, None] = b"",
                 headers: Optional[Dict[str, str]] = None,
                 stream: Optional[AsyncIterator[str]] = None):
        self.status = status
        self.body = body.encode() if isinstance(body, str) else (body or b"")
        self.headers = headers or {}
        self.stream = stream  # async iterator of SSE event strings
        # direct-write SSE session (local engine fast path): an object
        # with `async run()` that writes frames straight to the socket
        # via the server's raw-write extension — no per-token generator
        # resumption, no per-event ASGI messages (providers/local.py
        # DirectStream)
        self.direct = None

    @staticmethod
    def json(obj, status: int = 200,
             headers: Optional[Dict[str, str]] = None) -> "Response":
        h = {"content-type": "application/json"}
        if headers:
            h.update(headers)
        return Response(status, json.dumps(obj).encode(), h)

    @staticmethod
    def error(status: int, message: str) -> "Response":
        return Response.json({"error": message}, status)

    @staticmethod
    def sse(stream: AsyncIterator[str]) -> "Response":
        return Response(
            200, b"",
            {
                "content-type": "text/event-stream",
                "cache-control": "no-cache",
                "connection": "keep-alive",
                "x-accel-buffering": "no",
            },
            stream=stream,
        )


Handler = Callable  # async (Request) -> Response


def parse_scope(scope, body: bytes) -> Request:
    headers = {k.decode("latin1").lower(): v.decode("latin1")
               for k, v in scope.get("headers", [])}
    raw_q = scope.get("query_string", b"").decode("latin1")
    query = {k: v[0] for k, v in parse_qs(raw_q).items()}
    return Request(scope["method"], scope["path"], headers, query, body)


async def send_response(send, resp: Response) -> None:
    headers = [(k.encode(), v.encode()) for k, v in resp.headers.items()]
    await send({
        "type": "http.response.start",
        "status": resp.status,
        "headers": headers,
    })
    if resp.direct is not None:
        # the session writes every frame itself through the raw socket
        # extension (including the terminating chunk)
        await resp.direct.run()
        return
    if resp.stream is None:
        await send({"type": "http.response.body", "body": resp.body})
        return
    try:
        async for event in resp.stream:
            data = event if isinstance(event, bytes) else event.encode()
            if not data.endswith(b"\n\n"):
                data = data + b"\n\n"
            await send({
                "type": "http.response.body",
                "body": data,
                "more_body": True,
            })
    finally:
        await send({"type": "http.response.body", "body": b""})
