"""Distributed tracing: W3C trace context + OTLP/HTTP span export.

The analog of the reference's opt-in OTLP tracing: otelgin
server spans with /health filtered, otelhttp client spans named
"METHOD /path", W3C traceparent injected on every outbound hop
(core/provider.go:115, mcp/transport.go:63), explicit `execute_tool
<name>` spans around MCP tool execution (agent.go:402-429), and the
exporter's /v1/traces path fix (otel/otel.go:86-96).

No OTel SDK dependency: spans are tiny dicts, exported as OTLP/HTTP JSON
by a background flusher when TELEMETRY_TRACING_ENABLED and an OTLP
endpoint are configured; tests use the in-memory recorder directly."""
from __future__ import annotations

import asyncio
import contextvars
import random
import time
from typing import Dict, List, Optional

# the active request span (set by tracing_middleware); outbound client
# code parents on it and injects its traceparent without API plumbing
current_span: contextvars.ContextVar = contextvars.ContextVar(
    "igw_current_span", default=None)


def _hex(nbytes: int) -> str:
    return "".join(f"{random.getrandbits(8):02x}" for _ in range(nbytes))


def parse_traceparent(header: str):
    """'00-<32 hex trace id>-<16 hex span id>-<2 hex flags>' -> (trace_id,
    span_id) or None."""
    try:
        parts = header.strip().split("-")
        if len(parts) != 4 or parts[0] != "00":
            return None
        tid, sid = parts[1], parts[2]
        if len(tid) != 32 or len(sid) != 16:
            return None
        int(tid, 16), int(sid, 16)
        return tid, sid
    except Exception:
        return None


def format_traceparent(trace_id: str, span_id: str,
                       sampled: bool = True) -> str:
    return f"00-{trace_id}-{span_id}-{'01' if sampled else '00'}"


class Span:
    __slots__ = ("name", "trace_id", "span_id", "parent_id", "start_ns",
                 "end_ns", "attributes", "kind", "status_ok")

    def __init__(self, name: str, trace_id: str, parent_id: Optional[str],
                 kind: str = "internal"):
        self.name = name
        self.trace_id = trace_id
        self.span_id = _hex(8)
        self.parent_id = parent_id
        self.start_ns = time.time_ns()
        self.end_ns = 0
        self.attributes: Dict[str, object] = {}
        self.kind = kind
        self.status_ok = True

    def set(self, key: str, value) -> "Span":
        self.attributes[key] = value
        return self

    def end(self) -> None:
        self.end_ns = time.time_ns()

    def traceparent(self) -> str:
        return format_traceparent(self.trace_id, self.span_id)


_KIND = {"internal": 1, "server": 2, "client": 3}


class Tracer:
    """Collects finished spans; flushes OTLP/HTTP JSON batches when an
    endpoint + client are configured (endpoint gets '/v1/traces' appended
    when it has no path, like the reference)."""

    def __init__(self, service_name: str = "inference-gateway-amd",
                 endpoint: str = "", client=None, max_buffer: int = 4096):
        self.service_name = service_name
        self.client = client
        self.endpoint = self._fix_path(endpoint) if endpoint else ""
        self.spans: List[Span] = []
        self.max_buffer = max_buffer

    @staticmethod
    def _fix_path(endpoint: str) -> str:
        from urllib.parse import urlparse

        p = urlparse(endpoint)
        if not p.path or p.path == "/":
            return endpoint.rstrip("/") + "/v1/traces"
        return endpoint

    # -- span lifecycle -------------------------------------------------
    def start_span(self, name: str, parent: Optional[Span] = None,
                   traceparent: Optional[str] = None,
                   kind: str = "internal") -> Span:
        if parent is not None:
            return Span(name, parent.trace_id, parent.span_id, kind)
        if traceparent:
            ctx = parse_traceparent(traceparent)
            if ctx:
                return Span(name, ctx[0], ctx[1], kind)
        return Span(name, _hex(16), None, kind)

    def finish(self, span: Span) -> None:
        if not span.end_ns:
            span.end()
        self.spans.append(span)
        if len(self.spans) > self.max_buffer:
            del self.spans[: len(self.spans) - self.max_buffer]

    def drain(self) -> List[Span]:
        out, self.spans = self.spans, []
        return out

    # -- OTLP export ----------------------------------------------------
    def _otlp_payload(self, spans: List[Span]) -> dict:
        return {"resourceSpans": [{
            "resource": {"attributes": [
                {"key": "service.name",
                 "value": {"stringValue": self.service_name}}]},
            "scopeSpans": [{
                "scope": {"name": "inference_gateway_amd"},
                "spans": [{
                    "traceId": s.trace_id,
                    "spanId": s.span_id,
                    **({"parentSpanId": s.parent_id} if s.parent_id else {}),
                    "name": s.name,
                    "kind": _KIND.get(s.kind, 1),
                    "startTimeUnixNano": str(s.start_ns),
                    "endTimeUnixNano": str(s.end_ns),
                    "attributes": [
                        {"key": k, "value": {"stringValue": str(v)}}
                        for k, v in s.attributes.items()],
                    "status": {"code": 1 if s.status_ok else 2},
                } for s in spans],
            }],
        }]}

    async def flush(self) -> int:
        spans = self.drain()
        if not spans or not self.endpoint or self.client is None:
            return 0
        try:
            await self.client.post(self.endpoint,
                                   json=self._otlp_payload(spans))
        except Exception:
            pass  # tracing must never take the gateway down
        return len(spans)

    async def run_flusher(self, interval: float = 5.0) -> None:
        while True:
            await asyncio.sleep(interval)
            await self.flush()


def tracing_middleware(tracer: Tracer, skip=("/health", "/v1/metrics")):
    """Server-span middleware (otelgin analog): one span per request,
    named 'METHOD /path', /health and /v1/metrics filtered; the active
    span rides in req.state['span'] for handlers to parent on."""

    async def mw(req, nxt):
        if req.path in skip:
            return await nxt(req)
        span = tracer.start_span(
            f"{req.method} {req.path}",
            traceparent=req.headers.get("traceparent"), kind="server")
        req.state["span"] = span
        token = current_span.set(span)
        deferred = False
        try:
            resp = await nxt(req)
            span.set("http.status_code", resp.status)
            span.status_ok = resp.status < 500
            # streaming responses: the span must cover the WHOLE
            # stream (otelgin semantics — the Go handler blocks until
            # the stream ends), not just time-to-headers
            if resp.direct is not None:
                prev = resp.direct.on_end
                sess = resp.direct

                def _end(*a, **kw):
                    if prev is not None:
                        prev(*a, **kw)
                    tracer.finish(span)

                sess.on_end = _end
                deferred = True
            elif resp.stream is not None:
                orig = resp.stream

                async def _wrapped():
                    try:
                        async for chunk in orig:
                            yield chunk
                    finally:
                        tracer.finish(span)

                resp.stream = _wrapped()
                deferred = True
            return resp
        except Exception:
            span.status_ok = False
            raise
        finally:
            current_span.reset(token)
            if not deferred:
                tracer.finish(span)

    return mw


def inject_headers(headers: dict) -> dict:
    """Inject the active span's W3C traceparent (outbound hop)."""
    span = current_span.get()
    if span is not None:
        headers = dict(headers)
        headers["traceparent"] = span.traceparent()
    return headers


def install_client_hooks(client, tracer: Tracer) -> None:
    """Client spans for every outbound HTTP call, named 'METHOD /path'
    (the reference's otelhttp SpanNameFormatter on the shared client,
    providers/client/client.go:23-30), parented on the active request
    span.  Implemented as httpx event hooks so every provider/MCP call
    is covered centrally."""

    async def on_request(request) -> None:
        span = tracer.start_span(
            f"{request.method} {request.url.path}",
            parent=current_span.get(), kind="client")
        request.extensions["igw_span"] = span

    async def on_response(response) -> None:
        span = response.request.extensions.get("igw_span")
        if span is not None:
            span.set("http.status_code", response.status_code)
            span.status_ok = response.status_code < 500
            tracer.finish(span)

    client.event_hooks.setdefault("request", []).append(on_request)
    client.event_hooks.setdefault("response", []).append(on_response)
