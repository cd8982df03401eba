"""Gateway route handlers.

Re-implements the reference's 12-handler router (api/routes.go:44-57) over
the direct-call provider layer — same surface and semantics, no self-proxy
hop (SURVEY.md §3.2 note: the hop doubled middleware cost by design; the
rebuild short-circuits it in-process).

Handlers: health, list_models, chat_completions (stream + non-stream),
messages (Anthropic passthrough), responses (OpenAI passthrough), images
(generations/edits/variations passthrough), proxy, mcp tools, metrics
ingestion (otel/ingest.py).
"""
from __future__ import annotations

import asyncio
import json
from typing import Optional

from ..providers import routing
from ..providers.constants import (ANTHROPIC_MESSAGES_ENDPOINT,
                                   LOCAL_PROVIDER_ID,
                                   OPENAI_IMAGES_EDITS_ENDPOINT,
                                   OPENAI_IMAGES_ENDPOINT,
                                   OPENAI_IMAGES_VARIATIONS_ENDPOINT,
                                   OPENAI_RESPONSES_ENDPOINT)
from ..providers.core import HTTPError
from .http import Request, Response



async def _relay_sse_events(line_iter):
    """Group upstream SSE lines into WHOLE events before re-framing.

    The SSE writer appends the blank separator per yielded chunk, so
    yielding line-by-line would split multi-line frames — Anthropic's
    `event:` + `data:` pairs would become two frames and a
    spec-compliant EventSource client would see the data under the
    default event type instead of the named one."""
    buf = []
    async for line in line_iter:
        if line == "":
            if buf:
                yield "\n".join(buf)
                buf = []
        else:
            buf.append(line)
    if buf:
        yield "\n".join(buf)


class Router:
    def __init__(self, cfg, registry, logger, selector=None, mcp=None,
                 telemetry=None):
        self.cfg = cfg
        self.registry = registry
        self.logger = logger
        self.selector = selector     # routing.Selector or None
        self.mcp = mcp               # mcp layer (client + agent) or None
        self.telemetry = telemetry

    # ---------------- /health ----------------
    async def health(self, req: Request) -> Response:
        return Response.json({"status": "healthy"})

    # ---------------- GET /v1/models ----------------
    async def list_models(self, req: Request) -> Response:
        provider_q = req.query.get("provider")
        if provider_q:
            try:
                p = self.registry.build(provider_q)
            except HTTPError as e:
                return Response.error(e.status, e.message)
            try:
                result = await p.list_models()
            except HTTPError as e:
                return Response.error(502, f"upstream error: {e.message}")
            except Exception as e:
                return Response.error(502, f"upstream error: {e}")
            models = result["data"]
        else:
            # fan-out over all configured providers with a shared timeout
            # (reference routes.go:503-539); failures are skipped
            async def fetch(pid):
                try:
                    p = self.registry.build(pid)
                    return (await p.list_models())["data"]
                except Exception:
                    return []

            results = await asyncio.gather(
                *(fetch(pid) for pid in self.registry.configured_ids()),
                return_exceptions=False)
            models = [m for r in results for m in r]
        models = routing.filter_models(
            models, self.cfg.general.allowed_models,
            self.cfg.general.disallowed_models)
        include = [s.strip() for s in req.query.get("include", "").split(",")
                   if s.strip()]
        if include:
            from ..providers import metadata

            models = await metadata.enrich_models(models, include,
                                                  self.registry)
        return Response.json(
            {"object": "list", "data": models})

    # ---------------- POST /v1/chat/completions ----------------
    def _resolve(self, req_model: str, provider_q: Optional[str]):
        """alias pools -> prefix routing -> ?provider= (reference
        routes.go:669-693). Returns (provider_id, model) or raises."""
        if self.selector is not None:
            dep = self.selector.select(req_model)
            if dep is not None:
                return dep.provider, dep.model
        pid, model = routing.determine_provider_and_model(req_model)
        if pid is None:
            if not provider_q:
                raise HTTPError(
                    400,
                    "unable to determine provider for model; use "
                    "'provider/model' or the ?provider= query parameter")
            pid = provider_q.lower()
        return pid, model

    async def chat_completions(self, req: Request) -> Response:
        body = req.state.get("parsed_body")
        if body is None:
            try:
                body = req.json()
            except Exception:
                return Response.error(400, "invalid JSON body")
        if not isinstance(body, dict) or not body.get("model"):
            return Response.error(400, "model is required")
        from .validate import validate_chat_request

        verr = validate_chat_request(body)
        if verr is not None:
            return Response.error(400, verr)
        try:
            pid, model = self._resolve(body["model"],
                                       req.query.get("provider"))
        except HTTPError as e:
            return Response.error(e.status, e.message)
        if routing.model_denied(body["model"],
                                self.cfg.general.allowed_models,
                                self.cfg.general.disallowed_models):
            return Response.error(403, f"model not allowed: {body['model']}")
        try:
            provider = self.registry.build(pid)
        except HTTPError as e:
            return Response.error(e.status, e.message)
        body = dict(body)
        body["model"] = model
        # vision gate (reference routes.go:710-740): strip image parts when
        # vision is disabled OR the target model is not image-capable per
        # the community modalities table (ModelAcceptsImages)
        if not self.cfg.general.enable_vision:
            self._strip_images(body)
        else:
            from ..providers import metadata

            if not metadata.model_accepts_images(f"{pid}/{model}"):
                self._strip_images(body)

        stream = bool(body.get("stream"))
        req.state["provider_id"] = pid
        req.state["model"] = model

        # request-shape validation BEFORE the SSE response opens (the
        # local engine's context-length check: OpenAI answers a plain
        # 400, not an in-stream error frame)
        validate = getattr(provider, "validate_request", None)
        if validate is not None:
            try:
                validate(body)
            except HTTPError as e:
                return Response.error(e.status, e.message)

        if self.mcp is not None and self.mcp.applies(req, body):
            return await self.mcp.run_chat(provider, body, stream, req)

        if stream:
            # direct-write SSE fast path: local engine + native server +
            # n==1 — the session writes frames straight to the socket
            # (providers/local.py DirectStream).  Any other combination
            # takes the generator relay below.
            direct_io = (req.state.get("asgi_extensions") or {}).get(
                "igw.direct_sse")
            opener = getattr(provider, "open_direct_stream", None)
            if (direct_io is not None and opener is not None
                    and int(body.get("n") or 1) == 1):
                try:
                    session = opener(body, *direct_io)
                except HTTPError as e:
                    return Response.error(e.status, e.message)
                resp = Response.sse(None)
                resp.direct = session
                return resp
            try:
                gen = self._relay_stream(provider, body)
                return Response.sse(gen)
            except HTTPError as e:
                return Response.error(e.status, e.message)
        try:
            result = await provider.chat_completions(body)
        except HTTPError as e:
            status = 502 if e.status >= 500 else e.status
            return Response.error(status, e.message)
        except asyncio.TimeoutError:
            return Response.error(504, "upstream timeout")
        except Exception as e:
            return Response.error(502, f"upstream error: {e}")
        return Response.json(result)

    @staticmethod
    def _strip_images(body: dict) -> None:
        for msg in body.get("messages") or []:
            c = msg.get("content")
            if isinstance(c, list):
                text = [p for p in c if not (isinstance(p, dict)
                                             and p.get("type") == "image_url")]
                msg["content"] = text

    async def _relay_stream(self, provider, body):
        """Relay upstream SSE lines; guarantees exactly one [DONE]."""
        done_sent = False
        try:
            async for line in provider.stream_chat_completions(body):
                if not line:
                    continue
                if line.startswith("data:"):
                    if line[5:].strip() == "[DONE]":
                        done_sent = True
                    yield line
        except HTTPError as e:
            yield ("data: " + json.dumps(
                {"error": {"message": e.message, "code": e.status}}))
        except Exception as e:
            yield ("data: " + json.dumps({"error": {"message": str(e)}}))
        finally:
            if not done_sent:
                yield "data: [DONE]"

    # ---------------- byte passthrough handlers ----------------
    async def messages(self, req: Request) -> Response:
        """Anthropic-native /v1/messages (reference routes.go:842-1005):
        byte-for-byte passthrough; only `model` is sniffed/rewritten."""
        return await self._passthrough(
            req, required_provider="anthropic",
            endpoint=ANTHROPIC_MESSAGES_ENDPOINT,
            error_shape="anthropic")

    async def responses(self, req: Request) -> Response:
        """OpenAI /v1/responses passthrough (routes.go:1018-1179)."""
        return await self._passthrough(
            req, required_provider="openai",
            endpoint=OPENAI_RESPONSES_ENDPOINT, error_shape="openai")

    async def _passthrough(self, req: Request, required_provider: str,
                           endpoint: str, error_shape: str) -> Response:
        def err(status, msg):
            if error_shape == "anthropic":
                return Response.json(
                    {"type": "error",
                     "error": {"type": "invalid_request_error",
                               "message": msg}}, status)
            return Response.json({"error": {"message": msg}}, status)

        try:
            body = req.json()
        except Exception:
            return err(400, "invalid JSON body")
        if not isinstance(body, dict) or not body.get("model"):
            return err(400, "model is required")
        pid, model = routing.determine_provider_and_model(body["model"])
        if pid is None:
            pid = req.query.get("provider", required_provider).lower()
        if pid != required_provider:
            return err(400,
                       f"this endpoint only supports the {required_provider} "
                       f"provider")
        try:
            provider = self.registry.build(pid)
        except HTTPError as e:
            return err(e.status, e.message)
        raw = req.body
        if model != body["model"]:
            body = dict(body)
            body["model"] = model
            raw = json.dumps(body).encode()
        stream = bool(body.get("stream"))
        headers = {"content-type": "application/json"}
        if stream:
            async def gen():
                async with provider.stream_raw("POST", endpoint, raw,
                                               headers) as r:
                    if r.status_code >= 400:
                        payload = (await r.aread()).decode(
                            errors="replace")
                        yield "data: " + payload
                        return
                    async for ev in _relay_sse_events(
                            r.aiter_lines()):
                        yield ev
            return Response.sse(gen())
        try:
            r = await provider.forward_raw("POST", endpoint, raw, headers)
        except Exception as e:
            return err(502, f"upstream error: {e}")
        return Response(r.status_code, r.content,
                        {"content-type": r.headers.get("content-type",
                                                       "application/json")})

    async def images(self, req: Request) -> Response:
        if not self.cfg.general.enable_images:
            return Response.error(404, "images API is disabled")
        endpoint = {
            "/v1/images/generations": OPENAI_IMAGES_ENDPOINT,
            "/v1/images/edits": OPENAI_IMAGES_EDITS_ENDPOINT,
            "/v1/images/variations": OPENAI_IMAGES_VARIATIONS_ENDPOINT,
        }[req.path]
        ct = req.headers.get("content-type", "application/json")
        body = req.body
        pid = "openai"
        if ct.startswith("multipart/form-data"):
            # multipart (edits/variations with binary image parts): sniff
            # the model field, strip a provider prefix, re-encode in place
            # (reference handleImagesMultipart, routes.go:1388-1575)
            body, pid = self._rewrite_multipart_model(body, ct) or (body, pid)
        else:
            try:
                parsed = json.loads(body or b"{}")
            except Exception:
                parsed = {}
            model = parsed.get("model") if isinstance(parsed, dict) else None
            if model:
                p, m = routing.determine_provider_and_model(model)
                if p is not None:
                    pid = p
                    parsed["model"] = m
                    body = json.dumps(parsed).encode()
        try:
            provider = self.registry.build(pid)
        except HTTPError as e:
            return Response.error(e.status, e.message)
        try:
            r = await provider.forward_raw("POST", endpoint, body,
                                           {"content-type": ct})
        except Exception as e:
            return Response.error(502, f"upstream error: {e}")
        return Response(r.status_code, r.content,
                        {"content-type": r.headers.get("content-type",
                                                       "application/json")})

    @staticmethod
    def _rewrite_multipart_model(body: bytes, content_type: str):
        """Strip a provider prefix from the multipart `model` field.
        Returns (new_body, provider_id) or None if no prefixed model."""
        marker = "boundary="
        i = content_type.find(marker)
        if i < 0:
            return None
        boundary = content_type[i + len(marker):].split(";")[0].strip()
        boundary = boundary.strip('"')
        delim = ("--" + boundary).encode()
        parts = body.split(delim)
        for pi, part in enumerate(parts):
            if b'name="model"' not in part:
                continue
            head, sep, value = part.partition(b"\r\n\r\n")
            if not sep:
                continue
            value_body = value.rstrip(b"\r\n")
            model = value_body.decode(errors="replace")
            pid, bare = routing.determine_provider_and_model(model)
            if pid is None:
                return None
            parts[pi] = head + sep + bare.encode() + b"\r\n"
            return delim.join(parts), pid
        return None

    # ---------------- /proxy/{provider}/... ----------------
    async def proxy(self, req: Request) -> Response:
        """Generic reverse proxy (reference ProxyHandler routes.go:102-330):
        caller's Authorization is stripped; provider credential injected."""
        pid = req.path_params.get("provider", "")
        rest = req.path_params.get("rest", "")
        try:
            provider = self.registry.build(pid)
        except HTTPError as e:
            return Response.error(e.status, e.message)
        if pid == LOCAL_PROVIDER_ID:
            return Response.error(400, "local provider has no proxy path")
        endpoint = "/" + rest if rest else "/"
        if req.query:
            # the caller's query string travels with the proxied path
            # (reference ProxyHandler forwards the full RequestURI)
            from urllib.parse import urlencode

            endpoint += "?" + urlencode(req.query)
        headers = {}
        ct = req.headers.get("content-type")
        if ct:
            headers["content-type"] = ct
        accept = req.headers.get("accept", "")
        want_stream = "text/event-stream" in accept
        try:
            if want_stream:
                async def gen():
                    async with provider.stream_raw(req.method, endpoint,
                                                   req.body, headers) as r:
                        async for ev in _relay_sse_events(
                                r.aiter_lines()):
                            yield ev
                return Response.sse(gen())
            r = await provider.forward_raw(req.method, endpoint, req.body,
                                           headers)
        except Exception as e:
            return Response.error(502, f"upstream error: {e}")
        return Response(r.status_code, r.content,
                        {"content-type": r.headers.get("content-type",
                                                       "application/json")})

    # ---------------- GET /v1/mcp/tools ----------------
    async def list_tools(self, req: Request) -> Response:
        if self.mcp is None or not self.cfg.mcp.expose:
            return Response.error(
                403, "MCP tools endpoint is not exposed")
        tools = self.mcp.client.chat_tools()
        return Response.json({"object": "list",
                              "data": [t["function"] | {
                                  "server": self.mcp.client.tool_server(
                                      t["function"]["name"]) or ""}
                                  for t in tools]})
