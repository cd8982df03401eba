"""Gateway composition root (the analog of cmd/gateway/main.go:46-402).

Builds config -> logger -> telemetry -> middleware chain -> router ->
providers registry (+ in-process local engine provider) and exposes an ASGI
callable. Middleware order matches the reference (main.go:287-308):
tracing/logging -> telemetry -> auth -> guardrails -> MCP -> handler.
"""
from __future__ import annotations

import asyncio
from typing import Callable, List, Optional

from .. import config as config_mod
from ..logger import Logger
from ..providers.core import ProviderRegistry, build_http_client
from .http import Request, Response, parse_scope, send_response
from .routes import Router

Middleware = Callable  # async (Request, next) -> Response


class Gateway:
    def __init__(self, cfg: Optional[config_mod.Config] = None,
                 logger: Optional[Logger] = None,
                 local_provider=None,
                 http_client=None,
                 mcp_layer=None,
                 guardrails=None,
                 authenticator=None,
                 telemetry=None):
        self.cfg = cfg or config_mod.load()
        self.logger = logger or Logger(self.cfg.general.environment)
        self.client = http_client or build_http_client(self.cfg.client)
        if local_provider is None and self.cfg.local.enabled:
            from ..providers.local import LocalProvider

            local_provider = LocalProvider(self.cfg.local)
        self.local_provider = local_provider
        self.registry = ProviderRegistry(self.cfg, self.client,
                                         local_provider=local_provider)
        self.telemetry = telemetry
        if telemetry is None and self.cfg.telemetry.enabled:
            from ..otel.metrics import Telemetry

            self.telemetry = Telemetry()
        if (self.telemetry is not None and self.local_provider is not None
                and hasattr(self.local_provider, "engine")):
            self.telemetry.register_engine_gauges(
                self.local_provider.engine.engine)
        self.tracer = None
        if self.cfg.telemetry.tracing_enabled:
            from ..otel.tracing import Tracer, install_client_hooks

            self.tracer = Tracer(
                endpoint=self.cfg.telemetry.tracing_otlp_endpoint,
                client=self.client)
            install_client_hooks(self.client, self.tracer)
        self.guardrails = guardrails
        if guardrails is None and self.cfg.guardrails.enabled:
            from ..guardrails.engine import Evaluator

            self.guardrails = Evaluator.from_config(self.cfg.guardrails,
                                                    self.logger)
        self.authenticator = authenticator
        if authenticator is None and self.cfg.auth.enabled:
            from .middlewares.auth import OIDCAuthenticator

            self.authenticator = OIDCAuthenticator(self.cfg.auth,
                                                   self.client, self.logger)
        self.mcp = mcp_layer
        if mcp_layer is None and self.cfg.mcp.enabled:
            from ..mcp.layer import MCPLayer

            self.mcp = MCPLayer(self.cfg.mcp, self.client, self.logger,
                                guardrails=self.guardrails,
                                telemetry=self.telemetry,
                                tracer=self.tracer)
        selector = None
        if self.cfg.routing.enabled:
            from ..providers import routing as routing_mod

            pools = routing_mod.load_pools_config(self.cfg.routing.config_path)
            selector = routing_mod.new_selector(pools)
        self.router = Router(self.cfg, self.registry, self.logger,
                             selector=selector, mcp=self.mcp,
                             telemetry=self.telemetry)
        self._routes = self._build_routes()
        self._chain = self._build_chain()

    # ------------------------------------------------------------------
    def _build_routes(self):
        r = self.router
        routes = {
            ("GET", "/health"): r.health,
            ("GET", "/v1/models"): r.list_models,
            ("POST", "/v1/chat/completions"): r.chat_completions,
            ("POST", "/v1/messages"): r.messages,
            ("POST", "/v1/responses"): r.responses,
            ("POST", "/v1/images/generations"): r.images,
            ("POST", "/v1/images/edits"): r.images,
            ("POST", "/v1/images/variations"): r.images,
            ("GET", "/v1/mcp/tools"): r.list_tools,
        }
        if self.cfg.telemetry.metrics_push_enabled and self.telemetry:
            from ..otel.ingest import MetricsIngestion

            self._ingest = MetricsIngestion(self.telemetry, self.logger)
            routes[("POST", "/v1/metrics")] = self._ingest.handle
        return routes

    def _build_chain(self):
        from .middlewares.logging_mw import logging_middleware
        from .middlewares.telemetry_mw import telemetry_middleware

        mws: List[Middleware] = []
        if self.tracer is not None:
            from ..otel.tracing import tracing_middleware

            mws.append(tracing_middleware(self.tracer))
        mws.append(logging_middleware(self.logger))
        if self.cfg.general.environment == "development":
            from .middlewares.dev_mw import dev_logging_middleware

            mws.append(dev_logging_middleware(self.logger))
        if self.telemetry is not None:
            mws.append(telemetry_middleware(self.telemetry))
        if self.authenticator is not None:
            mws.append(self.authenticator.middleware())
        if self.guardrails is not None:
            from .middlewares.guardrails_mw import guardrails_middleware

            mws.append(guardrails_middleware(self.guardrails, self.logger))
        return mws

    def _lookup(self, method: str, path: str):
        h = self._routes.get((method, path))
        if h is not None:
            return h, {}
        if path.startswith("/proxy/"):
            rest = path[len("/proxy/"):]
            provider, _, tail = rest.partition("/")
            return self.router.proxy, {"provider": provider, "rest": tail}
        return None, {}

    async def handle(self, req: Request) -> Response:
        handler, params = self._lookup(req.method, req.path)
        if handler is None:
            return Response.error(404, "not found")
        req.path_params = params

        async def terminal(r: Request) -> Response:
            return await handler(r)

        nxt = terminal
        for mw in reversed(self._chain):
            nxt = self._wrap(mw, nxt)
        return await nxt(req)

    @staticmethod
    def _wrap(mw, nxt):
        async def wrapped(r: Request) -> Response:
            return await mw(r, nxt)

        return wrapped

    # ------------------------------------------------------------------
    async def __call__(self, scope, receive, send):
        if scope["type"] == "lifespan":
            while True:
                msg = await receive()
                if msg["type"] == "lifespan.startup":
                    await self.startup()
                    await send({"type": "lifespan.startup.complete"})
                elif msg["type"] == "lifespan.shutdown":
                    await self.shutdown()
                    await send({"type": "lifespan.shutdown.complete"})
                    return
        if scope["type"] != "http":
            return
        body = b""
        limit = self.cfg.server.max_request_body_size
        while True:
            msg = await receive()
            if msg["type"] == "http.request":
                body += msg.get("body", b"")
                if len(body) > limit:
                    await send_response(
                        send, Response.error(413, "request body too large"))
                    return
                if not msg.get("more_body"):
                    break
            elif msg["type"] == "http.disconnect":
                return
        req = parse_scope(scope, body)
        req.state["asgi_extensions"] = scope.get("extensions") or {}
        try:
            resp = await self.handle(req)
        except Exception as e:  # last-resort 500
            self.logger.error("handler panic", "error", str(e),
                              "path", req.path)
            resp = Response.error(500, "internal error")
        await send_response(send, resp)

    # ------------------------------------------------------------------
    async def startup(self) -> None:
        if self.mcp is not None:
            await self.mcp.initialize()
        if self.authenticator is not None:
            try:
                await self.authenticator.initialize()
            except Exception as e:
                self.logger.error("OIDC discovery failed", "error", str(e))
        # async provider validation (reference main.go:352-382, warn-only)
        asyncio.ensure_future(self._validate_providers())
        if self.tracer is not None and self.tracer.endpoint:
            self._flusher = asyncio.ensure_future(self.tracer.run_flusher())

    async def _validate_providers(self) -> None:
        for pid in self.registry.configured_ids():
            try:
                p = self.registry.build(pid)
                await p.list_models()
                self.logger.debug("provider ok", "provider", pid)
            except Exception as e:
                self.logger.warn("provider validation failed",
                                 "provider", pid, "error", str(e))

    async def shutdown(self) -> None:
        if self.mcp is not None:
            await self.mcp.stop()
        if self.tracer is not None:
            fl = getattr(self, "_flusher", None)
            if fl is not None:
                fl.cancel()
            await self.tracer.flush()
        try:
            await self.client.aclose()
        except Exception:
            pass
        if self.local_provider is not None:
            try:
                self.local_provider.engine.shutdown()
            except Exception:
                pass
