"""Provider abstraction.

The reference reaches every upstream through a self-proxy HTTP hop
(providers/core/provider.go:83-298 + client/client.go:70-79);
this rebuild keeps the provider interface and auth semantics but calls
upstreams DIRECTLY over a shared async HTTP client — the middleware chain
runs once, not twice — and the `local` provider is an in-process engine
call with no HTTP at all (SURVEY.md §7 design stance).

Auth injection parity (api/routes.go:287-312): the caller's Authorization
header is ALWAYS stripped before talking to an upstream; the provider's own
credential is attached according to its auth type.
"""
from __future__ import annotations

import json
import time
from typing import AsyncIterator, Dict, Optional

import httpx

from .constants import AUTH_BEARER, AUTH_NONE, AUTH_QUERY, AUTH_XHEADER, ProviderMeta
from .transformers import transform_list_models


class HTTPError(Exception):
    def __init__(self, status: int, message: str):
        super().__init__(message)
        self.status = status
        self.message = message


def build_http_client(cfg=None) -> httpx.AsyncClient:
    """Shared outbound client, tuned like the reference's http.Transport
    (providers/client/client.go:45-56): keep-alives, no compression for
    streaming latency."""
    timeout = httpx.Timeout(cfg.timeout if cfg else 30.0, connect=10.0)
    limits = httpx.Limits(
        max_keepalive_connections=cfg.max_idle_conns if cfg else 20,
        keepalive_expiry=cfg.idle_conn_timeout if cfg else 30.0,
    )
    return httpx.AsyncClient(timeout=timeout, limits=limits,
                             headers={"Accept-Encoding": "identity"})


class RemoteProvider:
    """A remote upstream (OpenAI, Anthropic, Groq, Ollama, ...)."""

    def __init__(self, meta: ProviderMeta, url: str, token: str,
                 client: httpx.AsyncClient):
        self.meta = meta
        self.id = meta.id
        self.url = url.rstrip("/")
        self.token = token
        self.client = client

    # ---- auth -----------------------------------------------------------
    def auth_headers(self) -> Dict[str, str]:
        # every outbound hop carries the active request's W3C traceparent
        # (reference core/provider.go:115)
        from ..otel.tracing import inject_headers

        h = inject_headers(dict(self.meta.extra_headers))
        if self.meta.auth_type == AUTH_BEARER and self.token:
            h["Authorization"] = f"Bearer {self.token}"
        elif self.meta.auth_type == AUTH_XHEADER and self.token:
            h["x-api-key"] = self.token
        return h

    def auth_params(self) -> Dict[str, str]:
        if self.meta.auth_type == AUTH_QUERY and self.token:
            return {"key": self.token}
        return {}

    # ---- API ------------------------------------------------------------
    async def list_models(self) -> dict:
        url = self.url + self.meta.models_endpoint
        r = await self.client.get(url, headers=self.auth_headers(),
                                  params=self.auth_params() or None)
        if r.status_code >= 400:
            raise HTTPError(r.status_code, r.text[:2048])
        return transform_list_models(self.id, r.json())

    async def chat_completions(self, req: dict,
                               extra_headers: Optional[dict] = None) -> dict:
        url = self.url + self.meta.chat_endpoint
        headers = self.auth_headers()
        if extra_headers:
            headers.update(extra_headers)
        r = await self.client.post(url, json=req, headers=headers,
                                   params=self.auth_params() or None)
        if r.status_code >= 400:
            raise HTTPError(r.status_code, r.text[:2048])
        return r.json()

    async def stream_chat_completions(
            self, req: dict,
            extra_headers: Optional[dict] = None) -> AsyncIterator[bytes]:
        """Yields raw SSE lines (without trailing newlines) from upstream."""
        url = self.url + self.meta.chat_endpoint
        headers = self.auth_headers()
        headers["Accept"] = "text/event-stream"
        if extra_headers:
            headers.update(extra_headers)
        async with self.client.stream("POST", url, json=req, headers=headers,
                                      params=self.auth_params() or None) as r:
            if r.status_code >= 400:
                body = await r.aread()
                raise HTTPError(r.status_code,
                                body.decode(errors="replace")[:2048])
            async for line in r.aiter_lines():
                yield line

    # raw passthrough (Messages / Responses / Images; reference
    # routes.go:842-1575 forwards bytes verbatim)
    async def forward_raw(self, method: str, endpoint: str, body: bytes,
                          headers: Dict[str, str]):
        url = self.url + endpoint
        h = self.auth_headers()
        for k, v in headers.items():
            if k.lower() in ("content-type", "accept"):
                h[k] = v
        return await self.client.request(method, url, content=body, headers=h,
                                         params=self.auth_params() or None)

    def stream_raw(self, method: str, endpoint: str, body: bytes,
                   headers: Dict[str, str]):
        url = self.url + endpoint
        h = self.auth_headers()
        for k, v in headers.items():
            if k.lower() in ("content-type", "accept"):
                h[k] = v
        return self.client.stream(method, url, content=body, headers=h,
                                  params=self.auth_params() or None)


class ProviderRegistry:
    """Builds provider objects from config (reference registry.go:25-63:
    rejects providers whose auth type needs a token that is not set)."""

    def __init__(self, cfg, client: httpx.AsyncClient,
                 local_provider=None):
        from .constants import PROVIDERS

        self._providers: Dict[str, RemoteProvider] = {}
        self._local = local_provider
        for pid, meta in PROVIDERS.items():
            entry = cfg.providers.get(pid)
            url = entry.url if entry else meta.url
            token = entry.token if entry else ""
            self._providers[pid] = RemoteProvider(meta, url, token, client)

    @property
    def local(self):
        return self._local

    def ids(self):
        ids = list(self._providers)
        if self._local is not None:
            ids.append(self._local.id)
        return ids

    def configured_ids(self):
        """Providers usable right now: local + remotes with credentials (or
        credential-free auth)."""
        out = []
        for pid, p in self._providers.items():
            if p.meta.auth_type == AUTH_NONE or p.token:
                out.append(pid)
        if self._local is not None:
            out.append(self._local.id)
        return out

    def build(self, pid: str):
        """Returns a provider or raises HTTPError(400/401) like the
        reference's BuildProvider."""
        if self._local is not None and pid == self._local.id:
            return self._local
        p = self._providers.get(pid)
        if p is None:
            raise HTTPError(400, f"unknown provider: {pid}")
        if p.meta.auth_type != AUTH_NONE and not p.token:
            raise HTTPError(
                401, f"provider {pid} requires an API key and none is configured")
        return p
