"""MCP client: JSON-RPC over streamable-HTTP with SSE fallback.

Parity with the reference's internal/mcp/{client,init,transport,health}.go:

* per-server connections with retry + exponential backoff capped at
  RetryInterval (init.go:150-228);
* streamable-HTTP first, '/mcp' -> '/sse' fallback URL on failure
  (transport.go:229-237, init.go:176-193);
* mcp-session-id capture/replay (transport.go:85-87,119-123);
* caller credentials are never forwarded to MCP servers (transport.go:59-61);
* SSE-framed JSON-RPC responses are unwrapped to plain JSON
  (transport.go:131-154);
* tool discovery via tools/list, include/exclude filtering (filter.go),
  cached chat-tool conversion with the mcp_ prefix (tools.go:92-114);
* background health polling using tools/list as the probe (health.go:67-106)
  and automatic reconnection (init.go:330-408).
"""
from __future__ import annotations

import asyncio
import itertools
import json
import time
from typing import Dict, List, Optional


def normalize_tool_name(name: str) -> str:
    n = name.strip().lower()
    if n.startswith("mcp_"):
        n = n[4:]
    return n


def parse_tool_list(items) -> set:
    if isinstance(items, str):
        items = items.split(",")
    return {normalize_tool_name(x) for x in items or [] if
            normalize_tool_name(x)}


def is_tool_allowed(name: str, include, exclude) -> bool:
    n = normalize_tool_name(name)
    inc = parse_tool_list(include)
    if inc:
        return n in inc
    exc = parse_tool_list(exclude)
    if exc:
        return n not in exc
    return True


def build_sse_fallback_url(url: str) -> str:
    """'/mcp' -> '/sse' (transport.go:229-237)."""
    if url.rstrip("/").endswith("/mcp"):
        return url.rstrip("/")[: -len("/mcp")] + "/sse"
    return url.rstrip("/") + "/sse"


class MCPServerConnection:
    def __init__(self, url: str, cfg, client, logger):
        self.url = url
        self.active_url = url
        self.cfg = cfg
        self.client = client
        self.logger = logger
        self.session_id: Optional[str] = None
        self.available = False
        self.tools: List[dict] = []
        self._rpc_id = itertools.count(1)

    async def rpc(self, method: str, params: Optional[dict] = None,
                  notification: bool = False) -> Optional[dict]:
        payload = {"jsonrpc": "2.0", "method": method}
        if params is not None:
            payload["params"] = params
        if not notification:
            payload["id"] = next(self._rpc_id)
        from ..otel.tracing import inject_headers

        # W3C trace context on every MCP hop (reference transport.go:63);
        # note caller credentials are never forwarded here (the MCP client
        # builds its own header set — transport.go:59-61)
        headers = inject_headers({
            "content-type": "application/json",
            "accept": "application/json, text/event-stream",
        })
        if self.session_id:
            headers["mcp-session-id"] = self.session_id
        r = await self.client.post(self.active_url, json=payload,
                                   headers=headers,
                                   timeout=self.cfg.request_timeout)
        sid = r.headers.get("mcp-session-id")
        if sid:
            self.session_id = sid
        if r.status_code >= 400:
            # mid-flight 4xx -> SSE fallback URL retry (transport.go:125-129)
            fallback = build_sse_fallback_url(self.url)
            if self.active_url != fallback and 400 <= r.status_code < 500:
                self.active_url = fallback
                return await self.rpc(method, params, notification)
            raise RuntimeError(
                f"MCP rpc {method} failed: HTTP {r.status_code}")
        if notification:
            return None
        return self._parse_response(r)

    @staticmethod
    def _parse_response(r) -> dict:
        ct = r.headers.get("content-type", "")
        text = r.text
        if "text/event-stream" in ct:
            # unwrap SSE frames: take the first data: payload that parses
            for line in text.splitlines():
                line = line.strip()
                if line.startswith("data:"):
                    data = line[5:].strip()
                    if data and data != "[DONE]":
                        try:
                            obj = json.loads(data)
                        except json.JSONDecodeError:
                            continue
                        if "result" in obj or "error" in obj:
                            return obj
            raise RuntimeError("no JSON-RPC payload in SSE response")
        return json.loads(text) if text else {}

    async def initialize(self) -> None:
        resp = await self.rpc("initialize", {
            "protocolVersion": "2024-11-05",
            "capabilities": {},
            "clientInfo": {"name": "inference-gateway-amd",
                           "version": "0.1.0"},
        })
        if resp is None or "error" in resp:
            raise RuntimeError(f"initialize failed: {resp}")
        await self.rpc("notifications/initialized", notification=True)
        await self.refresh_tools()
        self.available = True

    async def refresh_tools(self) -> None:
        resp = await self.rpc("tools/list", {})
        if resp is None or "error" in resp:
            raise RuntimeError(f"tools/list failed: {resp}")
        self.tools = resp.get("result", {}).get("tools", []) or []

    async def call_tool(self, name: str, arguments: dict) -> dict:
        resp = await self.rpc("tools/call",
                              {"name": name, "arguments": arguments})
        if resp is None:
            raise RuntimeError("empty tools/call response")
        if "error" in resp:
            raise RuntimeError(str(resp["error"]))
        return resp.get("result", {})


class MCPClient:
    def __init__(self, cfg, http_client, logger):
        self.cfg = cfg
        self.client = http_client
        self.logger = logger
        self.servers: Dict[str, MCPServerConnection] = {}
        self._chat_tools_cache: Optional[List[dict]] = None
        self._tasks: List[asyncio.Task] = []
        self._stopped = False

    # ---- lifecycle ------------------------------------------------------
    async def initialize_all(self) -> None:
        results = await asyncio.gather(
            *(self._init_server(url) for url in self.cfg.servers),
            return_exceptions=True)
        up = sum(1 for conn in self.servers.values() if conn.available)
        if up == 0 and self.cfg.servers and not self.cfg.enable_reconnect:
            raise RuntimeError("no MCP servers available at startup")
        if self.cfg.polling_enabled:
            self._tasks.append(asyncio.ensure_future(self._poll_loop()))
        if self.cfg.enable_reconnect:
            self._tasks.append(asyncio.ensure_future(self._reconnect_loop()))
        self.logger.info("MCP initialized", "servers", len(self.cfg.servers),
                         "available", up)
        del results

    async def _init_server(self, url: str) -> None:
        conn = MCPServerConnection(url, self.cfg, self.client, self.logger)
        self.servers[url] = conn
        backoff = self.cfg.initial_backoff
        for attempt in range(max(1, self.cfg.max_retries)):
            try:
                await conn.initialize()
                self._chat_tools_cache = None
                return
            except Exception as e:
                self.logger.warn("MCP server init failed", "server", url,
                                 "attempt", attempt + 1, "error", str(e))
                if attempt + 1 < self.cfg.max_retries:
                    await asyncio.sleep(
                        min(backoff, self.cfg.retry_interval))
                    backoff *= 2

    async def _poll_loop(self) -> None:
        while not self._stopped:
            await asyncio.sleep(self.cfg.polling_interval)
            for conn in self.servers.values():
                if not conn.available:
                    continue
                try:
                    await asyncio.wait_for(conn.refresh_tools(),
                                           self.cfg.polling_timeout)
                    self._chat_tools_cache = None
                except Exception:
                    conn.available = False
                    if not self.cfg.disable_healthcheck_logs:
                        self.logger.warn("MCP server unavailable",
                                         "server", conn.url)

    async def _reconnect_loop(self) -> None:
        while not self._stopped:
            await asyncio.sleep(self.cfg.reconnect_interval)
            for conn in self.servers.values():
                if conn.available:
                    continue
                try:
                    conn.active_url = conn.url
                    conn.session_id = None
                    await conn.initialize()
                    self._chat_tools_cache = None
                    self.logger.info("MCP server reconnected",
                                     "server", conn.url)
                except Exception:
                    pass

    async def stop(self) -> None:
        self._stopped = True
        for t in self._tasks:
            t.cancel()

    # ---- tools ----------------------------------------------------------
    def all_tools(self) -> List[dict]:
        """Filtered MCP tool dicts across available servers."""
        out = []
        for conn in self.servers.values():
            if not conn.available:
                continue
            for t in conn.tools:
                if is_tool_allowed(t.get("name", ""),
                                   self.cfg.include_tools,
                                   self.cfg.exclude_tools):
                    out.append(t)
        return out

    def chat_tools(self) -> List[dict]:
        """MCP tools converted to chat-completion function tools with the
        mcp_ prefix (tools.go:92-114)."""
        if self._chat_tools_cache is None:
            self._chat_tools_cache = [{
                "type": "function",
                "function": {
                    "name": f"mcp_{t.get('name', '')}",
                    "description": t.get("description", ""),
                    "parameters": t.get("inputSchema") or
                    {"type": "object", "properties": {}},
                },
            } for t in self.all_tools()]
        return self._chat_tools_cache

    def tool_server(self, name: str) -> Optional[str]:
        n = normalize_tool_name(name)
        for conn in self.servers.values():
            if not conn.available:
                continue
            for t in conn.tools:
                if normalize_tool_name(t.get("name", "")) == n:
                    return conn.url
        return None

    async def execute_tool(self, name: str, arguments: dict) -> str:
        """tools/call + content marshaling (tools.go:12-58)."""
        n = normalize_tool_name(name)
        server = self.tool_server(n)
        if server is None:
            raise RuntimeError(f"unknown MCP tool: {name}")
        conn = self.servers[server]
        # resolve the server's exact tool name (case preserved)
        exact = next((t["name"] for t in conn.tools
                      if normalize_tool_name(t.get("name", "")) == n), n)
        result = await conn.call_tool(exact, arguments)
        contents = result.get("content") or []
        parts = []
        for c in contents:
            if isinstance(c, dict) and c.get("type") == "text":
                parts.append(c.get("text", ""))
            else:
                parts.append(json.dumps(c))
        if result.get("isError"):
            return json.dumps({"error": "\n".join(parts) or "tool error"})
        return "\n".join(parts)
