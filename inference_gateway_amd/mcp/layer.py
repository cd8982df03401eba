"""MCP layer facade: the analog of the reference's MCP middleware
(api/middlewares/mcp.go:86-308) in the rebuild's direct-call architecture —
tool injection (selector meta-tools by default, every schema in direct
mode), then the agent loop for both streaming and non-streaming chat."""
from __future__ import annotations


from ..gateway.http import Request, Response
from .agent import Agent
from .client import MCPClient
from .selector import selector_tool_definitions

BYPASS_HEADER = "x-mcp-bypass"


class MCPLayer:
    def __init__(self, cfg, http_client, logger, guardrails=None,
                 telemetry=None, tracer=None):
        self.cfg = cfg
        self.logger = logger
        self.client = MCPClient(cfg, http_client, logger)
        self.agent = Agent(self.client, logger, guardrails=guardrails,
                           telemetry=telemetry, tracer=tracer)

    async def initialize(self) -> None:
        await self.client.initialize_all()

    async def stop(self) -> None:
        await self.client.stop()

    # ------------------------------------------------------------------
    def applies(self, req: Request, body: dict) -> bool:
        if req.headers.get(BYPASS_HEADER):
            return False
        return bool(self.client.chat_tools())

    def inject_tools(self, body: dict) -> dict:
        """selector mode injects 2 meta-tools; direct mode injects every
        tool schema (mcp.go:128-139). The caller's own tools are kept."""
        body = dict(body)
        own = list(body.get("tools") or [])
        if self.cfg.tool_mode == "direct":
            injected = self.client.chat_tools()
        else:
            injected = selector_tool_definitions()
        existing = {(t.get("function") or {}).get("name") for t in own}
        body["tools"] = own + [t for t in injected
                               if (t["function"]["name"] not in existing)]
        return body

    async def run_chat(self, provider, body: dict, stream: bool,
                       req: Request) -> Response:
        body = self.inject_tools(body)
        pid = req.state.get("provider_id", "")
        model = req.state.get("model", "")
        if stream:
            return Response.sse(
                self.agent.run_with_stream(provider, body, pid, model))
        result = await self.agent.run(provider, body, pid, model)
        return Response.json(result)
