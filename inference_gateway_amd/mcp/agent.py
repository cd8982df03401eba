"""MCP tool-call agent loop (reference internal/mcp/agent.go).

Non-streaming Run (agent.go:90-139): while the provider response contains
tool calls, execute them, extend the message history, re-ask — at most 10
iterations (agent.go:24). Streaming RunWithStream (agent.go:151-313): per
iteration the provider stream is forwarded to the client in real time while
tool-call deltas are accumulated by index (types/toolcalls.go:11-64 —
unnamed calls are dropped); `data: [DONE]` is emitted exactly once at the
very end. The two selector meta-tools are handled gateway-side
(agent.go:319-341); every dispatch runs the guardrail tool hooks
(agent.go:396-449).
"""
from __future__ import annotations

import json
import time
from typing import AsyncIterator, Dict, List

from .selector import (SELECTOR_TOOL_EXECUTE, SELECTOR_TOOL_GET,
                       handle_tools_get)

MAX_AGENT_ITERATIONS = 10


def accumulate_streaming_tool_calls(lines: List[str]) -> List[dict]:
    """Reconstruct complete tool calls from SSE data lines
    (types/toolcalls.go:11-64): accumulate by delta index; a call that never
    received a function name is dropped."""
    acc: Dict[int, dict] = {}
    for line in lines:
        data = line.strip()
        if data.startswith("data:"):
            data = data[5:].strip()
        if not data or data == "[DONE]":
            continue
        try:
            chunk = json.loads(data)
        except json.JSONDecodeError:
            continue
        choices = chunk.get("choices") or []
        if not choices:
            continue
        deltas = (choices[0].get("delta") or {}).get("tool_calls")
        if not deltas:
            continue
        for d in deltas:
            idx = d.get("index", 0)
            tc = acc.setdefault(idx, {
                "id": "",
                "type": "function",
                "function": {"name": "", "arguments": ""},
            })
            if d.get("id"):
                tc["id"] = d["id"]
            if d.get("type"):
                tc["type"] = d["type"]
            fn = d.get("function") or {}
            if fn.get("name"):
                tc["function"]["name"] = fn["name"]
            if fn.get("arguments"):
                tc["function"]["arguments"] += fn["arguments"]
    out = []
    for i in sorted(acc):        # sparse indices must not drop calls
        tc = acc[i]
        if tc["function"]["name"]:
            out.append(tc)
    return out


class Agent:
    def __init__(self, client, logger, guardrails=None, telemetry=None,
                 tracer=None):
        self.client = client
        self.logger = logger
        self.guardrails = guardrails
        self.telemetry = telemetry
        self.tracer = tracer

    # ---- tool execution -------------------------------------------------
    async def _dispatch(self, name: str, args: dict) -> str:
        """Guardrail pre/post hooks around one tool execution
        (agent.go:396-449)."""
        if self.guardrails is not None:
            decision = await self.guardrails.evaluate_tool_call(name, args)
            if decision.action == "block":
                return json.dumps({
                    "error": decision.message or "tool call blocked by "
                    "guardrails"})
        from ..otel.tracing import current_span

        start = time.monotonic()
        # explicit `execute_tool <name>` span (reference agent.go:402-429)
        parent = current_span.get()
        span = None
        if self.tracer is not None:
            span = self.tracer.start_span(f"execute_tool {name}",
                                          parent=parent, kind="internal")
            span.set("tool.name", name)
            token = current_span.set(span)
        try:
            output = await self.client.execute_tool(name, args)
        except Exception as e:
            output = json.dumps({"error": str(e)})
            if span is not None:
                span.status_ok = False
        finally:
            if span is not None:
                current_span.reset(token)
                self.tracer.finish(span)
        if self.telemetry is not None:
            self.telemetry.tool_duration.labels(name, "", "").observe(
                time.monotonic() - start)
        if self.guardrails is not None:
            decision = await self.guardrails.evaluate_tool_output(name,
                                                                  output)
            if decision.action == "block":
                return json.dumps({
                    "error": decision.message or "tool output blocked by "
                    "guardrails"})
            if decision.action == "redact" and decision.content is not None:
                # decision content includes the "name " prefix; strip it
                output = decision.content.split(" ", 1)[-1]
        return output

    async def execute_tools(self, tool_calls: List[dict], provider_id: str,
                            model: str) -> List[dict]:
        """Run each tool call; selector meta-tools are served gateway-side
        (agent.go:319-341). Returns role=tool messages."""
        messages = []
        for tc in tool_calls:
            fn = tc.get("function") or {}
            name = fn.get("name", "")
            try:
                args = json.loads(fn.get("arguments") or "{}")
            except json.JSONDecodeError:
                args = {}
            if name == SELECTOR_TOOL_GET:
                content = handle_tools_get(self.client, args)
            elif name == SELECTOR_TOOL_EXECUTE:
                inner = args.get("name", "")
                content = await self._dispatch(inner,
                                               args.get("arguments") or {})
            else:
                content = await self._dispatch(name, args)
            if self.telemetry is not None:
                self.telemetry.record_tool_call(provider_id, model, "mcp",
                                                name)
            messages.append({
                "role": "tool",
                "tool_call_id": tc.get("id") or name,
                "content": content,
            })
        return messages

    # ---- non-streaming loop ---------------------------------------------
    async def run(self, provider, req: dict, provider_id: str = "",
                  model: str = "") -> dict:
        req = dict(req)
        messages = list(req.get("messages") or [])
        resp: dict = {}
        for _ in range(MAX_AGENT_ITERATIONS):
            req["messages"] = messages
            resp = await provider.chat_completions(req)
            choice = (resp.get("choices") or [{}])[0]
            msg = choice.get("message") or {}
            tool_calls = msg.get("tool_calls") or []
            if not tool_calls:
                return resp
            messages.append(msg)
            tool_msgs = await self.execute_tools(tool_calls, provider_id,
                                                 model)
            messages.extend(tool_msgs)
        return resp

    # ---- streaming loop -------------------------------------------------
    async def run_with_stream(self, provider, req: dict,
                              provider_id: str = "",
                              model: str = "") -> AsyncIterator[str]:
        req = dict(req)
        messages = list(req.get("messages") or [])
        try:
            for _ in range(MAX_AGENT_ITERATIONS):
                req["messages"] = messages
                lines: List[str] = []
                content_parts: List[str] = []
                async for line in provider.stream_chat_completions(req):
                    if not line:
                        continue
                    data = line[5:].strip() if line.startswith("data:") \
                        else line.strip()
                    if data == "[DONE]":
                        continue  # emitted exactly once at the very end
                    lines.append(line)
                    yield line
                    try:
                        chunk = json.loads(data)
                        delta = (chunk.get("choices") or [{}])[0].get(
                            "delta") or {}
                        if delta.get("content"):
                            content_parts.append(delta["content"])
                    except Exception:
                        pass
                tool_calls = accumulate_streaming_tool_calls(lines)
                if not tool_calls:
                    break
                messages.append({
                    "role": "assistant",
                    "content": "".join(content_parts) or None,
                    "tool_calls": tool_calls,
                })
                tool_msgs = await self.execute_tools(tool_calls, provider_id,
                                                     model)
                messages.extend(tool_msgs)
        except Exception as e:
            yield "data: " + json.dumps({"error": {"message": str(e)}})
        # exactly one [DONE], after the loop or an error frame — never from
        # a finally block (yield-in-finally breaks GeneratorExit on client
        # disconnect)
        yield "data: [DONE]"
