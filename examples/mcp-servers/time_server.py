#!/usr/bin/env python3
"""Example time MCP server (the analog of the reference's mcp/time-server):
two tools — current_time and elapsed_days."""
from __future__ import annotations

import datetime
import json

TOOLS = [
    {"name": "current_time", "description": "Current UTC time (ISO-8601)",
     "inputSchema": {"type": "object", "properties": {}}},
    {"name": "elapsed_days",
     "description": "Days elapsed since an ISO date",
     "inputSchema": {"type": "object", "required": ["since"],
                     "properties": {"since": {"type": "string"}}}},
]


def call_tool(name: str, args: dict) -> str:
    now = datetime.datetime.now(datetime.timezone.utc)
    if name == "current_time":
        return now.isoformat()
    if name == "elapsed_days":
        then = datetime.datetime.fromisoformat(args["since"])
        if then.tzinfo is None:
            then = then.replace(tzinfo=datetime.timezone.utc)
        return str((now - then).days)
    raise ValueError(f"unknown tool: {name}")


def handle_rpc(body: dict) -> dict:
    rid, method = body.get("id"), body.get("method", "")
    if method == "initialize":
        result = {"protocolVersion": "2024-11-05",
                  "capabilities": {"tools": {}},
                  "serverInfo": {"name": "time-server", "version": "1.0.0"}}
    elif method == "notifications/initialized":
        return {}
    elif method == "tools/list":
        result = {"tools": TOOLS}
    elif method == "tools/call":
        p = body.get("params") or {}
        try:
            text = call_tool(p.get("name", ""), p.get("arguments") or {})
            result = {"content": [{"type": "text", "text": text}],
                      "isError": False}
        except Exception as e:
            result = {"content": [{"type": "text", "text": str(e)}],
                      "isError": True}
    else:
        return {"jsonrpc": "2.0", "id": rid,
                "error": {"code": -32601, "message": f"unknown: {method}"}}
    return {"jsonrpc": "2.0", "id": rid, "result": result}


async def app(scope, receive, send):
    if scope["type"] != "http":
        return
    body = b""
    while True:
        msg = await receive()
        body += msg.get("body", b"")
        if not msg.get("more_body"):
            break
    try:
        resp = handle_rpc(json.loads(body or b"{}"))
    except Exception as e:
        resp = {"jsonrpc": "2.0", "id": None,
                "error": {"code": -32700, "message": str(e)}}
    await send({"type": "http.response.start", "status": 200,
                "headers": [(b"content-type", b"application/json")]})
    await send({"type": "http.response.body",
                "body": json.dumps(resp).encode()})
