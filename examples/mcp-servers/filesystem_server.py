#!/usr/bin/env python3
"""Example filesystem MCP server (streamable-HTTP JSON-RPC), the analog of
the reference's examples/docker-compose/mcp/filesystem-server (7 tools,
main.go:192-447).  Serves the MCP `initialize` / `tools/list` /
`tools/call` methods over plain ASGI; run with:

    SANDBOX_DIR=/tmp/fs python -m uvicorn \
        --app-dir examples/mcp-servers filesystem_server:app --port 8083

The gateway's MCP tests also mount `app` in-process (httpx ASGITransport)
so the client/agent loop is exercised against a real server
implementation, not a canned mock.
"""
from __future__ import annotations

import json
import os
import shutil
from pathlib import Path

PROTOCOL_VERSION = "2024-11-05"


def _sandbox() -> Path:
    root = Path(os.environ.get("SANDBOX_DIR", "/tmp/mcp-fs")).resolve()
    root.mkdir(parents=True, exist_ok=True)
    return root


def _safe(path: str) -> Path:
    root = _sandbox()
    p = (root / path.lstrip("/")).resolve()
    if not str(p).startswith(str(root)):
        raise ValueError("path escapes the sandbox")
    return p


TOOLS = [
    {"name": "read_file",
     "description": "Read a file's contents from the sandbox",
     "inputSchema": {"type": "object", "required": ["path"],
                     "properties": {"path": {"type": "string"}}}},
    {"name": "write_file",
     "description": "Write content to a file in the sandbox",
     "inputSchema": {"type": "object", "required": ["path", "content"],
                     "properties": {"path": {"type": "string"},
                                    "content": {"type": "string"}}}},
    {"name": "list_directory",
     "description": "List a directory in the sandbox",
     "inputSchema": {"type": "object",
                     "properties": {"path": {"type": "string"}}}},
    {"name": "create_directory",
     "description": "Create a directory in the sandbox",
     "inputSchema": {"type": "object", "required": ["path"],
                     "properties": {"path": {"type": "string"}}}},
    {"name": "delete_file",
     "description": "Delete a file from the sandbox",
     "inputSchema": {"type": "object", "required": ["path"],
                     "properties": {"path": {"type": "string"}}}},
    {"name": "move_file",
     "description": "Move or rename a file inside the sandbox",
     "inputSchema": {"type": "object", "required": ["source", "destination"],
                     "properties": {"source": {"type": "string"},
                                    "destination": {"type": "string"}}}},
    {"name": "search_files",
     "description": "Search for files matching a substring",
     "inputSchema": {"type": "object", "required": ["pattern"],
                     "properties": {"pattern": {"type": "string"}}}},
]


def call_tool(name: str, args: dict) -> str:
    if name == "read_file":
        return _safe(args["path"]).read_text()
    if name == "write_file":
        p = _safe(args["path"])
        p.parent.mkdir(parents=True, exist_ok=True)
        p.write_text(args["content"])
        return f"wrote {len(args['content'])} bytes to {args['path']}"
    if name == "list_directory":
        p = _safe(args.get("path", "."))
        return "\n".join(sorted(
            ("[dir] " if c.is_dir() else "[file] ") + c.name
            for c in p.iterdir()))
    if name == "create_directory":
        _safe(args["path"]).mkdir(parents=True, exist_ok=True)
        return f"created {args['path']}"
    if name == "delete_file":
        _safe(args["path"]).unlink()
        return f"deleted {args['path']}"
    if name == "move_file":
        src, dst = _safe(args["source"]), _safe(args["destination"])
        dst.parent.mkdir(parents=True, exist_ok=True)
        shutil.move(str(src), str(dst))
        return f"moved {args['source']} -> {args['destination']}"
    if name == "search_files":
        root = _sandbox()
        hits = [str(p.relative_to(root)) for p in root.rglob("*")
                if args["pattern"] in p.name]
        return "\n".join(sorted(hits)) or "no matches"
    raise ValueError(f"unknown tool: {name}")


def handle_rpc(body: dict) -> dict:
    rid = body.get("id")
    method = body.get("method", "")
    if method == "initialize":
        result = {"protocolVersion": PROTOCOL_VERSION,
                  "capabilities": {"tools": {}},
                  "serverInfo": {"name": "filesystem-server",
                                 "version": "1.0.0"}}
    elif method == "notifications/initialized":
        return {}
    elif method == "tools/list":
        result = {"tools": TOOLS}
    elif method == "tools/call":
        params = body.get("params") or {}
        try:
            text = call_tool(params.get("name", ""),
                             params.get("arguments") or {})
            result = {"content": [{"type": "text", "text": text}],
                      "isError": False}
        except Exception as e:  # tool errors are results, not RPC errors
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
    payload = json.dumps(resp).encode()
    await send({"type": "http.response.start", "status": 200,
                "headers": [(b"content-type", b"application/json"),
                            (b"mcp-session-id", b"fs-session-1")]})
    await send({"type": "http.response.body", "body": payload})
