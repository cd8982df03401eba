#!/usr/bin/env python3
"""Example search MCP server (the analog of the reference's mcp/search
server, examples/docker-compose/tools): keyword search over a small
in-memory document corpus — two tools, `search` and `get_document`.

Run:  uvicorn search_server:app --port 8084
Then point the gateway at it:  MCP_SERVERS=http://localhost:8084/mcp
"""
from __future__ import annotations

import json

DOCS = {
    "mi355x-overview": (
        "The AMD Instinct MI355X accelerator has 256 compute units across "
        "8 XCDs, 288 GB of HBM3E at 8 TB/s, and MFMA matrix cores for "
        "bf16/fp8 GEMMs."),
    "gateway-quickstart": (
        "Start the inference gateway with server.py, set provider API "
        "keys via <ID>_API_KEY environment variables, and send OpenAI "
        "compatible requests to /v1/chat/completions."),
    "paged-kv-cache": (
        "The local engine stores keys and values in fixed-size blocks "
        "indexed by a per-sequence block table, so long contexts grow "
        "without fragmentation and preempted requests free whole blocks."),
    "mcp-agent-loop": (
        "When MCP tools are exposed, the gateway runs an agent loop of at "
        "most ten iterations: call the model, execute any requested "
        "tools, append the results, and re-ask until no tool calls "
        "remain."),
}

TOOLS = [
    {"name": "search",
     "description": "Keyword search over the document corpus; returns "
                    "matching document ids ranked by hit count",
     "inputSchema": {"type": "object", "required": ["query"],
                     "properties": {
                         "query": {"type": "string"},
                         "limit": {"type": "integer", "default": 5}}}},
    {"name": "get_document",
     "description": "Fetch the full text of a document by id",
     "inputSchema": {"type": "object", "required": ["id"],
                     "properties": {"id": {"type": "string"}}}},
]


def call_tool(name: str, args: dict) -> str:
    if name == "search":
        terms = [t for t in str(args["query"]).lower().split() if t]
        limit = int(args.get("limit", 5))
        scored = []
        for doc_id, text in DOCS.items():
            lower = text.lower()
            hits = sum(lower.count(t) for t in terms)
            if hits:
                scored.append((hits, doc_id))
        scored.sort(reverse=True)
        return json.dumps([{"id": d, "score": s}
                           for s, d in scored[:limit]])
    if name == "get_document":
        doc_id = str(args["id"])
        if doc_id not in DOCS:
            raise ValueError(f"no such document: {doc_id}")
        return DOCS[doc_id]
    raise ValueError(f"unknown tool: {name}")


def handle_rpc(body: dict) -> dict:
    rid, method = body.get("id"), body.get("method", "")
    if method == "initialize":
        result = {"protocolVersion": "2024-11-05",
                  "capabilities": {"tools": {}},
                  "serverInfo": {"name": "search-server",
                                 "version": "1.0.0"}}
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


if __name__ == "__main__":
    import uvicorn

    uvicorn.run(app, host="127.0.0.1", port=8084)
