"""Selector meta-tools (reference internal/mcp/selector.go:20-146).

The default MCP_TOOL_MODE=selector injects only two meta-tools instead of
every tool schema (request-size reduction is a reference benchmark bar,
tests/mcp_selector_bench_test.go): mcp_tools_get (catalog query / full
schemas by name) and mcp_tools_execute (dispatch by name).
"""
from __future__ import annotations

import json
from typing import List

SELECTOR_TOOL_GET = "mcp_tools_get"
SELECTOR_TOOL_EXECUTE = "mcp_tools_execute"


def selector_tool_definitions() -> List[dict]:
    return [
        {
            "type": "function",
            "function": {
                "name": SELECTOR_TOOL_GET,
                "description": (
                    "List available MCP tools or get their full input "
                    "schemas. Call without arguments for a compact catalog "
                    "(name + one-line description). Call with names to get "
                    "full schemas before executing."),
                "parameters": {
                    "type": "object",
                    "properties": {
                        "query": {
                            "type": "string",
                            "description": "Optional substring/keyword "
                            "filter on tool names and descriptions",
                        },
                        "names": {
                            "type": "array",
                            "items": {"type": "string"},
                            "description": "Return full input schemas for "
                            "these tools",
                        },
                    },
                },
            },
        },
        {
            "type": "function",
            "function": {
                "name": SELECTOR_TOOL_EXECUTE,
                "description": ("Execute an MCP tool by name. Look up the "
                                "tool and its schema with mcp_tools_get "
                                "first."),
                "parameters": {
                    "type": "object",
                    "properties": {
                        "name": {"type": "string",
                                 "description": "Tool name from "
                                 "mcp_tools_get"},
                        "arguments": {"type": "object",
                                      "description": "Arguments for the "
                                      "tool"},
                    },
                    "required": ["name"],
                },
            },
        },
    ]


def handle_tools_get(client, args: dict) -> str:
    """Catalog query by substring, or full schemas by name
    (selector.go:104-146)."""
    names = args.get("names") or []
    query = (args.get("query") or "").lower()
    entries = []
    for conn in client.servers.values():
        if not conn.available:
            continue
        for t in conn.tools:
            name = t.get("name", "")
            from .client import is_tool_allowed, normalize_tool_name

            if not is_tool_allowed(name, client.cfg.include_tools,
                                   client.cfg.exclude_tools):
                continue
            if names:
                wanted = {normalize_tool_name(x) for x in names}
                if normalize_tool_name(name) not in wanted:
                    continue
                entries.append({
                    "name": f"mcp_{name}",
                    "description": t.get("description", ""),
                    "server": conn.url,
                    "input_schema": t.get("inputSchema") or {},
                })
            else:
                desc = (t.get("description") or "").split("\n")[0]
                if query and query not in name.lower() \
                        and query not in desc.lower():
                    continue
                entries.append({
                    "name": f"mcp_{name}",
                    "description": desc,
                    "server": conn.url,
                })
    return json.dumps({"tools": entries})
