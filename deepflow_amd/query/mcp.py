"""MCP (Model Context Protocol) server — LLM tool surface.

Reference counterpart: server/mcp (mcp.go:39-66) exposes a single profile
analysis tool; ours exposes profile analysis plus DF-SQL query. Transport:
JSON-RPC 2.0 over HTTP POST /mcp (streamable-http style single endpoint).
"""
from __future__ import annotations

import json
from typing import Dict

from fastapi import Request

PROTOCOL_VERSION = "2024-11-05"

TOOLS = [
    {
        "name": "profile_analysis",
        "description": "Top-N hottest stacks from the continuous profiler "
                       "(CPU + GPU), optionally filtered by process name.",
        "inputSchema": {
            "type": "object",
            "properties": {
                "process_name": {"type": "string"},
                "top_n": {"type": "integer", "default": 10},
            },
        },
    },
    {
        "name": "query",
        "description": "Run a DF-SQL query over the observability store "
                       "(tables: l7_flow_log, l4_flow_log, application, "
                       "network; 'show tags from <table>' lists tags).",
        "inputSchema": {
            "type": "object",
            "properties": {"sql": {"type": "string"}},
            "required": ["sql"],
        },
    },
]


class McpServer:
    def __init__(self, engine, profile_pipeline):
        self.engine = engine
        self.profiles = profile_pipeline

    def _tool_profile(self, args: Dict) -> str:
        from ..ingest.profile_pipeline import build_flame
        st = self.profiles.store
        tree = build_flame(st.rows, st.id_to_loc,
                           process_name=args.get("process_name"))
        flat = []

        def walk(node, path):
            p = path + [node["name"]]
            if node["self"]:
                flat.append((node["self"], ";".join(p[1:])))
            for c in node["children"]:
                walk(c, p)

        walk(tree, [])
        flat.sort(reverse=True)
        top = flat[: int(args.get("top_n", 10))]
        total = tree["value"] or 1
        lines = [f"total samples: {tree['value']}"]
        for v, stack in top:
            lines.append(f"{v} ({100 * v / total:.1f}%)  {stack}")
        return "\n".join(lines)

    def _tool_query(self, args: Dict) -> str:
        r = self.engine.query(args["sql"])
        return json.dumps(r)

    def handle(self, req: Dict) -> Dict:
        rid = req.get("id")
        method = req.get("method", "")
        if method == "initialize":
            result = {"protocolVersion": PROTOCOL_VERSION,
                      "capabilities": {"tools": {}},
                      "serverInfo": {"name": "deepflow-amd-mcp",
                                     "version": "0.1.0"}}
        elif method == "tools/list":
            result = {"tools": TOOLS}
        elif method == "tools/call":
            params = req.get("params", {})
            name = params.get("name")
            args = params.get("arguments", {})
            try:
                if name == "profile_analysis":
                    text = self._tool_profile(args)
                elif name == "query":
                    text = self._tool_query(args)
                else:
                    return {"jsonrpc": "2.0", "id": rid,
                            "error": {"code": -32602,
                                      "message": f"unknown tool {name}"}}
                result = {"content": [{"type": "text", "text": text}]}
            except Exception as e:  # noqa: BLE001
                result = {"content": [{"type": "text",
                                       "text": f"error: {e}"}],
                          "isError": True}
        elif method == "notifications/initialized":
            return {}
        else:
            return {"jsonrpc": "2.0", "id": rid,
                    "error": {"code": -32601,
                              "message": f"unknown method {method}"}}
        return {"jsonrpc": "2.0", "id": rid, "result": result}

    def register(self, app) -> None:
        @app.post("/mcp")
        async def mcp(request: Request):
            body = await request.json()
            return self.handle(body)
