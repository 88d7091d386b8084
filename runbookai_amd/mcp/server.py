"""MCP server exposing the knowledge base over stdio JSON-RPC.

Parity with reference src/mcp/server.ts (558 LoC): 5 tools —
search_runbooks @77, get_known_issues @102, search_postmortems @127,
get_knowledge_stats @155, list_services @163; stdio JSON-RPC loop
run_stdio_server @480; initialize / tools/list @457 / tools/call @419-427.
"""
from __future__ import annotations

import json
import sys
from typing import Any, Optional

PROTOCOL_VERSION = "2024-11-05"


class MCPServer:
    def __init__(self, retriever: Any = None) -> None:
        if retriever is None:
            from ..knowledge.retriever.default import create_retriever

            retriever = create_retriever()
        self.retriever = retriever

    # -- tool surface -----------------------------------------------------------

    def tool_specs(self) -> list[dict[str, Any]]:
        q = {"type": "object",
             "properties": {"query": {"type": "string"},
                            "limit": {"type": "integer"},
                            "services": {"type": "array", "items": {"type": "string"}}},
             "required": ["query"]}
        issues = {"type": "object",
                  "properties": {"query": {"type": "string"},
                                 "limit": {"type": "integer"},
                                 "symptoms": {"type": "array", "items": {"type": "string"}}}}
        svc = {"type": "object", "properties": {"service": {"type": "string"},
                                                "type": {"type": "string"}}}
        return [
            {"name": "search_runbooks", "description": "Search operational runbooks.",
             "inputSchema": q},
            {"name": "get_known_issues",
             "description": "Known issues matching symptoms or a query.",
             "inputSchema": issues},
            {"name": "search_postmortems", "description": "Find similar past incidents.",
             "inputSchema": q},
            {"name": "get_knowledge_stats", "description": "Knowledge base statistics.",
             "inputSchema": {"type": "object", "properties": {}}},
            {"name": "list_services", "description": "Services known to the knowledge base.",
             "inputSchema": svc},
        ]

    @staticmethod
    def _filter_by_services(results: list[dict[str, Any]],
                            services: list[str]) -> list[dict[str, Any]]:
        if not services:
            return results
        wanted = {s.lower() for s in services}
        return [r for r in results
                if wanted & {str(s).lower() for s in r.get("services", [])}]

    def call_tool(self, name: str, args: dict[str, Any]) -> Any:
        query = str(args.get("query", ""))
        limit = int(args.get("limit", 5))
        services = [str(s) for s in args.get("services", []) or []]
        if name == "search_runbooks":
            hits = self.retriever.search(query, limit=limit, doc_type="runbook")
            return {"results": self._filter_by_services(hits, services)}
        if name == "get_known_issues":
            symptoms = [str(s) for s in args.get("symptoms", []) or []]
            q = " ".join([query] + symptoms).strip()
            return {"results": self.retriever.search(q, limit=limit, doc_type="known_issue")}
        if name == "search_postmortems":
            hits = self.retriever.search(query, limit=limit, doc_type="postmortem")
            return {"results": self._filter_by_services(hits, services)}
        if name == "get_knowledge_stats":
            return self.retriever.stats()
        if name == "list_services":
            type_filter = str(args.get("type", "")) or None
            svcs: set[str] = set()
            for doc in self.retriever.store.list_documents():
                if type_filter and doc.get("type") != type_filter:
                    continue
                svcs.update(doc.get("services", []))
            return {"services": sorted(svcs)}
        raise ValueError(f"unknown tool '{name}'")

    # -- JSON-RPC ----------------------------------------------------------------

    def handle(self, request: dict[str, Any]) -> Optional[dict[str, Any]]:
        rid = request.get("id")
        method = request.get("method", "")
        params = request.get("params", {}) or {}
        try:
            if method == "initialize":
                result: Any = {
                    "protocolVersion": PROTOCOL_VERSION,
                    "capabilities": {"tools": {}, "resources": {}},
                    "serverInfo": {"name": "runbook-knowledge", "version": "0.1.0"},
                }
            elif method == "notifications/initialized":
                return None
            elif method == "tools/list":
                result = {"tools": self.tool_specs()}
            elif method == "tools/call":
                out = self.call_tool(params.get("name", ""), params.get("arguments", {}) or {})
                result = {"content": [{"type": "text", "text": json.dumps(out, default=str)}]}
            elif method == "resources/list":
                result = {"resources": [
                    {"uri": "runbook://stats", "name": "knowledge stats",
                     "mimeType": "application/json"},
                ]}
            elif method == "resources/read":
                result = {"contents": [{"uri": params.get("uri", ""),
                                        "mimeType": "application/json",
                                        "text": json.dumps(self.retriever.stats())}]}
            elif method == "ping":
                result = {}
            else:
                return {"jsonrpc": "2.0", "id": rid,
                        "error": {"code": -32601, "message": f"method not found: {method}"}}
            return {"jsonrpc": "2.0", "id": rid, "result": result}
        except Exception as e:  # noqa: BLE001
            return {"jsonrpc": "2.0", "id": rid,
                    "error": {"code": -32000, "message": f"{type(e).__name__}: {e}"}}


def run_stdio_server(retriever: Any = None) -> None:
    """stdin JSON-RPC loop (reference runStdioServer @480)."""
    server = MCPServer(retriever)
    for line in sys.stdin:
        line = line.strip()
        if not line:
            continue
        try:
            request = json.loads(line)
        except json.JSONDecodeError:
            continue
        response = server.handle(request)
        if response is not None:
            sys.stdout.write(json.dumps(response, default=str) + "\n")
            sys.stdout.flush()
