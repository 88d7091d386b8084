"""MCP server tests, sized to the reference suite
(src/mcp/__tests__/server.test.ts, 16 cases): tool listing + schemas,
per-tool behavior with filters, error handling, JSON-RPC framing."""
from __future__ import annotations

import json

import pytest

from runbookai_amd.mcp.server import PROTOCOL_VERSION, MCPServer


class FakeStore:
    def __init__(self, docs):
        self.docs = docs

    def list_documents(self):
        return self.docs


class FakeRetriever:
    def __init__(self):
        self.docs = [
            {"title": "Redis runbook", "type": "runbook", "services": ["redis", "checkout-api"],
             "content": "raise pool size"},
            {"title": "DB failover runbook", "type": "runbook", "services": ["orders-db"],
             "content": "promote replica"},
            {"title": "Known: pool exhaustion", "type": "known_issue", "services": ["redis"],
             "content": "pool cap 100"},
            {"title": "PM-42 checkout outage", "type": "postmortem", "services": ["checkout-api"],
             "content": "bad deploy"},
        ]
        self.store = FakeStore(self.docs)
        self.calls = []

    def search(self, query, limit=5, doc_type=None, service=None):
        self.calls.append({"query": query, "doc_type": doc_type, "limit": limit})
        out = [d for d in self.docs if doc_type is None or d["type"] == doc_type]
        q = query.lower()
        out = [d for d in out if not q or any(
            w in (d["title"] + " " + d["content"]).lower() for w in q.split())]
        return out[:limit]

    def stats(self):
        return {"documents": len(self.docs),
                "byType": {"runbook": 2, "known_issue": 1, "postmortem": 1}}


@pytest.fixture()
def srv():
    return MCPServer(retriever=FakeRetriever())


class TestToolSurface:
    def test_lists_five_tools(self, srv):
        names = [t["name"] for t in srv.tool_specs()]
        assert names == ["search_runbooks", "get_known_issues", "search_postmortems",
                         "get_knowledge_stats", "list_services"]

    def test_every_tool_has_schema_and_description(self, srv):
        for t in srv.tool_specs():
            assert t["description"]
            assert t["inputSchema"]["type"] == "object"

    def test_search_runbooks_schema_requires_query(self, srv):
        spec = next(t for t in srv.tool_specs() if t["name"] == "search_runbooks")
        assert "query" in spec["inputSchema"]["required"]
        assert "services" in spec["inputSchema"]["properties"]


class TestTools:
    def test_search_runbooks(self, srv):
        out = srv.call_tool("search_runbooks", {"query": "pool"})
        assert [r["title"] for r in out["results"]] == ["Redis runbook"]

    def test_search_runbooks_service_filter(self, srv):
        out = srv.call_tool("search_runbooks", {"query": "runbook",
                                                "services": ["orders-db"]})
        assert [r["title"] for r in out["results"]] == ["DB failover runbook"]

    def test_empty_results(self, srv):
        assert srv.call_tool("search_runbooks", {"query": "zzznope"})["results"] == []

    def test_known_issues(self, srv):
        out = srv.call_tool("get_known_issues", {"query": "pool"})
        assert out["results"][0]["type"] == "known_issue"

    def test_known_issues_symptom_filter(self, srv):
        out = srv.call_tool("get_known_issues", {"symptoms": ["exhaustion"]})
        assert out["results"] and "exhaustion" in out["results"][0]["title"]

    def test_search_postmortems(self, srv):
        out = srv.call_tool("search_postmortems", {"query": "outage"})
        assert out["results"][0]["title"].startswith("PM-42")

    def test_stats(self, srv):
        out = srv.call_tool("get_knowledge_stats", {})
        assert out["documents"] == 4 and out["byType"]["runbook"] == 2

    def test_list_services(self, srv):
        out = srv.call_tool("list_services", {})
        assert out["services"] == ["checkout-api", "orders-db", "redis"]

    def test_list_services_type_filter(self, srv):
        out = srv.call_tool("list_services", {"type": "postmortem"})
        assert out["services"] == ["checkout-api"]

    def test_unknown_tool_raises(self, srv):
        with pytest.raises(ValueError):
            srv.call_tool("nope", {})


class TestJsonRpc:
    def test_initialize(self, srv):
        resp = srv.handle({"jsonrpc": "2.0", "id": 1, "method": "initialize"})
        assert resp["result"]["protocolVersion"] == PROTOCOL_VERSION
        assert resp["result"]["serverInfo"]["name"]

    def test_tools_list_mcp_format(self, srv):
        resp = srv.handle({"jsonrpc": "2.0", "id": 2, "method": "tools/list"})
        tools = resp["result"]["tools"]
        assert len(tools) == 5 and all("inputSchema" in t for t in tools)

    def test_tools_call_wraps_text_content(self, srv):
        resp = srv.handle({"jsonrpc": "2.0", "id": 3, "method": "tools/call",
                           "params": {"name": "get_knowledge_stats", "arguments": {}}})
        content = resp["result"]["content"]
        assert content[0]["type"] == "text"
        assert json.loads(content[0]["text"])["documents"] == 4

    def test_unknown_tool_is_rpc_error(self, srv):
        resp = srv.handle({"jsonrpc": "2.0", "id": 4, "method": "tools/call",
                           "params": {"name": "nope", "arguments": {}}})
        assert resp["error"]["code"] == -32000

    def test_unknown_method(self, srv):
        resp = srv.handle({"jsonrpc": "2.0", "id": 5, "method": "bogus/method"})
        assert resp["error"]["code"] == -32601

    def test_initialized_notification_silent(self, srv):
        assert srv.handle({"jsonrpc": "2.0", "method": "notifications/initialized"}) is None
