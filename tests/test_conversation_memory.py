"""ConversationMemory behavior tests, sized to the reference suite
(src/agent/__tests__/conversation-memory.test.ts, 34 cases): message
accessors, investigation tracking, search/recall, reference resolution,
prompt context, summary, management, serialization, compression."""
from __future__ import annotations

import json

import pytest

from runbookai_amd.agent.conversation_memory import ConversationMemory, Message


@pytest.fixture()
def mem():
    return ConversationMemory()


class TestMessages:
    def test_add_user_message(self, mem):
        m = mem.add_message("user", "hello")
        assert m.role == "user" and mem.get_messages()[0].content == "hello"

    def test_add_assistant_and_system(self, mem):
        mem.add_message("assistant", "hi")
        mem.add_message("system", "ctx")
        assert [m.role for m in mem.get_messages()] == ["assistant", "system"]

    def test_metadata(self, mem):
        m = mem.add_message("user", "x", metadata={"investigationId": "inv-1"})
        assert m.metadata["investigationId"] == "inv-1"

    def test_unique_ids(self, mem):
        ids = {mem.add_message("user", f"m{i}").id for i in range(10)}
        assert len(ids) == 10

    def test_recent_messages(self, mem):
        for i in range(6):
            mem.add_message("user", f"m{i}")
        assert [m.content for m in mem.recent_messages(2)] == ["m4", "m5"]

    def test_last_message(self, mem):
        assert mem.last_message() is None
        mem.add_message("user", "a")
        mem.add_message("assistant", "b")
        assert mem.last_message().content == "b"

    def test_last_user_message(self, mem):
        mem.add_message("user", "question")
        mem.add_message("assistant", "answer")
        assert mem.last_user_message().content == "question"

    def test_messages_since(self, mem):
        first = mem.add_message("user", "a")
        mem.add_message("assistant", "b")
        mem.add_message("user", "c")
        assert [m.content for m in mem.messages_since(first.id)] == ["b", "c"]

    def test_messages_since_unknown_id(self, mem):
        mem.add_message("user", "a")
        assert mem.messages_since("msg-nope") == []

    def test_max_messages_cap(self):
        mem = ConversationMemory(summarize_after_messages=1000, max_messages=5)
        for i in range(10):
            mem.add_message("user", f"m{i}")
        assert len(mem.get_messages()) == 5


class TestInvestigations:
    def test_store_and_get(self, mem):
        mem.add_investigation("API perf issue", "Memory leak in worker", ["api"])
        invs = mem.get_investigations()
        assert invs[0].query == "API perf issue" and invs[0].services == ["api"]

    def test_recent_investigations(self, mem):
        for i in range(7):
            mem.add_investigation(f"q{i}", f"a{i}")
        assert [s.query for s in mem.recent_investigations(2)] == ["q5", "q6"]

    def test_search_by_query_digest_service(self, mem):
        mem.add_investigation("DB slowness", "connection pool exhausted", ["orders-db"])
        mem.add_investigation("CDN errors", "cache misconfig", ["cdn-edge"])
        assert len(mem.search_investigations("slowness")) == 1
        assert len(mem.search_investigations("pool")) == 1
        assert len(mem.search_investigations("orders-db")) == 1
        assert mem.search_investigations("zzz") == []

    def test_cap(self):
        mem = ConversationMemory(max_investigations=3)
        for i in range(5):
            mem.add_investigation(f"q{i}", "a")
        assert [s.query for s in mem.get_investigations()] == ["q2", "q3", "q4"]


class TestSearchAndRecall:
    def test_search_messages_by_content(self, mem):
        mem.add_message("user", "the database connection dropped")
        mem.add_message("assistant", "weather is nice")
        hits = mem.search("database connection issues")
        assert hits and "database" in hits[0].content

    def test_search_case_insensitive(self, mem):
        mem.add_message("user", "Database Connection Reset")
        assert mem.search("database connection reset")

    def test_get_related_context(self, mem):
        mem.add_message("user", "Database is slow today")
        mem.add_investigation("Database slowness", "pool exhausted", ["orders-db"])
        rel = mem.get_related_context("database slow")
        assert rel["messages"] and rel["investigations"]

    def test_service_mention_recall(self, mem):
        mem.add_investigation("checkout failures", "bad deploy", ["payments-api"])
        rel = mem.get_related_context("what is wrong with payments-api?")
        assert rel["investigations"]


class TestReferences:
    def test_reference_from_investigation(self, mem):
        mem.add_investigation("API performance issue", "Memory leak", ["api"])
        ref = mem.get_reference("API")
        assert ref and "Memory leak" in ref

    def test_reference_from_messages(self, mem):
        mem.add_message("assistant", "The database connection was reset at 10:00 AM")
        ref = mem.get_reference("database connection")
        assert ref and "database connection was reset" in ref

    def test_reference_no_match(self, mem):
        mem.add_message("user", "Hello")
        assert mem.get_reference("nonexistent topic") is None


class TestPromptContext:
    def test_context_includes_recent(self, mem):
        mem.add_message("user", "checkout is failing")
        ctx = mem.get_context_for_prompt()
        assert "checkout is failing" in ctx

    def test_context_includes_investigations_on_query(self, mem):
        mem.add_investigation("checkout failures", "bad deploy on payments", ["payments"])
        ctx = mem.get_context_for_prompt(query="checkout failures again")
        assert "bad deploy" in ctx

    def test_token_budget_respected(self, mem):
        for i in range(50):
            mem.add_message("user", f"message number {i} " + "x" * 200)
        ctx = mem.get_context_for_prompt(token_budget=300)
        from runbookai_amd.utils.tokens import estimate_tokens

        assert estimate_tokens(ctx) < 500  # budget + headers slack


class TestSummary:
    def test_summarize_conversation(self, mem):
        mem.add_message("user", "why is checkout slow")
        mem.add_message("assistant", "investigating the payments service")
        s = mem.summarize()
        assert "checkout" in s["conversationSummary"]

    def test_summarize_investigations(self, mem):
        mem.add_investigation("Test query", "Test cause found", [])
        s = mem.summarize()
        assert "Test query" in s["investigationsSummary"]
        assert "Test cause" in s["investigationsSummary"]


class TestManagement:
    def test_clear_messages_keeps_investigations(self, mem):
        mem.add_message("user", "x")
        mem.add_investigation("q", "a")
        mem.clear_messages()
        assert mem.get_messages() == [] and len(mem.get_investigations()) == 1

    def test_clear_all(self, mem):
        mem.add_message("user", "x")
        mem.add_investigation("q", "a")
        mem.clear()
        assert mem.get_messages() == [] and mem.get_investigations() == []

    def test_stats(self, mem):
        mem.add_message("user", "Hello world")
        mem.add_message("assistant", "Hi there")
        mem.add_investigation("Test", "a")
        st = mem.stats()
        assert st["messageCount"] == 2 and st["investigationCount"] == 1
        assert st["estimatedTokens"] > 0


class TestSerialization:
    def test_round_trip(self, mem):
        mem.add_message("user", "Original message")
        mem.add_investigation("Original query", "answer")
        restored = ConversationMemory.from_json(mem.to_json())
        assert restored.get_messages()[0].content == "Original message"
        assert restored.get_investigations()[0].query == "Original query"

    def test_timestamps_and_ids_preserved(self, mem):
        m = mem.add_message("user", "Test")
        restored = ConversationMemory.from_json(mem.to_json())
        assert restored.get_messages()[0].timestamp == m.timestamp
        assert restored.get_messages()[0].id == m.id

    def test_json_is_plain(self, mem):
        mem.add_message("user", "x", metadata={"k": 1})
        parsed = json.loads(mem.to_json())
        assert parsed["messages"][0]["metadata"] == {"k": 1}


class TestCompression:
    def test_compress_at_threshold(self):
        mem = ConversationMemory(summarize_after_messages=5)
        for i in range(12):
            mem.add_message("user", f"Message {i}")
        assert len(mem.get_messages()) < 12
        assert mem.compressed_summary
        assert mem.stats()["compressed"]

    def test_compressed_context_retains_early_content(self):
        mem = ConversationMemory(summarize_after_messages=4)
        mem.add_message("user", "payments-api throwing 503s")
        for i in range(10):
            mem.add_message("assistant", f"step {i}")
        assert "payments-api" in mem.get_context_for_prompt()


class TestCorruptPersistence:
    def test_from_json_tolerates_malformed_entries(self):
        for raw in ('{"messages": "nope"}',
                    '{"messages": [{"content": 1}, "x", {"role": "user"}]}',
                    '{"investigations": [{"query": null}, "y"]}',
                    '{"investigations": "x", "compressedSummary": null}'):
            mem = ConversationMemory.from_json(raw)
            # skipped entries never crash, survivors keep working
            mem.stats()
            mem.get_context_for_prompt()

    def test_partial_message_survives(self):
        mem = ConversationMemory.from_json(
            '{"messages": [{"content": "kept", "role": "assistant"},'
            ' {"nope": 1}]}')
        assert [m.content for m in mem.get_messages()] == ["kept"]
