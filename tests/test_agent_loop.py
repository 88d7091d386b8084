"""Free-form agent loop tests with a scripted tool-calling mock LLM
(parity with reference agent behaviors: knowledge fast path, caching,
repeated-call suppression, parallel execution, tiered context,
compaction, final-answer appendices)."""
import json

import pytest

from runbookai_amd.agent.agent import Agent, is_procedural_runbook_query
from runbookai_amd.agent.types import AgentConfig, EventType, Tool
from runbookai_amd.model.client import MockLLMClient
from runbookai_amd.providers.simulation import SimScenario, set_scenario


@pytest.fixture(autouse=True)
def scenario():
    set_scenario(SimScenario.redis_exhaustion())
    yield
    set_scenario(None)


def make_tools(call_log):
    def alarms(state="", **_):
        call_log.append("cloudwatch_alarms")
        return {"alarms": [{"name": "redis-conns", "state": "ALARM",
                            "reason": "clients > 950", "service": "redis"}]}

    def logs(filter="", **_):
        call_log.append("cloudwatch_logs")
        return {"events": [{"message": "redis: connection pool exhausted",
                            "service": "checkout-api"}]}

    return [
        Tool(name="cloudwatch_alarms", description="alarms", parameters={}, execute=alarms),
        Tool(name="cloudwatch_logs", description="logs", parameters={}, execute=logs),
    ]


class FakeRetriever:
    def __init__(self):
        self.queries = []

    def retrieve(self, ctx):
        from runbookai_amd.agent.types import RetrievedKnowledge

        self.queries.append(ctx["query"])
        rk = RetrievedKnowledge()
        rk.runbooks.append({"title": "Redis runbook", "type": "runbook",
                            "content": "raise pool size to 500", "path": "rb.md"})
        return rk

    def search(self, q, limit=5, **kw):
        return [{"title": "Redis runbook", "type": "runbook", "content": "raise pool"}]

    def stats(self):
        return {"documents": 1}


def tool_call_response(*calls):
    return json.dumps({"thinking": "I found that redis pool exhausted in checkout-api.",
                       "toolCalls": [{"name": n, "arguments": a} for n, a in calls]})


class TestFreeFormLoop:
    def test_tools_then_final_answer_with_appendices(self):
        calls = []
        llm = MockLLMClient([
            tool_call_response(("cloudwatch_alarms", {"state": "ALARM"}),
                               ("cloudwatch_logs", {"filter": "redis"})),
            "The root cause is redis connection pool exhaustion. evidence: logs.",
            "Final: redis connection pool exhaustion caused the latency spike.",
        ])
        agent = Agent(llm=llm, tools=make_tools(calls), knowledge_retriever=FakeRetriever(),
                      config=AgentConfig(max_iterations=4))
        events = list(agent.run("why is checkout-api slow?"))
        types = [e.type for e in events]
        assert EventType.KNOWLEDGE_RETRIEVED in types
        assert types.count(EventType.TOOL_END) == 2
        final = next(e for e in events if e.type == EventType.ANSWER_FINAL)
        assert "redis" in final.data["text"].lower()
        assert "Sources" in final.data["text"]  # citation appendix
        assert set(calls) == {"cloudwatch_alarms", "cloudwatch_logs"}

    def test_cache_hits_on_repeat_call(self):
        calls = []
        resp = tool_call_response(("cloudwatch_alarms", {"state": "ALARM"}))
        llm = MockLLMClient([resp, resp, "done", "final answer"])
        agent = Agent(llm=llm, tools=make_tools(calls), config=AgentConfig(max_iterations=4))
        events = list(agent.run("alarms?"))
        # second identical call served from cache -> executor ran once
        assert calls.count("cloudwatch_alarms") == 1
        cached = [e for e in events if e.type == EventType.TOOL_END and e.data.get("cached")]
        assert len(cached) == 1

    def test_third_identical_call_suppressed(self):
        calls = []
        resp = tool_call_response(("cloudwatch_logs", {"filter": "x"}))
        llm = MockLLMClient([resp, resp, resp, "stop", "final"])
        agent = Agent(llm=llm, tools=make_tools(calls), config=AgentConfig(max_iterations=5))
        events = list(agent.run("logs?"))
        limits = [e for e in events if e.type == EventType.TOOL_LIMIT
                  and "suppressed" in e.data.get("reason", "")]
        assert limits, "3rd identical call must be suppressed"

    def test_knowledge_fast_path(self):
        assert is_procedural_runbook_query("how do I rotate redis credentials?")
        assert not is_procedural_runbook_query("checkout-api is slow")
        llm = MockLLMClient(["Follow the runbook: raise pool size. [1]"])
        agent = Agent(llm=llm, tools=[], knowledge_retriever=FakeRetriever())
        events = list(agent.run("what is the runbook for redis exhaustion?"))
        done = next(e for e in events if e.type == EventType.DONE)
        assert done.data.get("fastPath") is True
        assert len(llm.calls) == 1  # single LLM call, no tool loop

    def test_unknown_tool_surfaces_error(self):
        llm = MockLLMClient([tool_call_response(("bogus_tool", {})), "final"])
        agent = Agent(llm=llm, tools=make_tools([]), config=AgentConfig(max_iterations=2))
        events = list(agent.run("q"))
        errs = [e for e in events if e.type == EventType.TOOL_ERROR]
        assert any("unknown tool" in e.data.get("error", "") for e in errs)

    def test_compaction_triggers_on_context_budget(self):
        calls = []
        # unique args each round so nothing is cached/suppressed
        responses = [tool_call_response(("cloudwatch_logs", {"filter": f"f{i}"}))
                     for i in range(4)] + ["final"]
        llm = MockLLMClient(responses)

        def big_logs(filter="", **_):
            calls.append(filter)
            return {"events": [{"message": "x" * 400}] * 40}

        tools = [Tool(name="cloudwatch_logs", description="", parameters={},
                      execute=big_logs)]
        agent = Agent(llm=llm, tools=tools,
                      config=AgentConfig(max_iterations=5, context_threshold_tokens=1500))
        events = list(agent.run("dig through logs"))
        assert any(e.type == EventType.CONTEXT_CLEARED for e in events)

    def test_conversation_memory_records_turn(self):
        from runbookai_amd.agent.conversation_memory import ConversationMemory

        mem = ConversationMemory()
        llm = MockLLMClient(["no tools needed", "final answer text"])
        agent = Agent(llm=llm, tools=[], conversation_memory=mem)
        list(agent.run("first question"))
        assert len(mem.messages) == 2
        assert mem.investigations[0].query == "first question"


def test_final_answer_streams_chunks(make_agent=None):
    """ANSWER_CHUNK events precede ANSWER_FINAL when the client supports
    chat_stream; the final text equals the streamed body (+ appendices)."""
    from runbookai_amd.agent.agent import Agent
    from runbookai_amd.agent.types import AgentConfig, ChatResponse, EventType

    class StreamingMock:
        def chat(self, system, user, tools=None):
            return ChatResponse(content="no tools needed")

        def chat_stream(self, system, user, tools=None):
            yield "Root cause: "
            yield "redis pool exhausted."

    agent = Agent(llm=StreamingMock(), tools=[], knowledge_retriever=None,
                  config=AgentConfig(max_iterations=1))
    events = list(agent.run("why is checkout slow?"))
    chunks = [e for e in events if e.type == EventType.ANSWER_CHUNK]
    final = [e for e in events if e.type == EventType.ANSWER_FINAL][0]
    assert [c.data["text"] for c in chunks] == ["Root cause: ",
                                                "redis pool exhausted."]
    body = "".join(c.data["text"] for c in chunks)
    assert final.data["text"].startswith(body)
    assert final.data["streamedLen"] == len(body)


class TestCitations:
    """Reference agent/__tests__/agent-citations.test.ts (3 cases)."""

    def _dup_retriever(self):
        class R(FakeRetriever):
            def retrieve(self, ctx):
                from runbookai_amd.agent.types import RetrievedKnowledge

                rk = RetrievedKnowledge()
                # same doc retrieved twice: must cite once
                for _ in range(2):
                    rk.runbooks.append({"title": "Redis runbook", "type": "runbook",
                                        "content": "raise pool size", "path": "rb.md"})
                return rk
        return R()

    def test_knowledge_fast_path_dedupes_references(self):
        llm = MockLLMClient()
        llm.on(r"runbook knowledge only", "Follow the Redis runbook steps [1].")
        agent = Agent(llm=llm, tools=[], knowledge_retriever=self._dup_retriever(),
                      config=AgentConfig(max_iterations=2))
        final = [e for e in agent.run("how do I restart redis safely?")
                 if e.type == EventType.ANSWER_FINAL][-1]
        text = final.data["text"]
        assert "## Sources" in text
        assert text.count("Redis runbook") >= 1
        # deduplicated: exactly one [1] source row, no [2]
        assert "[2]" not in text.split("## Sources")[1]

    def test_final_synthesis_includes_references(self):
        log = []
        llm = MockLLMClient()
        llm.on(r"toolCalls", tool_call_response(("cloudwatch_alarms", {})))
        llm.add(tool_call_response(("cloudwatch_alarms", {})))
        llm.add(json.dumps({"content": "Redis pool exhausted."}))
        agent = Agent(llm=llm, tools=make_tools(log), knowledge_retriever=FakeRetriever(),
                      config=AgentConfig(max_iterations=2))
        final = [e for e in agent.run("why is checkout slow?")
                 if e.type == EventType.ANSWER_FINAL][-1]
        assert "## Sources" in final.data["text"]

    def test_no_references_without_knowledge(self):
        log = []
        llm = MockLLMClient()
        llm.add(json.dumps({"content": "All healthy."}))
        agent = Agent(llm=llm, tools=make_tools(log), knowledge_retriever=None,
                      config=AgentConfig(max_iterations=2))
        final = [e for e in agent.run("anything wrong?")
                 if e.type == EventType.ANSWER_FINAL][-1]
        assert "## Sources" not in final.data["text"]
