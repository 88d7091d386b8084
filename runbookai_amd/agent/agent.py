"""Free-form agent loop.

Parity with reference src/agent/agent.ts (918 LoC): Agent.run(query,
incident_id?) generator yielding AgentEvents; knowledge pre-retrieval
(L320-354); knowledge-only fast path for procedural queries (L356-390,
is_procedural_runbook_query L78-83); iteration loop with context-size
check + compaction (L401-442); LLM call (L488-493); repeated-call
suppression >2x via stable signature (L527-548); cache check (L589-604);
parallel vs sequential tool exec (L624-739); summarize + tiered append
(L742-787); final-answer synthesis + hypothesis-tree/citation appendices
(L799-851). Defaults (L47-57): max_iterations 10, max_hypothesis_depth 4,
context_threshold_tokens 100000, keep_tool_uses 5, tool_limits
{aws_query:10, search_knowledge:5, web_search:3}.
"""
from __future__ import annotations

import re
from typing import Any, Iterator, Optional

from ..utils.stable import call_signature
from ..utils.tokens import estimate_tokens
from .citation_context import CitationContext
from .context_compactor import create_compactor
from .conversation_memory import ConversationMemory
from .hypothesis import HypothesisEngine
from .investigation_memory import InvestigationMemory
from .knowledge_context import KnowledgeContextManager
from .parallel_executor import ParallelToolExecutor, execute_tool_call
from .prompts import (
    build_context_aware_system_prompt,
    build_final_answer_prompt,
    build_iteration_prompt,
    build_knowledge_prompt,
)
from .scratchpad import Scratchpad, set_active_scratchpad
from .tool_cache import ToolCache
from .tool_summarizer import ToolSummarizer
from .types import (
    AgentConfig,
    AgentEvent,
    EventType,
    LLMClient,
    RetrievedKnowledge,
    Tool,
    ToolCall,
    ToolResult,
    new_id,
)

_PROCEDURAL_RE = re.compile(
    r"\b(how (?:do|to|can) (?:i|we)?|what (?:is|are) the (?:steps|procedure|runbook)|"
    r"runbook for|procedure for|playbook for)\b",
    re.IGNORECASE,
)


def is_procedural_runbook_query(query: str) -> bool:
    """Reference agent.ts:78-83."""
    return bool(_PROCEDURAL_RE.search(query))


class Agent:
    def __init__(
        self,
        llm: LLMClient,
        tools: list[Tool],
        knowledge_retriever: Any = None,
        config: Optional[AgentConfig] = None,
        scratchpad_dir: Optional[str] = None,
        conversation_memory: Optional[ConversationMemory] = None,
        service_context: Any = None,
        infra_context: Any = None,
    ) -> None:
        self.llm = llm
        self.tools = tools
        self.tools_by_name = {t.name: t for t in tools}
        self.retriever = knowledge_retriever
        self.config = config or AgentConfig()
        self.scratchpad_dir = scratchpad_dir
        self.conversation_memory = conversation_memory
        self.service_context = service_context
        self.infra_context = infra_context
        self.cache = ToolCache()
        self.summarizer = ToolSummarizer()
        self.compactor = create_compactor("incident")
        self.parallel = ParallelToolExecutor(max_concurrent=5)
        self.hypothesis_engine = HypothesisEngine(max_depth=self.config.max_hypothesis_depth)
        self.last_answer: str = ""

    # -- cache stats (reference agent.ts:894-902) -----------------------------

    def get_cache_stats(self) -> dict[str, int]:
        return self.cache.stats()

    # -- main loop (reference run() L279-851) ---------------------------------

    def run(self, query: str, incident_id: Optional[str] = None) -> Iterator[AgentEvent]:
        session_id = new_id("sess-")
        pad = Scratchpad(session_id, self.scratchpad_dir)
        set_active_scratchpad(pad)
        memory = InvestigationMemory(session_id, self.scratchpad_dir)
        memory.init()
        citations = CitationContext(max_citations=self.config.max_citations)
        kctx = KnowledgeContextManager(self.retriever)
        pad.append("init", query=query, incidentId=incident_id)

        # knowledge pre-retrieval (reference L320-354)
        knowledge = self._retrieve_knowledge(query, incident_id)
        knowledge_md = ""
        if knowledge and not knowledge.is_empty():
            lines = []
            for item in knowledge.all_items()[:8]:
                ref = citations.ref(item.get("title", "?"), doc_type=item.get("type", ""),
                                    path=item.get("path", ""))
                snippet = str(item.get("content", ""))[:400]
                lines.append(f"{ref} **{item.get('title', '?')}**\n{snippet}")
            knowledge_md = "\n\n".join(lines)
            yield AgentEvent(EventType.KNOWLEDGE_RETRIEVED,
                             {"count": len(knowledge.all_items()), "citations": len(citations)})

        # knowledge-only fast path (reference L356-390)
        if knowledge_md and is_procedural_runbook_query(query):
            resp = self.llm.chat(
                "You are Runbook, an SRE assistant. Answer from the provided runbook "
                "knowledge only, citing sources.",
                build_knowledge_prompt(query, knowledge_md),
            )
            answer = resp.content + "\n\n" + citations.format_markdown()
            self.last_answer = answer
            yield AgentEvent(EventType.ANSWER_FINAL, {"text": answer})
            yield AgentEvent(EventType.DONE, {"iterations": 0, "fastPath": True})
            return

        kctx.build_index()
        tool_specs = [t.spec() for t in self.tools]
        call_counts: dict[str, int] = {}

        iteration = 0
        for iteration in range(1, self.config.max_iterations + 1):
            # context-size check + compaction (reference L401-442)
            context = pad.build_tiered_context()
            if estimate_tokens(context) > self.config.context_threshold_tokens:
                plan = self.compactor.compact(
                    pad, query=query,
                    hypotheses=[h.statement for h in self.hypothesis_engine.active()],
                    services=memory.discovered_services,
                )
                cleared = pad.apply_compaction_plan(plan)
                context = pad.build_tiered_context()
                yield AgentEvent(EventType.CONTEXT_CLEARED, {"cleared": cleared})

            system = build_context_aware_system_prompt(
                tools=tool_specs,
                skills=None,
                infra_overview=self.infra_context.prompt_overview() if self.infra_context else "",
                knowledge_section=kctx.prompt_section(),
                service_section=(
                    self.service_context.prompt_section(memory.discovered_services)
                    if self.service_context else ""
                ),
            )
            user = build_iteration_prompt(
                query, iteration, self.config.max_iterations, context,
                memory.build_context_summary(),
            )
            if knowledge_md and iteration == 1:
                user += "\n\n## Retrieved knowledge\n" + knowledge_md
            if self.conversation_memory:
                # related-context recall: past investigations / messages
                # matching THIS query surface alongside the rolling window
                conv = self.conversation_memory.get_context_for_prompt(
                    query=query)
                if conv:
                    user = conv + "\n\n" + user

            resp = self.llm.chat(system, user, tool_specs)  # (reference L488-493)

            if resp.thinking:
                memory.extract_from_thinking(resp.thinking)
                pad.append("thinking", text=resp.thinking[:2000])
                yield AgentEvent(EventType.THINKING, {"text": resp.thinking})
            if not resp.tool_calls:  # (reference L507-509)
                if resp.content:
                    self.last_answer = resp.content
                break

            # validate calls: repeat-suppression + graceful limits (L527-574)
            valid_calls: list[ToolCall] = []
            for call in resp.tool_calls:
                if call.name not in self.tools_by_name:
                    yield AgentEvent(EventType.TOOL_ERROR,
                                     {"tool": call.name, "error": "unknown tool"})
                    continue
                sig = call_signature(call.name, call.arguments)
                call_counts[sig] = call_counts.get(sig, 0) + 1
                if call_counts[sig] > 2:  # suppression >2x (reference L527-548)
                    yield AgentEvent(
                        EventType.TOOL_LIMIT,
                        {"tool": call.name, "reason": "repeated call suppressed"},
                    )
                    continue
                warning = pad.check_tool_limit(call.name, self.config.tool_limits)
                if warning:
                    yield AgentEvent(EventType.TOOL_LIMIT, {"tool": call.name, "reason": warning})
                valid_calls.append(call)
            if not valid_calls:
                continue

            # cache check (reference L589-604)
            uncached: list[ToolCall] = []
            for call in valid_calls:
                hit = self.cache.get(call.name, call.arguments)
                if hit is not None:
                    result = ToolResult(call=call, result=hit, cached=True)
                    yield from self._record_result(pad, memory, kctx, result)
                else:
                    uncached.append(call)

            # parallel vs sequential execution (reference L624-739)
            if uncached:
                if self.config.parallel_tools and len(uncached) > 1:
                    for call in uncached:
                        yield AgentEvent(EventType.TOOL_START,
                                         {"tool": call.name, "args": call.arguments})
                    results = self.parallel.execute_all(self.tools_by_name, uncached)
                else:
                    results = []
                    for call in uncached:
                        yield AgentEvent(EventType.TOOL_START,
                                         {"tool": call.name, "args": call.arguments})
                        results.append(
                            execute_tool_call(self.tools_by_name[call.name], call,
                                              self.parallel.timeout_s)
                        )
                for result in results:
                    if result.ok:
                        self.cache.put(result.call.name, result.call.arguments, result.result)
                    yield from self._record_result(pad, memory, kctx, result)

            # just-in-time knowledge on newly-discovered services/symptoms
            new_services, new_symptoms = memory.drain_new_discoveries()
            if new_services:
                kctx.query_for_new_services(new_services)
            if new_symptoms:
                kctx.query_for_new_symptoms(new_symptoms)

        # final answer synthesis (reference L799-851)
        final_prompt = build_final_answer_prompt(
            query,
            pad.build_tiered_context(),
            memory.build_final_summary(),
            self.hypothesis_engine.to_markdown(),
        )
        final_system = ("You are Runbook, an SRE investigation agent writing "
                        "a final incident answer.")
        if hasattr(self.llm, "chat_stream"):
            # REAL token streaming: ANSWER_CHUNK events as tokens sample
            # (reference's chatStream fakes this by re-chunking a finished
            # response, llm.ts:152-203; this engine streams live)
            chunks: list[str] = []
            for chunk in self.llm.chat_stream(final_system, final_prompt):
                chunks.append(chunk)
                yield AgentEvent(EventType.ANSWER_CHUNK, {"text": chunk})
            answer = "".join(chunks)
        else:
            resp = self.llm.chat(final_system, final_prompt)
            answer = resp.content
        answer = answer or self.last_answer or "Investigation produced no conclusive answer."
        appendices = []
        tree = self.hypothesis_engine.to_markdown()
        if tree:
            appendices.append(tree)
        mem_summary = memory.build_final_summary()
        if mem_summary:
            appendices.append(mem_summary)
        cite_md = citations.format_markdown()
        if cite_md:
            appendices.append(cite_md)
        if appendices:
            answer = answer + "\n\n" + "\n\n".join(appendices)
        self.last_answer = answer
        memory.save()
        pad.append("answer", text=answer[:4000])
        if self.conversation_memory:
            self.conversation_memory.add_message("user", query)
            self.conversation_memory.add_message("assistant", answer[:800])
            self.conversation_memory.add_investigation(query, answer, memory.discovered_services)
        streamed_len = len("".join(chunks)) if hasattr(self.llm, "chat_stream") else 0
        yield AgentEvent(EventType.ANSWER_FINAL, {"text": answer,
                                                  "streamedLen": streamed_len})
        yield AgentEvent(EventType.DONE, {"iterations": iteration,
                                          "cacheStats": self.cache.stats()})
        set_active_scratchpad(None)

    # -- helpers ---------------------------------------------------------------

    def _retrieve_knowledge(self, query: str, incident_id: Optional[str]) -> Optional[RetrievedKnowledge]:
        if self.retriever is None:
            return None
        try:
            retrieve = getattr(self.retriever, "retrieve", None)
            if retrieve is not None:
                return retrieve({"query": query, "incidentId": incident_id})
            hits = self.retriever.search(query, limit=6)
            if isinstance(hits, dict):
                hits = hits.get("results", [])
            rk = RetrievedKnowledge()
            for hit in hits or []:
                t = hit.get("type", "")
                bucket = {
                    "runbook": rk.runbooks, "postmortem": rk.postmortems,
                    "known_issue": rk.known_issues, "architecture": rk.architecture,
                }.get(t, rk.other)
                bucket.append(hit)
            return rk
        except Exception:  # noqa: BLE001
            return None

    def _record_result(
        self,
        pad: Scratchpad,
        memory: InvestigationMemory,
        kctx: KnowledgeContextManager,
        result: ToolResult,
    ) -> Iterator[AgentEvent]:
        compact = self.summarizer.summarize(
            result.call.name, result.call.arguments, result.result, result.error
        )
        rec = pad.append_tool_result(
            result.call.name, result.call.arguments, compact.one_liner(),
            result.result, has_errors=compact.has_errors,
        )
        for svc in compact.services:
            memory.track_service(svc)
        if result.error:
            yield AgentEvent(EventType.TOOL_ERROR,
                             {"tool": result.call.name, "error": result.error,
                              "resultId": rec.result_id})
        else:
            yield AgentEvent(
                EventType.TOOL_END,
                {"tool": result.call.name, "summary": compact.one_liner(),
                 "resultId": rec.result_id, "durationMs": result.duration_ms,
                 "cached": result.cached},
            )
