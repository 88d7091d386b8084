"""`runbook` CLI.

Parity with reference src/cli.tsx (2464 LoC) command surface: ask @1104,
chat @1119, investigate @1133 (--verbose/--auto-remediate/--learn/
--apply-runbook-updates, structured run @586-989 with event rendering and
free-form fallback @1169-1189), status @1194, init @1208, demo @1240,
knowledge sync/search/add/validate/stats @1250-1553, deploy @1556,
config @1587, integrations claude @1666-1995, webhook @1998,
slack-gateway @2056, mcp serve/tools @2182-2206, operability @2209-2350,
checkpoint list/show/delete @2353-2461. Ink UI is replaced by plain
ANSI terminal rendering. Beyond the reference: eval/eval-all, serve
(OpenAI-compatible endpoint), replay (session audit trails),
checkpoint resume, investigate --report, persistent chat memory.
"""
from __future__ import annotations

import json
import os
import sys
from typing import Any, Optional

import click

from .config.schema import Config, load_config, set_config_value, validate_config
from .model.client import create_llm_client

BOLD = "\033[1m"
DIM = "\033[2m"
RESET = "\033[0m"
GREEN = "\033[32m"
YELLOW = "\033[33m"
RED = "\033[31m"
CYAN = "\033[36m"


def _echo(s: str = "") -> None:
    click.echo(s)


def _build_runtime(config: Config, scenario: Optional[str] = None,
                   provider_override: Optional[str] = None) -> dict[str, Any]:
    """Build retriever + tools + llm (reference createRuntimeAgent cli.tsx:87-109)."""
    from .knowledge.indexer.embedder import create_embedder
    from .knowledge.retriever.default import create_retriever
    from .providers.simulation import load_scenario, set_scenario
    from .skills.registry import get_skill_registry
    from .tools.registry import ToolRegistry, get_runtime_tools

    if scenario:
        set_scenario(load_scenario(scenario))
    retriever = create_retriever(embedder=create_embedder({"backend": "auto"}))
    skills = get_skill_registry()
    skills.load_user_skills()
    llm_cfg = config.llm.model_dump(by_alias=False)
    if provider_override:
        llm_cfg["provider"] = provider_override
    llm = create_llm_client(llm_cfg)
    registry = ToolRegistry(knowledge_retriever=retriever, skill_registry=skills, llm=llm)
    tools = get_runtime_tools(registry, config.providers.model_dump())
    return {"retriever": retriever, "registry": registry, "tools": tools, "llm": llm,
            "skills": skills}


@click.group()
@click.option("--config", "config_path", default=None, help="Path to config.yaml")
@click.pass_context
def cli(ctx: click.Context, config_path: Optional[str]) -> None:
    """Runbook — MI355X-native AI SRE investigation agent."""
    ctx.ensure_object(dict)
    try:
        ctx.obj["config"] = load_config(config_path)
    except ValueError as e:
        raise click.ClickException(str(e)) from e


# -- ask / chat ----------------------------------------------------------------

@cli.command()
@click.argument("query")
@click.option("--provider", default=None, help="LLM provider override (local/mock)")
@click.option("--scenario", default=None, help="Simulated scenario name")
@click.pass_context
def ask(ctx: click.Context, query: str, provider: Optional[str], scenario: Optional[str]) -> None:
    """One-shot free-form investigation of a question."""
    from .agent.agent import Agent
    from .agent.types import AgentConfig, EventType

    rt = _build_runtime(ctx.obj["config"], scenario, provider)
    agent = Agent(llm=rt["llm"], tools=rt["tools"], knowledge_retriever=rt["retriever"],
                  config=AgentConfig(), scratchpad_dir=".runbook/scratchpad")
    for event in agent.run(query):
        _render_agent_event(event)


def _render_agent_event(event: Any) -> None:
    from .agent.types import EventType

    t = event.type
    if t == EventType.THINKING:
        _echo(f"{DIM}💭 {event.data.get('text', '')[:200]}{RESET}")
    elif t == EventType.TOOL_START:
        _echo(f"{CYAN}🔧 {event.data.get('tool')}{RESET} {DIM}{json.dumps(event.data.get('args', {}))[:120]}{RESET}")
    elif t == EventType.TOOL_END:
        _echo(f"   {GREEN}✓{RESET} {event.data.get('summary', '')[:160]} "
              f"{DIM}[{event.data.get('resultId', '')}]{RESET}")
    elif t == EventType.TOOL_ERROR:
        _echo(f"   {RED}✗ {event.data.get('error', '')[:160]}{RESET}")
    elif t == EventType.TOOL_LIMIT:
        _echo(f"   {YELLOW}⚠ {event.data.get('reason', '')[:160]}{RESET}")
    elif t == EventType.KNOWLEDGE_RETRIEVED:
        _echo(f"{DIM}📚 retrieved {event.data.get('count')} knowledge docs{RESET}")
    elif t == EventType.CONTEXT_CLEARED:
        _echo(f"{DIM}🧹 compacted context ({event.data.get('cleared')} results cleared){RESET}")
    elif t == EventType.ANSWER_CHUNK:
        # live token stream: print without newline as the engine samples
        print(event.data.get("text", ""), end="", flush=True)
    elif t == EventType.ANSWER_FINAL:
        text = event.data.get("text", "")
        streamed = event.data.get("streamedLen", 0)
        # when the body already streamed, print only the appendices
        _echo(("\n" if streamed else "\n" + "") + text[streamed:]
              if streamed else "\n" + text)
    elif t == EventType.DONE:
        _echo(f"\n{DIM}done in {event.data.get('iterations')} iterations{RESET}")


@cli.command()
@click.option("--provider", default=None)
@click.pass_context
def chat(ctx: click.Context, provider: Optional[str]) -> None:
    """Interactive chat REPL with conversation memory."""
    from .agent.agent import Agent
    from .agent.conversation_memory import ConversationMemory
    from .agent.types import AgentConfig

    rt = _build_runtime(ctx.obj["config"], None, provider)
    # conversation memory persists across chat sessions (beyond the
    # reference, whose memory is in-process only)
    mem_path = os.path.join(".runbook", "chat_memory.json")
    memory = None
    if os.path.exists(mem_path):
        try:
            with open(mem_path, encoding="utf-8") as f:
                memory = ConversationMemory.from_json(f.read())
            memory.llm = rt["llm"]
            _echo(f"{DIM}restored {memory.stats()['messageCount']} messages, "
                  f"{memory.stats()['investigationCount']} investigations{RESET}")
        except (OSError, ValueError, KeyError):
            memory = None
    if memory is None:
        memory = ConversationMemory(summarize_after_messages=16, llm=rt["llm"])
    agent = Agent(llm=rt["llm"], tools=rt["tools"], knowledge_retriever=rt["retriever"],
                  config=AgentConfig(), scratchpad_dir=".runbook/scratchpad",
                  conversation_memory=memory)
    _echo(f"{BOLD}runbook chat{RESET} — type 'exit' to quit")
    try:
        while True:
            try:
                query = input(f"{BOLD}> {RESET}").strip()
            except (EOFError, KeyboardInterrupt):
                break
            if query.lower() in ("exit", "quit", ""):
                if query:
                    break
                continue
            for event in agent.run(query):
                _render_agent_event(event)
    finally:
        try:
            os.makedirs(".runbook", exist_ok=True)
            with open(mem_path, "w", encoding="utf-8") as f:
                f.write(memory.to_json())
        except OSError:
            pass


# -- investigate ----------------------------------------------------------------

@cli.command()
@click.argument("incident_id")
@click.option("--verbose", is_flag=True)
@click.option("--auto-remediate", is_flag=True)
@click.option("--learn", is_flag=True, help="Run the learning loop afterwards")
@click.option("--apply-runbook-updates", is_flag=True)
@click.option("--provider", default=None)
@click.option("--scenario", default=None)
@click.option("--checkpoint/--no-checkpoint", "do_checkpoint", default=True)
@click.option("--report", default=None, help="write a markdown report to this path")
@click.pass_context
def investigate(ctx: click.Context, incident_id: str, verbose: bool, auto_remediate: bool,
                learn: bool, apply_runbook_updates: bool, provider: Optional[str],
                scenario: Optional[str], do_checkpoint: bool,
                report: Optional[str]) -> None:
    """Structured hypothesis-driven investigation of an incident."""
    from .agent.orchestrator import InvestigationOrchestrator
    from .session.checkpoint import CheckpointStore, checkpoint_from_machine

    config: Config = ctx.obj["config"]
    rt = _build_runtime(config, scenario or "redis-conn-exhaustion", provider)

    def render(e: Any) -> None:
        d = e.data
        if e.type == "phase":
            _echo(f"\n{BOLD}▶ {d['to'].upper()}{RESET}")
        elif e.type == "hypothesis":
            _echo(f"  💡 [{d.get('priority')}] {d.get('statement')}")
        elif e.type == "query":
            mark = GREEN + "✓" + RESET if d.get("ok") else RED + "✗" + RESET
            _echo(f"  {mark} {d.get('tool')} {DIM}({d.get('purpose', '')}){RESET}")
        elif e.type == "evaluated":
            _echo(f"  ⚖ {d.get('action')} (confidence {d.get('confidence'):.2f}) "
                  f"{DIM}{d.get('hypothesis', '')[:80]}{RESET}")
        elif e.type == "conclusion":
            _echo(f"\n{GREEN}✅ Root cause:{RESET} {d.get('rootCause')} "
                  f"{DIM}[{d.get('confidence')}]{RESET}")
        elif e.type == "remediation_plan":
            _echo(f"  🛠 plan: {d.get('summary')} ({d.get('steps')} steps)")
        elif verbose:
            _echo(f"{DIM}  {e.type}: {json.dumps(d, default=str)[:160]}{RESET}")

    try:
        orch = InvestigationOrchestrator(
            llm=rt["llm"], tool_executor=rt["registry"],
            available_tools=set(t.name for t in rt["tools"]),
            knowledge_retriever=rt["retriever"],
            max_iterations=ctx.obj["config"].agent.max_iterations * 2,
            auto_remediate=auto_remediate,
            approval_callback=lambda step: click.confirm(
                f"Approve {step.get('risk')}-risk step: {step.get('description')}?", default=False),
        )
        orch.on(render)
        result = orch.investigate(f"Investigate incident {incident_id}", incident_id=incident_id)
        if do_checkpoint:
            store = CheckpointStore()
            cp = checkpoint_from_machine(orch.machine, label="final")
            store.save(cp)
            _echo(f"{DIM}checkpoint saved: {cp.checkpoint_id}{RESET}")
        _echo("\n" + result.summary)
        _echo(f"\n{DIM}duration: {result.duration_ms} ms · LLM calls: "
              f"{orch.stats['llm_calls']} · tool calls: {orch.stats['tool_calls']}{RESET}")
        if report:
            from .session.report import write_investigation_report

            path = write_investigation_report(report, result, orch)
            _echo(f"{DIM}report written: {path}{RESET}")
        if learn:
            from .learning.loop import run_learning_loop

            _echo(f"\n{BOLD}▶ LEARNING{RESET}")
            artifacts = run_learning_loop(rt["llm"], result.to_dict(),
                                          retriever=rt["retriever"],
                                          apply_updates=apply_runbook_updates)
            _echo(f"  postmortem: {artifacts['postmortemPath']}")
            _echo(f"  suggestions: {len(artifacts['suggestions'])} "
                  f"(applied {len(artifacts['applied'])}, proposed {len(artifacts['proposed'])})")
    except Exception as e:  # noqa: BLE001 — fall back to free-form agent (reference cli.tsx:1169-1189)
        _echo(f"{YELLOW}structured investigation failed ({e}); falling back to free-form agent{RESET}")
        from .agent.agent import Agent
        from .agent.types import AgentConfig

        agent = Agent(llm=rt["llm"], tools=rt["tools"], knowledge_retriever=rt["retriever"],
                      config=AgentConfig(), scratchpad_dir=".runbook/scratchpad")
        for event in agent.run(f"Investigate incident {incident_id}", incident_id=incident_id):
            _render_agent_event(event)


@cli.command()
@click.argument("session_id", required=False)
@click.option("--dir", "directory", default=".runbook/scratchpad",
              help="scratchpad directory")
@click.option("--full", is_flag=True, help="show full tool results")
def replay(session_id: Optional[str], directory: str, full: bool) -> None:
    """Replay a recorded agent session's audit trail (scratchpad JSONL) —
    the investigation's timeline for debugging and handoff. Without a
    SESSION_ID, lists recorded sessions."""
    from .agent.scratchpad import Scratchpad

    if not session_id:
        if not os.path.isdir(directory):
            _echo(f"{DIM}(no sessions recorded under {directory}){RESET}")
            return
        files = sorted(fn for fn in os.listdir(directory) if fn.endswith(".jsonl"))
        for fn in files:
            path = os.path.join(directory, fn)
            n = sum(1 for _ in open(path, encoding="utf-8"))
            _echo(f"  {fn[:-6]}  {DIM}({n} entries){RESET}")
        if not files:
            _echo(f"{DIM}(no sessions recorded under {directory}){RESET}")
        return
    pad = Scratchpad.load(session_id, directory)
    if not pad.entries:
        _echo(f"{RED}no recorded session '{session_id}' in {directory}{RESET}")
        sys.exit(1)
    for e in pad.entries:
        d = e.data
        ts = ""
        if e.timestamp:
            import datetime as _dt

            secs = e.timestamp / 1000.0 if e.timestamp > 1e11 else e.timestamp
            ts = _dt.datetime.fromtimestamp(secs).strftime("%H:%M:%S ")
        if e.kind == "init":
            _echo(f"{BOLD}{ts}▶ session start:{RESET} {d.get('query', '')}")
        elif e.kind == "thinking":
            _echo(f"{DIM}{ts}💭 {str(d.get('text', ''))[:200]}{RESET}")
        elif e.kind == "tool_result":
            _echo(f"{CYAN}{ts}🔧 {d.get('tool')}{RESET} "
                  f"{DIM}{json.dumps(d.get('args', {}), default=str)[:100]}{RESET}")
            _echo(f"   → {str(d.get('summary', ''))[:160]} "
                  f"{DIM}[{d.get('resultId', '')}]{RESET}")
            if full and d.get("fullResult") is not None:
                _echo(f"{DIM}{json.dumps(d.get('fullResult'), default=str)[:800]}{RESET}")
        elif e.kind == "compaction":
            _echo(f"{DIM}{ts}🧹 compaction: cleared {d.get('cleared', '?')}{RESET}")
        elif e.kind == "answer":
            _echo(f"{GREEN}{ts}✅ answer:{RESET} {str(d.get('text', ''))[:400]}")
        else:
            _echo(f"{DIM}{ts}{e.kind}: {json.dumps(d, default=str)[:140]}{RESET}")


# -- demo / status / init --------------------------------------------------------

@cli.command()
@click.option("--fast", is_flag=True, help="3x speed")
def demo(fast: bool) -> None:
    """Scripted Redis-connection-exhaustion investigation (no GPU, no keys)."""
    from .demo.runner import format_step, run_demo

    for step in run_demo(fast=fast):
        _echo(format_step(step))


@cli.command()
@click.pass_context
def status(ctx: click.Context) -> None:
    """Show config, providers, knowledge and GPU status."""
    config: Config = ctx.obj["config"]
    problems = validate_config(config)
    _echo(f"{BOLD}runbook status{RESET}")
    _echo(f"  config: {'valid' if not problems else 'INVALID'}")
    for p in problems:
        _echo(f"    {RED}✗ {p}{RESET}")
    _echo(f"  llm: {config.llm.provider}/{config.llm.model} "
          f"TP={config.llm.tensor_parallel} dtype={config.llm.dtype}")
    try:
        import torch

        if torch.cuda.is_available():
            name = torch.cuda.get_device_name(0)
            count = torch.cuda.device_count()
            free, total = torch.cuda.mem_get_info(0)
            _echo(f"  gpu: {count}× {name} ({free / 2**30:.0f}/{total / 2**30:.0f} GiB free)")
        else:
            _echo("  gpu: none visible (CPU mode)")
    except Exception:  # noqa: BLE001
        _echo("  gpu: torch unavailable")
    try:
        from . import ops

        if ops.extension_loaded():
            _echo(f"  hip extension: {GREEN}loaded{RESET} (gfx950 kernels active)")
        else:
            import torch as _t

            mode = "CPU fp32 references" if not _t.cuda.is_available() \
                else f"{RED}NOT LOADED on a GPU — ops will raise{RESET}"
            _echo(f"  hip extension: {mode}")
    except Exception as e:  # noqa: BLE001
        _echo(f"  hip extension: probe failed ({type(e).__name__})")
    if os.path.isdir(".runbook"):
        from .knowledge.retriever.default import create_retriever

        r = create_retriever(in_memory=False)
        _echo(f"  knowledge: {json.dumps(r.stats().get('byType', {}))}")
    else:
        _echo("  knowledge: not initialized (run `runbook init`)")


@cli.command()
@click.option("--template", type=click.Choice(["ecs-rds", "serverless", "enterprise"]),
              default="ecs-rds")
@click.option("--interactive", is_flag=True, help="Run the setup wizard")
def init(template: str, interactive: bool) -> None:
    """Initialize .runbook/ with a config template (or the wizard)."""
    if interactive:
        from .config.wizard import run_wizard

        run_wizard()
        return
    from .config.onboarding import quick_setup

    paths = quick_setup(template)
    for p in paths:
        _echo(f"  wrote {p}")
    _echo(f"{GREEN}initialized .runbook/ ({template}){RESET}")


# -- knowledge -------------------------------------------------------------------

@cli.group()
def knowledge() -> None:
    """Knowledge base commands."""


@knowledge.command("sync")
@click.option("--since", type=float, default=None)
def knowledge_sync(since: Optional[float]) -> None:
    from .knowledge.indexer.embedder import create_embedder
    from .knowledge.retriever.default import create_retriever

    r = create_retriever(embedder=create_embedder())
    counts = r.sync(since=since)
    _echo(f"synced {counts['documents']} documents / {counts['chunks']} chunks")


@knowledge.command("search")
@click.argument("query")
@click.option("--limit", default=5)
@click.option("--type", "doc_type", default=None)
def knowledge_search(query: str, limit: int, doc_type: Optional[str]) -> None:
    from .knowledge.indexer.embedder import create_embedder
    from .knowledge.retriever.default import create_retriever

    r = create_retriever(embedder=create_embedder())
    for hit in r.search(query, limit=limit, doc_type=doc_type):
        _echo(f"{BOLD}{hit['title']}{RESET} {DIM}({hit['type']}, score {hit['score']:.3f}){RESET}")
        _echo(f"  {hit['content'][:160]}")


@knowledge.command("add")
@click.argument("path")
def knowledge_add(path: str) -> None:
    from .knowledge.retriever.default import create_retriever
    from .knowledge.sources.filesystem import load_markdown

    r = create_retriever()
    doc = load_markdown(path)
    r.store.upsert_document(doc)
    _echo(f"added '{doc.title}' ({len(doc.chunks)} chunks)")


@knowledge.command("validate")
def knowledge_validate() -> None:
    from .knowledge.retriever.default import create_retriever

    r = create_retriever()
    r.ensure_initialized()
    stats = r.stats()
    _echo(f"{GREEN}knowledge base valid{RESET}: {json.dumps(stats)}")


@knowledge.group("auth")
def knowledge_auth() -> None:
    """Knowledge-source authentication."""


@knowledge_auth.command("google")
@click.option("--client-id", envvar="GOOGLE_CLIENT_ID", default="")
@click.option("--client-secret", envvar="GOOGLE_CLIENT_SECRET", default="")
def knowledge_auth_google(client_id: str, client_secret: str) -> None:
    """Run the Google Drive OAuth loopback flow (requires network egress)."""
    from .knowledge.sources.google_auth import TokenStore, run_auth_flow

    store = TokenStore()
    if store.valid():
        _echo(f"{GREEN}already authenticated{RESET} (token at {store.path})")
        return
    if not client_id:
        _echo(f"{RED}set GOOGLE_CLIENT_ID / GOOGLE_CLIENT_SECRET first{RESET}")
        sys.exit(1)
    try:
        run_auth_flow(client_id, client_secret)
        _echo(f"{GREEN}authenticated{RESET}")
    except RuntimeError as e:
        _echo(f"{YELLOW}{e}{RESET}")
        sys.exit(1)


@knowledge.command("stats")
def knowledge_stats() -> None:
    from .knowledge.retriever.default import create_retriever

    r = create_retriever()
    r.ensure_initialized()
    _echo(json.dumps(r.stats(), indent=1))


# -- config ----------------------------------------------------------------------

@cli.command("config")
@click.option("--set", "set_kv", default=None, help="a.b.c=value dotted write")
@click.option("--show", is_flag=True)
@click.pass_context
def config_cmd(ctx: click.Context, set_kv: Optional[str], show: bool) -> None:
    """Show or modify configuration."""
    if set_kv:
        key, _, value = set_kv.partition("=")
        set_config_value(".runbook/config.yaml", key.strip(), value.strip())
        _echo(f"set {key.strip()} = {value.strip()}")
        return
    config: Config = ctx.obj["config"]
    _echo(json.dumps(config.model_dump(), indent=1, default=str))


# -- eval ------------------------------------------------------------------------

@cli.command("eval")
@click.option("--fixtures", default="examples/evals/investigation-fixtures.sample.json")
@click.option("--offline", is_flag=True, help="Score fixture mockResults only")
@click.option("--concurrency", default=1)
@click.option("--provider", default=None)
@click.option("--report", "report_path", default=None)
@click.pass_context
def eval_cmd(ctx: click.Context, fixtures: str, offline: bool, concurrency: int,
             provider: Optional[str], report_path: Optional[str]) -> None:
    """Run the investigation benchmark."""
    from .evals.benchmark import load_fixtures, run_benchmark

    fx = load_fixtures(fixtures)
    config: Config = ctx.obj["config"]
    llm_cfg = config.llm.model_dump(by_alias=False)
    if provider:
        llm_cfg["provider"] = provider

    def llm_factory() -> Any:
        return create_llm_client(llm_cfg)

    report = run_benchmark(fx, llm_factory=None if offline else llm_factory,
                           offline=offline, concurrency=concurrency)
    for case in report["cases"]:
        mark = GREEN + "PASS" + RESET if case["passed"] else RED + "FAIL" + RESET
        _echo(f"[{mark}] {case['id']}: {case['score']['overall']:.2f} "
              f"{DIM}({case['durationMs']} ms){RESET}")
    _echo(f"\npass rate {report['passRate']:.0%} · avg score {report['averageOverallScore']:.2f}"
          f" · wall {report['wallMs']} ms")
    if report_path:
        with open(report_path, "w", encoding="utf-8") as f:
            json.dump(report, f, indent=1)
        _echo(f"{DIM}report → {report_path}{RESET}")
    sys.exit(0 if report["failed"] == 0 else 1)


@cli.command("eval-all")
@click.option("--offline", is_flag=True)
@click.option("--concurrency", default=1)
@click.option("--provider", default=None)
@click.option("--out", "out_path", default=".runbook/evals/summary.json")
@click.pass_context
def eval_all_cmd(ctx: click.Context, offline: bool, concurrency: int,
                 provider: Optional[str], out_path: str) -> None:
    """Run every fixture suite (sample + converted datasets) and aggregate."""
    from .evals.run_all import run_all

    config: Config = ctx.obj["config"]
    llm_cfg = config.llm.model_dump(by_alias=False)
    if provider:
        llm_cfg["provider"] = provider

    summary = run_all(
        llm_factory=None if offline else (lambda: create_llm_client(llm_cfg)),
        offline=offline, concurrency=concurrency, out_path=out_path)
    for name, s in summary["suites"].items():
        if "skipped" in s:
            _echo(f"{DIM}{name}: skipped ({s['skipped']}){RESET}")
        else:
            _echo(f"{name}: {s['passed']}/{s['total']} pass "
                  f"(avg {s['averageOverallScore']:.2f}, {s['wallMs']} ms)")
    _echo(f"\noverall pass rate {summary['overallPassRate']:.0%} → {out_path}")


# -- checkpoint ------------------------------------------------------------------

@cli.group()
def checkpoint() -> None:
    """Investigation checkpoint management."""


@checkpoint.command("list")
@click.argument("investigation_id", required=False)
@click.option("--markdown", is_flag=True, help="render as a markdown table")
def checkpoint_list(investigation_id: Optional[str], markdown: bool) -> None:
    from .session.checkpoint import CheckpointStore, format_checkpoint_list_markdown

    store = CheckpointStore()
    if investigation_id is None:
        for s in store.investigations_summary():
            latest = s["latest"]
            _echo(f"{s['investigationId']}: {s['checkpointCount']} checkpoints "
                  f"(latest {latest['checkpointId']} · {latest['phase']} · "
                  f"{latest['hypothesisCount']} hypotheses)")
        return
    cps = store.list(investigation_id)
    if markdown:
        _echo(format_checkpoint_list_markdown(cps))
        return
    for cp in cps:
        _echo(cp.format())


@checkpoint.command("show")
@click.argument("investigation_id")
@click.argument("checkpoint_id", required=False)
def checkpoint_show(investigation_id: str, checkpoint_id: Optional[str]) -> None:
    from .session.checkpoint import CheckpointStore

    store = CheckpointStore()
    cp = (store.load(investigation_id, checkpoint_id) if checkpoint_id
          else store.load_latest(investigation_id))
    if cp is None:
        _echo(f"{RED}no checkpoint found{RESET}")
        sys.exit(1)
    _echo(cp.format())


@checkpoint.command("resume")
@click.argument("investigation_id")
@click.argument("checkpoint_id", required=False)
@click.option("--query", default="", help="updated incident description (defaults to the triage summary)")
@click.option("--scenario", default=None, help="simulated incident scenario")
@click.pass_context
def checkpoint_resume(ctx: click.Context, investigation_id: str,
                      checkpoint_id: Optional[str], query: str,
                      scenario: Optional[str]) -> None:
    """Rehydrate the saved state machine and continue the investigation
    from its checkpointed phase (beyond the reference, which only stores
    checkpoints)."""
    from .session.checkpoint import CheckpointStore, checkpoint_from_machine

    store = CheckpointStore()
    cp = (store.load(investigation_id, checkpoint_id) if checkpoint_id
          else store.load_latest(investigation_id))
    if cp is None:
        _echo(f"{RED}no checkpoint found for {investigation_id}{RESET}")
        sys.exit(1)
    config = ctx.obj["config"]
    rt = _build_runtime(config, scenario=scenario)
    from .agent.orchestrator import InvestigationOrchestrator

    orch = InvestigationOrchestrator(
        llm=rt["llm"], tool_executor=rt["registry"],
        available_tools=set(t.name for t in rt["tools"]),
        knowledge_retriever=rt["retriever"],
        max_iterations=config.agent.max_iterations * 2,
    )
    _echo(f"resuming {investigation_id} from checkpoint {cp.checkpoint_id} "
          f"(phase: {cp.phase}, {len(cp.hypotheses)} hypotheses)")
    result = orch.resume_from_checkpoint(
        query or f"resume: {'; '.join(cp.symptoms) or investigation_id}", cp)
    _echo(result.to_dict().get("summary", ""))
    new_cp = checkpoint_from_machine(orch.machine, label="resumed-final")
    store.save(new_cp)
    _echo(f"{DIM}checkpoint saved: {new_cp.checkpoint_id}{RESET}")
    if not result.success:
        sys.exit(1)


@checkpoint.command("delete")
@click.argument("investigation_id")
@click.argument("checkpoint_id", required=False)
def checkpoint_delete(investigation_id: str, checkpoint_id: Optional[str]) -> None:
    from .session.checkpoint import CheckpointStore

    removed = CheckpointStore().delete(investigation_id, checkpoint_id)
    _echo(f"deleted {removed} checkpoint(s)")


# -- metrics ---------------------------------------------------------------------

@cli.group()
def metrics() -> None:
    """Prometheus metrics for the serving engine."""


@metrics.command("serve")
@click.option("--port", type=int, default=9464)
@click.option("--addr", default="127.0.0.1")
@click.pass_context
def metrics_serve(ctx: click.Context, port: int, addr: str) -> None:
    """Expose engine counters + KV pool occupancy on /metrics (the
    reference ships no metrics backend — SURVEY §5)."""
    import time as _time

    from .engine.metrics import serve_metrics
    from .model.client import create_llm_client

    config = ctx.obj["config"]
    client = create_llm_client(config.llm.model_dump(by_alias=False))
    engine = getattr(client, "engine", None)
    if engine is None:
        _echo(f"{RED}metrics need the local engine (llm.provider=local){RESET}")
        sys.exit(1)
    serve_metrics(engine, port=port, addr=addr)
    _echo(f"metrics on http://{addr}:{port}/metrics (ctrl-c to stop)")
    try:
        while True:
            _time.sleep(3600)
    except KeyboardInterrupt:
        pass


# -- mcp -------------------------------------------------------------------------

@cli.group()
def mcp() -> None:
    """MCP server (stdio JSON-RPC)."""


@mcp.command("serve")
def mcp_serve() -> None:
    from .mcp.server import run_stdio_server

    run_stdio_server()


@mcp.command("tools")
def mcp_tools() -> None:
    from .mcp.server import MCPServer

    server = MCPServer()
    for t in server.tool_specs():
        _echo(f"{BOLD}{t['name']}{RESET}: {t['description']}")


# -- surfaces: slack gateway / webhook / integrations / operability ---------------

@cli.command("slack-gateway")
@click.option("--port", default=3030)
@click.option("--socket-mode", is_flag=True,
              help="Socket Mode (websocket envelopes) instead of HTTP events")
@click.pass_context
def slack_gateway(ctx: click.Context, port: int, socket_mode: bool) -> None:
    """Run the Slack events gateway (HTTP Events API or Socket Mode)."""
    from .slack.gateway import SlackGateway, SocketModeClient

    rt = _build_runtime(ctx.obj["config"])
    gw = SlackGateway(config=ctx.obj["config"].incident.slack, runtime=rt)
    if socket_mode:
        # framing/ack/dedupe/reconnect protocol is implemented
        # (slack/gateway.py SocketModeClient); the websocket transport
        # needs egress to slack.com, absent in this offline image
        app_token = (ctx.obj["config"].incident.slack or {}).get("appToken", "")
        if not app_token:
            raise click.ClickException(
                "socket mode needs incident.slack.appToken (xapp-...)")
        raise click.ClickException(
            "socket mode requires network egress to slack.com "
            "(apps.connections.open); this environment is offline — "
            "the protocol client is in place, supply a websocket "
            "transport where egress exists")
    _echo(f"slack gateway listening on :{port}")
    gw.serve(port=port)


@cli.command("webhook")
@click.option("--port", default=3031)
def webhook(port: int) -> None:
    """Run the Slack approval-button webhook server."""
    from .webhooks.slack_webhook import ApprovalWebhook

    _echo(f"approval webhook listening on :{port}")
    ApprovalWebhook().serve(port=port)


@cli.group()
def integrations() -> None:
    """Integration management (claude hooks, ...)."""


@integrations.group()
def claude() -> None:
    """Claude Code hook integration."""


@claude.command("enable")
@click.option("--scope", type=click.Choice(["project", "user"]), default="project")
def claude_enable(scope: str) -> None:
    from .integrations.claude_hooks import install_hooks

    result = install_hooks(scope)
    _echo(f"{GREEN}claude hooks installed{RESET} ({result['addedHooks']} added) "
          f"→ {result['settingsPath']}")


@claude.command("status")
def claude_status() -> None:
    from .integrations.claude_hooks import hooks_status

    _echo(json.dumps(hooks_status(), indent=1))


@claude.command("disable")
@click.option("--scope", type=click.Choice(["project", "user"]), default="project")
def claude_disable(scope: str) -> None:
    from .integrations.claude_hooks import uninstall_hooks

    uninstall_hooks(scope)
    _echo("claude hooks removed")


@claude.command("hook")
def claude_hook() -> None:
    """Hook entrypoint: reads a hook event from stdin, writes a response."""
    from .integrations.hook_handlers import handle_stdin

    handle_stdin()


@claude.command("learn")
@click.argument("session_id")
@click.pass_context
def claude_learn(ctx: click.Context, session_id: str) -> None:
    from .integrations.session_store import SessionStore
    from .learning.claude_session_ingestion import ingest_session

    rt = _build_runtime(ctx.obj["config"])
    result = ingest_session(SessionStore(), session_id, rt["llm"], rt["retriever"])
    _echo(json.dumps({k: v for k, v in result.items() if k != "postmortem"}, indent=1,
                     default=str))


@cli.group()
def operability() -> None:
    """Operability-context ingestion (agent change claims)."""


@operability.command("ingest")
@click.argument("phase", type=click.Choice(["start", "checkpoint", "end"]))
@click.option("--summary", default="")
@click.option("--files", default="")
@click.option("--services", default="")
def operability_ingest(phase: str, summary: str, files: str, services: str) -> None:
    from .integrations.operability_ingestion import ingest_claim

    claim = ingest_claim(phase, summary=summary,
                         files=[f for f in files.split(",") if f],
                         services=[s for s in services.split(",") if s])
    _echo(json.dumps(claim, indent=1, default=str))


@operability.command("replay")
def operability_replay() -> None:
    from .integrations.operability_ingestion import replay_spool

    n = replay_spool()
    _echo(f"replayed {n} spooled claims")


@operability.command("status")
def operability_status() -> None:
    from .integrations.operability_ingestion import spool_status

    _echo(json.dumps(spool_status(), indent=1))


@cli.command()
@click.argument("service")
@click.option("--version", default="latest")
@click.pass_context
def deploy(ctx: click.Context, service: str, version: str) -> None:
    """Deploy a service via the deploy-service skill (approval-gated)."""
    rt = _build_runtime(ctx.obj["config"])
    out = rt["registry"].execute("skill", {"action": "execute", "name": "deploy-service",
                                           "params": {"service": service, "version": version}})
    _echo(json.dumps(out, indent=1, default=str))


@cli.command()
@click.option("--model", default="tiny", help="engine model config or checkpoint name")
@click.option("--checkpoint", default=None, help="path to a trained checkpoint dir")
@click.option("--host", default="127.0.0.1")
@click.option("--port", default=8000, type=int)
@click.option("--tp", default=None, type=int, help="tensor-parallel degree")
@click.option("--api-key", default="", envvar="RUNBOOK_API_KEY",
              help="require Authorization: Bearer <key> on /v1 endpoints")
def serve(model: str, checkpoint: str, host: str, port: int, tp: int,
          api_key: str) -> None:
    """OpenAI-compatible serving endpoint over the local engine
    (/v1/chat/completions, /v1/completions, /metrics)."""
    from .engine.server import serve as _serve

    _echo(f"{GREEN}serving{RESET} {checkpoint or model} on http://{host}:{port}/v1")
    _serve(model=model, host=host, port=port, checkpoint=checkpoint, tp=tp,
           api_key=api_key)


def main() -> None:
    cli(obj={})


if __name__ == "__main__":
    main()
