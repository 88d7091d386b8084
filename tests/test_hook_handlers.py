"""Claude Code hook handler tests, sized to the reference suite
(src/integrations/__tests__/hook-handlers.test.ts, 30 cases): session start,
prompt-submit extraction + injection, pre-tool-use safety, post/stop
tracking, dispatch routing, stdin protocol."""
from __future__ import annotations

import io
import json

import pytest

from runbookai_amd.integrations.hook_handlers import (
    _PROMPT_COUNTS,
    dispatch,
    extract_services,
    extract_symptoms,
    handle_pre_tool_use,
    handle_session_start,
    handle_stop,
    handle_user_prompt_submit,
    prompt_count,
)
from runbookai_amd.integrations.session_store import SessionStore


class FakeRetriever:
    def __init__(self, hits=None, stats=None):
        self._hits = hits or []
        self._stats = stats or {}
        self.queries = []

    def search(self, query, limit=3):
        self.queries.append(query)
        return self._hits[:limit]

    def stats(self):
        return self._stats


@pytest.fixture(autouse=True)
def _reset_counts():
    _PROMPT_COUNTS.clear()


class TestSessionStart:
    def test_creates_session_state(self, tmp_path):
        store = SessionStore(directory=str(tmp_path))
        out = handle_session_start({"session_id": "s1"}, store=store)
        assert out["continue"]
        events = store.get_session_events("s1")
        assert events and events[0]["kind"] == "session_start"

    def test_includes_knowledge_stats(self):
        r = FakeRetriever(stats={"documents": 12, "byType": {"runbook": 7}})
        out = handle_session_start({"session_id": "s1"}, retriever=r)
        assert "12 docs" in out["systemMessage"]

    def test_no_stats_no_banner(self):
        out = handle_session_start({"session_id": "s1"})
        assert out == {"continue": True}


class TestPromptSubmit:
    HITS = [{"title": "Redis runbook", "content": "raise pool size"}]

    def test_extracts_services_into_query(self):
        r = FakeRetriever(hits=self.HITS)
        handle_user_prompt_submit({"prompt": "why is checkout-api slow"}, retriever=r)
        assert r.queries and "checkout-api" in r.queries[0]

    def test_extracts_symptoms_into_query(self):
        r = FakeRetriever(hits=self.HITS)
        handle_user_prompt_submit({"prompt": "seeing timeout and oom issues"}, retriever=r)
        assert "timeout" in r.queries[0] and "oom" in r.queries[0]

    def test_injects_system_message(self):
        r = FakeRetriever(hits=self.HITS)
        out = handle_user_prompt_submit({"prompt": "checkout-api timeout"}, retriever=r)
        assert "Redis runbook" in out["systemMessage"]

    def test_empty_prompt_noop(self):
        r = FakeRetriever(hits=self.HITS)
        out = handle_user_prompt_submit({"prompt": "   "}, retriever=r)
        assert out == {"continue": True} and r.queries == []

    def test_prompt_count_increments(self):
        for i in range(3):
            handle_user_prompt_submit({"prompt": "hello", "session_id": "s9"})
        assert prompt_count("s9") == 3

    def test_injection_disabled(self):
        r = FakeRetriever(hits=self.HITS)
        out = handle_user_prompt_submit({"prompt": "checkout-api timeout"},
                                        retriever=r, inject_context=False)
        assert "systemMessage" not in out and r.queries == []

    def test_no_signal_no_search(self):
        r = FakeRetriever(hits=self.HITS)
        out = handle_user_prompt_submit({"prompt": "hello there"}, retriever=r)
        assert "systemMessage" not in out


class TestExtractors:
    def test_service_patterns(self):
        assert "api-gateway" in extract_services("api-gateway is failing")
        assert "user-service" in extract_services("logs say service=user-service down")
        assert extract_services("no services here") == []

    def test_short_names_skipped(self):
        assert extract_services("a-b failed") == []

    def test_http_error_codes(self):
        out = extract_symptoms("seeing 503 and 404 responses")
        assert "http-5xx" in out and "http-4xx" in out

    def test_performance_phrases(self):
        assert "performance" in extract_symptoms("the API is very slow today")
        assert "performance" in extract_symptoms("high latency on p99")
        assert "performance" not in extract_symptoms("all good")

    def test_symptom_words(self):
        out = extract_symptoms("OOM crash with deadlock")
        assert {"oom", "crash", "deadlock"} <= set(out)


class TestPreToolUse:
    def test_safe_command_allowed(self):
        out = handle_pre_tool_use({"tool_input": {"command": "ls -la /tmp"}})
        assert out["continue"]

    def test_dangerous_rm_blocked(self):
        out = handle_pre_tool_use({"tool_input": {"command": "rm -rf /"}})
        assert not out["continue"] and "blocked" in out["stopReason"]

    def test_kubectl_delete_blocked(self):
        out = handle_pre_tool_use(
            {"tool_input": {"command": "kubectl delete deployment api"}})
        assert not out["continue"]

    def test_kubectl_get_allowed(self):
        out = handle_pre_tool_use({"tool_input": {"command": "kubectl get pods -A"}})
        assert out["continue"]

    def test_drop_table_blocked(self):
        out = handle_pre_tool_use({"tool_input": {"command": 'psql -c "DROP TABLE users"'}})
        assert not out["continue"]

    def test_missing_command_allowed(self):
        assert handle_pre_tool_use({})["continue"]


class TestStopAndPost:
    def test_stop_recorded(self, tmp_path):
        store = SessionStore(directory=str(tmp_path))
        out = handle_stop({"session_id": "s1"}, store=store)
        assert out["continue"]
        assert store.get_session_events("s1")[0]["kind"] == "stop"

    def test_post_tool_use_noop_without_store(self):
        assert dispatch({"hook_event_name": "PostToolUse"})["continue"]


class TestDispatch:
    def test_routes_by_event_name(self, tmp_path):
        store = SessionStore(directory=str(tmp_path))
        dispatch({"hook_event_name": "SessionStart", "session_id": "d1"}, store=store)
        dispatch({"hook_event_name": "Stop", "session_id": "d1"}, store=store)
        kinds = [e["kind"] for e in store.get_session_events("d1")]
        assert kinds == ["session_start", "stop"]

    def test_pre_tool_use_routed(self):
        out = dispatch({"hook_event_name": "PreToolUse",
                        "tool_input": {"command": "rm -rf /"}})
        assert not out["continue"]

    def test_unknown_event_graceful(self):
        assert dispatch({"hook_event_name": "SomethingNew"}) == {"continue": True}

    def test_missing_event_graceful(self):
        assert dispatch({}) == {"continue": True}


class TestStdinProtocol:
    def _run(self, monkeypatch, raw):
        from runbookai_amd.integrations import hook_handlers as hh

        monkeypatch.setattr("sys.stdin", io.StringIO(raw))
        buf = io.StringIO()
        monkeypatch.setattr("sys.stdout", buf)
        hh.handle_stdin()
        return buf.getvalue()

    def test_parses_json_and_responds(self, monkeypatch):
        out = self._run(monkeypatch, json.dumps(
            {"hook_event_name": "PreToolUse", "tool_input": {"command": "ls"}}))
        assert json.loads(out)["continue"] is True

    def test_blocked_over_stdin(self, monkeypatch):
        out = self._run(monkeypatch, json.dumps(
            {"hook_event_name": "PreToolUse", "tool_input": {"command": "rm -rf /"}}))
        assert json.loads(out)["continue"] is False

    def test_empty_input(self, monkeypatch):
        assert json.loads(self._run(monkeypatch, ""))["continue"] is True

    def test_invalid_json(self, monkeypatch):
        assert json.loads(self._run(monkeypatch, "{nope"))["continue"] is True


class TestClaudeHookInstall:
    """Reference integrations/__tests__/claude-hooks.test.ts:51-184."""

    def test_installs_seven_events_with_matchers(self, tmp_path):
        from runbookai_amd.integrations.claude_hooks import (
            HOOK_COMMAND,
            install_hooks,
        )

        result = install_hooks("project", cwd=str(tmp_path))
        assert result["addedHooks"] == 7
        assert result["eventsUpdated"] == [
            "SessionStart", "UserPromptSubmit", "PreToolUse", "PostToolUse",
            "Stop", "SubagentStop", "PreCompact"]
        settings = json.load(open(result["settingsPath"]))
        pre = settings["hooks"]["PreToolUse"][0]
        assert pre["matcher"] == ".*"
        assert pre["hooks"][0]["command"] == HOOK_COMMAND
        assert settings["hooks"]["SessionStart"][0]["matcher"] == ""

    def test_idempotent(self, tmp_path):
        from runbookai_amd.integrations.claude_hooks import install_hooks

        first = install_hooks("project", cwd=str(tmp_path))
        second = install_hooks("project", cwd=str(tmp_path))
        assert first["addedHooks"] == 7 and second["addedHooks"] == 0
        settings = json.load(open(first["settingsPath"]))
        total = sum(len(e["hooks"]) for entries in settings["hooks"].values()
                    for e in entries)
        assert total == 7

    def test_uninstall_preserves_user_hooks(self, tmp_path):
        import os

        from runbookai_amd.integrations.claude_hooks import (
            install_hooks,
            uninstall_hooks,
        )

        claude_dir = tmp_path / ".claude"
        claude_dir.mkdir()
        user_hook = {"matcher": "", "hooks": [{"type": "command", "command": "my-own-hook"}]}
        (claude_dir / "settings.json").write_text(json.dumps(
            {"hooks": {"Stop": [user_hook]}}))
        install_hooks("project", cwd=str(tmp_path))
        removed = uninstall_hooks("project", cwd=str(tmp_path))
        assert removed == 7
        settings = json.load(open(claude_dir / "settings.json"))
        assert settings["hooks"]["Stop"] == [user_hook]
        assert "PreToolUse" not in settings["hooks"]

    def test_status_reports_events(self, tmp_path):
        from runbookai_amd.integrations.claude_hooks import hooks_status, install_hooks

        install_hooks("project", cwd=str(tmp_path))
        status = hooks_status(cwd=str(tmp_path))
        assert status["project"]["enabled"]
        assert len(status["project"]["installedEvents"]) == 7


class TestSessionStoreBackends:
    """Reference integrations/__tests__/claude-session-store.test.ts (3 cases)."""

    def test_local_persist_and_retrieve_ndjson(self, tmp_path):
        from runbookai_amd.integrations.session_store import create_session_store

        store = create_session_store({"backend": "local", "directory": str(tmp_path)})
        store.append_event("s1", {"kind": "tool_use", "tool": "Bash"})
        store.append_event("s1", {"kind": "stop"})
        events = store.get_session_events("s1")
        assert [e["kind"] for e in events] == ["tool_use", "stop"]
        # NDJSON on disk: one JSON object per line
        lines = (tmp_path / "s1.jsonl").read_text().strip().splitlines()
        assert len(lines) == 2
        assert all(json.loads(l)["at"] > 0 for l in lines)

    def test_s3_requires_bucket(self, tmp_path):
        from runbookai_amd.integrations.session_store import create_session_store

        with pytest.raises(ValueError):
            create_session_store({"backend": "s3", "directory": str(tmp_path)})

    def test_s3_mirrors_locally_and_queues_uploads(self, tmp_path):
        from runbookai_amd.integrations.session_store import create_session_store

        uploads = []
        store = create_session_store({
            "backend": "s3", "bucket": "ops-sessions",
            "directory": str(tmp_path),
            "uploader": lambda b, k, payload: uploads.append((b, k, payload))})
        store.append_event("s2", {"kind": "tool_use"})
        # local mirror is the source of truth
        assert store.get_session_events("s2")
        assert uploads and uploads[0][0] == "ops-sessions"
        assert uploads[0][1].endswith("s2.jsonl")

    def test_s3_without_uploader_queues(self, tmp_path):
        from runbookai_amd.integrations.session_store import create_session_store

        store = create_session_store({"backend": "s3", "bucket": "b",
                                      "directory": str(tmp_path)})
        store.append_event("s3x", {"kind": "stop"})
        assert store.pending_uploads
        assert store.pending_uploads[0]["key"] == "claude-sessions/s3x.jsonl"


class TestLifecycleEvents:
    def test_subagent_stop_and_precompact_recorded(self, tmp_path):
        store = SessionStore(directory=str(tmp_path))
        for ev in ("SubagentStop", "PreCompact"):
            out = dispatch({"hook_event_name": ev, "session_id": "lc"}, store=store)
            assert out["continue"]
        kinds = [e["kind"] for e in store.get_session_events("lc")]
        assert kinds == ["subagentstop", "precompact"]
