"""Slack-pending-store approval glue: request blocks until the webhook
resolves the pending file (reference Slack approval path)."""
import threading
import time

from runbookai_amd.agent.approval import ApprovalManager, ApprovalPolicy
from runbookai_amd.providers.simulation import SimScenario, get_scenario, set_scenario
from runbookai_amd.webhooks.slack_webhook import ApprovalWebhook, PendingApprovalStore


def test_slack_approval_roundtrip(tmp_path):
    set_scenario(SimScenario.redis_exhaustion())
    try:
        pending = str(tmp_path / "pending")
        mgr = ApprovalManager.with_slack_pending_store(
            policy=ApprovalPolicy(), pending_dir=pending, timeout_s=20.0)
        store = PendingApprovalStore(pending)
        webhook = ApprovalWebhook(store)

        def approve_when_posted():
            deadline = time.time() + 10
            while time.time() < deadline:
                items = store.list_pending()
                if items:
                    webhook.handle_interaction({
                        "user": {"username": "oncall"},
                        "actions": [{"value": f"approve:{items[0]['id']}"}]})
                    return
                time.sleep(0.05)

        t = threading.Thread(target=approve_when_posted)
        t.start()
        rec = mgr.request_approval("update-service", "cart-service", "raise pool size")
        t.join()
        assert rec.approved
        assert rec.approver == "slack"
        # the request was announced on slack
        assert any("Approval needed" in m["text"] for m in get_scenario().slack_messages)
    finally:
        set_scenario(None)


def test_slack_approval_timeout_denies(tmp_path):
    set_scenario(SimScenario.redis_exhaustion())
    try:
        mgr = ApprovalManager.with_slack_pending_store(
            pending_dir=str(tmp_path / "p2"), timeout_s=0.2)
        rec = mgr.request_approval("delete-cluster", "prod-db")
        # nobody resolved it -> wait_for returns None -> fall through to
        # terminal (absent) -> denied by default
        assert not rec.approved
    finally:
        set_scenario(None)


def test_webhook_rejects_unsigned_request(tmp_path):
    """With a signing secret configured, an unsigned/forged POST must not
    resolve an approval (advisor finding: the approval gate was forgeable)."""
    import hashlib
    import hmac as hmac_mod
    import time as time_mod

    store = PendingApprovalStore(str(tmp_path / "pending"))
    approval_id = store.create({"operation": "scale-down", "resource": "prod"})
    webhook = ApprovalWebhook(store, signing_secret="s3cret")

    body = b'{"actions":[{"value":"approve:' + approval_id.encode() + b'"}]}'
    ts = str(int(time_mod.time()))

    # forged: bad signature
    assert not webhook.verify(ts, body, "v0=deadbeef")
    # stale timestamp: valid HMAC but older than the replay window
    old_ts = str(int(time_mod.time()) - 3600)
    stale_sig = "v0=" + hmac_mod.new(
        b"s3cret", f"v0:{old_ts}:".encode() + body, hashlib.sha256).hexdigest()
    assert not webhook.verify(old_ts, body, stale_sig)
    # properly signed: passes
    good_sig = "v0=" + hmac_mod.new(
        b"s3cret", f"v0:{ts}:".encode() + body, hashlib.sha256).hexdigest()
    assert webhook.verify(ts, body, good_sig)
    # and without a secret (local dev) verification is a no-op
    assert ApprovalWebhook(store).verify(ts, body, "")


class TestWebhookEnvConfig:
    """Reference slack-webhook.test.ts:40-76."""

    def test_none_without_secret(self, monkeypatch):
        from runbookai_amd.webhooks.slack_webhook import get_webhook_config_from_env

        monkeypatch.delenv("SLACK_SIGNING_SECRET", raising=False)
        assert get_webhook_config_from_env() is None

    def test_defaults_with_secret(self, monkeypatch):
        from runbookai_amd.webhooks.slack_webhook import get_webhook_config_from_env

        monkeypatch.setenv("SLACK_SIGNING_SECRET", "test-secret")
        monkeypatch.delenv("SLACK_WEBHOOK_PORT", raising=False)
        cfg = get_webhook_config_from_env()
        assert cfg["signingSecret"] == "test-secret" and cfg["port"] == 3000

    def test_custom_port_and_dir(self, monkeypatch):
        from runbookai_amd.webhooks.slack_webhook import get_webhook_config_from_env

        monkeypatch.setenv("SLACK_SIGNING_SECRET", "s")
        monkeypatch.setenv("SLACK_WEBHOOK_PORT", "8080")
        monkeypatch.setenv("RUNBOOK_PENDING_DIR", "/custom/path")
        cfg = get_webhook_config_from_env()
        assert cfg["port"] == 8080 and cfg["pendingDir"] == "/custom/path"

    def test_bad_port_falls_back(self, monkeypatch):
        from runbookai_amd.webhooks.slack_webhook import get_webhook_config_from_env

        monkeypatch.setenv("SLACK_SIGNING_SECRET", "s")
        monkeypatch.setenv("SLACK_WEBHOOK_PORT", "not-a-port")
        assert get_webhook_config_from_env()["port"] == 3000


class TestPendingStoreParity:
    """Reference slack-webhook.test.ts:79-186."""

    def test_empty_for_missing_and_empty_dir(self, tmp_path):
        from runbookai_amd.webhooks.slack_webhook import PendingApprovalStore

        assert PendingApprovalStore(str(tmp_path / "nope")).list_pending() == []
        assert PendingApprovalStore(str(tmp_path)).list_pending() == []

    def test_lists_pending_skips_resolved(self, tmp_path):
        from runbookai_amd.webhooks.slack_webhook import PendingApprovalStore

        store = PendingApprovalStore(str(tmp_path))
        a = store.create({"op": "restart"})
        b = store.create({"op": "scale"})
        store.resolve(b, approved=True, approver="alice")
        pending = store.list_pending()
        assert [p["id"] for p in pending] == [a]

    def test_malformed_json_skipped(self, tmp_path):
        from runbookai_amd.webhooks.slack_webhook import PendingApprovalStore

        store = PendingApprovalStore(str(tmp_path))
        store.create({"op": "x"})
        (tmp_path / "broken.json").write_text("{nope")
        assert len(store.list_pending()) == 1

    def test_cleanup_missing_dir(self, tmp_path):
        from runbookai_amd.webhooks.slack_webhook import PendingApprovalStore

        assert PendingApprovalStore(str(tmp_path / "nope")).cleanup_old(0) == 0

    def test_cleanup_keeps_recent(self, tmp_path):
        from runbookai_amd.webhooks.slack_webhook import PendingApprovalStore

        store = PendingApprovalStore(str(tmp_path))
        store.create({"op": "x"})
        assert store.cleanup_old(max_age_s=3600) == 0
        assert len(store.list_pending()) == 1

    def test_cleanup_age_zero_removes_all(self, tmp_path):
        from runbookai_amd.webhooks.slack_webhook import PendingApprovalStore

        store = PendingApprovalStore(str(tmp_path))
        store.create({"op": "x"})
        store.create({"op": "y"})
        assert store.cleanup_old(max_age_s=0) == 2
        assert store.list_pending() == []

    def test_resolution_record_format(self, tmp_path):
        from runbookai_amd.webhooks.slack_webhook import PendingApprovalStore

        store = PendingApprovalStore(str(tmp_path))
        aid = store.create({"op": "restart"})
        store.resolve(aid, approved=False, approver="bob")
        data = store.get(aid)
        assert data["status"] == "denied" and data["approver"] == "bob"
        assert data["resolvedAt"] > 0


class TestActionParsing:
    """Reference slack-webhook.test.ts:266-292."""

    def test_value_format(self):
        from runbookai_amd.webhooks.slack_webhook import ApprovalWebhook

        assert ApprovalWebhook.parse_action({"value": "approve:abc123"}) == ("approve", "abc123")
        assert ApprovalWebhook.parse_action({"value": "deny:abc123"}) == ("deny", "abc123")

    def test_action_id_format(self):
        from runbookai_amd.webhooks.slack_webhook import ApprovalWebhook

        assert ApprovalWebhook.parse_action({"action_id": "approve_mut_123"}) == ("approve", "mut_123")
        assert ApprovalWebhook.parse_action({"action_id": "reject_mut_123"}) == ("deny", "mut_123")

    def test_unparseable(self):
        from runbookai_amd.webhooks.slack_webhook import ApprovalWebhook

        assert ApprovalWebhook.parse_action({"action_id": "other_thing"}) == ("", "")

    def test_block_actions_payload(self, tmp_path):
        from runbookai_amd.webhooks.slack_webhook import ApprovalWebhook, PendingApprovalStore

        store = PendingApprovalStore(str(tmp_path))
        aid = store.create({"op": "restart"})
        wh = ApprovalWebhook(store=store)
        out = wh.handle_interaction({
            "type": "block_actions",
            "user": {"username": "carol"},
            "actions": [{"action_id": f"approve_{aid}", "value": f"approve:{aid}"}],
        })
        assert out["ok"] and out["status"] == "approved"
        assert store.get(aid)["approver"] == "carol"


class TestSignatures:
    """Reference slack-webhook.test.ts:189-218."""

    def test_signature_format_and_uniqueness(self):
        import hashlib
        import hmac as hmac_mod

        from runbookai_amd.slack.gateway import sign_request

        sig1 = sign_request("secret", "123", b"body-a")
        sig2 = sign_request("secret", "123", b"body-b")
        sig3 = sign_request("other", "123", b"body-a")
        assert sig1.startswith("v0=") and len(sig1) == 3 + 64
        assert sig1 != sig2 and sig1 != sig3


class TestBuildSlackRequest:
    """Reference slack/__tests__/gateway.test.ts:35-113."""

    def test_deploy_request_query(self):
        from runbookai_amd.slack.gateway import build_slack_request, parse_command

        parsed = parse_command("<@Ubot> deploy checkout-api to production")
        req = build_slack_request(parsed, {"channel": "C123", "user": "U1", "ts": "123.4"})
        assert req["command"] == "deploy"
        assert "Deploy checkout-api to production" in req["query"]
        assert req["threadTs"] == "123.4"

    def test_incident_id_captured(self):
        from runbookai_amd.slack.gateway import build_slack_request, parse_command

        parsed = parse_command("<@Ubot> investigate PD-777 redis latency spike")
        req = build_slack_request(parsed, {"ts": "123.4", "thread_ts": "100.2"})
        assert req["incidentId"] == "PD-777"
        assert req["threadTs"] == "100.2"  # reply stays in the thread

    def test_no_incident_id(self):
        from runbookai_amd.slack.gateway import build_slack_request, parse_command

        req = build_slack_request(parse_command("investigate redis is down"), {"ts": "1"})
        assert req["incidentId"] is None

    def test_thread_required_rejection(self):
        from runbookai_amd.slack.gateway import SlackGateway

        gw = SlackGateway(config={"requireThread": True})
        out = gw.handle_event({"channel": "C1", "user": "U1", "ts": "1.0",
                               "text": "<@UBOT> status"})
        assert not out["ok"] and "thread" in out["error"]

    def test_threaded_mention_accepted(self):
        from runbookai_amd.slack.gateway import SlackGateway

        gw = SlackGateway(config={"requireThread": True})
        out = gw.handle_event({"channel": "C1", "user": "U1", "ts": "2.0",
                               "thread_ts": "1.0", "text": "<@UBOT> status"})
        assert out["ok"] and out["request"]["threadTs"] == "1.0"
