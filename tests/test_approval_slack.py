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
