"""Human-in-the-loop approval queue + gateway escalation."""

from __future__ import annotations

from datetime import datetime, timedelta, timezone

import pytest

from agentbom_amd.runtime.hitl import (
    APPROVED,
    CONSUMED,
    DENIED,
    EXPIRED,
    PENDING,
    ApprovalQueue,
    frame_fingerprint,
)

FRAME = {"jsonrpc": "2.0", "id": 1, "method": "tools/call",
         "params": {"name": "delete_repo", "arguments": {"repo": "prod"}}}


class TestQueue:
    def test_park_decide_consume_single_use(self):
        q = ApprovalQueue()
        req = q.park("up", "bot", FRAME, alerts=[{"message": "suspicious"}])
        assert req.status == PENDING and req.summary == "tools/call delete_repo"
        assert q.decide(req.request_id, approve=True, actor="secops",
                        reason="verified with owner").status == APPROVED
        # consuming authorizes exactly one replay of the SAME frame
        assert q.consume(req.request_id, {**FRAME, "id": 99})  # id ignored
        assert q.get(req.request_id).status == CONSUMED
        assert not q.consume(req.request_id, FRAME)  # burnt

    def test_frame_mismatch_refused(self):
        q = ApprovalQueue()
        req = q.park("up", "bot", FRAME)
        q.decide(req.request_id, approve=True, actor="secops")
        other = {**FRAME, "params": {"name": "delete_repo",
                                     "arguments": {"repo": "OTHER"}}}
        assert not q.consume(req.request_id, other)
        assert q.get(req.request_id).status == APPROVED  # not burnt by a miss

    def test_deny_and_double_decide(self):
        q = ApprovalQueue()
        req = q.park("up", "bot", FRAME)
        assert q.decide(req.request_id, approve=False, actor="secops",
                        reason="no").status == DENIED
        assert q.decide(req.request_id, approve=True, actor="x") is None
        assert not q.consume(req.request_id, FRAME)

    def test_expiry_fails_closed(self):
        q = ApprovalQueue(default_ttl_minutes=0.0)
        req = q.park("up", "bot", FRAME)
        assert q.get(req.request_id).status == EXPIRED
        assert q.decide(req.request_id, approve=True, actor="x") is None
        assert not q.consume(req.request_id, FRAME)
        assert q.list(status=EXPIRED)

    def test_fingerprint_stable(self):
        assert frame_fingerprint(FRAME) == frame_fingerprint({**FRAME, "id": 7})
        assert frame_fingerprint(FRAME) != frame_fingerprint(
            {**FRAME, "method": "tools/list"})


class TestGatewayEscalation:
    def _gw(self):
        from agentbom_amd.runtime.gateway import Gateway, Upstream

        q = ApprovalQueue()
        gw = Gateway(hitl_queue=q, hitl_escalate=True)
        gw.register(Upstream(name="up", handler=lambda f: {
            "jsonrpc": "2.0", "id": f.get("id"), "result": {"ok": True}}))
        return gw, q

    def _warn_frame(self):
        # path traversal trips the argument analyzer at warn level (not block)
        return {"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                "params": {"name": "read_file", "arguments": {
                    "path": "../../etc/passwd"}}}

    def test_warn_parks_then_approval_releases(self):
        gw, q = self._gw()
        frame = self._warn_frame()
        out = gw.relay("up", frame, principal="bot")
        assert out["error"]["code"] == -32009
        rid = out["error"]["data"]["approval_request_id"]
        assert q.get(rid).status == PENDING
        # retry without approval parks AGAIN (new request)
        out2 = gw.relay("up", frame, principal="bot")
        assert out2["error"]["code"] == -32009
        # approve, then the SAME frame passes exactly once
        q.decide(rid, approve=True, actor="secops")
        ok = gw.relay("up", frame, principal="bot", approval_id=rid)
        assert ok.get("result") == {"ok": True}
        again = gw.relay("up", frame, principal="bot", approval_id=rid)
        assert again["error"]["code"] == -32009  # single-use grant burnt

    def test_escalation_off_keeps_legacy_warn_passthrough(self):
        from agentbom_amd.runtime.gateway import Gateway, Upstream

        gw = Gateway()  # no queue
        gw.register(Upstream(name="up", handler=lambda f: {
            "jsonrpc": "2.0", "id": f.get("id"), "result": {"ok": True}}))
        out = gw.relay("up", self._warn_frame(), principal="bot")
        assert "error" not in out or out["error"]["code"] != -32009


class TestApi:
    def test_approval_endpoints(self):
        from starlette.testclient import TestClient

        from agentbom_amd.api.server import create_app

        client = TestClient(create_app())
        q = client.app.state.abom.approvals
        req = q.park("up", "bot", FRAME, alerts=[{"message": "sus"}])
        rows = client.get("/v1/approvals?status=pending").json()
        assert rows["total"] == 1
        a = client.post(f"/v1/approvals/{req.request_id}/approve",
                        json={"reason": "checked"}).json()
        assert a["status"] == "approved"
        assert client.post(
            f"/v1/approvals/{req.request_id}/deny").status_code == 409
