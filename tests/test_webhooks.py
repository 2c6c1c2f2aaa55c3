"""Webhook subscription + signed delivery tests."""

import hashlib
import hmac
import json

import pytest
from fastapi.testclient import TestClient

from agentbom_amd.api.server import create_app
from agentbom_amd.api.webhooks import WebhookRegistry


class Recorder:
    def __init__(self, fail_times: int = 0):
        self.calls = []
        self.fail_times = fail_times

    def __call__(self, url, body, headers):
        self.calls.append({"url": url, "body": body, "headers": headers})
        if self.fail_times > 0:
            self.fail_times -= 1
            return False
        return True


class TestRegistry:
    def test_signed_delivery(self):
        rec = Recorder()
        reg = WebhookRegistry(transport=rec)
        sub = reg.subscribe("https://sink.example/hook", ["scan.completed"],
                            secret="s3cret")
        n = reg.emit("scan.completed", {"job_id": "j1"})
        assert n == 1 and len(rec.calls) == 1
        call = rec.calls[0]
        sig = call["headers"]["X-AgentBom-Signature"].removeprefix("sha256=")
        expect = hmac.new(b"s3cret", call["body"], hashlib.sha256).hexdigest()
        assert hmac.compare_digest(sig, expect)
        assert call["headers"]["X-AgentBom-Event"] == "scan.completed"
        assert call["headers"]["X-AgentBom-Delivery"].startswith("evt-")
        assert json.loads(call["body"])["payload"]["job_id"] == "j1"
        assert sub["webhook_id"]

    def test_event_filtering(self):
        rec = Recorder()
        reg = WebhookRegistry(transport=rec)
        reg.subscribe("https://x/h", ["shield.action"])
        assert reg.emit("scan.completed", {}) == 0
        assert rec.calls == []

    def test_retry_then_success(self):
        rec = Recorder(fail_times=2)
        reg = WebhookRegistry(transport=rec, backoff_s=0.001)
        reg.subscribe("https://x/h", ["scan.completed"])
        assert reg.emit("scan.completed", {}) == 1
        assert len(rec.calls) == 3
        assert reg.deliveries[-1]["attempts"] == 3
        assert not reg.dead_letters

    def test_dead_letter_after_exhausted_retries(self):
        rec = Recorder(fail_times=99)
        reg = WebhookRegistry(transport=rec, max_attempts=3, backoff_s=0.001)
        reg.subscribe("https://x/h", ["scan.completed"])
        assert reg.emit("scan.completed", {"job_id": "j"}) == 0
        assert len(reg.dead_letters) == 1
        assert json.loads(reg.dead_letters[0]["body"])["payload"]["job_id"] == "j"

    def test_unknown_event_kind(self):
        reg = WebhookRegistry(transport=Recorder())
        with pytest.raises(ValueError):
            reg.subscribe("https://x/h", ["nope"])
        with pytest.raises(ValueError):
            reg.emit("nope", {})


class TestWebhookRoutes:
    def test_crud_and_scan_fire(self):
        app = create_app()
        c = TestClient(app)
        rec = Recorder()
        app.state.abom.webhooks.transport = rec
        sub = c.post("/v1/webhooks", json={
            "url": "https://sink/h", "events": ["scan.completed"]}).json()
        assert sub["secret"]  # shown once
        listed = c.get("/v1/webhooks").json()
        assert listed["webhooks"][0]["webhook_id"] == sub["webhook_id"]
        assert "secret" not in listed["webhooks"][0]

        job = c.post("/v1/scan", json={"demo": True}).json()
        import time

        for _ in range(100):
            if c.get(f"/v1/scan/{job['job_id']}").json()["status"] == "done":
                break
            time.sleep(0.05)
        assert any(json.loads(call["body"])["payload"]["job_id"]
                   == job["job_id"] for call in rec.calls)

        assert c.delete(f"/v1/webhooks/{sub['webhook_id']}").status_code == 200
        assert c.delete("/v1/webhooks/nope").status_code == 404

    def test_bad_event_rejected(self):
        c = TestClient(create_app())
        resp = c.post("/v1/webhooks", json={"url": "https://x", "events": ["zz"]})
        assert resp.status_code == 400
