"""REST identity-lifecycle route tests (/v1/identities)."""

import pytest
from fastapi.testclient import TestClient

from agentbom_amd.api.server import create_app

ADMIN_HEADERS = {
    "X-Operator-Role": "admin",
    "X-Operator-Scopes": "identity:write",
    "X-Audit-Reason": "integration test issue",
}


@pytest.fixture()
def client():
    return TestClient(create_app())


def _issue(client, name="bot", **kw):
    resp = client.post("/v1/identities",
                       json={"agent_name": name, **kw}, headers=ADMIN_HEADERS)
    assert resp.status_code == 201, resp.text
    return resp.json()


class TestIdentityRoutes:
    def test_issue_list_get(self, client):
        out = _issue(client, scopes=["scan:read"])
        assert out["token"].startswith("abi_")
        iid = out["identity"]["identity_id"]
        listed = client.get("/v1/identities").json()
        assert listed["total"] == 1
        one = client.get(f"/v1/identities/{iid}").json()
        assert one["identity"]["agent_name"] == "bot"
        assert one["active_scopes"] == ["scan:read"]

    def test_writes_require_gate(self, client):
        resp = client.post("/v1/identities", json={"agent_name": "x"})
        assert resp.status_code == 403
        assert resp.json()["detail"]["status"] == "blocked"
        # wrong scope
        resp = client.post("/v1/identities", json={"agent_name": "x"},
                           headers={**ADMIN_HEADERS,
                                    "X-Operator-Scopes": "scan:read"})
        assert resp.status_code == 403

    def test_verify_route(self, client):
        out = _issue(client, allowed_tools=["read_file"])
        ok = client.post("/v1/identities/verify",
                         json={"token": out["token"], "tool": "read_file"}).json()
        assert ok["valid"]
        bad = client.post("/v1/identities/verify",
                          json={"token": out["token"], "tool": "exec"}).json()
        assert not bad["valid"]

    def test_rotate_and_revoke(self, client):
        out = _issue(client)
        iid = out["identity"]["identity_id"]
        rotated = client.post(f"/v1/identities/{iid}/rotate",
                              headers=ADMIN_HEADERS)
        assert rotated.status_code == 200
        assert rotated.json()["identity"]["rotated_from"] == iid
        resp = client.delete(f"/v1/identities/{iid}", headers=ADMIN_HEADERS)
        assert resp.status_code == 200
        # rotate a revoked identity -> 404
        resp = client.post(f"/v1/identities/{iid}/rotate", headers=ADMIN_HEADERS)
        assert resp.status_code == 404

    def test_jit_routes(self, client):
        iid = _issue(client)["identity"]["identity_id"]
        g = client.post(f"/v1/identities/{iid}/jit",
                        json={"scopes": ["shield:write"], "ttl_minutes": 5},
                        headers=ADMIN_HEADERS)
        assert g.status_code == 201
        grant_id = g.json()["grant"]["grant_id"]
        one = client.get(f"/v1/identities/{iid}").json()
        assert "shield:write" in one["active_scopes"]
        resp = client.delete(f"/v1/identities/jit/{grant_id}",
                             headers=ADMIN_HEADERS)
        assert resp.status_code == 200

    def test_access_review_and_audit(self, client):
        _issue(client, scopes=["*"])
        review = client.get("/v1/identities/reviews/access").json()
        assert review["live_identities"] == 1
        assert review["wildcard_or_unscoped"]
        audit = client.get("/v1/identities/audit").json()
        assert audit["chain_valid"] and audit["total"] >= 1

    def test_nhi_discover_inline_export(self, client):
        payload = {"okta": {"apps": [
            {"id": "a1", "label": "svc", "signOnMode": "OPENID_CONNECT",
             "settings": {"oauthClient": {"application_type": "service"}}}],
            "api_tokens": []}}
        out = client.post("/v1/identities/nhi/discover", json=payload).json()
        assert out["okta"]["status"] == "ok"
        assert len(out["okta"]["identities"]) == 1


class TestRbac:
    @pytest.fixture()
    def rbac_client(self, monkeypatch):
        monkeypatch.setenv("AGENT_BOM_API_KEYS",
                           "adm-key:admin,op-key:operator,view-key:viewer")
        monkeypatch.delenv("AGENT_BOM_API_KEY", raising=False)
        return TestClient(create_app())

    def test_viewer_reads_only(self, rbac_client):
        assert rbac_client.get(
            "/v1/fleet", headers={"X-API-Key": "view-key"}).status_code == 200
        resp = rbac_client.post("/v1/scan", json={"demo": True},
                                headers={"X-API-Key": "view-key"})
        assert resp.status_code == 403

    def test_operator_can_scan_not_identity(self, rbac_client):
        resp = rbac_client.post("/v1/scan", json={"demo": True},
                                headers={"X-API-Key": "op-key"})
        assert resp.status_code == 201
        resp = rbac_client.post(
            "/v1/identities", json={"agent_name": "x"},
            headers={"X-API-Key": "op-key", **ADMIN_HEADERS})
        assert resp.status_code == 403

    def test_admin_all_access(self, rbac_client):
        resp = rbac_client.post(
            "/v1/identities", json={"agent_name": "x"},
            headers={"X-API-Key": "adm-key", **ADMIN_HEADERS})
        assert resp.status_code == 201

    def test_unknown_key_rejected(self, rbac_client):
        assert rbac_client.get(
            "/v1/fleet", headers={"X-API-Key": "nope"}).status_code == 401

    def test_legacy_single_key_is_admin(self, monkeypatch):
        monkeypatch.setenv("AGENT_BOM_API_KEY", "legacy")
        monkeypatch.delenv("AGENT_BOM_API_KEYS", raising=False)
        c = TestClient(create_app())
        assert c.get("/v1/fleet",
                     headers={"X-API-Key": "legacy"}).status_code == 200
        assert c.get("/v1/fleet").status_code == 401


class TestJobLifecycle:
    def test_cancel_finished_job_is_noop(self, client):
        job = client.post("/v1/scan", json={"demo": True}).json()
        import time

        for _ in range(100):
            if client.get(f"/v1/scan/{job['job_id']}").json()["status"] == "done":
                break
            time.sleep(0.05)
        out = client.post(f"/v1/scan/{job['job_id']}/cancel").json()
        assert out["status"] == "done" and "already finished" in out["note"]

    def test_cancel_unknown_404(self, client):
        assert client.post("/v1/scan/nope/cancel").status_code == 404

    def test_stuck_job_reaped(self, client):
        app = client.app
        state = app.state.abom
        # forge a stuck running job (worker died without updating status)
        import time

        state.jobs["stuck-1"] = {"id": "stuck-1", "status": "running",
                                 "submitted_at": "x", "steps": [],
                                 "result": None, "error": None,
                                 "started_monotonic": time.monotonic() - 10_000}
        reaped = app.state.reap_stuck_jobs(600.0)
        assert "stuck-1" in reaped
        job = client.get("/v1/scan/stuck-1").json()
        assert job["status"] == "failed" and "reaped" in job["error"]

    def test_fresh_jobs_not_reaped(self, client):
        app = client.app
        state = app.state.abom
        import time

        state.jobs["fresh-1"] = {"id": "fresh-1", "status": "running",
                                 "submitted_at": "x", "steps": [],
                                 "result": None, "error": None,
                                 "started_monotonic": time.monotonic()}
        assert app.state.reap_stuck_jobs(600.0) == []
        assert state.jobs["fresh-1"]["status"] == "running"

    def test_cooperative_cancel_flag(self):
        """A pending job with cancel_requested set reports the request and
        the worker exits 'cancelled' at its first phase boundary."""
        app = create_app()
        c = TestClient(app)
        state = app.state.abom
        state.jobs["c1"] = {"id": "c1", "status": "pending",
                            "submitted_at": "x", "steps": [], "result": None,
                            "error": None, "cancel_requested": True}
        resp = c.post("/v1/scan/c1/cancel").json()
        assert resp["cancel_requested"] is True


class TestPostureRemediationTrust:
    def test_routes(self, client):
        import time

        job = client.post("/v1/scan", json={"demo": True}).json()
        for _ in range(100):
            if client.get(f"/v1/scan/{job['job_id']}").json()["status"] == "done":
                break
            time.sleep(0.05)
        rem = client.get("/v1/remediation").json()
        assert rem["commands"]
        script = client.get("/v1/remediation?script=true")
        assert script.text.startswith("#!/bin/sh")
        posture = client.get("/v1/posture").json()
        assert "mcp_auth_posture" in posture and "self_posture" in posture
        trust = client.get("/v1/trust/pypi/reqeusts").json()
        assert trust["grade"] == "F"


class TestApiRobustness:
    """Hostile payloads to every POST route must 4xx, never 500."""

    JUNK = [
        {}, {"x": 1}, {"url": 123, "events": "no"},
        {"agent_name": ["list"]}, {"demo": "yes-ish"},
        {"scopes": {"a": 1}}, {"token": None},
    ]

    def test_posts_never_500(self, client):
        from agentbom_amd.api.server import create_app

        app = create_app()
        c = TestClient(app)
        post_paths = []
        for route in app.routes:
            methods = getattr(route, "methods", None) or set()
            if "POST" in methods:
                path = route.path
                # fill path params with junk ids
                path = (path.replace("{job_id}", "nope")
                            .replace("{identity_id}", "nope")
                            .replace("{grant_id}", "nope")
                            .replace("{webhook_id}", "nope"))
                post_paths.append(path)
        assert post_paths
        for path in post_paths:
            for junk in self.JUNK:
                resp = c.post(path, json=junk, headers=ADMIN_HEADERS)
                assert resp.status_code < 500, (path, junk, resp.text)

    def test_gets_never_500(self, client):
        from agentbom_amd.api.server import create_app

        app = create_app()
        c = TestClient(app)
        for route in app.routes:
            methods = getattr(route, "methods", None) or set()
            if "GET" not in methods:
                continue
            path = route.path
            for param in ("job_id", "identity_id", "node_id", "container_id",
                          "framework", "ecosystem", "name", "grant_id",
                          "webhook_id"):
                path = path.replace("{%s}" % param, "nope").replace(
                    "{%s:path}" % param, "nope")
            resp = c.get(path + "?limit=nope&severity=')--&view=zzz")
            assert resp.status_code < 500, (path, resp.text)
