"""Vulnerability exception workflow: lifecycle, matching, suppression, API."""

from __future__ import annotations

from datetime import datetime, timedelta, timezone

import pytest

from agentbom_amd.api.exceptions_store import (
    APPROVED,
    EXPIRED,
    PENDING,
    ExceptionStore,
    VulnException,
    apply_exceptions_to_report,
)


def _exc(**kw):
    base = dict(vuln_id="CVE-2024-1", package_name="pyyaml",
                reason="accepted risk until Q4 migration",
                requested_by="alice")
    base.update(kw)
    return VulnException(**base)


class TestLifecycle:
    def test_request_approve_expiry_default(self):
        store = ExceptionStore()
        exc = store.request(_exc())
        assert exc.status == PENDING
        ok = store.approve(exc.exception_id, actor="secops", ttl_days=30)
        assert ok.status == APPROVED and ok.approved_by == "secops"
        assert ok.expires_at  # time-boxed by default
        got = store.get(exc.exception_id)
        assert got.status == APPROVED

    def test_invalid_transitions(self):
        store = ExceptionStore()
        exc = store.request(_exc())
        assert store.revoke(exc.exception_id, "x") is None   # pending->revoked
        store.reject(exc.exception_id, "secops", "not justified")
        assert store.approve(exc.exception_id, "secops") is None
        assert store.approve("exc-nope", "secops") is None

    def test_expired_listed_as_expired_and_never_matches(self):
        store = ExceptionStore()
        past = (datetime.now(timezone.utc) - timedelta(days=1)).isoformat()
        exc = store.request(_exc(expires_at=past))
        store.approve(exc.exception_id, "secops")
        rows = store.list()
        assert rows[0].status == EXPIRED
        assert store.active_for("default", "CVE-2024-1", "pyyaml") is None

    def test_audit_chain(self):
        store = ExceptionStore()
        exc = store.request(_exc())
        store.approve(exc.exception_id, "secops")
        store.revoke(exc.exception_id, "secops", "vuln now exploited")
        assert store.audit_chain_valid()


class TestMatching:
    def test_wildcards_and_server_scope(self):
        e = _exc(vuln_id="*", server_name="srv-a", status=APPROVED)
        assert e.matches("CVE-X", "pyyaml", "srv-a")
        assert not e.matches("CVE-X", "pyyaml", "srv-b")
        assert not e.matches("CVE-X", "requests", "srv-a")
        allp = _exc(package_name="*", status=APPROVED)
        assert allp.matches("CVE-2024-1", "anything", "any-server")
        assert not _exc(status=PENDING).matches("CVE-2024-1", "pyyaml")


class TestReportSuppression:
    def test_apply_to_demo_report(self):
        from agentbom_amd.scan.orchestrator import run_demo_scan

        report = run_demo_scan()
        target = next(br for br in report.blast_radii if not br.suppressed)
        store = ExceptionStore()
        exc = store.request(_exc(vuln_id=target.vulnerability.id,
                                 package_name=target.package.name))
        assert apply_exceptions_to_report(report, store) == 0  # pending
        store.approve(exc.exception_id, "secops")
        n = apply_exceptions_to_report(report, store)
        assert n >= 1
        assert target.suppressed
        assert target.suppression_state == "exception"
        assert target.suppression_id == exc.exception_id
        assert target.unsuppressed_risk_score is not None
        # idempotent: already-suppressed rows are not double-counted
        assert apply_exceptions_to_report(report, store) == 0


class TestApi:
    @pytest.fixture()
    def client(self):
        from starlette.testclient import TestClient

        from agentbom_amd.api.server import create_app

        return TestClient(create_app())

    def test_workflow_endpoints(self, client):
        r = client.post("/v1/exceptions", json={
            "vuln_id": "CVE-2024-9", "package_name": "requests",
            "reason": "compensating control"})
        assert r.status_code == 201
        eid = r.json()["exception_id"]
        assert client.get("/v1/exceptions").json()["total"] == 1
        a = client.post(f"/v1/exceptions/{eid}/approve",
                        json={"ttl_days": 7}).json()
        assert a["status"] == "approved" and a["expires_at"]
        assert client.post(f"/v1/exceptions/{eid}/reject").status_code == 409
        v = client.post(f"/v1/exceptions/{eid}/revoke",
                        json={"reason": "incident"}).json()
        assert v["status"] == "revoked"
        assert client.get("/v1/exceptions/audit").json()["chain_valid"]
        assert client.post("/v1/exceptions", json={}).status_code == 400


def test_cli_exceptions_db(tmp_path):
    """--exceptions-db applies approved waivers in a CLI scan."""
    import json
    import subprocess
    import sys

    db = tmp_path / "waivers.db"
    store = ExceptionStore(str(db))
    # find a demo finding to waive
    from agentbom_amd.scan.orchestrator import run_demo_scan

    target = run_demo_scan().blast_radii[0]
    exc = store.request(_exc(vuln_id=target.vulnerability.id,
                             package_name=target.package.name))
    store.approve(exc.exception_id, "secops")
    out = subprocess.run(
        [sys.executable, "-m", "agentbom_amd.cli", "scan", "--demo",
         "--exceptions-db", str(db), "--format", "json", "--exit-zero"],
        capture_output=True, text=True, timeout=240)
    # demo contains a malicious package: that gate fails closed even with
    # --exit-zero, so only the waiver mechanics are asserted here
    assert out.returncode in (0, 1), out.stderr[-500:]
    assert "suppressed by" in out.stderr
    doc = json.loads(out.stdout)
    waived = [r for r in doc["blast_radius"]
              if r["suppression_id"] == exc.exception_id]
    assert waived and all(r["suppression_state"] == "exception"
                          for r in waived)


def test_consolidated_audit_verify():
    from starlette.testclient import TestClient

    from agentbom_amd.api.server import create_app

    client = TestClient(create_app())
    # exercise both chains
    r = client.post("/v1/exceptions", json={
        "vuln_id": "CVE-1", "package_name": "p", "reason": "r"})
    client.post(f"/v1/exceptions/{r.json()['exception_id']}/approve")
    client.post("/v1/identities", json={"agent_name": "bot"})
    out = client.get("/v1/audit/verify").json()
    assert out["all_valid"] and out["chains"]["exceptions"]
    assert out["chains"].get("identities", True)
