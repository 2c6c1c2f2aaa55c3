"""Compliance Hub: ingest, absent reconciliation, deltas, overview cache."""

from __future__ import annotations

import pytest

from agentbom_amd.api.hub import ComplianceHub, finding_key


def _f(fid, sev="high", title="x"):
    return {"finding_id": fid, "severity": sev, "title": title,
            "cve_id": "CVE-2024-1"}


@pytest.fixture
def hub():
    return ComplianceHub()


class TestIngest:
    def test_new_changed_resolved_lifecycle(self, hub):
        d1 = hub.ingest("t1", "wiz", [_f("a"), _f("b")])
        assert d1["new"] == ["a", "b"] and not d1["resolved"]
        # same batch again: nothing new, nothing changed
        d2 = hub.ingest("t1", "wiz", [_f("a"), _f("b")])
        assert not d2["new"] and not d2["changed"] and not d2["resolved"]
        # b disappears, a changes severity -> changed + resolved
        d3 = hub.ingest("t1", "wiz", [_f("a", sev="critical")])
        assert d3["changed"] == ["a"] and d3["resolved"] == ["b"]
        rows = {r["finding_key"]: r for r in
                hub.findings("t1", status="")}
        assert rows["a"]["status"] == "open"
        assert rows["a"]["severity"] == "critical"
        assert rows["b"]["status"] == "resolved"
        # b comes back -> reopened (reported as changed, status open again)
        d4 = hub.ingest("t1", "wiz", [_f("a", sev="critical"), _f("b")])
        assert d4["changed"] == ["b"]
        assert {r["finding_key"] for r in hub.findings("t1")} == {"a", "b"}

    def test_sources_isolated(self, hub):
        hub.ingest("t1", "wiz", [_f("a")])
        d = hub.ingest("t1", "prisma", [_f("p1")])
        # ingest from prisma must not resolve wiz's findings
        assert not d["resolved"]
        assert len(hub.findings("t1")) == 2
        assert len(hub.findings("t1", source="wiz")) == 1

    def test_tenant_isolation(self, hub):
        hub.ingest("t1", "wiz", [_f("a")])
        assert hub.findings("t2") == []
        assert hub.overview("t2")["open_total"] == 0

    def test_finding_key_fallback_hash(self):
        row = {"cve_id": "CVE-1", "resource": "db-1", "title": "open port"}
        k1, k2 = finding_key(dict(row)), finding_key(dict(row))
        assert k1 == k2 and k1.startswith("hub-")
        assert finding_key({"id": "X-9"}) == "X-9"


class TestOverview:
    def test_aggregates_and_cache_generation(self, hub):
        hub.ingest("t1", "wiz", [_f("a", "critical"), _f("b", "high")])
        hub.ingest("t1", "prisma", [_f("p1", "high")])
        o1 = hub.overview("t1")
        assert o1["open_total"] == 3
        assert o1["open_by_severity"] == {"critical": 1, "high": 2}
        assert o1["sources"]["wiz"]["open"] == 2
        # cached: same generation object until a write lands
        assert hub.overview("t1") is o1
        hub.ingest("t1", "wiz", [_f("a", "critical")])  # resolves b
        o2 = hub.overview("t1")
        assert o2 is not o1 and o2["open_total"] == 2
        assert o2["resolved_total"] == 1

    def test_ledger(self, hub):
        hub.ingest("t1", "wiz", [_f("a")])
        hub.ingest("t1", "wiz", [])
        rows = hub.ledger("t1")
        assert len(rows) == 2
        assert rows[0]["resolved"] == 1  # latest first: the empty batch
        assert rows[1]["new"] == 1


class TestApi:
    def test_endpoints(self):
        from starlette.testclient import TestClient

        from agentbom_amd.api.server import create_app

        client = TestClient(create_app())
        r = client.post("/v1/compliance/ingest", json={
            "source": "wiz", "findings": [
                {"finding_id": "w1", "severity": "critical",
                 "title": "public bucket"}]})
        assert r.status_code == 200 and r.json()["new"] == ["w1"]
        o = client.get("/v1/compliance/hub/overview").json()
        assert o["open_total"] == 1
        rows = client.get("/v1/compliance/hub/findings?source=wiz").json()
        assert rows["total"] == 1
        assert client.post("/v1/compliance/ingest",
                           json={"findings": []}).status_code == 400
