"""ClickHouse analytics tier over a mock HTTP interface."""

from __future__ import annotations

import json

import httpx
import pytest

from agentbom_amd.api.clickhouse_store import ClickHouseAnalyticsStore
from agentbom_amd.utils.http_client import OfflineError, set_offline


@pytest.fixture(autouse=True)
def _online():
    set_offline(False)
    yield
    set_offline(False)


class _FakeCH:
    """Captures queries; replays canned JSONEachRow answers."""

    def __init__(self):
        self.queries: list[str] = []
        self.inserts: list[str] = []
        self.answers: dict[str, str] = {}

    def handler(self, request: httpx.Request) -> httpx.Response:
        q = request.url.params.get("query", "")
        body = request.content.decode()
        if q:  # insert with FORMAT JSONEachRow body
            self.queries.append(q)
            self.inserts.append(body)
            return httpx.Response(200, text="")
        self.queries.append(body)
        for marker, answer in self.answers.items():
            if marker in body:
                return httpx.Response(200, text=answer)
        return httpx.Response(200, text="")


@pytest.fixture
def fake():
    return _FakeCH()


@pytest.fixture
def store(fake):
    client = httpx.Client(transport=httpx.MockTransport(fake.handler))
    return ClickHouseAnalyticsStore("http://ch.example:8123", client=client)


def test_schema_and_insert(store, fake):
    from agentbom_amd.scan.orchestrator import run_demo_scan

    store.ensure_schema()
    assert any("CREATE DATABASE IF NOT EXISTS agentbom" in q for q in fake.queries)
    assert any("MergeTree" in q and "scan_findings" in q for q in fake.queries)

    report = run_demo_scan()
    n = store.insert_findings(report, tenant_id="acme")
    assert n == len(report.blast_radii) > 0
    rows = [json.loads(line) for line in fake.inserts[-1].splitlines()]
    assert len(rows) == n
    assert all(r["tenant_id"] == "acme" for r in rows)
    assert any(r["vuln_id"].startswith("CVE-") for r in rows)
    assert all(0.0 <= r["risk_score"] <= 10.0 for r in rows)


def test_trend_queries_parse(store, fake):
    fake.answers["GROUP BY day, severity"] = (
        '{"day":"2026-09-01","severity":"critical","findings":3}\n'
        '{"day":"2026-09-02","severity":"high","findings":7}\n')
    trend = store.severity_trend("acme", days=7)
    assert trend[0]["findings"] == 3 and trend[1]["severity"] == "high"
    assert any("INTERVAL 7 DAY" in q for q in fake.queries)

    fake.answers["max_risk"] = '{"package":"pyyaml@5.3","max_risk":10,"findings":4}\n'
    top = store.top_risk_packages("acme", limit=5)
    assert top[0]["package"] == "pyyaml@5.3"
    assert any("LIMIT 5" in q for q in fake.queries)


def test_tenant_escaping(store, fake):
    store.severity_trend("o'brien")
    assert any("o\\'brien" in q for q in fake.queries)


def test_offline_refused():
    set_offline(True)
    with pytest.raises(OfflineError):
        ClickHouseAnalyticsStore("http://ch.example:8123")


def test_error_raises(fake):
    def bad(request):
        return httpx.Response(500, text="boom")

    client = httpx.Client(transport=httpx.MockTransport(bad))
    store = ClickHouseAnalyticsStore("http://ch.example:8123", client=client)
    with pytest.raises(RuntimeError, match="clickhouse"):
        store.ensure_schema()
