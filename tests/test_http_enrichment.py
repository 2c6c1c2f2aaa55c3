"""HTTP client (retry/breaker/offline), enrichment, CLI evidence/mesh."""

import json

import httpx
import pytest
from click.testing import CliRunner

from agentbom_amd.utils import http_client as hc


@pytest.fixture(autouse=True)
def _reset():
    hc.set_offline(False)
    hc.reset_rate_limit_breaker()
    yield
    hc.set_offline(False)
    hc.reset_rate_limit_breaker()


class TestHttpClient:
    def _client(self, handler):
        return httpx.Client(transport=httpx.MockTransport(handler))

    def test_retry_on_500_then_success(self):
        calls = []

        def handler(request):
            calls.append(1)
            return httpx.Response(500 if len(calls) < 3 else 200, json={"ok": True})

        resp = hc.request_with_retry(self._client(handler), "GET",
                                     "https://api.example/x", sleep=lambda s: None)
        assert resp.status_code == 200 and len(calls) == 3

    def test_4xx_not_retried(self):
        calls = []

        def handler(request):
            calls.append(1)
            return httpx.Response(404)

        resp = hc.request_with_retry(self._client(handler), "GET",
                                     "https://api.example/x", sleep=lambda s: None)
        assert resp.status_code == 404 and len(calls) == 1

    def test_breaker_trips_on_429(self):
        def handler(request):
            return httpx.Response(429)

        client = self._client(handler)
        hc.request_with_retry(client, "GET", "https://throttled.example/a",
                              sleep=lambda s: None)
        assert hc.registry_breaker_tripped("throttled.example")
        # breaker open -> short circuit without any call
        assert hc.request_with_retry(client, "GET", "https://throttled.example/b",
                                     sleep=lambda s: None) is None
        # other hosts unaffected
        assert not hc.registry_breaker_tripped("fine.example")

    def test_offline_guard(self):
        hc.set_offline(True)
        with pytest.raises(hc.OfflineError):
            hc.request_with_retry(self._client(lambda r: httpx.Response(200)),
                                  "GET", "https://api.example/x")

    def test_sanitize_url(self):
        assert hc.sanitize_url("https://user:pass@host/x") == "https://***@host/x"
        assert "secret123" not in hc.sanitize_url("https://h/x?api_key=secret123&b=1")


class TestEnrichment:
    def test_store_join_and_rescore(self, tmp_path):
        from agentbom_amd.db.store import AdvisoryStore
        from agentbom_amd.scan.enrichment import enrich_vulnerabilities
        from agentbom_amd.scan.orchestrator import run_demo_scan

        report = run_demo_scan()
        target = next(br for br in report.blast_radii
                      if br.vulnerability.id == "CVE-2020-14343")
        base_score = target.risk_score
        assert target.vulnerability.epss_score is None

        store = AdvisoryStore(tmp_path / "v.db")
        store.conn.execute(
            "INSERT INTO epss_scores(cve_id, probability, percentile, updated_at)"
            " VALUES ('CVE-2020-14343', 0.93, 99.1, 'now')")
        store.conn.commit()
        stats = enrich_vulnerabilities(report, store)
        assert stats["epss_hits"] >= 1
        assert target.vulnerability.epss_score == pytest.approx(0.93)
        assert target.vulnerability.exploit_likelihood == "likely_exploited"
        # epss >= 0.7 adds the EPSS boost unless already clamped at 10
        assert target.risk_score >= base_score

    def test_offline_bundle(self, tmp_path):
        from agentbom_amd.scan.enrichment import load_offline_bundle
        from agentbom_amd.scan.orchestrator import run_demo_scan

        bundle = tmp_path / "bundle"
        bundle.mkdir()
        (bundle / "epss.csv").write_text("cve,epss,percentile\nCVE-2023-45857,0.88,0.99\n")
        (bundle / "kev.json").write_text(json.dumps(
            {"vulnerabilities": [{"cveID": "CVE-2022-0235", "dateAdded": "2024-01-01"}]}))
        report = run_demo_scan()
        stats = load_offline_bundle(report, bundle)
        assert stats["epss"] == 1 and stats["kev"] == 1
        nf = next(br for br in report.blast_radii
                  if br.vulnerability.id == "CVE-2022-0235")
        assert nf.vulnerability.is_kev


class TestCliEvidenceMesh:
    def test_mesh_demo(self):
        res = CliRunner().invoke(
            __import__("agentbom_amd.cli", fromlist=["main"]).main,
            ["mesh", "--demo", "-f", "json"])
        assert res.exit_code == 0
        mesh = json.loads(res.output)
        assert len(mesh["agents"]) == 5
        assert "DATABASE_URL" in mesh["shared_credentials"]

    def test_graph_evidence(self, tmp_path, monkeypatch):
        from agentbom_amd.graph.builder import build_unified_graph_from_report
        from agentbom_amd.graph.store import SQLiteGraphStore
        from agentbom_amd.scan.orchestrator import run_demo_scan

        store = SQLiteGraphStore(tmp_path / "g.db")
        store.save_snapshot(build_unified_graph_from_report(run_demo_scan()), scan_id="s1")
        store.close()
        res = CliRunner().invoke(
            __import__("agentbom_amd.cli", fromlist=["main"]).main,
            ["graph-evidence", "--mode", "manifest", "--store", str(tmp_path / "g.db")])
        assert res.exit_code == 0
        doc = json.loads(res.output)
        assert doc["schema_version"] == "agent-bom.graph_evidence_manifest/v1"
