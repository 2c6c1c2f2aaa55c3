"""Live advisory acquisition (db/live.py) against httpx.MockTransport.

The offline default stays untouched: every entry point raises OfflineError
in offline mode (first test).  All protocol behavior — OSV batch chunking,
detail caching, GHSA range parsing, EPSS paging, KEV ingestion, NVD
checkpointing — is exercised against mock transports; tests that need the
real services carry the ``network`` marker and skip by default.
"""

from __future__ import annotations

import io
import json
import zipfile

import httpx
import pytest

from agentbom_amd.db import live
from agentbom_amd.db.store import AdvisoryStore
from agentbom_amd.utils.http_client import OfflineError, set_offline


@pytest.fixture(autouse=True)
def _online():
    set_offline(False)
    yield
    set_offline(False)


@pytest.fixture
def store(tmp_path):
    return AdvisoryStore(tmp_path / "vuln.db")


def _client(handler):
    return httpx.Client(transport=httpx.MockTransport(handler))


def test_offline_guard_blocks_every_entry_point(store, tmp_path):
    set_offline(True)
    with pytest.raises(OfflineError):
        live.query_osv_batch([("npm", "leftpad", "1.0.0")])
    with pytest.raises(OfflineError):
        live.fetch_osv_details(["OSV-1"])
    with pytest.raises(OfflineError):
        live.sync_osv_bulk(store, ["npm"])
    with pytest.raises(OfflineError):
        live.fetch_ghsa_advisories([("npm", "leftpad")])
    with pytest.raises(OfflineError):
        live.sync_epss_live(store)
    with pytest.raises(OfflineError):
        live.sync_kev_live(store)
    with pytest.raises(OfflineError):
        live.sync_nvd_live(store)


class TestOsvBatch:
    def test_batch_chunking_and_ids(self):
        calls = []

        def handler(request):
            body = json.loads(request.content)
            calls.append(len(body["queries"]))
            results = [{"vulns": [{"id": f"OSV-{i}"}]} if i % 2 == 0 else {}
                       for i in range(len(body["queries"]))]
            return httpx.Response(200, json={"results": results})

        queries = [("npm", f"pkg{i}", "1.0.0") for i in range(1500)]
        out = live.query_osv_batch(queries, client=_client(handler))
        assert calls == [1000, 500]  # <=1000 per chunk (OSV API limit)
        assert len(out) == 1500
        assert out[0] == ["OSV-0"] and out[1] == []

    def test_ecosystem_spelling(self):
        seen = {}

        def handler(request):
            body = json.loads(request.content)
            seen["eco"] = body["queries"][0]["package"]["ecosystem"]
            return httpx.Response(200, json={"results": [{}]})

        live.query_osv_batch([("pypi", "requests", "2.0.0")], client=_client(handler))
        assert seen["eco"] == "PyPI"

    def test_failed_chunk_fails_open(self):
        def handler(request):
            return httpx.Response(400)

        out = live.query_osv_batch([("npm", "a", "1")], client=_client(handler))
        assert out == [[]]

    def test_details_cached(self, tmp_path):
        hits = {"n": 0}

        def handler(request):
            hits["n"] += 1
            vid = request.url.path.rsplit("/", 1)[-1]
            return httpx.Response(200, json={
                "id": vid, "summary": "s",
                "affected": [{"package": {"ecosystem": "npm", "name": "x"},
                              "ranges": [{"type": "SEMVER", "events": [
                                  {"introduced": "0"}, {"fixed": "2.0.0"}]}]}],
            })

        cache = live.ScanCache(root=tmp_path / "cache")
        recs = live.fetch_osv_details(["OSV-9"], client=_client(handler), cache=cache)
        assert recs[0]["id"] == "OSV-9" and hits["n"] == 1
        recs2 = live.fetch_osv_details(["OSV-9"], client=_client(handler), cache=cache)
        assert recs2[0]["id"] == "OSV-9" and hits["n"] == 1  # cache hit

    def test_windows_for_packages_end_to_end(self, tmp_path):
        def handler(request):
            if request.url.path.endswith("querybatch"):
                return httpx.Response(200, json={"results": [
                    {"vulns": [{"id": "OSV-42"}]}]})
            return httpx.Response(200, json={
                "id": "OSV-42", "summary": "RCE",
                "database_specific": {"severity": "HIGH"},
                "affected": [{"package": {"ecosystem": "npm", "name": "leftpad"},
                              "ranges": [{"type": "SEMVER", "events": [
                                  {"introduced": "0"}, {"fixed": "1.3.0"}]}]}],
            })

        wins = live.osv_windows_for_packages(
            [("npm", "leftpad", "1.0.0")], client=_client(handler),
            cache=live.ScanCache(root=tmp_path / "c"))
        assert len(wins) == 1
        w = wins[0]
        assert (w.vuln_id, w.package_name, w.fixed) == ("OSV-42", "leftpad", "1.3.0")


def test_osv_bulk_zip_ingest(store):
    rec = {
        "id": "OSV-100", "summary": "bulk",
        "affected": [{"package": {"ecosystem": "npm", "name": "bulkpkg"},
                      "ranges": [{"type": "SEMVER", "events": [
                          {"introduced": "1.0.0"}, {"fixed": "1.2.0"}]}]}],
    }
    buf = io.BytesIO()
    with zipfile.ZipFile(buf, "w") as zf:
        zf.writestr("OSV-100.json", json.dumps(rec))
        zf.writestr("README.md", "not json")

    def handler(request):
        assert request.url.path == "/npm/all.zip"
        return httpx.Response(200, content=buf.getvalue())

    n = live.sync_osv_bulk(store, ["npm"], client=_client(handler))
    assert n == 1
    st = store.status()
    assert st["counts"]["affected_windows"] == 1
    assert "osv:npm" in st["sync"]


class TestGhsa:
    def test_fetch_and_window_parse(self):
        def handler(request):
            assert request.url.params["ecosystem"] == "pip"
            return httpx.Response(200, json=[{
                "ghsa_id": "GHSA-xxxx", "cve_id": "CVE-2024-1111",
                "severity": "high", "summary": "ssrf",
                "cvss": {"score": 8.1},
                "cwes": [{"cwe_id": "CWE-918"}],
                "vulnerabilities": [{
                    "package": {"ecosystem": "pip", "name": "requests"},
                    "vulnerable_version_range": ">= 2.3.0, < 2.31.0",
                    "first_patched_version": "2.31.0",
                }],
            }])

        advs = live.fetch_ghsa_advisories([("pypi", "requests")],
                                          client=_client(handler))
        wins = live.ghsa_windows(advs)
        assert len(wins) == 1
        w = wins[0]
        assert w.vuln_id == "CVE-2024-1111"
        assert (w.introduced, w.fixed) == ("2.3.0", "2.31.0")
        assert w.ecosystem == "pypi" and w.cwe_ids == ("CWE-918",)

    @pytest.mark.parametrize("vrange,expect", [
        (">= 1.0.0, < 2.0.0", ("1.0.0", "2.0.0", None)),
        ("< 3.1.0", ("0", "3.1.0", None)),
        ("<= 2.2.2", ("0", None, "2.2.2")),
        ("= 1.5.0", ("1.5.0", None, "1.5.0")),
    ])
    def test_range_grammar(self, vrange, expect):
        assert live._parse_ghsa_range(vrange) == expect


def test_epss_paged_sync(store):
    pages = {"n": 0}

    def handler(request):
        pages["n"] += 1
        offset = int(request.url.params["offset"])
        if offset == 0:
            data = [{"cve": f"CVE-2024-{i}", "epss": "0.5", "percentile": "0.9"}
                    for i in range(3)]
        else:
            data = []
        return httpx.Response(200, json={"data": data})

    n = live.sync_epss_live(store, client=_client(handler), page_size=3)
    assert n == 3 and pages["n"] == 2
    row = store.conn.execute(
        "SELECT probability, percentile FROM epss_scores WHERE cve_id='CVE-2024-1'"
    ).fetchone()
    assert row == (0.5, 90.0)


def test_kev_live_sync(store):
    def handler(request):
        return httpx.Response(200, json={"vulnerabilities": [
            {"cveID": "CVE-2023-1", "dateAdded": "2023-01-01",
             "dueDate": "2023-01-15", "product": "P", "vendorProject": "V"},
        ]})

    assert live.sync_kev_live(store, client=_client(handler)) == 1
    assert store.status()["counts"]["kev_entries"] == 1


def test_nvd_checkpointed_sync(store):
    store.conn.execute(
        "INSERT INTO vulns(id, summary, severity, source) VALUES"
        " ('CVE-2024-7', 's', 'unknown', 'osv')")
    store.conn.commit()
    calls = []

    def handler(request):
        calls.append(dict(request.url.params))
        return httpx.Response(200, json={
            "totalResults": 1,
            "vulnerabilities": [{"cve": {
                "id": "CVE-2024-7", "lastModified": "2026-01-02T00:00:00",
                "metrics": {"cvssMetricV31": [{"cvssData": {
                    "baseScore": 9.8, "baseSeverity": "CRITICAL"}}]},
            }}],
        })

    n = live.sync_nvd_live(store, client=_client(handler))
    assert n == 1
    row = store.conn.execute(
        "SELECT cvss_score, severity FROM vulns WHERE id='CVE-2024-7'").fetchone()
    assert row == (9.8, "critical")
    # second sync resumes from the checkpoint
    live.sync_nvd_live(store, client=_client(handler))
    assert "lastModStartDate" in calls[-1]
    assert calls[-1]["lastModStartDate"].startswith("2026-01-02")


def test_orchestrator_live_osv_merges_windows():
    """scan_agents(live_osv=True) matches against live windows; transport
    failure degrades to a warning, never a crash."""
    from unittest.mock import patch

    from agentbom_amd.db.arena import AdvisoryWindow
    from agentbom_amd.models.core import Severity
    from agentbom_amd.scan.demo import DEMO_INVENTORY
    from agentbom_amd.scan.orchestrator import ScanOptions, inventory_to_agents, scan_agents

    agents = inventory_to_agents(DEMO_INVENTORY)
    live_win = AdvisoryWindow(
        ecosystem="pypi", package_name="pyyaml", vuln_id="OSV-LIVE-1",
        introduced="0", fixed="99.0", severity=Severity.HIGH)
    with patch("agentbom_amd.db.live.osv_windows_for_packages",
               return_value=[live_win]):
        report = scan_agents(agents, [], ScanOptions(live_osv=True))
    assert any(v.id == "OSV-LIVE-1"
               for br in report.blast_radii for v in [br.vulnerability])

    def boom(*a, **k):
        raise RuntimeError("transport down")

    with patch("agentbom_amd.db.live.osv_windows_for_packages", side_effect=boom):
        report = scan_agents(agents, [], ScanOptions(live_osv=True))
    assert any("live OSV query unavailable" in w for w in report.warnings)


@pytest.mark.network
def test_real_osv_roundtrip():  # pragma: no cover - requires egress
    out = live.query_osv_batch([("pypi", "pyyaml", "5.3")])
    assert out and out[0]
