"""Vulnerability enrichment: EPSS / KEV / NVD from the local store +
offline bundles.

Reference: src/agent_bom/enrichment.py:677 enrich_vulnerabilities (NVD
CVSS/CWE, EPSS bulk, CISA KEV; disk-cached with staleness windows;
offline bundles).  The live-fetch layer plugs into db/osv_ingest sync_*;
enrichment itself is a pure join from the AdvisoryStore onto a report,
followed by rescoring — at estate scale the same join is the device-side
gather in the match pipeline (arena columns epss/kev travel with every
window).
"""

from __future__ import annotations

from pathlib import Path
from typing import Optional

from agentbom_amd.db.store import AdvisoryStore, default_db_path
from agentbom_amd.models import AIBOMReport


def enrich_vulnerabilities(report: AIBOMReport,
                           store: Optional[AdvisoryStore] = None) -> dict:
    """Stamp EPSS scores and KEV membership onto every vulnerability from
    the local advisory DB; rescore blast radii.  Returns coverage stats."""
    own_store = False
    if store is None:
        path = default_db_path()
        if not path.exists():
            return {"enriched": 0, "epss_hits": 0, "kev_hits": 0,
                    "note": "no local advisory DB"}
        store = AdvisoryStore(path)
        own_store = True
    try:
        epss = {
            row[0]: (row[1], row[2])
            for row in store.conn.execute(
                "SELECT cve_id, probability, percentile FROM epss_scores")
        }
        kev = {
            row[0]: (row[1], row[2])
            for row in store.conn.execute(
                "SELECT cve_id, date_added, due_date FROM kev_entries")
        }
    finally:
        if own_store:
            store.close()

    def ids_of(vuln) -> list[str]:
        return [vuln.id, *vuln.aliases]

    enriched = epss_hits = kev_hits = 0
    seen: set[int] = set()
    all_vulns = [
        v for agent in report.agents for s in agent.mcp_servers
        for p in s.packages for v in p.vulnerabilities
    ] + [br.vulnerability for br in report.blast_radii]
    for vuln in all_vulns:
        if id(vuln) in seen:
            continue
        seen.add(id(vuln))
        hit = False
        for vid in ids_of(vuln):
            if vid in epss and vuln.epss_score is None:
                vuln.epss_score, vuln.epss_percentile = epss[vid]
                epss_hits += 1
                hit = True
            if vid in kev and not vuln.is_kev:
                vuln.is_kev = True
                vuln.kev_date_added, vuln.kev_due_date = kev[vid]
                kev_hits += 1
                hit = True
        if hit:
            enriched += 1

    for br in report.blast_radii:
        br.calculate_risk_score()
    report.blast_radii.sort(key=lambda b: -b.risk_score)
    report.vuln_data_freshness = _freshness(store if not own_store else None)
    return {"enriched": enriched, "epss_hits": epss_hits, "kev_hits": kev_hits}


def _freshness(store: Optional[AdvisoryStore]) -> dict:
    if store is None:
        path = default_db_path()
        if not path.exists():
            return {"sources": {}, "note": "no local advisory DB"}
        store = AdvisoryStore(path)
        try:
            return _freshness(store)
        finally:
            store.close()
    return {"sources": store.status()["sync"]}


def load_offline_bundle(report: AIBOMReport, bundle_dir: str | Path) -> dict:
    """Air-gapped enrichment bundle: epss.csv + kev.json + osv/*.json in a
    directory, ingested into a temp store and applied to the report."""
    import tempfile

    from agentbom_amd.db.osv_ingest import sync_epss, sync_kev, sync_osv

    bundle = Path(bundle_dir)
    with tempfile.TemporaryDirectory() as tmp:
        store = AdvisoryStore(Path(tmp) / "bundle.db")
        stats = {"osv": 0, "epss": 0, "kev": 0}
        osv_dir = bundle / "osv"
        if osv_dir.exists():
            stats["osv"] = sync_osv(store, osv_dir)
        if (bundle / "epss.csv").exists():
            stats["epss"] = sync_epss(store, bundle / "epss.csv")
        if (bundle / "kev.json").exists():
            stats["kev"] = sync_kev(store, bundle / "kev.json")
        result = enrich_vulnerabilities(report, store)
        store.close()
        return {**stats, **result}
