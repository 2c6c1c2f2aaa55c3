"""Graph store (snapshots/diff/evidence), toxic combos, OCSF/HTML outputs."""

import json

import pytest

from agentbom_amd.graph.builder import build_unified_graph_from_report
from agentbom_amd.graph.store import SQLiteGraphStore
from agentbom_amd.graph.toxic_combos import (
    detect_toxic_combinations,
    toxic_combination_to_finding,
)
from agentbom_amd.scan.orchestrator import run_demo_scan


@pytest.fixture(scope="module")
def report():
    return run_demo_scan()


@pytest.fixture(scope="module")
def graph(report):
    return build_unified_graph_from_report(report)


class TestGraphStore:
    def test_save_load_roundtrip(self, tmp_path, graph):
        store = SQLiteGraphStore(tmp_path / "g.db")
        sid = store.save_snapshot(graph, scan_id="scan-1")
        loaded = store.load_snapshot(sid)
        assert loaded.node_count == graph.node_count
        assert loaded.edge_count == graph.edge_count
        # traversal works on the rehydrated graph
        assert loaded.shortest_path("agent:cursor", "vuln:CVE-2020-14343")

    def test_pagination(self, tmp_path, graph):
        store = SQLiteGraphStore(tmp_path / "g.db")
        sid = store.save_snapshot(graph)
        page1 = store.nodes_page(sid, limit=10)
        assert len(page1["nodes"]) == 10 and page1["next_cursor"]
        page2 = store.nodes_page(sid, cursor=page1["next_cursor"], limit=1000)
        ids1 = {n["id"] for n in page1["nodes"]}
        ids2 = {n["id"] for n in page2["nodes"]}
        assert not ids1 & ids2
        assert len(ids1 | ids2) == graph.node_count

    def test_search(self, tmp_path, graph):
        store = SQLiteGraphStore(tmp_path / "g.db")
        sid = store.save_snapshot(graph)
        hits = store.search_nodes(sid, "pyyaml")
        assert hits and all("pyyaml" in h["id"].lower() or "pyyaml" in h["label"].lower()
                            for h in hits)

    def test_diff_and_history(self, tmp_path, graph, report):
        store = SQLiteGraphStore(tmp_path / "g.db")
        s1 = store.save_snapshot(graph, scan_id="a")
        # second snapshot without the data-pipeline agent
        from agentbom_amd.scan.demo import DEMO_INVENTORY, demo_advisory_windows
        from agentbom_amd.scan.orchestrator import ScanOptions, inventory_to_agents, scan_agents

        inv = {"agents": [a for a in DEMO_INVENTORY["agents"] if a["name"] != "data-pipeline"]}
        r2 = scan_agents(inventory_to_agents(inv), demo_advisory_windows(), ScanOptions())
        g2 = build_unified_graph_from_report(r2)
        s2 = store.save_snapshot(g2, scan_id="b")
        diff = store.diff_snapshots(s1, s2)
        assert "agent:data-pipeline" in diff["nodes_removed"]
        assert diff["summary"]["nodes_removed"] > 0
        history = store.graph_history()
        assert len(history["snapshots"]) == 2
        assert history["snapshots"][0]["diff_vs_previous"]

    def test_retention(self, tmp_path, graph):
        store = SQLiteGraphStore(tmp_path / "g.db", retention=2)
        for i in range(4):
            store.save_snapshot(graph, scan_id=f"s{i}")
        assert len(store.list_snapshots()) == 2

    def test_evidence_manifest(self, tmp_path, graph):
        store = SQLiteGraphStore(tmp_path / "g.db")
        store.save_snapshot(graph, scan_id="x")
        m = store.evidence_manifest()
        assert m["schema_version"] == "agent-bom.graph_evidence_manifest/v1"
        assert m["graph_digest"]
        assert m["counts"]["nodes"] == graph.node_count
        assert m["retention_policy"] == {"snapshots": 10}


class TestToxicCombos:
    def test_hero_combo_detected(self, graph):
        combos = detect_toxic_combinations(graph)
        assert combos
        names = {c.name for c in combos}
        # shell-runner-server: PyYAML critical + AWS creds + run_shell
        assert "exploitable-server-with-credentials-and-execution" in names
        combo = next(c for c in combos
                     if c.name == "exploitable-server-with-credentials-and-execution"
                     and "server:shell-runner-server" in c.nodes)
        assert combo.severity == "critical"

    def test_malicious_cred_combo(self, graph):
        combos = detect_toxic_combinations(graph)
        mal = [c for c in combos if c.name == "malicious-package-with-credential-reach"]
        assert mal and mal[0].score == pytest.approx(9.8)

    def test_shared_cred_combo(self, graph):
        combos = detect_toxic_combinations(graph)
        assert any(c.name == "shared-credential-lateral-blast" for c in combos)

    def test_deterministic(self, graph):
        a = detect_toxic_combinations(graph)
        b = detect_toxic_combinations(graph)
        assert [c.id for c in a] == [c.id for c in b]

    def test_to_finding(self, graph):
        combo = detect_toxic_combinations(graph)[0]
        f = toxic_combination_to_finding(combo)
        assert f.finding_type.value == "COMBINATION"
        assert f.source.value == "GRAPH_ANALYSIS"
        assert f.id == combo.id


class TestOcsfHtml:
    def test_ocsf_events(self, report):
        from agentbom_amd.output.ocsf import to_ocsf

        doc = to_ocsf(report)
        vuln_events = [e for e in doc["events"] if e["class_uid"] == 2002]
        assert len(vuln_events) == len(report.blast_radii)
        ev = next(e for e in vuln_events
                  if e["vulnerabilities"][0]["cve"]["uid"] == "CVE-2023-4863")
        assert ev["vulnerabilities"][0]["is_exploit_available"] is True
        assert ev["severity_id"] == 4
        inv_events = [e for e in doc["events"] if e["class_uid"] == 5020]
        assert len(inv_events) == report.total_agents
        json.dumps(doc)  # serializable

    def test_html(self, report):
        from agentbom_amd.output.html_fmt import to_html

        html_text = to_html(report)
        assert html_text.startswith("<!DOCTYPE html>")
        assert "CVE-2020-14343" in html_text
        assert "MALICIOUS" in html_text
