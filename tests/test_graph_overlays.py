"""Cost / evidence / endpoint overlay tests (shared overlay contract)."""

import copy

import pytest

from agentbom_amd.graph.builder import build_unified_graph_from_report
from agentbom_amd.graph.overlays import (
    apply_cost_overlay,
    apply_endpoint_overlay,
    apply_evidence_overlay,
)
from agentbom_amd.scan.orchestrator import run_demo_scan


@pytest.fixture(scope="module")
def report():
    return run_demo_scan()


@pytest.fixture()
def graph(report):
    return build_unified_graph_from_report(report)


class TestCostOverlay:
    def test_attach_rollup_fuse(self, graph, report):
        agent_label = report.agents[0].name
        stats = apply_cost_overlay(graph, [
            {"target": agent_label, "usd": 120.5},
            {"target": "no-such-node", "usd": 1.0}])
        assert stats["matched"] == 1
        assert stats["unmatched"] == ["no-such-node"]
        node = next(n for n in graph.nodes.values()
                    if n.label == agent_label)
        assert node.properties["cost_usd"] == 120.5
        assert node.properties["subtree_cost_usd"] >= 120.5

    def test_noop_and_idempotent(self, graph):
        before = {nid: copy.deepcopy(n.properties)
                  for nid, n in graph.nodes.items()}
        assert apply_cost_overlay(graph, []) == {
            "matched": 0, "unmatched": [], "fused": 0}
        assert {nid: n.properties for nid, n in graph.nodes.items()} == before

    def test_idempotent_with_data(self, graph, report):
        recs = [{"target": report.agents[0].name, "usd": 10}]
        apply_cost_overlay(graph, recs)
        snap = {nid: copy.deepcopy(n.properties) for nid, n in graph.nodes.items()}
        apply_cost_overlay(graph, recs)
        assert {nid: n.properties for nid, n in graph.nodes.items()} == snap


class TestEvidenceOverlay:
    def test_stamps_and_phantoms(self, graph, report):
        tool = next(t.name for b in report.blast_radii for t in b.exposed_tools)
        rows = [{"method": "tools/call", "tool": tool, "action": "allow",
                 "ts": 100.0}] * 3
        rows.append({"method": "tools/call", "tool": "ghost_tool",
                     "action": "block", "ts": 101.0})
        stats = apply_evidence_overlay(graph, rows)
        assert stats["stamped"] >= 1
        assert stats["phantom_tools"] == [{"tool": "ghost_tool", "calls": 1}]
        stamped = [n for n in graph.nodes.values()
                   if n.properties.get("runtime_confirmed")]
        assert stamped and stamped[0].properties["observed_calls"] == 3

    def test_noop(self, graph):
        assert apply_evidence_overlay(graph, []) == {
            "stamped": 0, "phantom_tools": []}


class TestEndpointOverlay:
    def test_links_running_agents(self, graph, report):
        agent_label = report.agents[0].name
        inv = {"processes": [{"name": agent_label.lower(), "pid": 42}],
               "listening_ports": [{"port": 8080}]}
        stats = apply_endpoint_overlay(graph, inv, host_name="workstation-1")
        assert stats["nodes_added"] == 1
        assert stats["edges_added"] >= 1
        host = graph.nodes["endpoint:workstation-1"]
        assert host.properties["listening_ports"] == [8080]
        agent = next(n for n in graph.nodes.values() if n.label == agent_label)
        assert agent.properties.get("endpoint_confirmed")

    def test_noop(self, graph):
        assert apply_endpoint_overlay(graph, {}) == {
            "nodes_added": 0, "edges_added": 0}


class TestEffectivePermissions:
    def test_direct_and_identity_paths(self, graph, report):
        from agentbom_amd.graph.effective_permissions import (
            compute_effective_permissions,
        )
        from agentbom_amd.graph.nhi_overlay import apply_issued_identity_overlay
        from agentbom_amd.identity import AgentIdentityStore

        store = AgentIdentityStore()
        store.issue(report.agents[0].name, scopes=["*"])
        apply_issued_identity_overlay(graph, store)

        perms = compute_effective_permissions(graph)
        assert perms["summary"]["total_agents"] == len(report.agents)
        first = next(a for a in perms["agents"]
                     if a["agent"] == report.agents[0].name)
        assert first["effective_tools"] > 0
        kinds = {p["kind"] for g in first["grants"] for p in g["paths"]}
        assert "direct" in kinds
        assert perms["summary"]["wildcard_identities"]

    def test_evidence_informed_unused_grants(self, graph, report):
        from agentbom_amd.graph.effective_permissions import (
            compute_effective_permissions,
            least_privilege_recommendations,
        )

        called_tool = next(t.name for b in report.blast_radii
                           for t in b.exposed_tools)
        apply_evidence_overlay(graph, [
            {"method": "tools/call", "tool": called_tool, "ts": 1.0}])
        perms = compute_effective_permissions(graph)
        assert perms["evidence_informed"]
        # every granted tool except the called one is an unused grant
        assert perms["summary"]["total_unused_grants"] > 0
        for agent in perms["agents"]:
            assert called_tool not in agent["unused_grants"] or not any(
                g["tool"] == called_tool and g["observed_calls"] > 0
                for g in agent["grants"])
        recs = least_privilege_recommendations(graph)
        assert any(r["kind"] == "revoke_unused_grant" for r in recs)

    def test_no_evidence_means_no_unused_claims(self, graph):
        from agentbom_amd.graph.effective_permissions import (
            compute_effective_permissions,
        )

        perms = compute_effective_permissions(graph)
        assert not perms["evidence_informed"]
        assert perms["summary"]["total_unused_grants"] is None
