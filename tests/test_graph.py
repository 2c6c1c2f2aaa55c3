"""UnifiedGraph container, builder, reach, attack paths, rollup."""

import pytest

from agentbom_amd.graph.attack_paths import compute_fused_attack_paths, path_rank_tuple
from agentbom_amd.graph.builder import (
    build_unified_graph_from_report,
    build_unified_graph_from_report_json,
)
from agentbom_amd.graph.container import UnifiedEdge, UnifiedGraph, UnifiedNode
from agentbom_amd.graph.dependency_reach import (
    apply_dependency_reachability_to_blast_radii,
    compute_dependency_reach,
)
from agentbom_amd.graph.rollup import drill_down, rollup_view
from agentbom_amd.graph.types import (
    DEPENDENCY_REACH_MASK,
    EntityType,
    GraphSemanticLayer,
    NodeStatus,
    RelationshipType,
    rel_mask,
)
from agentbom_amd.scan.orchestrator import run_demo_scan


@pytest.fixture(scope="module")
def report():
    return run_demo_scan()


@pytest.fixture(scope="module")
def graph(report):
    return build_unified_graph_from_report(report)


class TestTypes:
    def test_enum_counts(self):
        assert len(EntityType) == 42
        assert len(RelationshipType) == 49
        assert len(GraphSemanticLayer) == 14
        assert len(NodeStatus) == 4

    def test_rel_mask(self):
        m = rel_mask(RelationshipType.USES, RelationshipType.CONTAINS)
        assert bin(m).count("1") == 2
        assert DEPENDENCY_REACH_MASK != 0


class TestContainer:
    def _tiny(self):
        g = UnifiedGraph()
        for nid, et in [("a", EntityType.AGENT), ("s", EntityType.SERVER),
                        ("p", EntityType.PACKAGE), ("v", EntityType.VULNERABILITY)]:
            g.add_node(UnifiedNode(id=nid, entity_type=et, label=nid))
        g.add_edge(UnifiedEdge("a", "s", RelationshipType.USES))
        g.add_edge(UnifiedEdge("s", "p", RelationshipType.DEPENDS_ON))
        g.add_edge(UnifiedEdge("p", "v", RelationshipType.VULNERABLE_TO))
        return g

    def test_add_node_merge_union(self):
        g = UnifiedGraph()
        g.add_node(UnifiedNode(id="x", entity_type=EntityType.AGENT, label="x",
                               properties={"a": 1}, tags=["t1"]))
        g.add_node(UnifiedNode(id="x", entity_type=EntityType.AGENT, label="x",
                               properties={"b": 2}, tags=["t2"],
                               status=NodeStatus.VULNERABLE))
        node = g.nodes["x"]
        assert node.properties == {"a": 1, "b": 2}
        assert node.tags == ["t1", "t2"]
        assert node.status == NodeStatus.VULNERABLE
        assert g.node_count == 1

    def test_add_edge_dedup(self):
        g = self._tiny()
        before = g.edge_count
        assert not g.add_edge(UnifiedEdge("a", "s", RelationshipType.USES))
        assert g.edge_count == before

    def test_bfs_and_shortest_path(self):
        g = self._tiny()
        dist = g.bfs("a")
        assert dist == {"a": 0, "s": 1, "p": 2, "v": 3}
        assert g.shortest_path("a", "v") == ["a", "s", "p", "v"]
        assert g.shortest_path("v", "a") is None  # directed

    def test_reverse_bfs(self):
        g = self._tiny()
        assert g.bfs("v", reverse=True) == {"v": 0, "p": 1, "s": 2, "a": 3}

    def test_bidirectional_edge(self):
        g = self._tiny()
        g.add_node(UnifiedNode(id="b", entity_type=EntityType.AGENT, label="b"))
        g.add_edge(UnifiedEdge("a", "b", RelationshipType.SHARES_SERVER, bidirectional=True))
        assert "a" in g.bfs("b")

    def test_node_budget_truncation(self):
        g = UnifiedGraph(node_budget=2)
        for i in range(5):
            g.add_node(UnifiedNode(id=f"n{i}", entity_type=EntityType.AGENT, label=f"n{i}"))
        assert g.node_count == 2
        assert g.truncated
        assert g.completeness()["dropped_nodes"] == 3

    def test_traverse_subgraph_bounded(self, graph):
        sub = graph.traverse_subgraph("agent:cursor", max_depth=2, max_nodes=5)
        assert len(sub["nodes"]) <= 5
        assert sub["truncated"] is True
        full = graph.traverse_subgraph("agent:cursor", max_depth=6, max_nodes=10_000)
        assert full["complete"]

    def test_impact_of_bounded(self, graph):
        imp = graph.impact_of("pkg:pypi:pyyaml@5.3")
        assert imp["total_impacted"] > 0
        assert max(imp["hops"].values()) <= 4
        assert "vulnerability" in imp["by_entity_type"]

    def test_search(self, graph):
        hits = graph.search(query="pyyaml")
        assert hits and all("pyyaml" in (n.label + n.id).lower() for n in hits)
        vulns = graph.search(entity_types=[EntityType.VULNERABILITY], limit=5)
        assert all(n.entity_type == EntityType.VULNERABILITY for n in vulns)

    def test_degree_centrality(self, graph):
        top = graph.degree_centrality(5)
        assert len(top) == 5
        degrees = [d for _n, d in top]
        assert degrees == sorted(degrees, reverse=True)

    def test_bottlenecks(self, graph):
        b = graph.bottlenecks(5)
        assert b and all(isinstance(score, float) for _n, score in b)

    def test_views(self, graph):
        inv = graph.view("inventory")
        assert inv["nodes"] and inv["edges"]
        lat = graph.view("lateral")
        assert all(e["relationship"] in ("shares_server", "shares_cred",
                                         "lateral_path", "delegated_to")
                   for e in lat["edges"])
        with pytest.raises(ValueError):
            graph.view("nope")

    def test_roundtrip_dict(self, graph):
        g2 = UnifiedGraph.from_dict(graph.to_dict())
        assert g2.node_count == graph.node_count
        assert g2.edge_count == graph.edge_count

    def test_exports(self, graph):
        assert "digraph" in graph.export("dot")
        assert graph.export("mermaid").startswith("graph LR")
        assert "<graphml" in graph.export("graphml")
        assert "MERGE" in graph.export("cypher")

    def test_to_csr_shape(self, graph):
        order, row_off, col, etype = graph.to_csr()
        assert len(order) == graph.node_count
        assert row_off[-1] == len(col) == len(etype)


class TestBuilder:
    def test_from_report(self, graph, report):
        kinds = {n.entity_type for n in graph.nodes.values()}
        assert {EntityType.AGENT, EntityType.SERVER, EntityType.PACKAGE,
                EntityType.TOOL, EntityType.CREDENTIAL,
                EntityType.VULNERABILITY} <= kinds
        rels = {e.relationship for e in graph.edges}
        assert {RelationshipType.USES, RelationshipType.DEPENDS_ON,
                RelationshipType.VULNERABLE_TO, RelationshipType.AFFECTS,
                RelationshipType.EXPLOITABLE_VIA, RelationshipType.EXPOSES_CRED,
                RelationshipType.SHARES_CRED} <= rels

    def test_from_report_json(self, report):
        from agentbom_amd.output.json_fmt import to_json

        g = build_unified_graph_from_report_json(to_json(report))
        assert g.node_count > 50
        assert g.shortest_path("agent:cursor", "vuln:CVE-2020-14343")


class TestDependencyReach:
    def test_all_demo_packages_reachable(self, graph, report):
        reach = compute_dependency_reach(graph, use_gpu=False)
        pkg_nodes = [nid for nid, n in graph.nodes.items()
                     if n.entity_type == EntityType.PACKAGE]
        assert set(reach.package_reach) == set(pkg_nodes)
        # vulns joined via AFFECTS
        assert "vuln:CVE-2020-14343" in reach.vuln_reach
        v = reach.vuln_reach["vuln:CVE-2020-14343"]
        assert v["min_hops"] == 2
        assert "agent:cursor" in v["reachable_from"]

    def test_stamping_rescores(self, report, graph):
        r = run_demo_scan()
        g = build_unified_graph_from_report(r)
        base = {br.vulnerability.id: br.risk_score for br in r.blast_radii}
        apply_dependency_reachability_to_blast_radii(r, g)
        for br in r.blast_radii:
            assert br.dependency_reachable is True
            assert br.dependency_min_hop_distance == 2
            assert br.dependency_reachable_from_agents


class TestAttackPaths:
    def test_paths_found_and_scored(self, graph):
        paths = compute_fused_attack_paths(graph, max_paths=20)
        assert paths
        scores = [p.score for p in paths]
        assert scores == sorted(scores, reverse=True)
        p = paths[0]
        assert p.entry.startswith("agent:")
        assert p.nodes[0] == p.entry
        assert p.nodes[-1] == p.target
        assert len(p.relationships) == len(p.nodes) - 1

    def test_techniques_mapped(self, graph):
        paths = compute_fused_attack_paths(graph, max_paths=50)
        assert any(p.techniques for p in paths)

    def test_deterministic(self, graph):
        a = compute_fused_attack_paths(graph, max_paths=10)
        b = compute_fused_attack_paths(graph, max_paths=10)
        assert [p.id for p in a] == [p.id for p in b]

    def test_rank_tuple(self, graph):
        paths = compute_fused_attack_paths(graph, max_paths=5)
        ranked = sorted(paths, key=path_rank_tuple)
        assert ranked[0].score >= ranked[-1].score


class TestRollup:
    def test_rollup_view(self, graph):
        rv = rollup_view(graph)
        assert rv["containers"]
        top = rv["containers"][0]
        assert top["worst_severity"] == "critical"
        assert sum(top["severity_histogram"].values()) > 0

    def test_toxic_flag(self, graph):
        rv = rollup_view(graph)
        shell = next(c for c in rv["containers"] if c["id"] == "server:shell-runner-server")
        assert shell["toxic"]  # critical finding + exposed credentials

    def test_drill_down(self, graph):
        dd = drill_down(graph, "agent:cursor")
        ids = [c["id"] for c in dd["children"]]
        assert "server:shell-runner-server" in ids
        assert drill_down(graph, "nope")["error"] == "not_found"
