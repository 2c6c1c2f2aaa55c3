"""Graph analytics backend (Protocol) + store-backed UnifiedGraph
(SURVEY §2.4 'Graph backend protocol' + 'Store-backed graph' rows)."""

import math

import pytest

from agentbom_amd.graph.backend import (
    GraphBackendProtocol,
    NativeBackend,
    get_backend,
)
from agentbom_amd.graph.container import UnifiedEdge, UnifiedGraph, UnifiedNode
from agentbom_amd.graph.store_backed import (
    AUTO_SPILL_THRESHOLD,
    StoreBackedUnifiedGraph,
    maybe_store_backed,
)
from agentbom_amd.graph.types import EntityType, RelationshipType


def _node(i: str) -> UnifiedNode:
    return UnifiedNode(id=i, entity_type=EntityType.PACKAGE, label=i)


def _chain_graph(n: int, cls=UnifiedGraph, **kw) -> UnifiedGraph:
    g = cls(**kw)
    for i in range(n):
        g.add_node(_node(f"n{i:04d}"))
    for i in range(n - 1):
        g.add_edge(UnifiedEdge(f"n{i:04d}", f"n{i+1:04d}",
                               RelationshipType.DEPENDS_ON))
    return g


class TestNativeBackend:
    def test_protocol_conformance(self):
        assert isinstance(NativeBackend(), GraphBackendProtocol)

    def test_pagerank_star(self):
        """Hub receiving all edges outranks the spokes; ranks sum to ~1."""
        g = UnifiedGraph()
        g.add_node(_node("hub"))
        for i in range(5):
            g.add_node(_node(f"s{i}"))
            g.add_edge(UnifiedEdge(f"s{i}", "hub", RelationshipType.DEPENDS_ON))
        ranks = g.pagerank()
        assert math.isclose(sum(ranks.values()), 1.0, rel_tol=1e-6)
        assert ranks["hub"] > max(ranks[f"s{i}"] for i in range(5)) * 3

    def test_pagerank_empty_and_top_n(self):
        assert UnifiedGraph().pagerank() == {}
        g = _chain_graph(10)
        top = g.pagerank(top_n=3)
        assert len(top) == 3

    def test_betweenness_chain_center(self):
        """Middle of a path has the highest betweenness, endpoints zero."""
        g = _chain_graph(7)
        bc = g.betweenness(sample=100)
        assert bc["n0003"] == max(bc.values())
        assert bc["n0000"] == 0.0

    def test_betweenness_tiny_graph(self):
        g = _chain_graph(2)
        assert set(g.betweenness().values()) == {0.0}

    def test_get_backend_fallback(self, monkeypatch):
        # networkx absent in this image -> request falls back to native
        b = get_backend("networkx")
        assert b.name in ("native", "networkx")
        monkeypatch.setenv("AGENT_BOM_GRAPH_BACKEND", "native")
        assert get_backend().name == "native"

    def test_nontraversable_edges_ignored(self):
        g = UnifiedGraph()
        for i in ("a", "b"):
            g.add_node(_node(i))
        e = UnifiedEdge("a", "b", RelationshipType.DEPENDS_ON)
        e.traversable = False
        g.add_edge(e)
        ranks = g.pagerank()
        assert math.isclose(ranks["a"], ranks["b"], rel_tol=1e-9)


class TestStoreBacked:
    def test_same_interface_and_traversals(self, tmp_path):
        g = _chain_graph(50, cls=StoreBackedUnifiedGraph,
                         db_path=tmp_path / "g.sqlite", cache_capacity=16)
        # Mapping contract
        assert len(g.nodes) == 50
        assert "n0010" in g.nodes and "zz" not in g.nodes
        assert g.nodes["n0010"].label == "n0010"
        # BFS + impact work unchanged through the spilling map
        hops = g.bfs("n0000", max_depth=5)
        assert len(hops) >= 5
        g.close()

    def test_eviction_and_writeback(self, tmp_path):
        g = StoreBackedUnifiedGraph(db_path=tmp_path / "g.sqlite",
                                    cache_capacity=16)
        for i in range(100):
            g.add_node(_node(f"n{i:04d}"))
        # far beyond capacity: early nodes were evicted to SQLite
        assert len(g.nodes._cache) <= 16
        assert g.nodes["n0000"].label == "n0000"  # reloaded from disk
        g.close()

    def test_mutation_survives_eviction(self, tmp_path):
        g = StoreBackedUnifiedGraph(db_path=tmp_path / "g.sqlite",
                                    cache_capacity=8)
        for i in range(10):
            g.add_node(_node(f"n{i:04d}"))
        g.nodes["n0001"].tags.append("stamped")
        for i in range(10, 40):  # force eviction of n0001
            g.add_node(_node(f"n{i:04d}"))
        assert "stamped" in g.nodes["n0001"].tags
        g.close()

    def test_merge_semantics_preserved(self, tmp_path):
        g = StoreBackedUnifiedGraph(db_path=tmp_path / "g.sqlite")
        g.add_node(_node("x"))
        dup = _node("x")
        dup.tags = ["extra"]
        assert g.add_node(dup)  # merge, not replace
        assert "extra" in g.nodes["x"].tags
        assert len(g.nodes) == 1
        g.close()

    def test_to_memory_roundtrip(self, tmp_path):
        g = _chain_graph(30, cls=StoreBackedUnifiedGraph,
                         db_path=tmp_path / "g.sqlite", cache_capacity=8)
        m = g.to_memory()
        assert type(m) is UnifiedGraph
        assert len(m.nodes) == 30 and len(m.edges) == 29
        g.close()

    def test_auto_wire_threshold(self):
        small = maybe_store_backed(100)
        big = maybe_store_backed(AUTO_SPILL_THRESHOLD)
        assert type(small) is UnifiedGraph
        assert isinstance(big, StoreBackedUnifiedGraph)
        big.close()

    def test_builder_small_reports_stay_in_ram(self):
        from agentbom_amd.graph.builder import build_unified_graph_from_report
        from agentbom_amd.scan.orchestrator import run_demo_scan

        g = build_unified_graph_from_report(run_demo_scan())
        assert type(g) is UnifiedGraph  # demo estate is tiny

    def test_analytics_work_on_store_backed(self, tmp_path):
        g = _chain_graph(20, cls=StoreBackedUnifiedGraph,
                         db_path=tmp_path / "g.sqlite", cache_capacity=4)
        ranks = g.pagerank()
        assert len(ranks) == 20
        g.close()
