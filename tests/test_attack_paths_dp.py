"""Attack/exposure path DP: oracle semantics, DFS equivalence, engine paths.

The GPU kernel (ops/csrc/paths.hip) is bit-checked against
cpu_ref.path_relax in tests/test_ops_gpu.py; here the CPU oracle itself is
validated: hand-built graphs with known best paths, gate propagation,
reconstruction exactness, and best-per-target equivalence with a DFS
enumeration on acyclic graphs (where both semantics provably coincide).
"""

from __future__ import annotations

import numpy as np
import pytest

from agentbom_amd.graph.path_engine import PathHit, run_path_dp
from agentbom_amd.scan.synth import generate_estate


def _tables(boosts: dict, trav: list, gates: list = ()):
    eb = np.zeros(256, dtype=np.float32)
    for k, v in boosts.items():
        eb[k] = v
    tv = np.zeros(256, dtype=np.uint8)
    tv[trav] = 1
    eg = np.zeros(256, dtype=np.uint8)
    for g in gates:
        eg[g] = 1
    return eb, tv, eg


class TestHandBuiltGraphs:
    def test_linear_gated_path(self):
        # 0 -e0-> 1 -e1(gate)-> 2 -e2-> 3(target)
        src = np.array([0, 1, 2], dtype=np.int32)
        dst = np.array([1, 2, 3], dtype=np.int32)
        et = np.array([0, 1, 0], dtype=np.uint8)
        eb, tv, eg = _tables({0: 0.5, 1: 2.0}, [0, 1], gates=[1])
        nb = np.zeros(4, dtype=np.float32)
        tm = np.zeros(4, dtype=np.uint8)
        tm[3] = 1
        hits = run_path_dp(src, dst, et, None, 4, [0], nb, eb, tv, eg, None,
                           tm, max_depth=6, k=10)
        assert len(hits) == 1
        h = hits[0]
        assert h.nodes == [0, 1, 2, 3]
        assert h.etypes == [0, 1, 0]
        assert h.score == pytest.approx(0.5 + 2.0 + 0.5, abs=1e-4)

    def test_ungated_path_not_reported(self):
        # target reachable only via non-gate edges -> no gated label -> no hit
        src = np.array([0, 1], dtype=np.int32)
        dst = np.array([1, 2], dtype=np.int32)
        et = np.array([0, 0], dtype=np.uint8)
        eb, tv, eg = _tables({0: 1.0}, [0])
        tm = np.zeros(3, dtype=np.uint8)
        tm[2] = 1
        hits = run_path_dp(src, dst, et, None, 3, [0],
                           np.zeros(3, dtype=np.float32), eb, tv, eg, None,
                           tm, max_depth=6, k=10)
        assert hits == []

    def test_node_gate_triggers(self):
        # gate via touching a gated NODE (vulnerable server), no gate edges
        src = np.array([0, 1], dtype=np.int32)
        dst = np.array([1, 2], dtype=np.int32)
        et = np.array([0, 0], dtype=np.uint8)
        eb, tv, eg = _tables({0: 1.0}, [0])
        ng = np.array([0, 1, 0], dtype=np.uint8)
        tm = np.zeros(3, dtype=np.uint8)
        tm[2] = 1
        nb = np.array([0.0, 1.5, 0.0], dtype=np.float32)
        hits = run_path_dp(src, dst, et, None, 3, [0], nb, eb, tv, eg, ng,
                           tm, max_depth=6, k=10)
        assert len(hits) == 1
        assert hits[0].nodes == [0, 1, 2]
        assert hits[0].score == pytest.approx(1.0 + 1.5 + 1.0, abs=1e-4)

    def test_best_of_two_paths_wins(self):
        # entry 0 -> target 3 via 1 (cheap) or 2 (boosted): best must win and
        # reconstruction must walk the winning route
        src = np.array([0, 0, 1, 2], dtype=np.int32)
        dst = np.array([1, 2, 3, 3], dtype=np.int32)
        et = np.array([1, 1, 1, 1], dtype=np.uint8)
        eb, tv, eg = _tables({1: 1.0}, [1], gates=[1])
        nb = np.array([0, 0.2, 3.0, 0], dtype=np.float32)
        tm = np.zeros(4, dtype=np.uint8)
        tm[3] = 1
        hits = run_path_dp(src, dst, et, None, 4, [0], nb, eb, tv, eg, None,
                           tm, max_depth=6, k=10)
        assert hits[0].nodes == [0, 2, 3]
        assert hits[0].score == pytest.approx(1.0 + 3.0 + 1.0, abs=1e-4)

    def test_hop_bound_enforced(self):
        # chain longer than max_depth: target unreachable within the bound
        n = 8
        src = np.arange(n - 1, dtype=np.int32)
        dst = np.arange(1, n, dtype=np.int32)
        et = np.ones(n - 1, dtype=np.uint8)
        eb, tv, eg = _tables({1: 1.0}, [1], gates=[1])
        tm = np.zeros(n, dtype=np.uint8)
        tm[n - 1] = 1
        hits = run_path_dp(src, dst, et, None, n, [0],
                           np.zeros(n, dtype=np.float32), eb, tv, eg, None,
                           tm, max_depth=3, k=10)
        assert hits == []
        hits = run_path_dp(src, dst, et, None, n, [0],
                           np.zeros(n, dtype=np.float32), eb, tv, eg, None,
                           tm, max_depth=n - 1, k=10)
        assert len(hits) == 1 and len(hits[0].edges) == n - 1

    def test_cycle_rejected_not_crash(self):
        # positive-weight cycle 1<->2 on the way to target 3: the DP may
        # label a non-simple walk; reconstruction must reject it and fall
        # back to the simple path
        src = np.array([0, 1, 2, 2], dtype=np.int32)
        dst = np.array([1, 2, 1, 3], dtype=np.int32)
        et = np.array([1, 1, 1, 1], dtype=np.uint8)
        eb, tv, eg = _tables({1: 1.0}, [1], gates=[1])
        tm = np.zeros(4, dtype=np.uint8)
        tm[3] = 1
        hits = run_path_dp(src, dst, et, None, 4, [0],
                           np.zeros(4, dtype=np.float32), eb, tv, eg, None,
                           tm, max_depth=6, k=10)
        for h in hits:
            assert len(set(h.nodes)) == len(h.nodes)


def _dfs_best_per_target(src, dst, et, nb, eb, tv, eg, ng, entries, tm,
                         max_depth):
    """Reference DFS enumeration (per-entry, simple paths), best per target —
    the attack_paths.py walk semantics on numeric arrays (float64)."""
    from collections import defaultdict

    out_edges = defaultdict(list)
    for e in range(len(src)):
        if tv[et[e]]:
            out_edges[int(src[e])].append(e)
    best: dict[int, float] = {}
    for entry in entries:
        stack = [(int(entry), 0.0, False, frozenset([int(entry)]), 0)]
        while stack:
            u, score, gated, visited, depth = stack.pop()
            if gated and tm[u] and depth > 0:
                if u not in best or score > best[u]:
                    best[u] = score
                # targets are sinks in these graphs; continue anyway
            if depth >= max_depth:
                continue
            for e in out_edges[u]:
                v = int(dst[e])
                if v in visited:
                    continue
                step = float(eb[et[e]]) + float(nb[v])
                g2 = gated or bool(eg[et[e]]) or (ng is not None and bool(ng[v]))
                stack.append((v, score + step, g2, visited | {v}, depth + 1))
    return best


def test_dp_equals_dfs_on_acyclic_estate():
    """On a DAG (the estate forward edges are acyclic: agent -> server ->
    pkg/cred/tool) the DP best-gated-score per target must equal the DFS
    enumeration's best over all entries."""
    est = generate_estate(n_agents=12, n_servers=40, n_packages=400,
                          name_catalog=100, seed=77)
    src = est.edge_src.astype(np.int32)
    dst = est.edge_dst.astype(np.int32)
    et = est.edge_type
    eb, tv, eg = _tables({0: 0.3, 1: 0.3, 2: 1.5, 3: 0.3}, [0, 1, 2, 3])
    rng = np.random.default_rng(5)
    nb = (rng.random(est.num_nodes) * 2).astype(np.float32)
    ng = (rng.random(est.num_nodes) < 0.15).astype(np.uint8)
    tm = np.zeros(est.num_nodes, dtype=np.uint8)
    tm[est.cred_base:est.cred_base + est.n_creds] = 1
    tm[est.tool_base:est.tool_base + est.n_tools] = 1
    entries = np.arange(est.n_agents)

    hits = run_path_dp(src, dst, et, None, est.num_nodes, entries, nb, eb,
                       tv, eg, ng, tm, max_depth=6, k=10_000)
    dp_best = {h.target: h.score for h in hits}

    dfs_best = _dfs_best_per_target(src, dst, et, nb, eb, tv, eg, ng,
                                    entries, tm, 6)
    assert set(dp_best) == set(dfs_best)
    for t, s in dfs_best.items():
        assert dp_best[t] == pytest.approx(round(min(s, 100.0), 2), abs=0.01), t


def test_engine_attack_paths_cpu():
    from agentbom_amd.graph.gpu_engine import EstateEngine

    est = generate_estate(n_agents=40, n_servers=160, n_packages=3000,
                          name_catalog=500, seed=11)
    eng = EstateEngine(est, device="cpu")
    res = eng.step()
    hits = eng.attack_paths(step_res=res, k=25)
    assert hits, "seeded estate with findings must yield attack paths"
    assert len(hits) <= 25
    scores = [h.score for h in hits]
    assert scores == sorted(scores, reverse=True)
    for h in hits[:5]:
        # entry is an agent, target a cred or tool
        assert 0 <= h.entry < est.n_agents
        assert (est.cred_base <= h.target < est.cred_base + est.n_creds
                or est.tool_base <= h.target < est.tool_base + est.n_tools)
        assert len(set(h.nodes)) == len(h.nodes)
        assert isinstance(h, PathHit)


def test_dp_adapter_on_unified_graph():
    """compute_attack_paths_dp on the demo report graph: well-formed
    AttackPath objects, gate relationship present, score-descending, and
    per-target scores never below the DFS best (the DP maximizes over a
    superset of the DFS's per-entry simple paths)."""
    from agentbom_amd.graph.attack_paths import (
        compute_attack_paths_dp,
        compute_fused_attack_paths,
    )
    from agentbom_amd.graph.builder import build_unified_graph_from_report
    from agentbom_amd.scan.orchestrator import run_demo_scan

    graph = build_unified_graph_from_report(run_demo_scan())
    dp = compute_attack_paths_dp(graph, max_paths=50)
    assert dp, "demo graph has vulnerable packages -> gated paths must exist"
    scores = [p.score for p in dp]
    assert scores == sorted(scores, reverse=True)
    gate_rels = {"vulnerable_to", "exploitable_via", "exposes_cred"}
    for p in dp:
        assert p.nodes[0] == p.entry and p.nodes[-1] == p.target
        assert len(p.relationships) == len(p.nodes) - 1
        assert gate_rels & set(p.relationships), "ungated path reported"
        assert len(set(p.nodes)) == len(p.nodes)

    dfs = compute_fused_attack_paths(graph, max_paths=200)
    dfs_best: dict[str, float] = {}
    for p in dfs:
        dfs_best[p.target] = max(dfs_best.get(p.target, 0.0), p.score)
    dp_best = {p.target: p.score for p in dp}
    for tgt, s in dfs_best.items():
        if tgt in dp_best:
            assert dp_best[tgt] >= s - 0.02, (tgt, dp_best[tgt], s)


def test_dp_threshold_switch(monkeypatch):
    """compute_fused_attack_paths must route big graphs to the DP."""
    from agentbom_amd.graph import attack_paths as ap
    from agentbom_amd.graph.builder import build_unified_graph_from_report
    from agentbom_amd.scan.orchestrator import run_demo_scan

    graph = build_unified_graph_from_report(run_demo_scan())
    monkeypatch.setenv("AGENT_BOM_PATH_DP_THRESHOLD", "1")
    routed = ap.compute_fused_attack_paths(graph, max_paths=10)
    dp = ap.compute_attack_paths_dp(graph, max_paths=10)
    assert [p.id for p in routed] == [p.id for p in dp]
