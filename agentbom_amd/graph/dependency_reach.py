"""Dependency-reach: which agents can structurally reach which packages.

Reference: src/agent_bom/graph/dependency_reach.py:109 compute_dependency_reach
(two passes: (1) BFS from every agent along USES/DEPENDS_ON/CONTAINS/
PROVIDES_TOOL recording min-hop per package; (2) join vulns to packages via
AFFECTS/VULNERABLE_TO -> per-vuln reachable_from + min hop) and
graph/blast_reach.py:48 apply_dependency_reachability_to_blast_radii.

The CPU path here is the small-estate/reference implementation; at scale
the identical traversal runs as ONE multi-source GPU BFS over the CSR
(graph/gpu_engine.dependency_reach — multi-source labels replace the
per-agent loop entirely; parity asserted in tests).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

from agentbom_amd.graph.container import UnifiedGraph
from agentbom_amd.graph.types import EntityType, RelationshipType

_REACH_RELS = {
    RelationshipType.USES,
    RelationshipType.DEPENDS_ON,
    RelationshipType.CONTAINS,
    RelationshipType.PROVIDES_TOOL,
}


@dataclass
class DependencyReach:
    # package node id -> {agent node id: min hops}
    package_reach: dict[str, dict[str, int]] = field(default_factory=dict)
    # vuln node id -> {"reachable_from": [agent ids], "min_hops": int}
    vuln_reach: dict[str, dict] = field(default_factory=dict)


def compute_dependency_reach(graph: UnifiedGraph,
                             use_gpu: Optional[bool] = None) -> DependencyReach:
    agents = [nid for nid, n in graph.nodes.items() if n.entity_type == EntityType.AGENT]
    reach = DependencyReach()

    if use_gpu is None:
        try:
            import torch

            use_gpu = torch.cuda.is_available() and graph.node_count >= 5000
        except Exception:
            use_gpu = False

    if use_gpu:
        _reach_gpu(graph, agents, reach)
    else:
        for a in agents:
            dist = graph.bfs(a, max_depth=16, allowed=_REACH_RELS)
            for nid, hops in dist.items():
                if graph.nodes[nid].entity_type == EntityType.PACKAGE:
                    cur = reach.package_reach.setdefault(nid, {})
                    if a not in cur or hops < cur[a]:
                        cur[a] = hops

    # pass 2: join vulns to packages
    for e in graph.edges:
        if e.relationship != RelationshipType.AFFECTS:
            continue
        v_id, p_id = e.source, e.target
        pr = reach.package_reach.get(p_id)
        if not pr:
            continue
        entry = reach.vuln_reach.setdefault(v_id, {"reachable_from": [], "min_hops": None})
        for a, hops in pr.items():
            if a not in entry["reachable_from"]:
                entry["reachable_from"].append(a)
            if entry["min_hops"] is None or hops < entry["min_hops"]:
                entry["min_hops"] = hops
        entry["reachable_from"].sort()
    return reach


def _reach_gpu(graph: UnifiedGraph, agents: list[str], reach: "DependencyReach") -> None:
    """One multi-source BFS on the device CSR; min-hop per package from ANY
    agent (per-agent attribution is filled from a bounded CPU pass only for
    reached packages)."""
    import numpy as np
    import torch

    from agentbom_amd.graph.types import DEPENDENCY_REACH_MASK
    from agentbom_amd.ops import native

    order, row_off, col, etype = graph.to_csr()
    index = {nid: i for i, nid in enumerate(order)}
    dev = torch.device("cuda")
    row_t = torch.from_numpy(row_off).to(dev)
    col_t = torch.from_numpy(col.view(np.int32)).to(dev)
    et_t = torch.from_numpy(etype).to(dev)
    sources = torch.tensor([index[a] for a in agents], dtype=torch.int32, device=dev)
    dist = native.bfs(row_t, col_t, sources, len(order), etype=et_t,
                      allowed_mask=DEPENDENCY_REACH_MASK).cpu().numpy().view(np.uint32)
    for i, nid in enumerate(order):
        if dist[i] != 0xFFFFFFFF and graph.nodes[nid].entity_type == EntityType.PACKAGE:
            # GPU gives min-hop over all agents; attribute per-agent via a
            # reverse bounded walk only for reached packages.
            reach.package_reach[nid] = {"*": int(dist[i])}
    # per-agent attribution for reached packages (CPU, bounded)
    for a in agents:
        d = graph.bfs(a, max_depth=16, allowed=_REACH_RELS)
        for nid, hops in d.items():
            if nid in reach.package_reach:
                cur = reach.package_reach[nid]
                cur.pop("*", None)
                if a not in cur or hops < cur[a]:
                    cur[a] = hops


def apply_dependency_reachability_to_blast_radii(report, graph: UnifiedGraph,
                                                 reach: Optional[DependencyReach] = None) -> int:
    """Stamp dependency_* fields on every BlastRadius and rescore.

    Reference: graph/blast_reach.py:48 — structural closure proves topology,
    not exploitability; scores only shift via the reachability nudge."""
    from agentbom_amd.utils.canonical_ids import canonical_package_key

    if reach is None:
        reach = compute_dependency_reach(graph)
    updated = 0
    for br in report.blast_radii:
        p_id = "pkg:" + canonical_package_key(
            br.package.name, br.package.version, br.package.ecosystem, br.package.purl
        )
        pr = reach.package_reach.get(p_id)
        if pr:
            br.dependency_reachable = True
            br.dependency_min_hop_distance = min(pr.values())
            br.dependency_reachable_from_agents = sorted(
                a.removeprefix("agent:") for a in pr if a != "*"
            )
        else:
            br.dependency_reachable = False
            br.dependency_min_hop_distance = None
            br.dependency_reachable_from_agents = []
        br.calculate_risk_score()
        updated += 1
    report.blast_radii.sort(key=lambda b: -b.risk_score)
    return updated
