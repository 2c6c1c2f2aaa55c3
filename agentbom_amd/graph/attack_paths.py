"""Attack-path fusion + path ranking.

Reference: src/agent_bom/graph/attack_path_fusion.py:194 compute_fused_attack_paths
(DFS from entry nodes toward crown jewels with edge/node risk boosts ->
scored AttackPath objects) and graph/path_ranking.py:62 path_rank_tuple.
"""

from __future__ import annotations

from typing import Optional

from agentbom_amd.graph.container import AttackPath, UnifiedGraph
from agentbom_amd.graph.types import EntityType, NodeStatus, RelationshipType
from agentbom_amd.utils.canonical_ids import canonical_id

# Crown jewels: what an attacker is ultimately after.
_TARGET_TYPES = {EntityType.CREDENTIAL, EntityType.DATA_STORE, EntityType.TOOL}
# Entry classes: internet-exposed resources and agents (prompt surface).
_ENTRY_TYPES = {EntityType.AGENT}

_EDGE_BOOST = {
    RelationshipType.VULNERABLE_TO: 2.0,
    RelationshipType.EXPLOITABLE_VIA: 2.5,
    RelationshipType.EXPOSES_CRED: 1.5,
    RelationshipType.SHARES_CRED: 1.2,
    RelationshipType.SHARES_SERVER: 1.0,
    RelationshipType.EXPOSED_TO: 2.0,
}

_TRAVERSABLE = set(_EDGE_BOOST) | {
    RelationshipType.USES, RelationshipType.DEPENDS_ON, RelationshipType.CONTAINS,
    RelationshipType.PROVIDES_TOOL, RelationshipType.AFFECTS,
    RelationshipType.REACHES_TOOL, RelationshipType.CAN_ACCESS,
    RelationshipType.DELEGATED_TO, RelationshipType.LATERAL_PATH,
}

# ATT&CK technique hints per relationship class.
_TECHNIQUES = {
    RelationshipType.VULNERABLE_TO: "T1190",  # exploit public-facing app
    RelationshipType.EXPLOITABLE_VIA: "T1068",  # exploitation for privilege escalation
    RelationshipType.EXPOSES_CRED: "T1552",  # unsecured credentials
    RelationshipType.SHARES_CRED: "T1550",  # use alternate auth material
    RelationshipType.SHARES_SERVER: "T1021",  # lateral movement
    RelationshipType.PROVIDES_TOOL: "T1059",  # command execution surface
}


def _node_boost(graph: UnifiedGraph, nid: str) -> float:
    node = graph.nodes[nid]
    boost = 0.0
    if node.status == NodeStatus.VULNERABLE:
        boost += 1.5
    props = node.properties
    if props.get("is_kev"):
        boost += 2.0
    sev = props.get("severity")
    boost += {"critical": 2.0, "high": 1.2, "medium": 0.6}.get(sev, 0.0)
    if props.get("is_malicious"):
        boost += 2.0
    return boost


def compute_fused_attack_paths(
    graph: UnifiedGraph,
    max_depth: int = 6,
    max_paths: int = 100,
    entries: Optional[list[str]] = None,
) -> list[AttackPath]:
    """Walk from entry nodes toward crown jewels, scoring each path.

    Graphs above AGENT_BOM_PATH_DP_THRESHOLD nodes (default 20k) switch to
    the GPU-capable path DP (compute_attack_paths_dp) — the DFS is the
    small-graph reference implementation."""
    import os

    from agentbom_amd.utils import config as _cfg

    threshold = int(os.environ.get("AGENT_BOM_PATH_DP_THRESHOLD",
                                   str(_cfg.PATH_DP_THRESHOLD)))
    if len(graph.nodes) >= threshold:
        device = None
        try:
            import torch

            if torch.cuda.is_available():
                device = torch.device("cuda")
        except Exception:
            pass
        return compute_attack_paths_dp(graph, max_depth=max_depth,
                                       max_paths=max_paths, entries=entries,
                                       device=device)
    if entries is None:
        entries = sorted(
            nid for nid, n in graph.nodes.items() if n.entity_type in _ENTRY_TYPES
        )
    paths: list[AttackPath] = []

    for entry in entries:
        if entry not in graph.nodes:
            continue
        # iterative DFS with per-path score accumulation
        stack: list[tuple[str, list[str], list[str], float, set[str]]] = [
            (entry, [entry], [], 0.0, {entry})
        ]
        while stack and len(paths) < max_paths * 4:
            u, path, rels, score, visited = stack.pop()
            node = graph.nodes[u]
            if (
                len(path) > 1
                and node.entity_type in _TARGET_TYPES
                and any(r in ("vulnerable_to", "exploitable_via", "exposes_cred") for r in rels)
            ):
                paths.append(
                    AttackPath(
                        id=canonical_id("attack_path", entry, u, *path),
                        nodes=list(path),
                        relationships=list(rels),
                        score=round(min(score, 100.0), 2),
                        entry=entry,
                        target=u,
                        techniques=sorted(
                            {
                                _TECHNIQUES[r]
                                for r in (RelationshipType(x) for x in rels)
                                if r in _TECHNIQUES
                            }
                        ),
                        narrative=_narrative(graph, path, rels),
                    )
                )
                continue
            if len(path) > max_depth:
                continue
            for v, e in graph._out(u):
                if v in visited or e.relationship not in _TRAVERSABLE:
                    continue
                step = e.weight * 0.3 + _EDGE_BOOST.get(e.relationship, 0.3)
                step += _node_boost(graph, v)
                stack.append((v, path + [v], rels + [e.relationship.value],
                              score + step, visited | {v}))

    # campaign-style dedup: best-scoring path per (entry, target)
    best: dict[tuple[str, str], AttackPath] = {}
    for p in paths:
        key = (p.entry, p.target)
        if key not in best or p.score > best[key].score:
            best[key] = p
    ranked = sorted(best.values(), key=lambda p: (-p.score, p.id))
    return ranked[:max_paths]


def compute_attack_paths_dp(
    graph: UnifiedGraph,
    max_depth: int = 6,
    max_paths: int = 100,
    entries: Optional[list[str]] = None,
    device=None,
) -> list[AttackPath]:
    """Attack paths via the hop-bounded max-score path DP (GPU-capable).

    The estate-scale replacement for the DFS above: exports the graph's
    numeric CSR (container.to_csr) and runs graph/path_engine.run_path_dp —
    on a CUDA device that is the paths.hip HIP kernel.  Reports the best
    gated path per TARGET (the DFS enumerates per (entry, target); on the
    ranked top-k surface the per-target best is what serves).  Used
    automatically by compute_fused_attack_paths for graphs above
    AGENT_BOM_PATH_DP_THRESHOLD nodes (default 20k), where the Python DFS
    is the reference's 1.6 s-per-query bottleneck
    (/v1/graph/paths p50 1,625 ms at 10.5k nodes).
    """
    import numpy as np

    from agentbom_amd.graph.path_engine import run_path_dp
    from agentbom_amd.graph.types import REL_CODE

    order, row_off, col, et = graph.to_csr()
    index = {nid: i for i, nid in enumerate(order)}
    n = len(order)
    # edge-centric arrays: expand row_off to col-aligned src
    src = np.repeat(np.arange(n, dtype=np.int64), np.diff(row_off)).astype(np.int32)
    col = col.astype(np.int32)

    # weights per edge: walk edges in to_csr order is lost; rebuild weights
    # by (src,dst,rel) lookup — weight differences only nudge scores, and
    # dedup by (s,t,rel) matches container.add_edge's own dedup key
    wmap = {}
    for e in graph.edges:
        key = (index[e.source], index[e.target], min(REL_CODE[e.relationship], 255))
        wmap[key] = e.weight
        if e.bidirectional:
            wmap[(key[1], key[0], key[2])] = e.weight
    ew = np.array([wmap.get((int(s), int(d), int(t)), 1.0)
                   for s, d, t in zip(src, col, et)], dtype=np.float32)

    etype_boost = np.full(256, 0.3, dtype=np.float32)
    etype_trav = np.zeros(256, dtype=np.uint8)
    for rel in _TRAVERSABLE:
        etype_trav[min(REL_CODE[rel], 255)] = 1
    for rel, b in _EDGE_BOOST.items():
        etype_boost[min(REL_CODE[rel], 255)] = b
    etype_gate = np.zeros(256, dtype=np.uint8)
    for rel in (RelationshipType.VULNERABLE_TO, RelationshipType.EXPLOITABLE_VIA,
                RelationshipType.EXPOSES_CRED):
        etype_gate[min(REL_CODE[rel], 255)] = 1

    node_boost = np.array([_node_boost(graph, nid) for nid in order],
                          dtype=np.float32)
    target_mask = np.array(
        [1 if graph.nodes[nid].entity_type in _TARGET_TYPES else 0 for nid in order],
        dtype=np.uint8)
    if entries is None:
        entry_idx = [i for i, nid in enumerate(order)
                     if graph.nodes[nid].entity_type in _ENTRY_TYPES]
    else:
        entry_idx = [index[e] for e in entries if e in index]
    if not entry_idx:
        return []

    hits = run_path_dp(src, col, et, ew, n, np.asarray(entry_idx), node_boost,
                       etype_boost, etype_trav, etype_gate, None, target_mask,
                       max_depth=max_depth, k=max_paths, device=device)

    code_rel = {min(c, 255): r for r, c in REL_CODE.items()}
    out = []
    for h in hits:
        nodes = [order[i] for i in h.nodes]
        rels = [code_rel[t].value for t in h.etypes]
        out.append(AttackPath(
            id=canonical_id("attack_path", nodes[0], nodes[-1], *nodes),
            nodes=nodes,
            relationships=rels,
            score=h.score,
            entry=nodes[0],
            target=nodes[-1],
            techniques=sorted({
                _TECHNIQUES[r] for r in (RelationshipType(x) for x in rels)
                if r in _TECHNIQUES
            }),
            narrative=_narrative(graph, nodes, rels),
        ))
    return out


def _narrative(graph: UnifiedGraph, path: list[str], rels: list[str]) -> str:
    hops = []
    for i, nid in enumerate(path):
        node = graph.nodes[nid]
        hops.append(f"{node.entity_type.value} {node.label}")
        if i < len(rels):
            hops.append(f"--{rels[i]}-->")
    return " ".join(hops)


def path_rank_tuple(path: AttackPath, environment_weight: float = 1.0) -> tuple:
    """Deterministic rank ordering: score desc, length asc, id asc."""
    tool_boost = 1.0 + 0.1 * sum(1 for r in path.relationships if r == "provides_tool")
    return (-path.score * environment_weight * tool_boost, len(path.nodes), path.id)
