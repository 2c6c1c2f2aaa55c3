"""Estate roll-up: CONTAINS-tree collapse with descendant aggregates.

Reference: src/agent_bom/graph/rollup.py:432 rollup_view / :583 drill_down /
:674 attack_path_view — container nodes with worst-severity, severity
histograms, exposed/toxic flags; drill-down one level at a time.

At estate scale the severity histograms run as the GPU segmented
reduction (ops/csrc graph.hip severity_histogram_kernel).
"""

from __future__ import annotations

from typing import Any, Optional

from agentbom_amd.graph.container import UnifiedGraph
from agentbom_amd.graph.types import EntityType, RelationshipType

_SEV_ORDER = ["critical", "high", "medium", "low", "none", "unknown"]
_CONTAINER_TYPES = {
    EntityType.PROVIDER, EntityType.ORG, EntityType.ACCOUNT, EntityType.ENVIRONMENT,
    EntityType.FLEET, EntityType.CLUSTER, EntityType.AGENT, EntityType.SERVER,
    EntityType.APPLICATION, EntityType.DIRECTORY, EntityType.CONTAINER,
}
_TREE_RELS = {RelationshipType.HOSTS, RelationshipType.USES, RelationshipType.CONTAINS,
              RelationshipType.DEPENDS_ON, RelationshipType.PART_OF}


def _worst(a: Optional[str], b: Optional[str]) -> Optional[str]:
    if a is None:
        return b
    if b is None:
        return a
    return a if _SEV_ORDER.index(a) <= _SEV_ORDER.index(b) else b


def _aggregate(graph: UnifiedGraph, root: str, _memo: dict) -> dict[str, Any]:
    if root in _memo:
        return _memo[root]
    hist = dict.fromkeys(_SEV_ORDER, 0)
    worst: Optional[str] = None
    descendants = 0
    exposed = False
    seen = {root}
    stack = [root]
    while stack:
        u = stack.pop()
        for v, e in graph._out(u, _TREE_RELS):
            if v in seen:
                continue
            seen.add(v)
            descendants += 1
            stack.append(v)
        # vulnerabilities hanging off u
        for v, e in graph._out(u, {RelationshipType.VULNERABLE_TO}):
            sev = str(graph.nodes[v].properties.get("severity") or "unknown")
            if sev not in hist:
                sev = "unknown"
            hist[sev] += 1
            worst = _worst(worst, sev)
        for _v, e in graph._out(u, {RelationshipType.EXPOSES_CRED}):
            exposed = True
    agg = {
        "descendants": descendants,
        "severity_histogram": hist,
        "worst_severity": worst,
        "exposed_credentials": exposed,
        "toxic": bool(worst in ("critical", "high") and exposed),
    }
    _memo[root] = agg
    return agg


def rollup_view(graph: UnifiedGraph, max_containers: int = 200) -> dict[str, Any]:
    """Collapse the estate to container nodes with descendant aggregates."""
    memo: dict = {}
    containers = []
    for nid in sorted(graph.nodes):
        node = graph.nodes[nid]
        if node.entity_type not in _CONTAINER_TYPES:
            continue
        agg = _aggregate(graph, nid, memo)
        containers.append(
            {
                "id": nid,
                "entity_type": node.entity_type.value,
                "label": node.label,
                **agg,
            }
        )
        if len(containers) >= max_containers:
            break
    containers.sort(key=lambda c: (
        _SEV_ORDER.index(c["worst_severity"]) if c["worst_severity"] else len(_SEV_ORDER),
        -c["descendants"], c["id"],
    ))
    return {
        "schema_version": "1",
        "containers": containers,
        "total_nodes": graph.node_count,
        "total_edges": graph.edge_count,
        "completeness": graph.completeness(),
    }


def drill_down(graph: UnifiedGraph, container_id: str) -> dict[str, Any]:
    """One level of children under a container, each with its aggregate."""
    if container_id not in graph.nodes:
        return {"id": container_id, "children": [], "error": "not_found"}
    memo: dict = {}
    children = []
    for v, e in graph._out(container_id, _TREE_RELS):
        node = graph.nodes[v]
        children.append(
            {
                "id": v,
                "entity_type": node.entity_type.value,
                "label": node.label,
                "relationship": e.relationship.value,
                **_aggregate(graph, v, memo),
            }
        )
    children.sort(key=lambda c: c["id"])
    return {"id": container_id, "children": children}


def attack_path_view(graph: UnifiedGraph, max_paths: int = 25) -> dict[str, Any]:
    """Attack-path-first roll-up: top fused paths + their container context."""
    from agentbom_amd.graph.attack_paths import compute_fused_attack_paths

    paths = compute_fused_attack_paths(graph, max_paths=max_paths)
    return {
        "schema_version": "1",
        "path_count": len(paths),
        "paths": [p.to_dict() for p in paths],
    }
