"""Graph fusion overlays: cost (FinOps), runtime evidence, endpoint.

Reference parity: src/agent_bom/graph/{cost_overlay,evidence_overlay,
endpoint_overlay}.py.  Shared contract for every overlay here:

- **pure in-place mutation** — no network, no stores, caller supplies data;
- **idempotent** — applying twice yields identical attributes;
- **deterministic** — all iteration sorted;
- **no-op on empty input** — the graph stays byte-identical.
"""

from __future__ import annotations

from collections import defaultdict
from typing import Any, Iterable, Optional

from agentbom_amd.graph.container import UnifiedEdge, UnifiedGraph, UnifiedNode
from agentbom_amd.graph.types import (
    EntityType,
    GraphSemanticLayer,
    NodeStatus,
    RelationshipType,
)

_TREE_RELS = {RelationshipType.HOSTS, RelationshipType.USES,
              RelationshipType.CONTAINS, RelationshipType.RUNS}


# ── cost (FinOps) ───────────────────────────────────────────────────────────


def apply_cost_overlay(graph: UnifiedGraph,
                       cost_records: Iterable[dict[str, Any]]) -> dict[str, Any]:
    """Attach per-node spend, roll it up the CONTAINS tree, fuse cost×risk.

    ``cost_records``: [{"target": <node label or id>, "usd": float}, ...] —
    typically produced from OTel GenAI spans by the cost tools.
    """
    records = sorted((dict(r) for r in cost_records),
                     key=lambda r: str(r.get("target", "")))
    if not records:
        return {"matched": 0, "unmatched": [], "fused": 0}

    label_index = {n.label.strip().lower(): n.id for n in graph.nodes.values()}
    spend: dict[str, float] = defaultdict(float)
    unmatched = []
    for rec in records:
        target = str(rec.get("target", ""))
        usd = float(rec.get("usd", 0.0))
        node_id = target if target in graph.nodes \
            else label_index.get(target.strip().lower())
        if node_id is None:
            unmatched.append(target)
            continue
        spend[node_id] += usd
    for node_id, usd in sorted(spend.items()):
        graph.nodes[node_id].properties["cost_usd"] = round(usd, 4)

    # roll up along tree edges: parent -> child accumulates child subtree
    children: dict[str, list[str]] = defaultdict(list)
    for e in graph.edges:
        if e.relationship in _TREE_RELS:
            children[e.source].append(e.target)

    memo: dict[str, float] = {}

    def subtree(nid: str, stack: frozenset) -> float:
        if nid in memo:
            return memo[nid]
        if nid in stack:  # cycle guard
            return 0.0
        total = spend.get(nid, 0.0)
        for child in sorted(set(children.get(nid, []))):
            total += subtree(child, stack | {nid})
        memo[nid] = total
        return total

    for nid in sorted(graph.nodes):
        total = subtree(nid, frozenset())
        if total > 0:
            graph.nodes[nid].properties["subtree_cost_usd"] = round(total, 4)

    # fuse: expensive AND risky nodes get a priority signal
    fused = 0
    costs = sorted((spend.get(n, 0.0) for n in spend), reverse=True)
    threshold = costs[max(len(costs) // 4 - 1, 0)] if costs else 0.0
    for nid in sorted(spend):
        node = graph.nodes[nid]
        risky = (node.status == NodeStatus.VULNERABLE
                 or node.properties.get("internet_exposed")
                 or node.properties.get("toxic_combination"))
        if risky and spend[nid] >= threshold and spend[nid] > 0:
            node.properties["cost_risk_priority"] = "high"
            fused += 1
    return {"matched": len(spend), "unmatched": sorted(set(unmatched)),
            "fused": fused}


# ── runtime evidence ────────────────────────────────────────────────────────


def apply_evidence_overlay(graph: UnifiedGraph,
                           audit_rows: Iterable[dict[str, Any]]) -> dict[str, Any]:
    """Stamp observed runtime traffic from a proxy audit log onto tool nodes.

    A vulnerable tool that was ACTUALLY called is confirmed attack surface;
    tools invoked at runtime with no graph node are phantom tools.
    """
    calls: dict[str, int] = defaultdict(int)
    blocked: dict[str, int] = defaultdict(int)
    last_seen: dict[str, float] = {}
    for row in audit_rows:
        if row.get("method") != "tools/call" or not row.get("tool"):
            continue
        tool = str(row["tool"])
        calls[tool] += 1
        if row.get("action") == "block":
            blocked[tool] += 1
        ts = row.get("ts")
        if isinstance(ts, (int, float)):
            last_seen[tool] = max(last_seen.get(tool, 0.0), float(ts))
    if not calls:
        return {"stamped": 0, "phantom_tools": []}

    tool_index: dict[str, list[str]] = defaultdict(list)
    for node in graph.nodes.values():
        if node.entity_type == EntityType.TOOL:
            tool_index[node.label.strip().lower()].append(node.id)

    stamped = 0
    phantom = []
    for tool in sorted(calls):
        node_ids = tool_index.get(tool.strip().lower())
        if not node_ids:
            phantom.append({"tool": tool, "calls": calls[tool]})
            continue
        for nid in sorted(node_ids):
            props = graph.nodes[nid].properties
            props["observed_calls"] = calls[tool]
            props["observed_blocked"] = blocked.get(tool, 0)
            props["runtime_confirmed"] = True
            if tool in last_seen:
                props["last_called_ts"] = last_seen[tool]
            stamped += 1
    return {"stamped": stamped, "phantom_tools": phantom}


# ── endpoint ────────────────────────────────────────────────────────────────


def apply_endpoint_overlay(graph: UnifiedGraph, inventory: dict[str, Any],
                           host_name: str = "localhost") -> dict[str, Any]:
    """Project a workstation endpoint inventory as a HOST node RUNNING the
    agents whose processes appear in it."""
    processes = inventory.get("processes") or []
    ports = inventory.get("listening_ports") or []
    if not processes and not ports:
        return {"nodes_added": 0, "edges_added": 0}

    host_id = f"endpoint:{host_name}"
    added_nodes = 0
    if host_id not in graph.nodes:
        added_nodes += 1
    graph.add_node(UnifiedNode(
        id=host_id, entity_type=EntityType.RESOURCE, label=host_name,
        layer=GraphSemanticLayer.INFRA,
        properties={
            "process_count": len(processes),
            "listening_ports": sorted({int(p.get("port", 0)) for p in ports
                                       if p.get("port")}),
            "overlay_source": "endpoint-inventory",
        },
        tags=["endpoint"]))

    agent_index = {n.label.strip().lower(): n.id for n in graph.nodes.values()
                   if n.entity_type == EntityType.AGENT}
    added_edges = 0
    proc_names = sorted({str(p.get("name", "")).strip().lower()
                         for p in processes})
    for pname in proc_names:
        for label, agent_id in sorted(agent_index.items()):
            if pname and (pname in label or label.split()[0] in pname):
                if graph.add_edge(UnifiedEdge(
                        source=host_id, target=agent_id,
                        relationship=RelationshipType.RUNS,
                        weight=3.0, evidence=f"process {pname!r}")):
                    added_edges += 1
                graph.nodes[agent_id].properties["endpoint_confirmed"] = True
    return {"nodes_added": added_nodes, "edges_added": added_edges}
