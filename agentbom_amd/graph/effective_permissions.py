"""Effective-permissions engine: what can each agent ACTUALLY touch, and why.

Reference parity: src/agent_bom/graph/effective_permissions.py +
governance_overlay.py — the governance question is not "what tools exist"
but "through which grant does this agent reach this tool, and is the
grant used".  Two grant paths are computed per agent:

- **direct**    — agent → server (USES) → tool (PROVIDES_TOOL/REACHES_TOOL);
- **identity**  — agent → AUTHENTICATES_AS → managed_identity → SCOPED_TO →
  tool (issued identities and discovered NHIs alike, via the overlays).

Fused with the runtime-evidence overlay: when the graph carries observed
call counts, every granted-but-never-called tool is an **unused grant**
(least-privilege reduction candidate), and wildcard-scoped identities are
flagged.  Pure read over the graph; deterministic output.
"""

from __future__ import annotations

from typing import Any, Optional

from agentbom_amd.graph.container import UnifiedGraph
from agentbom_amd.graph.types import EntityType, RelationshipType

_DIRECT_RELS = {RelationshipType.USES, RelationshipType.PROVIDES_TOOL,
                RelationshipType.REACHES_TOOL, RelationshipType.CONTAINS}
_IDENTITY_REL = RelationshipType.AUTHENTICATES_AS
_SCOPE_REL = RelationshipType.SCOPED_TO


def compute_effective_permissions(graph: UnifiedGraph) -> dict[str, Any]:
    """Per-agent effective tool permissions with grant provenance."""
    agents = sorted((n for n in graph.nodes.values()
                     if n.entity_type == EntityType.AGENT),
                    key=lambda n: n.id)
    evidence_present = any(
        n.properties.get("runtime_confirmed")
        for n in graph.nodes.values() if n.entity_type == EntityType.TOOL)

    out_agents = []
    all_unused = 0
    wildcard_identities: set[str] = set()
    for agent in agents:
        grants: dict[str, dict[str, Any]] = {}

        # direct reach through servers
        for node_id, hops in graph.bfs(agent.id, max_depth=3,
                                       allowed=_DIRECT_RELS).items():
            node = graph.nodes[node_id]
            if node.entity_type != EntityType.TOOL:
                continue
            grants.setdefault(node_id, {
                "tool_id": node_id, "tool": node.label,
                "paths": [], "observed_calls":
                    int(node.properties.get("observed_calls", 0))})
            grants[node_id]["paths"].append({"kind": "direct", "hops": hops})

        # identity-scoped reach
        for edge in graph.edges:
            if edge.relationship is not _IDENTITY_REL or edge.source != agent.id:
                continue
            ident = graph.nodes.get(edge.target)
            if ident is None or ident.entity_type != EntityType.MANAGED_IDENTITY:
                continue
            scopes = [str(s) for s in ident.properties.get("scopes", [])]
            if "*" in scopes:
                wildcard_identities.add(ident.id)
            for e2 in graph.edges:
                if e2.relationship is not _SCOPE_REL or e2.source != ident.id:
                    continue
                tool = graph.nodes.get(e2.target)
                if tool is None or tool.entity_type != EntityType.TOOL:
                    continue
                grants.setdefault(e2.target, {
                    "tool_id": e2.target, "tool": tool.label,
                    "paths": [], "observed_calls":
                        int(tool.properties.get("observed_calls", 0))})
                grants[e2.target]["paths"].append(
                    {"kind": "identity", "identity": ident.id,
                     "identity_label": ident.label})

        rows = sorted(grants.values(), key=lambda g: g["tool_id"])
        unused = [g["tool"] for g in rows
                  if evidence_present and g["observed_calls"] == 0]
        all_unused += len(unused)
        out_agents.append({
            "agent_id": agent.id,
            "agent": agent.label,
            "effective_tools": len(rows),
            "grants": rows,
            "unused_grants": sorted(unused),
        })

    return {
        "schema_version": "1",
        "evidence_informed": evidence_present,
        "agents": out_agents,
        "summary": {
            "total_agents": len(out_agents),
            "total_unused_grants": all_unused if evidence_present else None,
            "wildcard_identities": sorted(wildcard_identities),
        },
    }


def least_privilege_recommendations(graph: UnifiedGraph,
                                    max_items: int = 50) -> list[dict[str, Any]]:
    """Actionable reductions: unused grants + wildcard identities, ranked."""
    perms = compute_effective_permissions(graph)
    recs = []
    for ident_id in perms["summary"]["wildcard_identities"]:
        recs.append({"kind": "narrow_wildcard_identity", "target": ident_id,
                     "detail": "identity carries '*' scope; pin to the scopes "
                               "actually exercised"})
    if perms["evidence_informed"]:
        for agent in perms["agents"]:
            for tool in agent["unused_grants"]:
                recs.append({
                    "kind": "revoke_unused_grant",
                    "target": f"{agent['agent']} -> {tool}",
                    "detail": "granted but never called in observed runtime "
                              "traffic"})
    return recs[:max_items]
