"""Project discovered + issued non-human identities into the unified graph.

Discovered NHIs (``identity.nhi``) and agent-bom-issued identities
(``identity.lifecycle``) both become ``managed_identity`` nodes so the
effective-permissions / governance traversals treat them uniformly:

    agent → managed_identity → tool → vulnerable package

Reference parity: src/agent_bom/graph/nhi_overlay.py (keyed on
provider+identity id, idempotent re-runs, tool matching by label, unmatched
scopes recorded on node attributes).
"""

from __future__ import annotations

from collections import defaultdict
from typing import Iterable, Optional

from agentbom_amd.graph.container import UnifiedEdge, UnifiedGraph, UnifiedNode
from agentbom_amd.graph.types import EntityType, GraphSemanticLayer, RelationshipType
from agentbom_amd.identity.lifecycle import AgentIdentityStore
from agentbom_amd.identity.nhi import DiscoveredNonHumanIdentity

_OVERLAY_SOURCE = "nhi-discovery"


def _tool_label_index(graph: UnifiedGraph) -> dict[str, list[str]]:
    index: dict[str, list[str]] = defaultdict(list)
    for node in graph.nodes.values():
        if node.entity_type == EntityType.TOOL:
            index[node.label.strip().lower()].append(node.id)
    return index


def _agent_label_index(graph: UnifiedGraph) -> dict[str, str]:
    return {n.label.strip().lower(): n.id for n in graph.nodes.values()
            if n.entity_type == EntityType.AGENT}


def apply_nhi_overlay(graph: UnifiedGraph,
                      identities: Iterable[DiscoveredNonHumanIdentity]) -> dict:
    """Add one ``managed_identity`` node per discovered NHI. Idempotent."""
    tool_index = _tool_label_index(graph)
    added_nodes = added_edges = 0
    for nhi in identities:
        node_id = f"nhi:{nhi.provider}:{nhi.identity_id}"
        is_new = node_id not in graph.nodes
        graph.add_node(UnifiedNode(
            id=node_id,
            entity_type=EntityType.MANAGED_IDENTITY,
            label=nhi.name,
            layer=GraphSemanticLayer.IDENTITY,
            properties={
                "discovered": True,
                "provider": nhi.provider,
                "identity_type": nhi.identity_type,
                "identity_status": nhi.status,
                "credential_expires_at": nhi.credential_expires_at,
                "scopes": list(nhi.scopes),
                "overlay_source": _OVERLAY_SOURCE,
            },
            tags=["nhi", nhi.provider],
        ))
        if is_new:
            added_nodes += 1
        unmatched = []
        for scope in nhi.scopes:
            targets = tool_index.get(scope.strip().lower())
            if not targets:
                unmatched.append(scope)
                continue
            for tid in targets:
                if graph.add_edge(UnifiedEdge(
                        source=node_id, target=tid,
                        relationship=RelationshipType.SCOPED_TO,
                        weight=4.0, evidence=f"nhi scope {scope!r}")):
                    added_edges += 1
        if unmatched:
            graph.nodes[node_id].properties["unmatched_scopes"] = unmatched
    return {"nodes_added": added_nodes, "edges_added": added_edges}


def apply_issued_identity_overlay(graph: UnifiedGraph,
                                  store: AgentIdentityStore) -> dict:
    """Project agent-bom-issued identities; link each to its agent by label."""
    agent_index = _agent_label_index(graph)
    added_nodes = added_edges = 0
    for ident in store.list():
        node_id = f"nhi:agent-bom:{ident.identity_id}"
        if node_id not in graph.nodes:
            added_nodes += 1
        graph.add_node(UnifiedNode(
            id=node_id,
            entity_type=EntityType.MANAGED_IDENTITY,
            label=f"{ident.agent_name} identity ({ident.token_prefix})",
            layer=GraphSemanticLayer.IDENTITY,
            properties={
                "discovered": False,
                "provider": "agent-bom",
                "identity_type": "issued",
                "identity_status": "active" if ident.is_live() else "inactive",
                "scopes": store.active_scopes(ident.identity_id),
                "expires_at": ident.expires_at,
                "overlay_source": "identity-lifecycle",
            },
            tags=["nhi", "issued"],
        ))
        agent_id = agent_index.get(ident.agent_name.strip().lower())
        if agent_id and graph.add_edge(UnifiedEdge(
                source=agent_id, target=node_id,
                relationship=RelationshipType.AUTHENTICATES_AS,
                weight=5.0, evidence="issued identity")):
            added_edges += 1
    return {"nodes_added": added_nodes, "edges_added": added_edges}


def nhi_posture(graph: UnifiedGraph) -> dict:
    """Summarize identity posture from the overlaid graph."""
    nhis = [n for n in graph.nodes.values()
            if n.entity_type == EntityType.MANAGED_IDENTITY]
    by_provider: dict[str, int] = defaultdict(int)
    stale, unscoped = [], []
    for n in nhis:
        by_provider[str(n.properties.get("provider", "unknown"))] += 1
        if n.properties.get("credential_expires_at") is None \
                and n.properties.get("identity_type") in ("api_token", "service_principal"):
            stale.append(n.id)
        if not n.properties.get("scopes"):
            unscoped.append(n.id)
    return {
        "total_identities": len(nhis),
        "by_provider": dict(by_provider),
        "no_expiry_credentials": stale,
        "unscoped": unscoped,
    }
