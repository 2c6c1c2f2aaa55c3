"""Graph builder: scan report -> UnifiedGraph.

Reference: src/agent_bom/graph/builder.py:43 build_unified_graph_from_report —
provider -> agent -> server -> package -> tool -> credential nodes/edges;
vuln nodes with VULNERABLE_TO/AFFECTS/EXPLOITABLE_VIA; lateral
SHARES_SERVER / SHARES_CRED edges.
"""

from __future__ import annotations

from typing import Any

from agentbom_amd.graph.container import UnifiedEdge, UnifiedGraph, UnifiedNode
from agentbom_amd.graph.types import EntityType, GraphSemanticLayer, NodeStatus, RelationshipType
from agentbom_amd.models import AIBOMReport
from agentbom_amd.utils.canonical_ids import canonical_package_key


def _n(nid: str, et: EntityType, label: str, layer: GraphSemanticLayer,
       status: NodeStatus = NodeStatus.ACTIVE, **props) -> UnifiedNode:
    return UnifiedNode(id=nid, entity_type=et, label=label, layer=layer,
                       status=status, properties=props)


def build_unified_graph_from_report(report: AIBOMReport,
                                    node_budget: int = 250_000) -> UnifiedGraph:
    # estate-scale builds spill node payloads to a SQLite workspace
    # (reference auto-wire: store_backed.py for >= 5,000 entities)
    from agentbom_amd.graph.store_backed import maybe_store_backed

    expected = sum(
        1 + len(a.mcp_servers) + sum(len(s.packages) + len(s.tools)
                                     for s in a.mcp_servers)
        for a in report.agents) + len(report.blast_radii)
    g = maybe_store_backed(expected, node_budget=node_budget)
    provider_id = "provider:local"
    g.add_node(_n(provider_id, EntityType.PROVIDER, "local", GraphSemanticLayer.INFRA))

    # lateral bookkeeping
    server_agents: dict[str, list[str]] = {}
    cred_agents: dict[str, set[str]] = {}

    for agent in report.agents:
        a_id = f"agent:{agent.name}"
        g.add_node(_n(a_id, EntityType.AGENT, agent.name, GraphSemanticLayer.APP,
                      agent_type=agent.agent_type.value, canonical_id=agent.canonical_id))
        g.add_edge(UnifiedEdge(provider_id, a_id, RelationshipType.HOSTS))

        for server in agent.mcp_servers:
            s_id = f"server:{server.name}"
            g.add_node(_n(s_id, EntityType.SERVER, server.name, GraphSemanticLayer.MCP_SERVER,
                          transport=server.transport.value, surface=server.surface.value,
                          canonical_id=server.canonical_id))
            g.add_edge(UnifiedEdge(a_id, s_id, RelationshipType.USES))
            server_agents.setdefault(s_id, []).append(a_id)

            for cred in server.credential_names:
                c_id = f"credential:{cred}"
                g.add_node(_n(c_id, EntityType.CREDENTIAL, cred, GraphSemanticLayer.IDENTITY))
                g.add_edge(UnifiedEdge(s_id, c_id, RelationshipType.EXPOSES_CRED, weight=6.0))
                cred_agents.setdefault(c_id, set()).add(a_id)

            for tool in server.tools:
                t_id = f"tool:{server.name}/{tool.name}"
                g.add_node(_n(t_id, EntityType.TOOL, tool.name, GraphSemanticLayer.TOOL,
                              risk_score=tool.risk_score))
                g.add_edge(UnifiedEdge(s_id, t_id, RelationshipType.PROVIDES_TOOL))

            for pkg in server.packages:
                p_id = f"pkg:{canonical_package_key(pkg.name, pkg.version, pkg.ecosystem, pkg.purl)}"
                status = NodeStatus.VULNERABLE if pkg.has_vulnerabilities else NodeStatus.ACTIVE
                g.add_node(_n(p_id, EntityType.PACKAGE, f"{pkg.name}@{pkg.version}",
                              GraphSemanticLayer.PACKAGE, status=status,
                              ecosystem=pkg.ecosystem, is_malicious=pkg.is_malicious,
                              canonical_id=pkg.canonical_id))
                g.add_edge(UnifiedEdge(s_id, p_id, RelationshipType.DEPENDS_ON))

                for v in pkg.vulnerabilities:
                    v_id = f"vuln:{v.id}"
                    g.add_node(_n(v_id, EntityType.VULNERABILITY, v.id,
                                  GraphSemanticLayer.FINDING, status=NodeStatus.ACTIVE,
                                  severity=v.severity.value, cvss_score=v.cvss_score,
                                  is_kev=v.is_kev, epss_score=v.epss_score))
                    g.add_edge(UnifiedEdge(p_id, v_id, RelationshipType.VULNERABLE_TO,
                                           weight=_sev_weight(v.severity.value)))
                    g.add_edge(UnifiedEdge(v_id, p_id, RelationshipType.AFFECTS))

    # exploitable_via edges: vuln -> creds/tools of its blast radius
    for br in report.blast_radii:
        v_id = f"vuln:{br.vulnerability.id}"
        if v_id not in g.nodes:
            continue
        for cred in br.exposed_credentials:
            c_id = f"credential:{cred}"
            if c_id in g.nodes:
                g.add_edge(UnifiedEdge(v_id, c_id, RelationshipType.EXPLOITABLE_VIA,
                                       weight=7.0, evidence="blast_radius"))
        for srv in br.affected_servers:
            for tool in br.exposed_tools:
                t_id = f"tool:{srv.name}/{tool.name}"
                if t_id in g.nodes:
                    g.add_edge(UnifiedEdge(v_id, t_id, RelationshipType.EXPLOITABLE_VIA,
                                           weight=6.0, evidence="blast_radius"))

    # lateral edges
    for s_id, agents in server_agents.items():
        if len(agents) > 1:
            for i, a in enumerate(agents):
                for b in agents[i + 1:]:
                    g.add_edge(UnifiedEdge(a, b, RelationshipType.SHARES_SERVER,
                                           bidirectional=True, evidence=s_id))
    for c_id, agents in cred_agents.items():
        ag = sorted(agents)
        if len(ag) > 1:
            for i, a in enumerate(ag):
                for b in ag[i + 1:]:
                    g.add_edge(UnifiedEdge(a, b, RelationshipType.SHARES_CRED,
                                           bidirectional=True, evidence=c_id))
    return g


def _sev_weight(sev: str) -> float:
    return {"critical": 9.0, "high": 7.0, "medium": 5.0, "low": 3.0}.get(sev, 1.0)


def build_unified_graph_from_report_json(data: dict[str, Any],
                                         node_budget: int = 250_000) -> UnifiedGraph:
    """Build from a persisted report JSON (the to_json shape)."""
    g = UnifiedGraph(node_budget=node_budget)
    provider_id = "provider:local"
    g.add_node(_n(provider_id, EntityType.PROVIDER, "local", GraphSemanticLayer.INFRA))

    def _rows(value, key):
        """Fail-soft row iteration: persisted/hand-edited documents may
        carry malformed entries — skip anything that is not a dict with
        the join key instead of crashing the whole build."""
        if not isinstance(value, list):
            return []
        return [r for r in value if isinstance(r, dict) and r.get(key)]

    for agent in _rows(data.get("agents"), "name"):
        a_id = f"agent:{agent['name']}"
        g.add_node(_n(a_id, EntityType.AGENT, agent["name"], GraphSemanticLayer.APP,
                      agent_type=agent.get("agent_type")))
        g.add_edge(UnifiedEdge(provider_id, a_id, RelationshipType.HOSTS))
        for server in _rows(agent.get("mcp_servers"), "name"):
            s_id = f"server:{server['name']}"
            g.add_node(_n(s_id, EntityType.SERVER, server["name"], GraphSemanticLayer.MCP_SERVER,
                          transport=server.get("transport")))
            g.add_edge(UnifiedEdge(a_id, s_id, RelationshipType.USES))
            creds = server.get("credential_env_vars")
            for cred in (creds if isinstance(creds, list) else []):
                if not isinstance(cred, str) or not cred:
                    continue
                c_id = f"credential:{cred}"
                g.add_node(_n(c_id, EntityType.CREDENTIAL, cred, GraphSemanticLayer.IDENTITY))
                g.add_edge(UnifiedEdge(s_id, c_id, RelationshipType.EXPOSES_CRED, weight=6.0))
            for tool in _rows(server.get("tools"), "name"):
                t_id = f"tool:{server['name']}/{tool['name']}"
                g.add_node(_n(t_id, EntityType.TOOL, tool["name"], GraphSemanticLayer.TOOL))
                g.add_edge(UnifiedEdge(s_id, t_id, RelationshipType.PROVIDES_TOOL))
            for pkg in _rows(server.get("packages"), "name"):
                p_id = f"pkg:{pkg.get('ecosystem','')}:{pkg['name']}@{pkg.get('version','')}"
                vulns = _rows(pkg.get("vulnerabilities"), "id")
                g.add_node(_n(p_id, EntityType.PACKAGE,
                              f"{pkg['name']}@{pkg.get('version','')}",
                              GraphSemanticLayer.PACKAGE,
                              status=NodeStatus.VULNERABLE if vulns else NodeStatus.ACTIVE,
                              ecosystem=pkg.get("ecosystem")))
                g.add_edge(UnifiedEdge(s_id, p_id, RelationshipType.DEPENDS_ON))
                for v in vulns:
                    v_id = f"vuln:{v['id']}"
                    g.add_node(_n(v_id, EntityType.VULNERABILITY, v["id"],
                                  GraphSemanticLayer.FINDING,
                                  severity=v.get("severity"), is_kev=v.get("is_kev", False)))
                    g.add_edge(UnifiedEdge(p_id, v_id, RelationshipType.VULNERABLE_TO,
                                           weight=_sev_weight(v.get("severity", ""))))
                    g.add_edge(UnifiedEdge(v_id, p_id, RelationshipType.AFFECTS))
    apply_report_overlays(g, data)
    return g


def apply_report_overlays(graph, data: dict[str, Any]) -> dict[str, dict]:
    """Phase-B overlays over a built graph, driven by the report JSON blocks
    (reference: builder.py Phase-B overlay appliers, :1188-1253).

    Order matters: repo structure places the file/dir nodes the code-graph
    and CI overlays stitch onto; CNAPP sets exposure flags the ASPM
    reachability verdicts read.  Every overlay no-ops on absent input."""
    from agentbom_amd.graph.overlays_code import (
        apply_aspm_overlay,
        apply_ci_graph_overlay,
        apply_cnapp_overlay,
        apply_code_graph_overlay,
        apply_repo_structure_overlay,
    )

    return {
        "repo_structure": apply_repo_structure_overlay(graph, data),
        "code_graph": apply_code_graph_overlay(graph, data),
        "ci_graph": apply_ci_graph_overlay(graph, data),
        "cnapp": apply_cnapp_overlay(graph),
        "aspm": apply_aspm_overlay(graph, data),
    }
