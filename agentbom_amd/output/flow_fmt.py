"""React-Flow graph exports: per-CVE attack flow + agent-mesh topology.

Reference parity: src/agent_bom/output/attack_flow.py (columnar
CVE → package → server → agent/credential/tool layout) and
src/agent_bom/output/agent_mesh.py (multi-agent topology with
vulnerability-count coloring and shared-server merging).  Both emit
``{"nodes": [...], "edges": [...], "stats"/"filters": ...}`` consumable
by any @xyflow/react UI — the reference's Next.js UI reads exactly this
shape over REST, so a user pointing that UI at this engine gets the same
visuals.
"""

from __future__ import annotations

from typing import Any, Optional

from agentbom_amd.models import AIBOMReport

# column X positions, left-to-right kill-chain layout
_X_CVE, _X_PACKAGE, _X_SERVER, _X_RIGHT = 0, 350, 700, 1050
_Y_SPACING = 120


def _vuln_color(count: int) -> str:
    if count == 0:
        return "#10b981"  # green
    if count <= 3:
        return "#eab308"  # yellow
    if count <= 8:
        return "#f97316"  # orange
    return "#ef4444"  # red


def _severity_color(severity: str) -> str:
    return {"critical": "#ef4444", "high": "#f97316", "medium": "#eab308",
            "low": "#3b82f6"}.get(severity, "#6b7280")


def _node(node_id: str, label: str, x: int, y: int, kind: str,
          **data: Any) -> dict:
    return {"id": node_id, "type": "default",
            "position": {"x": x, "y": y},
            "data": {"label": label, "kind": kind, **data}}


def _edge(src: str, dst: str, label: str = "", animated: bool = False) -> dict:
    return {"id": f"e:{src}->{dst}", "source": src, "target": dst,
            "label": label, "animated": animated}


def build_attack_flow(report: AIBOMReport, cve: Optional[str] = None,
                      min_severity: Optional[str] = None,
                      agent: Optional[str] = None,
                      max_findings: int = 50) -> dict:
    """CVE → package → server → {agent, credential, tool} chains."""
    sev_order = {"critical": 4, "high": 3, "medium": 2, "low": 1}
    floor = sev_order.get(min_severity or "", 0)

    nodes: dict[str, dict] = {}
    edges: dict[str, dict] = {}
    y = 0
    rendered = 0
    for br in report.blast_radii:
        sev = br.vulnerability.severity.value
        if cve and br.vulnerability.id != cve:
            continue
        if floor and sev_order.get(sev, 0) < floor:
            continue
        if agent and agent not in {a.name for a in br.affected_agents}:
            continue
        if rendered >= max_findings:
            break
        rendered += 1

        cve_id = f"cve:{br.vulnerability.id}"
        pkg_id = f"pkg:{br.package.ecosystem}:{br.package.name}@{br.package.version}"
        nodes.setdefault(cve_id, _node(
            cve_id, br.vulnerability.id, _X_CVE, y, "cve",
            severity=sev, color=_severity_color(sev),
            risk_score=br.risk_score, is_kev=br.vulnerability.is_kev))
        nodes.setdefault(pkg_id, _node(
            pkg_id, f"{br.package.name}@{br.package.version}", _X_PACKAGE, y,
            "package", ecosystem=br.package.ecosystem,
            malicious=br.package.is_malicious))
        edges.setdefault(f"e:{cve_id}->{pkg_id}", _edge(
            cve_id, pkg_id, label="affects", animated=br.vulnerability.is_kev))

        for si, server in enumerate(br.affected_servers):
            srv_id = f"srv:{server.name}"
            nodes.setdefault(srv_id, _node(
                srv_id, server.name, _X_SERVER, y + si * _Y_SPACING, "server"))
            edges.setdefault(f"e:{pkg_id}->{srv_id}",
                             _edge(pkg_id, srv_id, label="runs in"))
            right_y = y + si * _Y_SPACING
            for ag in br.affected_agents:
                ag_id = f"agent:{ag.name}"
                nodes.setdefault(ag_id, _node(
                    ag_id, ag.name, _X_RIGHT, right_y, "agent"))
                edges.setdefault(f"e:{srv_id}->{ag_id}",
                                 _edge(srv_id, ag_id, label="serves"))
                right_y += _Y_SPACING
            for cred in br.exposed_credentials[:8]:
                cr_id = f"cred:{cred}"
                nodes.setdefault(cr_id, _node(
                    cr_id, cred, _X_RIGHT, right_y, "credential",
                    color="#ef4444"))
                edges.setdefault(f"e:{srv_id}->{cr_id}",
                                 _edge(srv_id, cr_id, label="exposes",
                                       animated=True))
                right_y += _Y_SPACING
            for tool in br.exposed_tools[:8]:
                t_id = f"tool:{tool.name}"
                nodes.setdefault(t_id, _node(
                    t_id, tool.name, _X_RIGHT, right_y, "tool"))
                edges.setdefault(f"e:{srv_id}->{t_id}",
                                 _edge(srv_id, t_id, label="reaches"))
                right_y += _Y_SPACING
        y += max(_Y_SPACING, (len(br.affected_servers) or 1) * _Y_SPACING)

    return {
        "nodes": list(nodes.values()),
        "edges": list(edges.values()),
        "filters": {"cve": cve, "min_severity": min_severity, "agent": agent},
        "stats": {"findings_rendered": rendered,
                  "truncated": rendered >= max_findings},
    }


def build_agent_mesh(report: AIBOMReport) -> dict:
    """Multi-agent topology: agents ↔ shared servers with vuln coloring."""
    pkg_vulns: dict[str, int] = {}
    for br in report.blast_radii:
        key = f"{br.package.name}@{br.package.version}"
        pkg_vulns[key] = pkg_vulns.get(key, 0) + 1

    nodes: list[dict] = []
    edges: list[dict] = []
    seen_servers: dict[str, str] = {}
    total_packages = 0
    ax = 0
    for agent in report.agents:
        ag_id = f"agent:{agent.name}"
        nodes.append(_node(ag_id, agent.name, ax, 0, "agent",
                           agent_type=agent.agent_type.value,
                           servers=len(agent.mcp_servers)))
        sy = _Y_SPACING
        for server in agent.mcp_servers:
            srv_id = seen_servers.get(server.name)
            shared = srv_id is not None
            if srv_id is None:
                srv_id = f"srv:{server.name}"
                seen_servers[server.name] = srv_id
                vulns = sum(pkg_vulns.get(f"{p.name}@{p.version}", 0)
                            for p in server.packages)
                total_packages += len(server.packages)
                nodes.append(_node(
                    srv_id, server.name, ax + 200, sy, "server",
                    vuln_count=vulns, color=_vuln_color(vulns),
                    credentials=len(server.credential_names),
                    tools=len(server.tools)))
            edges.append(_edge(ag_id, srv_id,
                               label="shared" if shared else "",
                               animated=shared))
            sy += _Y_SPACING
        ax += 500

    shared_count = sum(1 for e in edges if e["label"] == "shared")
    return {
        "nodes": nodes,
        "edges": edges,
        "stats": {
            "total_agents": len(report.agents),
            "total_servers": len(seen_servers),
            "total_packages": total_packages,
            "shared_server_links": shared_count,
        },
    }
