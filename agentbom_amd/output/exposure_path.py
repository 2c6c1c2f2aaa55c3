"""Bounded ExposurePath projection shared by JSON, MCP and API surfaces.

Reference: src/agent_bom/output/exposure_path.py:29-125 — one report-safe
path object per finding: id/rank/label/riskScore/hops/relationships/
nodeIds/edgeIds/fix/proof.
"""

from __future__ import annotations

import re
from typing import Any

from agentbom_amd.models.finding import Finding

_SLUG_RE = re.compile(r"[^a-z0-9]+")


def _slug(text: str) -> str:
    return _SLUG_RE.sub("-", str(text).lower()).strip("-") or "unknown"


def _ordered_unique(items: list[str]) -> list[str]:
    seen: set[str] = set()
    out = []
    for x in items:
        if x and x not in seen:
            seen.add(x)
            out.append(x)
    return out


def exposure_path_for_finding(
    finding: Finding,
    *,
    rank: int | None = None,
    provenance_source: str = "finding_output",
) -> dict[str, Any]:
    ev = finding.evidence if isinstance(finding.evidence, dict) else {}
    pkg_name = str(ev.get("package_name") or finding.asset.name)
    pkg_version = str(ev.get("package_version") or "")
    ecosystem = str(ev.get("ecosystem") or "")
    package_ref = f"pkg:{ecosystem}:{pkg_name}@{pkg_version or 'unknown'}"
    vuln_id = finding.cve_id or finding.title or finding.asset.name
    finding_ref = f"finding:{vuln_id}"

    if finding.affected_agents:
        source_ref = f"agent:{finding.affected_agents[0]}"
    elif finding.asset.asset_type == "mcp_server":
        source_ref = f"server:{finding.asset.name}"
    else:
        source_ref = package_ref

    server_refs = [f"server:{s}" for s in finding.affected_servers]
    tool_refs = [f"tool:{t}" for t in finding.exposed_tools]
    cred_refs = [f"credential:{c}" for c in finding.exposed_credentials]

    nodes = _ordered_unique(
        [source_ref, *server_refs[:3], package_ref, finding_ref, *tool_refs[:3], *cred_refs[:3]]
    )
    relationships: list[dict[str, Any]] = []

    def rel(src: str, tgt: str, kind: str) -> None:
        relationships.append(
            {"id": f"{_slug(src)}--{kind}--{_slug(tgt)}", "source": src, "target": tgt, "type": kind}
        )

    for s in server_refs[:3]:
        rel(source_ref, s, "uses")
        rel(s, package_ref, "contains")
    if not server_refs:
        rel(source_ref, package_ref, "contains")
    rel(package_ref, finding_ref, "vulnerable_to")
    for t in tool_refs[:3]:
        rel(finding_ref, t, "exposes")
    for c in cred_refs[:3]:
        rel(finding_ref, c, "exposes")

    fix = (
        f"Upgrade {pkg_name} to {finding.fixed_version}"
        if finding.fixed_version
        else "No upstream fix recorded; monitor advisory source"
    )
    proof: list[str] = []
    if finding.affected_agents:
        proof.append(f"{len(finding.affected_agents)} affected agent(s)")
    if finding.affected_servers:
        proof.append(f"{len(finding.affected_servers)} affected server(s)")
    if finding.exposed_tools:
        proof.append(f"{len(finding.exposed_tools)} reachable tool(s)")
    if finding.exposed_credentials:
        proof.append(f"{len(finding.exposed_credentials)} exposed credential reference(s)")
    if finding.is_kev:
        proof.append("CISA KEV")
    if finding.epss_score is not None:
        proof.append(f"EPSS {finding.epss_score:.4f}")

    reachability = finding.reachability or "unknown"
    path: dict[str, Any] = {
        "id": "finding:" + ":".join(_slug(p) for p in (vuln_id, ecosystem, pkg_name, pkg_version or "unknown")),
        "rank": rank,
        "label": f"{pkg_name}@{pkg_version or '?'} -> {vuln_id}",
        "summary": finding.attack_vector_summary
        or finding.ai_risk_context
        or f"{vuln_id} affects {pkg_name}@{pkg_version or '?'} with {reachability} reachability.",
        "riskScore": round(float(finding.risk_score or 0.0), 2),
        "severity": finding.severity,
        "source": source_ref,
        "target": finding_ref,
        "hops": nodes,
        "relationships": relationships,
        "nodeIds": nodes,
        "edgeIds": [r["id"] for r in relationships],
        "findings": [vuln_id],
        "affectedAgents": list(finding.affected_agents[:10]),
        "affectedServers": list(finding.affected_servers[:10]),
        "reachableTools": list(finding.exposed_tools[:10]),
        "exposedCredentials": list(finding.exposed_credentials[:10]),
        "dependencyContext": {
            "package": pkg_name,
            "version": pkg_version,
            "ecosystem": ecosystem,
            "direct": ev.get("package_is_direct"),
            "dependencyDepth": ev.get("package_dependency_depth"),
            "reachabilityEvidence": ev.get("package_reachability_evidence"),
        },
        "fix": fix,
        "evidence": proof,
        "provenance": {"source": provenance_source, "graphPersistence": False},
    }
    return {k: v for k, v in path.items() if v is not None}
