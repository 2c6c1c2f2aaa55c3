"""Finding views: grouped/compact projections of a scan report.

Reference parity: src/agent_bom/output/{finding_views,compact}.py — the
UI/SDK-facing groupings (by severity, package, agent, framework) and the
compact document (summary + top findings) used where the full ~95-key
blast_radius rows are too heavy.
"""

from __future__ import annotations

from collections import defaultdict
from typing import Any

from agentbom_amd.models import AIBOMReport

_SEV_ORDER = ("critical", "high", "medium", "low", "unknown", "none")


def _row(br) -> dict[str, Any]:
    return {
        "vulnerability_id": br.vulnerability.id,
        "package": f"{br.package.name}@{br.package.version}",
        "ecosystem": br.package.ecosystem,
        "severity": br.vulnerability.severity.value,
        "risk_score": round(float(br.risk_score), 2),
        "is_kev": bool(br.vulnerability.is_kev),
        "is_malicious": bool(br.package.is_malicious),
        "reachability": br.reachability,
        "fixed_version": br.vulnerability.fixed_version,
    }


def by_severity(report: AIBOMReport) -> dict[str, Any]:
    groups: dict[str, list] = defaultdict(list)
    for br in report.blast_radii:
        groups[br.vulnerability.severity.value].append(_row(br))
    return {"view": "by_severity",
            "groups": {sev: groups[sev] for sev in _SEV_ORDER if sev in groups}}


def by_package(report: AIBOMReport) -> dict[str, Any]:
    groups: dict[str, list] = defaultdict(list)
    for br in report.blast_radii:
        groups[f"{br.package.ecosystem}:{br.package.name}"].append(_row(br))
    ordered = sorted(groups.items(),
                     key=lambda kv: -max(r["risk_score"] for r in kv[1]))
    return {"view": "by_package",
            "groups": {k: v for k, v in ordered}}


def by_agent(report: AIBOMReport) -> dict[str, Any]:
    groups: dict[str, list] = defaultdict(list)
    for br in report.blast_radii:
        for agent in br.affected_agents or []:
            groups[agent.name].append(_row(br))
        if not br.affected_agents:
            groups["unattributed"].append(_row(br))
    return {"view": "by_agent", "groups": dict(sorted(groups.items()))}


def by_framework(report: AIBOMReport, framework_field: str = "owasp_tags") -> dict[str, Any]:
    groups: dict[str, list] = defaultdict(list)
    for br in report.blast_radii:
        for tag in getattr(br, framework_field, []) or []:
            groups[str(tag)].append(_row(br))
    return {"view": f"by_{framework_field}", "groups": dict(sorted(groups.items()))}


def to_compact(report: AIBOMReport, top: int = 10) -> dict[str, Any]:
    """Summary + top-N findings — the lightweight exchange document."""
    counts = report.severity_counts()
    return {
        "schema_version": "compact-1",
        "scan_id": report.scan_id,
        "summary": {
            "total_agents": report.total_agents,
            "total_mcp_servers": report.total_servers,
            "total_packages": report.total_packages,
            "total_findings": len(report.blast_radii),
            "severity_counts": counts,
            "kev_count": sum(1 for br in report.blast_radii
                             if br.vulnerability.is_kev),
            "malicious_count": sum(1 for br in report.blast_radii
                                   if br.package.is_malicious),
            "max_risk": max((br.risk_score for br in report.blast_radii),
                            default=0.0),
        },
        "top_findings": [_row(br) for br in report.blast_radii[:top]],
    }
