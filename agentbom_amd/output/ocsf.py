"""OCSF (Open Cybersecurity Schema Framework) event export for SIEM ingest.

Reference: src/agent_bom/output/ocsf.py + graph/node.py:179
(UnifiedNode.to_ocsf_event) — Vulnerability Finding (class_uid 2002) per
CVE row, Detection Finding (2004) for runtime/combination findings,
inventory as Device/Software Inventory Info (5001/5020-class events).
"""

from __future__ import annotations

from typing import Any

from agentbom_amd import __version__
from agentbom_amd.models import AIBOMReport

_SEVERITY_ID = {"unknown": 0, "none": 1, "low": 2, "medium": 3, "high": 4, "critical": 5}


def _metadata(report: AIBOMReport) -> dict[str, Any]:
    return {
        "version": "1.1.0",
        "product": {"name": "agent-bom", "vendor_name": "agent-bom",
                    "version": __version__},
        "logged_time": report.generated_at.isoformat(),
    }


def to_ocsf_events(report: AIBOMReport) -> list[dict[str, Any]]:
    events: list[dict[str, Any]] = []
    meta = _metadata(report)

    for finding in report.to_findings():
        sev = finding.severity
        base = {
            "category_uid": 2,
            "category_name": "Findings",
            "severity": sev.capitalize(),
            "severity_id": _SEVERITY_ID.get(sev, 0),
            "time": report.generated_at.isoformat(),
            "metadata": meta,
            "status": "New",
            "status_id": 1,
        }
        if finding.finding_type.value == "CVE":
            ev = dict(finding.evidence) if isinstance(finding.evidence, dict) else {}
            events.append({
                **base,
                "class_uid": 2002,
                "class_name": "Vulnerability Finding",
                "activity_id": 1,
                "activity_name": "Create",
                "finding_info": {
                    "uid": finding.id,
                    "title": finding.title,
                    "desc": finding.description,
                    "types": ["Vulnerability"],
                },
                "vulnerabilities": [
                    {
                        "cve": {"uid": finding.cve_id or "",
                                **({"cvss": [{"base_score": finding.cvss_score,
                                              "vector_string": finding.cvss_vector,
                                              "version": "3.1"}]}
                                   if finding.cvss_score else {})},
                        "severity": sev.capitalize(),
                        "is_exploit_available": bool(finding.is_kev),
                        "affected_packages": [
                            {
                                "name": ev.get("package_name", ""),
                                "version": ev.get("package_version", ""),
                                "package_manager": ev.get("ecosystem", ""),
                                **({"fixed_in_version": finding.fixed_version}
                                   if finding.fixed_version else {}),
                            }
                        ],
                        "references": (ev.get("references") or [])[:5],
                    }
                ],
                "resources": [
                    {"uid": finding.asset.stable_id, "name": finding.asset.name,
                     "type": finding.asset.asset_type}
                ],
                "unmapped": {
                    "risk_score": finding.risk_score,
                    "reachability": finding.reachability,
                    "epss_score": finding.epss_score,
                    "exposed_credentials": list(finding.exposed_credentials),
                    "affected_agents": list(finding.affected_agents),
                    "impact_category": finding.impact_category,
                },
            })
        else:
            events.append({
                **base,
                "class_uid": 2004,
                "class_name": "Detection Finding",
                "activity_id": 1,
                "activity_name": "Create",
                "finding_info": {
                    "uid": finding.id,
                    "title": finding.title,
                    "desc": finding.description,
                    "types": [finding.finding_type.value],
                },
                "resources": [
                    {"uid": finding.asset.stable_id, "name": finding.asset.name,
                     "type": finding.asset.asset_type}
                ],
                "unmapped": {"risk_score": finding.risk_score},
            })

    # inventory: one Software Inventory Info event per agent
    for agent in report.agents:
        events.append({
            "category_uid": 5,
            "category_name": "Discovery",
            "class_uid": 5020,
            "class_name": "Software Inventory Info",
            "activity_id": 2,
            "activity_name": "Collect",
            "severity": "Informational",
            "severity_id": 1,
            "time": report.generated_at.isoformat(),
            "metadata": meta,
            "device": {"uid": agent.stable_id, "name": agent.name,
                       "type": "AI Agent"},
            "software": [
                {"name": s.name, "type": "MCP Server",
                 "uid": s.stable_id}
                for s in agent.mcp_servers
            ],
        })
    return events


def to_ocsf(report: AIBOMReport) -> dict[str, Any]:
    events = to_ocsf_events(report)
    return {"ocsf_schema_version": "1.1.0", "event_count": len(events), "events": events}
