"""SARIF 2.1.0 export for code-scanning ingestion.

Reference: src/agent_bom/output/sarif.py — one run, one rule per advisory,
one result per finding; reachability + KEV + EPSS in properties.
"""

from __future__ import annotations

from typing import Any

from agentbom_amd import __version__
from agentbom_amd.models import AIBOMReport

_SARIF_SCHEMA = "https://raw.githubusercontent.com/oasis-tcs/sarif-spec/master/Schemata/sarif-schema-2.1.0.json"

_LEVEL = {"critical": "error", "high": "error", "medium": "warning",
          "low": "note", "none": "note", "unknown": "warning"}


def to_sarif(report: AIBOMReport) -> dict[str, Any]:
    rules: dict[str, dict] = {}
    results: list[dict] = []

    for finding in report.to_findings():
        rule_id = finding.vulnerability_id or finding.title or finding.id
        if rule_id not in rules:
            help_text = finding.remediation_guidance or finding.description or rule_id
            rules[rule_id] = {
                "id": rule_id,
                "name": rule_id.replace("-", "_"),
                "shortDescription": {"text": (finding.title or rule_id)[:120]},
                "fullDescription": {"text": finding.description or rule_id},
                "help": {"text": help_text},
                "properties": {
                    "tags": ["security", finding.finding_type.value.lower(),
                             *(["cisa-kev"] if finding.is_kev else [])],
                    "cwe_ids": list(finding.cwe_ids),
                    "security-severity": str(finding.cvss_score or 0.0),
                },
            }
        location = finding.asset.location or finding.asset.name or "unknown"
        results.append(
            {
                "ruleId": rule_id,
                "level": _LEVEL.get(finding.severity, "warning"),
                "message": {
                    "text": f"{finding.title}: {finding.description}"[:1000]
                    or rule_id,
                },
                "locations": [
                    {
                        "physicalLocation": {
                            "artifactLocation": {"uri": str(location).replace(" ", "_")},
                        },
                        "logicalLocations": [
                            {"name": finding.asset.name, "kind": finding.asset.asset_type}
                        ],
                    }
                ],
                "partialFingerprints": {"findingId": finding.id},
                "properties": {
                    "risk_score": finding.risk_score,
                    "reachability": finding.reachability,
                    "is_kev": finding.is_kev,
                    "epss_score": finding.epss_score,
                    "cvss_score": finding.cvss_score,
                    "impact_category": finding.impact_category,
                    "is_malicious": finding.is_malicious,
                    "exposed_credentials": list(finding.exposed_credentials),
                    "exposed_tools": list(finding.exposed_tools),
                    "affected_agents": list(finding.affected_agents),
                    "fixed_version": finding.fixed_version,
                    "triage_priority": finding.evidence.get("triage_priority")
                    if isinstance(finding.evidence, dict) else None,
                },
            }
        )

    return {
        "$schema": _SARIF_SCHEMA,
        "version": "2.1.0",
        "runs": [
            {
                "tool": {
                    "driver": {
                        "name": "agent-bom",
                        "version": __version__,
                        "informationUri": "https://github.com/agent-bom/agent-bom-mi355x",
                        "rules": sorted(rules.values(), key=lambda r: r["id"]),
                    }
                },
                "results": results,
                "properties": {
                    "scan_id": report.scan_id,
                    "generated_at": report.generated_at.isoformat(),
                    "total_findings": len(results),
                },
            }
        ],
    }
