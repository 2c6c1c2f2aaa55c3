"""CycloneDX 1.6 SBOM export.

Reference: src/agent_bom/output/cyclonedx_fmt.py — components for packages +
MCP servers/agents as services, vulnerabilities block with ratings/advisories.
"""

from __future__ import annotations

import uuid
from typing import Any

from agentbom_amd import __version__
from agentbom_amd.models import AIBOMReport

_SEV_CDX = {"critical": "critical", "high": "high", "medium": "medium",
            "low": "low", "none": "none", "unknown": "unknown"}


def _purl(pkg) -> str:
    if pkg.purl:
        return pkg.purl
    eco_map = {"pypi": "pypi", "npm": "npm", "go": "golang", "cargo": "cargo",
               "maven": "maven", "nuget": "nuget", "rubygems": "gem",
               "composer": "composer", "deb": "deb", "rpm": "rpm", "apk": "apk"}
    ptype = eco_map.get(pkg.ecosystem.lower(), "generic")
    return f"pkg:{ptype}/{pkg.name}@{pkg.version}"


def to_cyclonedx(report: AIBOMReport) -> dict[str, Any]:
    components: dict[str, dict] = {}
    services: list[dict] = []
    vulnerabilities: dict[str, dict] = {}

    for agent in report.agents:
        for server in agent.mcp_servers:
            if server.is_mcp_surface:
                services.append(
                    {
                        "bom-ref": server.stable_id,
                        "name": server.name,
                        "description": f"MCP server ({server.transport.value})",
                        "properties": [
                            {"name": "agent-bom:agent", "value": agent.name},
                            {"name": "agent-bom:transport", "value": server.transport.value},
                            {"name": "agent-bom:has_credentials", "value": str(server.has_credentials).lower()},
                        ],
                    }
                )
            for pkg in server.packages:
                ref = pkg.stable_id
                components.setdefault(
                    ref,
                    {
                        "bom-ref": ref,
                        "type": "library",
                        "name": pkg.name,
                        "version": pkg.version,
                        "purl": _purl(pkg),
                        **({"licenses": [{"license": {"id": pkg.license}}]} if pkg.license else {}),
                        "properties": [
                            {"name": "agent-bom:ecosystem", "value": pkg.ecosystem},
                            {"name": "agent-bom:is_direct", "value": str(pkg.is_direct).lower()},
                            *([{"name": "agent-bom:is_malicious", "value": "true"}] if pkg.is_malicious else []),
                        ],
                    },
                )
                for v in pkg.vulnerabilities:
                    entry = vulnerabilities.setdefault(
                        v.id,
                        {
                            "id": v.id,
                            "source": {"name": (v.all_advisory_sources[0] if v.all_advisory_sources else "osv")},
                            "description": v.summary,
                            "ratings": [
                                {
                                    "severity": _SEV_CDX.get(v.severity.value, "unknown"),
                                    **({"score": v.cvss_score, "method": "CVSSv3"} if v.cvss_score else {}),
                                }
                            ],
                            **({"cwes": [int(c.split("-")[1]) for c in v.cwe_ids if c.split("-")[-1].isdigit()]} if v.cwe_ids else {}),
                            "affects": [],
                            "properties": [
                                *([{"name": "agent-bom:is_kev", "value": "true"}] if v.is_kev else []),
                                *([{"name": "agent-bom:epss_score", "value": str(v.epss_score)}] if v.epss_score is not None else []),
                            ],
                        },
                    )
                    if not any(a["ref"] == ref for a in entry["affects"]):
                        entry["affects"].append({"ref": ref})

    return {
        "bomFormat": "CycloneDX",
        "specVersion": "1.6",
        "serialNumber": f"urn:uuid:{uuid.uuid5(uuid.NAMESPACE_URL, report.scan_id or 'agent-bom')}",
        "version": 1,
        "metadata": {
            "timestamp": report.generated_at.isoformat(),
            "tools": [{"vendor": "agent-bom", "name": "agent-bom", "version": __version__}],
        },
        "components": sorted(components.values(), key=lambda c: c["bom-ref"]),
        "services": services,
        "vulnerabilities": sorted(vulnerabilities.values(), key=lambda v: v["id"]),
    }
