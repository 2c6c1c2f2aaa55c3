"""The JSON report contract — schema_version 1.0.

Key-compatible rebuild of the reference's report serialization
(reference: src/agent_bom/output/json_fmt.py:997-1300 to_json,
:882-994 _blast_radius_json_entry).  Top-level keys: scan_id, scan_run,
warnings, scan_sources, ai_bom_entities, packages, summary, finding_summary,
assets, inventory_snapshot, agents[] (nested servers->tools/resources/
prompts/packages), blast_radius[], exposure_paths{}, findings[],
threat_framework_summary, scorecard_summary, remediation_plan.
"""

from __future__ import annotations

from collections import Counter
from typing import Any

from agentbom_amd import __version__
from agentbom_amd.models import (
    AIBOMReport,
    BlastRadius,
    FRAMEWORK_TAG_FIELDS,
    Finding,
    Severity,
    fused_triage_priority,
)
from agentbom_amd.models.finding import forward_fixed_version
from agentbom_amd.output.exposure_path import exposure_path_for_finding
from agentbom_amd.utils.canonical_ids import CANONICAL_ID_SCHEMA_VERSION

SCAN_REPORT_SCHEMA_VERSION = "1.0"
SCAN_RUN_SCHEMA_VERSION = "1"
BLAST_RADIUS_SCHEMA_VERSION = "1"

_FINDING_SEVERITIES = ("critical", "high", "medium", "low", "none", "unknown")


def _severity_state(severity: Severity) -> str:
    return "pending" if severity == Severity.UNKNOWN else "scored"


def _severity_label(severity: Severity) -> str:
    return "advisory" if severity == Severity.UNKNOWN else severity.value


def _tags_dict(br: BlastRadius) -> dict[str, list[str]]:
    return {fname: list(getattr(br, fname)) for fname, _slug in FRAMEWORK_TAG_FIELDS}


def _framework_qualified_tags(br: BlastRadius) -> list[str]:
    out = []
    for fname, slug in FRAMEWORK_TAG_FIELDS:
        out.extend(f"{slug}:{t}" for t in getattr(br, fname))
    return out


def _blast_radius_json_entry(
    br: BlastRadius, finding: Finding, rank: int, exposure_path: dict[str, Any]
) -> dict[str, Any]:
    asset = finding.asset
    v = br.vulnerability
    entry = {
        "schema_version": BLAST_RADIUS_SCHEMA_VERSION,
        "canonical_id": finding.canonical_id,
        "asset": asset.to_dict() | {"source_ids": {}},
        "exposure_path": exposure_path,
        "package_name": br.package.name,
        "package_version": br.package.version,
        "package_stable_id": br.package.stable_id,
        "package_canonical_id": br.package.canonical_id,
        **_tags_dict(br),
        "risk_score": br.risk_score,
        "reachability": br.reachability,
        "actionable": br.is_actionable,
        "vulnerability_id": finding.cve_id or v.id,
        "severity": v.severity.value,
        "severity_label": _severity_label(v.severity),
        "severity_state": _severity_state(v.severity),
        "advisory_sources": v.all_advisory_sources,
        "primary_advisory_source": v.all_advisory_sources[0] if v.all_advisory_sources else None,
        "advisory_coverage_state": v.advisory_coverage_state,
        "match_confidence_tier": v.match_confidence_tier,
        "cvss_score": v.cvss_score,
        "epss_score": v.epss_score,
        "is_kev": v.is_kev,
        "exploit_likelihood": v.exploit_likelihood,
        "published_at": v.published_at,
        "modified_at": v.modified_at,
        "nvd_status": v.nvd_status,
        "vex_status": v.vex_status,
        "vex_justification": v.vex_justification,
        "vex_suppressed": br.risk_score == 0.0 and v.vex_status in {"not_affected", "fixed"},
        "suppressed": br.suppressed,
        "suppression_id": br.suppression_id,
        "suppression_state": br.suppression_state,
        "suppression_reason": br.suppression_reason,
        "unsuppressed_risk_score": br.unsuppressed_risk_score,
        "compliance_tags": v.compliance_tags,
        "package": f"{br.package.name}@{br.package.version}",
        "ecosystem": br.package.ecosystem,
        "layer_attribution": [o.to_dict() for o in br.layer_attribution],
        "introduced_in_layer": (
            br.package.primary_occurrence.to_dict() if br.package.primary_occurrence else None
        ),
        "package_discovery_provenance": br.package.discovery_provenance,
        "package_version_provenance": {"source": br.package.version_source},
        "is_malicious": br.package.is_malicious,
        "malicious_reason": br.package.malicious_reason,
        "scorecard_score": br.package.scorecard_score,
        "scorecard_repo": br.package.scorecard_repo,
        "scorecard_lookup_state": br.package.scorecard_lookup_state,
        "affected_agents": [a.name for a in br.affected_agents],
        "affected_servers": [s.name for s in br.affected_servers],
        "exposed_credentials": list(br.exposed_credentials),
        "exposed_tools": [t.name for t in br.exposed_tools],
        "phantom_tools": [t.name for t in br.phantom_tools],
        "framework_tags": _framework_qualified_tags(br),
        "impact_category": br.impact_category,
        "cvss_vector": v.cvss_vector,
        "attack_vector": v.attack_vector,
        "attack_complexity": v.attack_complexity,
        "privileges_required": v.privileges_required,
        "user_interaction": v.user_interaction,
        "network_exploitable": v.network_exploitable,
        "triage_priority": fused_triage_priority(
            severity=v.severity.value,
            is_kev=bool(v.is_kev),
            epss_score=v.epss_score,
            network_exploitable=bool(v.network_exploitable),
            impact_category=br.impact_category,
            reachable=br.graph_reachable,
            exposed_credential_count=len(br.exposed_credentials),
            exposed_tool_count=len(br.exposed_tools),
        ),
        "all_server_credentials": list(br.all_server_credentials),
        "attack_vector_summary": br.attack_vector_summary,
        "fixed_version": forward_fixed_version(v.fixed_version, br.package.version, br.package.ecosystem),
        "vendor_severity": None,
        "cvss_severity": None,
        "ai_risk_context": br.ai_risk_context,
        "ai_summary": br.ai_summary,
        "hop_depth": br.hop_depth,
        "delegation_chain": list(br.delegation_chain),
        "transitive_agents": list(br.transitive_agents),
        "transitive_credentials": list(br.transitive_credentials),
        "transitive_risk_score": br.transitive_risk_score,
        "dependency_reachable": br.dependency_reachable,
        "dependency_min_hop_distance": br.dependency_min_hop_distance,
        "dependency_reachable_from_agents": list(br.dependency_reachable_from_agents),
        "graph_reachable": br.graph_reachable,
        "graph_min_hop_distance": br.graph_min_hop_distance,
        "graph_reachable_from_agents": list(br.graph_reachable_from_agents),
        "symbol_reachability": br.symbol_reachability,
        "reachable_affected_symbols": list(br.reachable_affected_symbols),
    }
    return entry


def _vuln_dict(v) -> dict[str, Any]:
    return {
        "id": v.id,
        "summary": v.summary,
        "severity": v.severity.value,
        "severity_label": _severity_label(v.severity),
        "severity_state": _severity_state(v.severity),
        "severity_source": v.severity_source,
        "advisory_sources": v.all_advisory_sources,
        "primary_advisory_source": v.all_advisory_sources[0] if v.all_advisory_sources else None,
        "advisory_coverage_state": v.advisory_coverage_state,
        "match_confidence_tier": v.match_confidence_tier,
        "confidence": v.confidence,
        "cvss_score": v.cvss_score,
        "epss_score": v.epss_score,
        "epss_percentile": v.epss_percentile,
        "is_kev": v.is_kev,
        "kev_date_added": v.kev_date_added,
        "kev_due_date": v.kev_due_date,
        "exploit_likelihood": v.exploit_likelihood,
        "published_at": v.published_at,
        "modified_at": v.modified_at,
        "aliases": v.aliases,
        "exploitability": v.exploitability,
        "cwe_ids": v.cwe_ids,
        "fixed_version": v.fixed_version,
        "references": v.references,
        "nvd_published": v.nvd_published,
        "nvd_modified": v.nvd_modified,
        "nvd_status": v.nvd_status,
        "vex_status": v.vex_status,
        "vex_justification": v.vex_justification,
        "compliance_tags": v.compliance_tags,
    }


def _package_dict(pkg) -> dict[str, Any]:
    return {
        "name": pkg.name,
        "stable_id": pkg.stable_id,
        "canonical_id": pkg.canonical_id,
        "version": pkg.version,
        "ecosystem": pkg.ecosystem,
        "purl": pkg.purl,
        "source_package": pkg.source_package,
        "distro_name": pkg.distro_name,
        "distro_version": pkg.distro_version,
        "occurrence_count": len(pkg.occurrences),
        "occurrences": [o.to_dict() for o in pkg.occurrences],
        "introduced_in_layer": pkg.primary_occurrence.to_dict() if pkg.primary_occurrence else None,
        "is_direct": pkg.is_direct,
        "parent_package": pkg.parent_package,
        "dependency_depth": pkg.dependency_depth,
        "dependency_scope": pkg.dependency_scope,
        "reachability_evidence": pkg.reachability_evidence,
        "resolved_from_registry": pkg.resolved_from_registry,
        "version_source": pkg.version_source,
        "declared_version": pkg.declared_version,
        "resolved_version": pkg.resolved_version,
        "version_confidence": pkg.version_confidence,
        "version_resolved_at": pkg.version_resolved_at,
        "version_evidence": pkg.version_evidence or None,
        "version_conflicts": pkg.version_conflicts or None,
        "floating_reference": pkg.floating_reference,
        "floating_reference_reason": pkg.floating_reference_reason,
        "is_malicious": pkg.is_malicious,
        "malicious_reason": pkg.malicious_reason,
        "registry_version": pkg.registry_version,
        "license": pkg.license,
        "license_expression": pkg.license_expression,
        "supplier": pkg.supplier,
        "author": pkg.author,
        "description": pkg.description,
        "homepage": pkg.homepage,
        "repository_url": pkg.repository_url,
        "download_url": pkg.download_url,
        "copyright_text": pkg.copyright_text,
        "deps_dev_resolved": pkg.deps_dev_resolved,
        "integrity_verified": pkg.integrity_verified,
        "provenance_attested": pkg.provenance_attested,
        "provenance_source": pkg.provenance_source,
        "provenance_status": pkg.provenance_status,
        "scorecard_score": pkg.scorecard_score,
        "scorecard_checks": pkg.scorecard_checks or None,
        "scorecard_repo": pkg.scorecard_repo,
        "scorecard_lookup_state": pkg.scorecard_lookup_state,
        "scorecard_lookup_reason": pkg.scorecard_lookup_reason,
        "vulnerability_count": len(pkg.vulnerabilities),
        "vulnerabilities": [_vuln_dict(v) for v in pkg.vulnerabilities],
    }


def _server_dict(server) -> dict[str, Any]:
    return {
        "name": server.name,
        "stable_id": server.stable_id,
        "canonical_id": server.canonical_id,
        "surface": server.surface.value,
        "fingerprint": server.fingerprint,
        "command": server.command,
        "args": list(server.args),
        "transport": server.transport.value,
        "url": server.url,
        "auth_mode": server.auth_mode,
        "mcp_version": server.mcp_version,
        "has_credentials": server.has_credentials,
        "credential_env_vars": server.credential_names,
        "identity_bindings": [],
        "registry_verified": server.registry_verified,
        "registry_badge": "verified" if server.registry_verified else "unknown",
        "security_blocked": server.security_blocked,
        "security_warnings": list(server.security_warnings),
        "security_intelligence": list(server.security_intelligence),
        "discovery_sources": list(server.discovery_sources),
        "discovery_provenance": server.discovery_provenance,
        "tools": [
            {
                "name": t.name,
                "stable_id": t.stable_id,
                "canonical_id": t.canonical_id,
                "fingerprint": t.fingerprint,
                "description": t.description,
                "discovery_source": t.discovery_source,
                "discovery_confidence": t.discovery_confidence,
                "schema_findings": t.schema_findings,
                "schema_rule_findings": t.schema_rule_findings,
                "risk_score": t.risk_score,
            }
            for t in server.tools
        ],
        "resources": [
            {
                "uri": r.uri,
                "stable_id": r.stable_id,
                "canonical_id": r.canonical_id,
                "fingerprint": r.fingerprint,
                "name": r.name,
                "description": r.description,
                "mime_type": r.mime_type,
                "content_findings": r.content_findings,
                "risk_score": r.risk_score,
            }
            for r in server.resources
        ],
        "prompts": [
            {
                "name": p.name,
                "stable_id": p.stable_id,
                "canonical_id": p.canonical_id,
                "fingerprint": p.fingerprint,
                "description": p.description,
                "arguments": p.arguments,
                "content_findings": p.content_findings,
                "risk_score": p.risk_score,
            }
            for p in server.prompts
        ],
        "packages": [_package_dict(pkg) for pkg in server.packages],
        "permission_profile": (
            {
                "runs_as_root": server.permission_profile.runs_as_root,
                "container_privileged": server.permission_profile.container_privileged,
                "privilege_level": server.permission_profile.privilege_level,
                "tool_permissions": server.permission_profile.tool_permissions,
                "capabilities": server.permission_profile.capabilities,
                "network_access": server.permission_profile.network_access,
                "filesystem_write": server.permission_profile.filesystem_write,
                "shell_access": server.permission_profile.shell_access,
            }
            if server.permission_profile
            else None
        ),
    }


def _build_finding_summary(findings: list[dict[str, Any]]) -> dict[str, Any]:
    by_severity = dict.fromkeys(_FINDING_SEVERITIES, 0)
    by_type: dict[str, int] = {}
    by_source: dict[str, int] = {}
    for f in findings:
        sev = str(f.get("severity") or "unknown").lower()
        if sev not in by_severity:
            sev = "unknown"
        by_severity[sev] += 1
        ftype = str(f.get("finding_type") or "UNKNOWN")
        src = str(f.get("source") or "UNKNOWN")
        by_type[ftype] = by_type.get(ftype, 0) + 1
        by_source[src] = by_source.get(src, 0) + 1
    return {
        "total": len(findings),
        "by_severity": by_severity,
        "by_type": dict(sorted(by_type.items())),
        "by_source": dict(sorted(by_source.items())),
    }


def _build_asset_inventory(findings: list[dict[str, Any]]) -> list[dict[str, Any]]:
    assets: dict[str, dict[str, Any]] = {}
    for f in findings:
        raw = f.get("asset")
        if not isinstance(raw, dict):
            continue
        sid = str(raw.get("stable_id") or raw.get("canonical_id") or "").strip()
        if not sid:
            continue
        asset = assets.setdefault(
            sid,
            {
                "schema_version": "1",
                "stable_id": sid,
                "canonical_id": str(raw.get("canonical_id") or sid),
                "name": raw.get("name"),
                "asset_type": raw.get("asset_type"),
                "identifier": raw.get("identifier"),
                "location": raw.get("location"),
                "provider": raw.get("provider"),
                "finding_ids": [],
            },
        )
        fid = f.get("id")
        if fid and fid not in asset["finding_ids"]:
            asset["finding_ids"].append(fid)
    return sorted(assets.values(), key=lambda a: a["stable_id"])


def _build_ai_bom_entities(report: AIBOMReport) -> dict[str, Any]:
    agents: list[dict] = []
    servers: dict[str, dict] = {}
    tools: dict[str, dict] = {}
    packages: dict[str, dict] = {}
    relationships: list[dict] = []
    for agent in report.agents:
        server_ids = []
        for server in agent.mcp_servers:
            sid = server.stable_id
            server_ids.append(sid)
            servers.setdefault(
                sid,
                {
                    "id": sid,
                    "canonical_id": server.canonical_id,
                    "name": server.name,
                    "transport": server.transport.value,
                    "surface": server.surface.value,
                    "has_credentials": server.has_credentials,
                },
            )
            relationships.append({"source": agent.stable_id, "target": sid, "type": "uses"})
            for t in server.tools:
                tools.setdefault(
                    t.stable_id,
                    {"id": t.stable_id, "name": t.name, "server_id": sid, "risk_score": t.risk_score},
                )
                relationships.append({"source": sid, "target": t.stable_id, "type": "provides_tool"})
            for pkg in server.packages:
                packages.setdefault(
                    pkg.stable_id,
                    {
                        "id": pkg.stable_id,
                        "canonical_id": pkg.canonical_id,
                        "name": pkg.name,
                        "version": pkg.version,
                        "ecosystem": pkg.ecosystem,
                        "is_malicious": pkg.is_malicious,
                        "vulnerability_count": len(pkg.vulnerabilities),
                    },
                )
                relationships.append({"source": sid, "target": pkg.stable_id, "type": "contains"})
        agents.append(
            {
                "id": agent.stable_id,
                "canonical_id": agent.canonical_id,
                "name": agent.name,
                "agent_type": agent.agent_type.value,
                "type": agent.agent_type.value,
                "status": agent.status.value,
                "discovered_at": agent.discovered_at,
                "mcp_server_ids": server_ids,
            }
        )
    return {
        "agents": agents,
        "mcp_servers": sorted(servers.values(), key=lambda s: s["id"]),
        "tools": sorted(tools.values(), key=lambda t: t["id"]),
        "resources": [],
        "prompts": [],
        "packages": sorted(packages.values(), key=lambda p: p["id"]),
        "relationships": relationships,
    }


def _build_inventory_snapshot(report: AIBOMReport) -> dict[str, Any]:
    agents = []
    for agent in report.agents:
        agents.append(
            {
                "name": agent.name,
                "agent_type": agent.agent_type.value,
                "config_path": agent.config_path or None,
                "source": agent.source,
                "discovered_at": agent.discovered_at,
                "last_seen": agent.last_seen,
                "mcp_servers": [
                    {
                        "name": s.name,
                        "command": s.command,
                        "args": list(s.args),
                        "transport": s.transport.value,
                        "env_keys": sorted(s.env.keys()),
                        "packages": [
                            {"name": p.name, "version": p.version, "ecosystem": p.ecosystem}
                            for p in s.packages
                        ],
                        "tools": [{"name": t.name} for t in s.tools],
                    }
                    for s in agent.mcp_servers
                ],
            }
        )
    return {"schema_version": "1", "agents": agents}


def _build_framework_summary(blast_radii: list[BlastRadius]) -> dict[str, Any]:
    counters: dict[str, Counter] = {slug: Counter() for _f, slug in FRAMEWORK_TAG_FIELDS}
    for br in blast_radii:
        for fname, slug in FRAMEWORK_TAG_FIELDS:
            counters[slug].update(getattr(br, fname))
    return {
        slug: {
            "tagged_findings": sum(c.values()),
            "unique_controls": len(c),
            "controls": dict(sorted(c.items())),
        }
        for slug, c in counters.items()
    }


def _estate_score(report: AIBOMReport) -> dict[str, Any]:
    from agentbom_amd.scan.risk import estate_exec_score

    return estate_exec_score(report)


def _build_remediation_json(report: AIBOMReport) -> list[dict[str, Any]]:
    """Group findings by (package, fix) into prioritized remediation items."""
    items: dict[tuple[str, str], dict[str, Any]] = {}
    for br in report.blast_radii:
        if br.suppressed:
            continue
        fix = forward_fixed_version(
            br.vulnerability.fixed_version, br.package.version, br.package.ecosystem
        )
        key = (br.package.stable_id, fix or "")
        item = items.setdefault(
            key,
            {
                "package": f"{br.package.name}@{br.package.version}",
                "ecosystem": br.package.ecosystem,
                "fix_version": fix,
                "action": (
                    f"Remove {br.package.name} immediately (malicious)" if br.package.is_malicious
                    else (f"Upgrade {br.package.name} to {fix}" if fix
                          else f"Review advisories for {br.package.name}")
                ),
                "vulns": [],
                "max_risk_score": 0.0,
                "agents": [],
                "creds": [],
                "tools": [],
            },
        )
        if br.vulnerability.id not in item["vulns"]:
            item["vulns"].append(br.vulnerability.id)
        item["max_risk_score"] = max(item["max_risk_score"], br.risk_score)
        for a in br.affected_agents:
            if a.name not in item["agents"]:
                item["agents"].append(a.name)
        for c in br.exposed_credentials:
            if c not in item["creds"]:
                item["creds"].append(c)
        for t in br.exposed_tools:
            if t.name not in item["tools"]:
                item["tools"].append(t.name)

    total_agents = report.total_agents or 1
    out = []
    for item in sorted(items.values(), key=lambda x: -x["max_risk_score"]):
        out.append(
            {
                **item,
                "priority": len(out) + 1,
                "agents_affected_pct": round(100.0 * len(item["agents"]) / total_agents, 1),
                "risk_narrative": _risk_narrative(item),
            }
        )
    return out


def _risk_narrative(item: dict) -> str:
    vuln_id = item["vulns"][0] if item["vulns"] else "this vulnerability"
    agents = ", ".join(item["agents"][:3]) or "affected agents"
    creds = ", ".join(item["creds"][:3])
    tools = ", ".join(item["tools"][:3])
    parts = [f"If not remediated, an attacker exploiting {vuln_id}"]
    if creds:
        parts.append(f"can exfiltrate {creds}")
    parts.append(f"via {agents}")
    if tools:
        parts.append(f"through {tools}")
    return " ".join(parts) + "."


def to_json(report: AIBOMReport) -> dict[str, Any]:
    """Serialize the whole report — the byte-compat JSON contract."""
    ai_bom_entities = _build_ai_bom_entities(report)
    export_findings = list(report.to_findings())
    scan_observed_at = report.generated_at.isoformat()
    for f in export_findings:
        if not f.first_seen:
            f.first_seen = scan_observed_at
    unified = [f.to_dict() for f in export_findings]
    from agentbom_amd.models.maestro import classify_finding, maestro_summary

    for row in unified:
        row["maestro_layer"] = classify_finding(row).value
    finding_summary = _build_finding_summary(unified)
    asset_inventory = _build_asset_inventory(unified)

    from agentbom_amd.models import blast_radius_to_finding

    cve_pairs = [(blast_radius_to_finding(br), br) for br in report.blast_radii]
    exposure_paths = [
        exposure_path_for_finding(f, rank=rank)
        for rank, (f, _br) in enumerate(cve_pairs, start=1)
    ]

    return {
        "schema_version": SCAN_REPORT_SCHEMA_VERSION,
        "canonical_id_schema_version": CANONICAL_ID_SCHEMA_VERSION,
        "document_type": "AI-BOM",
        "spec_version": SCAN_REPORT_SCHEMA_VERSION,
        "scan_id": report.scan_id,
        "ai_bom_version": report.tool_version or __version__,
        "generated_at": report.generated_at.isoformat(),
        "scan_run": {
            "schema_version": SCAN_RUN_SCHEMA_VERSION,
            "scan_id": report.scan_id,
            "generated_at": report.generated_at.isoformat(),
            "source_count": len(report.scan_sources),
            **report.scan_run.to_dict(),
        },
        "warnings": list(report.warnings),
        "scan_sources": list(report.scan_sources),
        "codeowners": [],
        "has_mcp_context": report.has_mcp_context,
        "has_agent_context": report.has_agent_context,
        "framework_catalogs": {"mitre_attack": {"source": "bundled", "version": "enterprise-bundled"}},
        "ai_bom_entities": {"schema_version": "1.0", **ai_bom_entities},
        "packages": ai_bom_entities.get("packages", []),
        "summary": {
            "total_agents": report.total_agents,
            "total_mcp_servers": report.total_servers,
            "total_packages": report.total_packages,
            "unique_packages": len(ai_bom_entities.get("packages", [])),
            "total_vulnerabilities": report.total_vulnerabilities,
            "critical_findings": len(report.critical_vulns),
            "total_findings": finding_summary["total"],
            "unique_assets": len(asset_inventory),
            "critical_unified_findings": finding_summary["by_severity"]["critical"],
            "high_unified_findings": finding_summary["by_severity"]["high"],
            "coverage_warnings": list(report.coverage_warnings),
        },
        "finding_summary": finding_summary,
        "assets": asset_inventory,
        "coverage_warnings": list(report.coverage_warnings),
        "inventory_snapshot": _build_inventory_snapshot(report),
        "agents": [
            {
                "name": agent.name,
                "stable_id": agent.stable_id,
                "canonical_id": agent.canonical_id,
                "previous_canonical_ids": [],
                "agent_type": agent.agent_type.value,
                "type": agent.agent_type.value,
                "config_path": agent.config_path or "",
                "source": agent.source,
                "status": agent.status.value,
                "discovered_at": agent.discovered_at,
                "last_seen": agent.last_seen,
                "discovery_provenance": agent.discovery_provenance,
                "metadata": agent.metadata,
                "automation_settings": agent.automation_settings,
                "mcp_servers": [_server_dict(s) for s in agent.mcp_servers],
            }
            for agent in report.agents
        ],
        "blast_radius": [
            _blast_radius_json_entry(br, f, rank, exposure_paths[rank - 1])
            for rank, (f, br) in enumerate(cve_pairs, start=1)
        ],
        "exposure_paths": {
            "schema_version": "1",
            "source": "blast_radius_output",
            "path_count": len(exposure_paths),
            "paths": exposure_paths,
        },
        "findings": unified,
        "threat_framework_summary": _build_framework_summary(report.blast_radii),
        "maestro_summary": maestro_summary(unified),
        "scorecard_summary": {
            "total_packages": report.total_packages,
            "with_scorecard": sum(
                1 for a in report.agents for s in a.mcp_servers for p in s.packages
                if p.scorecard_score is not None
            ),
        },
        "remediation_plan": _build_remediation_json(report),
        "scan_performance": report.scan_performance_data,
        "intel_matches": report.intel_matches or [],
        "estate_score": _estate_score(report),
        # optional side blocks travel under their reference key names; absent
        # blocks are omitted entirely (contract: no null-noise keys)
        **{key: value for key, value in (
            ("project_inventory", report.project_inventory_data),
            ("iac_findings", (report.iac_findings_data or {}).get("findings")
             if report.iac_findings_data else None),
            ("ai_inventory", report.ai_inventory_data),
            ("license_report", report.license_report),
            ("vex", report.vex_data),
            ("context_graph", report.context_graph_data),
            ("enforcement", report.enforcement_data),
            ("delta", report.delta_data),
        ) if value is not None},
        **dict(sorted(report.extra_data.items())),
    }
