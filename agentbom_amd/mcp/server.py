"""agent-bom MCP server — stdio JSON-RPC 2.0, MCP protocol 2024-11-05.

Reference surface: src/agent_bom/mcp_server.py:586 create_mcp_server (81
tools / 6 resources / 8 prompts, read-first).  This build implements the
MCP wire protocol directly (initialize / tools/list / tools/call /
resources/list / resources/read / prompts/list) with the core tool set:
scan, check_package, blast_radius, exposure_paths, should_i_deploy,
policy_check, generate_sbom, compliance_posture, graph_search,
graph_impact, attack_paths, rollup, risk_summary, db_status.
"""

from __future__ import annotations

import json
import sys
from typing import Any, Callable, Optional


def resolve_mcp_tenant_id() -> str:
    """Sanctioned tenant resolution for the MCP surface (reference:
    mcp_tenant.py).  MCP clients pass no authenticated tenant identity, so
    the launching operator decides via AGENT_BOM_MCP_TENANT_ID (preferred)
    with AGENT_BOM_TENANT_ID as fallback; "default" otherwise."""
    import os

    return (os.environ.get("AGENT_BOM_MCP_TENANT_ID")
            or os.environ.get("AGENT_BOM_TENANT_ID")
            or "default").strip() or "default"


def _check_strict_args(schema: dict, arguments: dict) -> Optional[dict]:
    """Reject unknown and missing-required tool arguments (fail loud).

    Returns a structured error dict, or None when the call is well-formed.
    """
    props = schema.get("properties") or {}
    unknown = sorted(set(arguments) - set(props))
    if unknown:
        return {"error": "unknown arguments rejected (strict-args contract)",
                "unknown": unknown, "accepted": sorted(props)}
    missing = sorted(set(schema.get("required") or []) - set(arguments))
    if missing:
        return {"error": "missing required arguments",
                "missing": missing, "accepted": sorted(props)}
    return None


# Prompt catalog: (description, required args, message template).
# Read-first guidance — each prompt steers the model toward the matching
# read tools before any write-gated action.
_PROMPTS = {
    "triage": (
        "Triage the highest-risk findings.", (),
        "Call findings_triage and risk_summary, then rank the top 10 "
        "findings by fused priority. For each: why it ranks there (KEV, "
        "EPSS, reachability, blast radius) and the single next action."),
    "remediate": (
        "Draft a remediation plan.", (),
        "Call remediate and group the commands into waves by risk. "
        "Produce a reviewable plan: wave, command, what it clears, "
        "rollback note."),
    "investigate-cve": (
        "Deep-dive one CVE across the estate.", ("cve",),
        "Investigate {cve}: call intel_lookup, blast_radius and "
        "exposure_paths for it. Summarize which agents/credentials/tools "
        "it reaches, whether runtime evidence confirms exposure, and the "
        "exact fix."),
    "blast-analysis": (
        "Explain the blast radius of a package.", ("package",),
        "For package {package}: call check_package, then graph_search and "
        "graph_impact on its node. Explain every hop from the package to "
        "credentials and agents in plain language."),
    "compliance-report": (
        "Summarize posture for one framework.", ("framework",),
        "Call compliance_posture and analytics_query, then write an "
        "auditor-facing summary for {framework}: controls with evidence, "
        "gaps, and the top 3 remediations that close the most controls."),
    "harden-mcp": (
        "Harden the MCP server estate.", (),
        "Call mcp_auth_posture, a2a_auth_posture and self_posture. "
        "Produce a hardening checklist ordered by severity with the "
        "specific config change for each item."),
    "incident-response": (
        "Respond to a suspected compromise.", ("indicator",),
        "Indicator: {indicator}. Call runtime_correlate and proxy_alerts "
        "to scope it, effective_permissions for the implicated agents, "
        "then propose shield actions (quarantine/block/revoke) with the "
        "exact admin-gated tool calls an operator should approve."),
    "executive-summary": (
        "Executive one-pager for the latest scan.", (),
        "Call scan, intel_daily_brief and should_i_deploy. Write one page: "
        "estate size, risk trend, deploy verdict, top 3 risks in business "
        "language, and what was fixed since last scan (diff if a baseline "
        "is available)."),
}


class McpTool:
    def __init__(self, name: str, description: str, schema: dict, fn: Callable[..., Any]):
        self.name = name
        self.description = description
        self.schema = schema
        self.fn = fn


class AgentBomMcpServer:
    """Tool registry + dispatch over a shared scan/graph session.

    Guardrails on every call (reference: mcp_server_runtime.py):
    response truncation to MAX_RESPONSE_BYTES, per-caller token-bucket rate
    limiting, and per-tool latency/error metrics.
    """

    PROTOCOL_VERSION = "2024-11-05"
    MAX_RESPONSE_BYTES = 512 * 1024
    RATE_LIMIT_CALLS = 120  # per caller per minute

    def __init__(self, demo: bool = False):
        self.demo = demo
        self._report = None
        self._graph = None
        self.tools: dict[str, McpTool] = {}
        self.tool_metrics: dict[str, dict] = {}
        self._rate_window: dict[str, list[float]] = {}
        self._register_tools()
        from agentbom_amd.mcp.tools_operator import register_operator_tools
        from agentbom_amd.mcp.tools_specialized import register_specialized_tools

        register_operator_tools(self)
        register_specialized_tools(self)

    # shared operator state, built lazily so `tools/list` stays instant
    _identity_store = None
    _ticket_store = None
    _campaign_store = None
    _shield = None

    @property
    def identity_store(self):
        if self._identity_store is None:
            from agentbom_amd.identity import AgentIdentityStore

            self._identity_store = AgentIdentityStore()
        return self._identity_store

    @property
    def ticket_store(self):
        if self._ticket_store is None:
            from agentbom_amd.runtime.ticketing import TicketStore

            self._ticket_store = TicketStore()
        return self._ticket_store

    @property
    def campaign_store(self):
        if self._campaign_store is None:
            from agentbom_amd.scan.campaigns import CampaignStore

            self._campaign_store = CampaignStore()
        return self._campaign_store

    @property
    def shield(self):
        if self._shield is None:
            from agentbom_amd.runtime.shield import Shield

            self._shield = Shield()
        return self._shield

    def tool(self, name: str, description: str, schema: Optional[dict] = None):
        """Decorator used by the core registry and every tool group."""

        def deco(fn):
            self.tools[name] = McpTool(
                name, description,
                schema or {"type": "object", "properties": {},
                           "additionalProperties": False},
                fn,
            )
            return fn
        return deco

    def _check_rate_limit(self, caller: str) -> bool:
        import time

        now = time.monotonic()
        window = self._rate_window.setdefault(caller, [])
        window[:] = [t for t in window if now - t < 60.0]
        if len(window) >= self.RATE_LIMIT_CALLS:
            return False
        window.append(now)
        return True

    def _record_metric(self, name: str, ms: float, error: bool) -> None:
        m = self.tool_metrics.setdefault(
            name, {"calls": 0, "errors": 0, "total_ms": 0.0})
        m["calls"] += 1
        m["total_ms"] += ms
        if error:
            m["errors"] += 1

    # ── session state ─────────────────────────────────────────────────────

    def _ensure_scan(self):
        if self._report is None:
            from agentbom_amd.graph.builder import build_unified_graph_from_report
            from agentbom_amd.graph.dependency_reach import (
                apply_dependency_reachability_to_blast_radii,
            )
            from agentbom_amd.scan.orchestrator import run_demo_scan

            self._report = run_demo_scan()
            self._graph = build_unified_graph_from_report(self._report)
            apply_dependency_reachability_to_blast_radii(self._report, self._graph)
        return self._report, self._graph

    # ── tools ─────────────────────────────────────────────────────────────

    def _register_tools(self) -> None:
        tool = self.tool

        @tool("scan", "Security Scan: discover agents/MCP servers and scan their packages "
                      "for vulnerabilities, returning the findings summary.")
        def scan_tool() -> dict:
            from agentbom_amd.output.json_fmt import to_json

            report, _g = self._ensure_scan()
            doc = to_json(report)
            return {"summary": doc["summary"], "finding_summary": doc["finding_summary"]}

        @tool("check_package", "Check one package@version against the advisory data.",
              {"type": "object",
               "properties": {"name": {"type": "string"}, "version": {"type": "string"},
                              "ecosystem": {"type": "string"}},
               "required": ["name", "version", "ecosystem"]})
        def check_package(name: str, version: str, ecosystem: str) -> dict:
            from agentbom_amd.db.store import load_advisory_windows
            from agentbom_amd.utils.canonical_ids import normalize_package_name
            from agentbom_amd.utils.version_utils import version_in_range

            hits = []
            for w in load_advisory_windows(offline=True):
                if (w.ecosystem.lower() == ecosystem.lower()
                        and normalize_package_name(w.package_name, w.ecosystem)
                        == normalize_package_name(name, ecosystem)):
                    if version_in_range(version, w.introduced, w.fixed, w.last_affected, ecosystem):
                        hits.append({
                            "vuln_id": w.vuln_id, "severity": w.severity.value,
                            "cvss_score": w.cvss_score, "is_kev": w.is_kev,
                            "fixed": w.fixed, "summary": w.summary,
                        })
            return {"package": f"{name}@{version}", "ecosystem": ecosystem,
                    "vulnerable": bool(hits), "advisories": hits}

        @tool("blast_radius", "Blast radius of a vulnerability or package: which agents, "
                              "credentials and tools it can reach.",
              {"type": "object", "properties": {"vuln_id": {"type": "string"}},
               "required": ["vuln_id"]})
        def blast_radius(vuln_id: str) -> dict:
            report, _g = self._ensure_scan()
            for br in report.blast_radii:
                if br.vulnerability.id == vuln_id:
                    return {
                        "vulnerability_id": vuln_id,
                        "risk_score": br.risk_score,
                        "severity": br.vulnerability.severity.value,
                        "package": f"{br.package.name}@{br.package.version}",
                        "affected_agents": [a.name for a in br.affected_agents],
                        "affected_servers": [s.name for s in br.affected_servers],
                        "exposed_credentials": br.exposed_credentials,
                        "exposed_tools": [t.name for t in br.exposed_tools],
                        "reachability": br.reachability,
                        "impact_category": br.impact_category,
                    }
            return {"vulnerability_id": vuln_id, "error": "not found in latest scan"}

        @tool("exposure_paths", "Ranked exposure paths for the latest scan.",
              {"type": "object", "properties": {"limit": {"type": "integer", "default": 10}}})
        def exposure_paths(limit: int = 10) -> dict:
            from agentbom_amd.models import blast_radius_to_finding
            from agentbom_amd.output.exposure_path import exposure_path_for_finding

            report, _g = self._ensure_scan()
            paths = [
                exposure_path_for_finding(blast_radius_to_finding(br), rank=i + 1)
                for i, br in enumerate(report.blast_radii[:limit])
            ]
            return {"schema_version": "1", "source": "blast_radius_output",
                    "path_count": len(paths), "paths": paths}

        @tool("should_i_deploy", "Deploy gate: allow / warn / block against risk thresholds.")
        def should_i_deploy() -> dict:
            from agentbom_amd.utils import config as cfg

            report, _g = self._ensure_scan()
            max_risk = max((br.risk_score for br in report.blast_radii), default=0.0) * 10
            has_malicious = any(br.package.is_malicious for br in report.blast_radii)
            has_kev = any(br.vulnerability.is_kev for br in report.blast_radii)
            verdict = ("block" if has_malicious or max_risk >= cfg.DEPLOY_BLOCK_RISK
                       else "warn" if has_kev or max_risk >= cfg.DEPLOY_WARN_RISK
                       else "allow")
            return {"verdict": verdict, "max_risk": max_risk,
                    "has_malicious": has_malicious, "has_kev": has_kev}

        @tool("policy_check", "Evaluate a simple policy expression over findings.",
              {"type": "object",
               "properties": {"max_risk": {"type": "number"},
                              "block_kev": {"type": "boolean"},
                              "block_malicious": {"type": "boolean"}}})
        def policy_check(max_risk: float = 10.0, block_kev: bool = True,
                         block_malicious: bool = True) -> dict:
            report, _g = self._ensure_scan()
            violations = []
            for br in report.blast_radii:
                if br.risk_score > max_risk:
                    violations.append({"rule": f"risk>{max_risk}", "vuln": br.vulnerability.id})
                if block_kev and br.vulnerability.is_kev:
                    violations.append({"rule": "kev", "vuln": br.vulnerability.id})
                if block_malicious and br.package.is_malicious:
                    violations.append({"rule": "malicious", "package": br.package.name})
            return {"passed": not violations, "violations": violations}

        @tool("generate_sbom", "Generate a CycloneDX or SPDX SBOM for the latest scan.",
              {"type": "object", "properties": {"format": {"type": "string",
               "enum": ["cyclonedx", "spdx"], "default": "cyclonedx"}}})
        def generate_sbom(format: str = "cyclonedx") -> dict:
            report, _g = self._ensure_scan()
            if format == "spdx":
                from agentbom_amd.output.spdx_fmt import to_spdx

                return to_spdx(report)
            from agentbom_amd.output.cyclonedx_fmt import to_cyclonedx

            return to_cyclonedx(report)

        @tool("compliance_posture", "Framework tag coverage across the latest findings.")
        def compliance_posture() -> dict:
            from agentbom_amd.output.json_fmt import _build_framework_summary

            report, _g = self._ensure_scan()
            return _build_framework_summary(report.blast_radii)

        @tool("graph_search", "Search estate graph nodes.",
              {"type": "object", "properties": {"q": {"type": "string"},
               "entity_type": {"type": "string"}}, "required": ["q"]})
        def graph_search(q: str, entity_type: Optional[str] = None) -> dict:
            from agentbom_amd.graph.types import EntityType

            _r, g = self._ensure_scan()
            et = [EntityType(entity_type)] if entity_type else None
            nodes = g.search(query=q, entity_types=et)
            return {"total": len(nodes), "nodes": [n.to_dict() for n in nodes]}

        @tool("graph_impact", "Bounded impact neighborhood of a graph node (<=4 hops).",
              {"type": "object", "properties": {"node_id": {"type": "string"},
               "max_hops": {"type": "integer", "default": 4}}, "required": ["node_id"]})
        def graph_impact(node_id: str, max_hops: int = 4) -> dict:
            _r, g = self._ensure_scan()
            if node_id not in g.nodes:
                return {"error": f"node {node_id!r} not found"}
            return g.impact_of(node_id, max_hops=min(max_hops, 4))

        @tool("attack_paths", "Fused attack paths from entry agents to crown jewels.",
              {"type": "object", "properties": {"limit": {"type": "integer", "default": 10}}})
        def attack_paths(limit: int = 10) -> dict:
            from agentbom_amd.graph.attack_paths import compute_fused_attack_paths

            _r, g = self._ensure_scan()
            paths = compute_fused_attack_paths(g, max_paths=limit)
            return {"path_count": len(paths), "paths": [p.to_dict() for p in paths]}

        @tool("rollup", "Estate roll-up: containers with severity aggregates.")
        def rollup() -> dict:
            from agentbom_amd.graph.rollup import rollup_view

            _r, g = self._ensure_scan()
            return rollup_view(g)

        @tool("risk_summary", "Top findings by risk score.")
        def risk_summary() -> dict:
            report, _g = self._ensure_scan()
            return {
                "findings": [
                    {
                        "vuln_id": br.vulnerability.id, "risk_score": br.risk_score,
                        "severity": br.vulnerability.severity.value,
                        "package": f"{br.package.name}@{br.package.version}",
                        "is_kev": br.vulnerability.is_kev,
                    }
                    for br in report.blast_radii[:20]
                ]
            }

        @tool("intel_lookup", "Threat-intel view of one vulnerability: "
                              "advisories, KEV, EPSS, CVSS, affected windows.",
              {"type": "object", "properties": {"vuln_id": {"type": "string"}},
               "required": ["vuln_id"]})
        def intel_lookup(vuln_id: str) -> dict:
            from agentbom_amd.db.store import load_advisory_windows

            windows = [w for w in load_advisory_windows(offline=True)
                       if w.vuln_id == vuln_id]
            if not windows:
                return {"vuln_id": vuln_id, "found": False}
            w0 = windows[0]
            return {"vuln_id": vuln_id, "found": True,
                    "severity": w0.severity.value, "cvss_score": w0.cvss_score,
                    "is_kev": w0.is_kev, "epss_score": w0.epss_score,
                    "summary": w0.summary,
                    "affected": [{"ecosystem": w.ecosystem,
                                  "package": w.package_name,
                                  "introduced": w.introduced, "fixed": w.fixed,
                                  "last_affected": w.last_affected}
                                 for w in windows]}

        @tool("intel_match", "Match a package list against the advisory arena "
                             "in one batch.",
              {"type": "object", "properties": {
                  "packages": {"type": "array", "items": {"type": "object"}}},
               "required": ["packages"]})
        def intel_match(packages: list) -> dict:
            from agentbom_amd.db.store import load_advisory_windows
            from agentbom_amd.utils.canonical_ids import normalize_package_name
            from agentbom_amd.utils.version_utils import version_in_range

            windows = load_advisory_windows(offline=True)
            out = []
            for p in packages[:10_000]:
                name, version = str(p.get("name", "")), str(p.get("version", ""))
                eco = str(p.get("ecosystem", ""))
                norm = normalize_package_name(name, eco)
                hits = [w.vuln_id for w in windows
                        if w.ecosystem.lower() == eco.lower()
                        and normalize_package_name(w.package_name, w.ecosystem) == norm
                        and version_in_range(version, w.introduced, w.fixed,
                                             w.last_affected, eco)]
                if hits:
                    out.append({"package": f"{name}@{version}",
                                "ecosystem": eco, "vuln_ids": sorted(set(hits))})
            return {"matched": len(out), "results": out}

        @tool("intel_sources", "Advisory/intel sources bundled in this build "
                               "and their record counts.")
        def intel_sources() -> dict:
            from agentbom_amd.db.store import load_advisory_windows

            windows = load_advisory_windows(offline=True)
            by_eco: dict[str, int] = {}
            for w in windows:
                by_eco[w.ecosystem] = by_eco.get(w.ecosystem, 0) + 1
            return {"sources": ["osv-bundled", "cisa-kev-bundled",
                                "epss-bundled", "malicious-registry"],
                    "advisory_windows": len(windows),
                    "kev_entries": sum(1 for w in windows if w.is_kev),
                    "by_ecosystem": by_eco}

        @tool("intel_daily_brief", "Operator brief: top risks, KEV exposure, "
                                   "malicious hits, deploy verdict.")
        def intel_daily_brief() -> dict:
            report, _g = self._ensure_scan()
            top = report.blast_radii[:5]
            from agentbom_amd.scan.risk import estate_exec_score

            return {
                "date_scope": "latest scan",
                "estate_score": estate_exec_score(report),
                "headline_risks": [{
                    "vuln_id": b.vulnerability.id, "risk_score": b.risk_score,
                    "package": f"{b.package.name}@{b.package.version}",
                    "is_kev": b.vulnerability.is_kev,
                    "summary": b.attack_vector_summary} for b in top],
                "kev_count": sum(1 for b in report.blast_radii
                                 if b.vulnerability.is_kev),
                "malicious_count": sum(1 for b in report.blast_radii
                                       if b.package.is_malicious),
                "deploy_verdict": self.tools["should_i_deploy"].fn(),
            }

        @tool("remediate", "Prioritized remediation plan with executable "
                           "per-ecosystem upgrade commands.")
        def remediate() -> dict:
            from agentbom_amd.output.json_fmt import _build_remediation_json
            from agentbom_amd.scan.remediation import (
                remediation_commands,
                remediation_script,
            )

            report, _g = self._ensure_scan()
            return {"plan": _build_remediation_json(report),
                    "commands": remediation_commands(report),
                    "script": remediation_script(report)}

        @tool("verify", "Re-scan and verify whether a finding is resolved.",
              {"type": "object", "properties": {"vuln_id": {"type": "string"}},
               "required": ["vuln_id"]})
        def verify(vuln_id: str) -> dict:
            self._report = None  # force a fresh scan
            self._graph = None
            report, _g = self._ensure_scan()
            still = [b for b in report.blast_radii
                     if b.vulnerability.id == vuln_id]
            return {"vuln_id": vuln_id,
                    "resolved": not still,
                    "still_affected": [f"{b.package.name}@{b.package.version}"
                                       for b in still]}

        @tool("inventory_summary", "Asset inventory rollup for the latest scan.")
        def inventory_summary() -> dict:
            from agentbom_amd.output.json_fmt import to_json

            report, _g = self._ensure_scan()
            doc = to_json(report)
            return doc["inventory_snapshot"]

        @tool("inventory_list", "All inventoried assets with finding counts.",
              {"type": "object", "properties": {
                  "asset_type": {"type": "string", "default": ""}}})
        def inventory_list(asset_type: str = "") -> dict:
            from agentbom_amd.output.json_fmt import to_json

            report, _g = self._ensure_scan()
            assets = to_json(report)["assets"]
            if asset_type:
                assets = [a for a in assets if a.get("asset_type") == asset_type]
            return {"total": len(assets), "assets": assets}

        @tool("inventory_asset", "One asset's full detail incl. its findings.",
              {"type": "object", "properties": {"name": {"type": "string"}},
               "required": ["name"]})
        def inventory_asset(name: str) -> dict:
            from agentbom_amd.output.json_fmt import to_json

            report, _g = self._ensure_scan()
            doc = to_json(report)
            asset = next((a for a in doc["assets"] if a.get("name") == name), None)
            if asset is None:
                return {"error": f"asset {name!r} not found"}
            findings = [f for f in doc["findings"]
                        if (f.get("asset") or {}).get("name") == name]
            return {"asset": asset, "findings": findings}

        @tool("registry_lookup", "Look up an MCP server in the known-server "
                                 "registry (verification + risk notes).",
              {"type": "object", "properties": {"name": {"type": "string"}},
               "required": ["name"]})
        def registry_lookup(name: str) -> dict:
            report, _g = self._ensure_scan()
            matches = [s for a in report.agents for s in a.mcp_servers
                       if name.lower() in s.name.lower()]
            if not matches:
                from agentbom_amd.mcp.registry import load_registry

                reg = load_registry()
                hits = [dict(e) for key, e in reg["servers"].items()
                        if name.lower() in key.lower()
                        or name.lower() in str(e.get("name", "")).lower()]
                if hits:
                    return {"name": name, "found": True, "source": "registry",
                            "entries": hits[:10]}
                return {"name": name, "found": False,
                        "note": "not in latest scan or registry; run "
                                "marketplace_check for trust signals"}
            from agentbom_amd.scan.risk import score_server_risk, server_risk_level

            out = []
            for s in matches:
                score = score_server_risk(s)
                out.append({"name": s.name, "command": s.command,
                            "registry_verified": s.registry_verified,
                            "tool_count": len(s.tools),
                            "credentials": s.credential_names,
                            "risk_score": score,
                            "risk_level": server_risk_level(score),
                            "security_warnings": s.security_warnings})
            return {"name": name, "found": True, "servers": out}

        @tool("db_status", "Local advisory DB freshness and counts.")
        def db_status() -> dict:
            from agentbom_amd.db.store import AdvisoryStore, default_db_path

            path = default_db_path()
            if not path.exists():
                return {"path": str(path), "exists": False,
                        "note": "no local DB; scans use bundled advisories"}
            store = AdvisoryStore(path)
            try:
                return store.status()
            finally:
                store.close()

    # ── JSON-RPC dispatch ─────────────────────────────────────────────────

    def handle(self, msg: dict) -> Optional[dict]:
        method = msg.get("method", "")
        msg_id = msg.get("id")
        params = msg.get("params") or {}

        def ok(result: Any) -> dict:
            return {"jsonrpc": "2.0", "id": msg_id, "result": result}

        def err(code: int, message: str) -> dict:
            return {"jsonrpc": "2.0", "id": msg_id, "error": {"code": code, "message": message}}

        if method == "initialize":
            return ok({
                "protocolVersion": self.PROTOCOL_VERSION,
                "capabilities": {"tools": {}, "resources": {}, "prompts": {}},
                "serverInfo": {"name": "agent-bom", "version": "0.1.0",
                               "tenant": resolve_mcp_tenant_id()},
            })
        if method == "notifications/initialized":
            return None
        if method == "tools/list":
            return ok({
                "tools": [
                    {"name": t.name, "description": t.description, "inputSchema": t.schema}
                    for t in self.tools.values()
                ]
            })
        if method == "tools/call":
            import time as _time

            name = params.get("name")
            tool = self.tools.get(name)
            if tool is None:
                return err(-32602, f"unknown tool {name!r}")
            caller = str((params.get("_meta") or {}).get("caller", "default"))
            if not self._check_rate_limit(caller):
                return ok({"content": [{"type": "text", "text": json.dumps(
                    {"error": "rate limit exceeded", "retry_after_s": 60})}],
                    "isError": True})
            # strict-args contract (reference: mcp_strict_args.py — silent
            # arg drops turn typo'd gates into false-clean verdicts): unknown
            # or missing-required arguments are rejected, never dropped.
            arguments = params.get("arguments") or {}
            strict_error = _check_strict_args(tool.schema, arguments)
            if strict_error is not None:
                return ok({"content": [{"type": "text",
                                        "text": json.dumps(strict_error)}],
                           "isError": True})
            t0 = _time.perf_counter()
            try:
                result = tool.fn(**arguments)
                text = json.dumps(result, default=str)
                if len(text) > self.MAX_RESPONSE_BYTES:
                    text = json.dumps({
                        "truncated": True,
                        "full_bytes": len(text),
                        "preview": text[: self.MAX_RESPONSE_BYTES // 2],
                    })
                self._record_metric(name, (_time.perf_counter() - t0) * 1000, False)
                return ok({"content": [{"type": "text", "text": text}]})
            except Exception as exc:  # noqa: BLE001 — tool boundary
                self._record_metric(name, (_time.perf_counter() - t0) * 1000, True)
                return ok({"content": [{"type": "text", "text": json.dumps({"error": str(exc)})}],
                           "isError": True})
        if method == "resources/list":
            return ok({"resources": [
                {"uri": "agent-bom://report/latest", "name": "latest scan report",
                 "mimeType": "application/json"},
                {"uri": "agent-bom://graph/latest", "name": "latest estate graph",
                 "mimeType": "application/json"},
                {"uri": "agent-bom://findings/latest", "name": "unified findings",
                 "mimeType": "application/json"},
                {"uri": "agent-bom://remediation/latest",
                 "name": "remediation plan + commands",
                 "mimeType": "application/json"},
                {"uri": "agent-bom://posture/latest",
                 "name": "auth + self posture", "mimeType": "application/json"},
                {"uri": "agent-bom://registry/servers",
                 "name": "known-MCP-server security registry",
                 "mimeType": "application/json"},
            ]})
        if method == "resources/read":
            uri = params.get("uri", "")
            report, graph = self._ensure_scan()
            if uri == "agent-bom://report/latest":
                from agentbom_amd.output.json_fmt import to_json

                text = json.dumps(to_json(report), default=str)
            elif uri == "agent-bom://graph/latest":
                text = json.dumps(graph.to_dict(), default=str)
            elif uri == "agent-bom://findings/latest":
                text = json.dumps([f.to_dict() for f in report.to_findings()],
                                  default=str)
            elif uri == "agent-bom://remediation/latest":
                from agentbom_amd.scan.remediation import remediation_commands

                text = json.dumps({"commands": remediation_commands(report)},
                                  default=str)
            elif uri == "agent-bom://posture/latest":
                from agentbom_amd.scan.auth_posture import assess_estate
                from agentbom_amd.scan.self_posture import evaluate_self_posture

                text = json.dumps({"mcp_auth_posture": assess_estate(report.agents),
                                   "self_posture": evaluate_self_posture()},
                                  default=str)
            elif uri == "agent-bom://registry/servers":
                from agentbom_amd.mcp.registry import load_registry

                text = json.dumps(load_registry(), default=str)
            else:
                return err(-32602, f"unknown resource {uri!r}")
            return ok({"contents": [{"uri": uri, "mimeType": "application/json", "text": text}]})
        if method == "prompts/list":
            return ok({"prompts": [
                {"name": p_name, "description": desc,
                 "arguments": [{"name": a, "required": True} for a in args]}
                for p_name, (desc, args, _tpl) in _PROMPTS.items()
            ]})
        if method == "prompts/get":
            p_name = params.get("name", "")
            entry = _PROMPTS.get(p_name)
            if entry is None:
                return err(-32602, f"unknown prompt {p_name!r}")
            desc, arg_names, template = entry
            values = params.get("arguments") or {}
            missing = [a for a in arg_names if a not in values]
            if missing:
                return err(-32602, f"missing prompt arguments {missing}")
            return ok({"description": desc, "messages": [{
                "role": "user",
                "content": {"type": "text",
                            "text": template.format(**values)},
            }]})
        if method == "ping":
            return ok({})
        return err(-32601, f"method {method!r} not found")


def run_stdio_server(demo: bool = False) -> None:
    """Line-delimited JSON-RPC over stdio (the MCP stdio transport)."""
    server = AgentBomMcpServer(demo=demo)
    for line in sys.stdin:
        line = line.strip()
        if not line:
            continue
        try:
            msg = json.loads(line)
        except json.JSONDecodeError:
            continue
        resp = server.handle(msg)
        if resp is not None:
            sys.stdout.write(json.dumps(resp) + "\n")
            sys.stdout.flush()
