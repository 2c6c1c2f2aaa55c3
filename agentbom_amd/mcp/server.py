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


class McpTool:
    def __init__(self, name: str, description: str, schema: dict, fn: Callable[..., Any]):
        self.name = name
        self.description = description
        self.schema = schema
        self.fn = fn


class AgentBomMcpServer:
    """Tool registry + dispatch over a shared scan/graph session."""

    PROTOCOL_VERSION = "2024-11-05"

    def __init__(self, demo: bool = False):
        self.demo = demo
        self._report = None
        self._graph = None
        self.tools: dict[str, McpTool] = {}
        self._register_tools()

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
        def tool(name: str, description: str, schema: Optional[dict] = None):
            def deco(fn):
                self.tools[name] = McpTool(
                    name, description,
                    schema or {"type": "object", "properties": {}, "additionalProperties": False},
                    fn,
                )
                return fn
            return deco

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
                "serverInfo": {"name": "agent-bom", "version": "0.1.0"},
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
            name = params.get("name")
            tool = self.tools.get(name)
            if tool is None:
                return err(-32602, f"unknown tool {name!r}")
            try:
                result = tool.fn(**(params.get("arguments") or {}))
                return ok({"content": [{"type": "text", "text": json.dumps(result, default=str)}]})
            except Exception as exc:  # noqa: BLE001 — tool boundary
                return ok({"content": [{"type": "text", "text": json.dumps({"error": str(exc)})}],
                           "isError": True})
        if method == "resources/list":
            return ok({"resources": [
                {"uri": "agent-bom://report/latest", "name": "latest scan report",
                 "mimeType": "application/json"},
                {"uri": "agent-bom://graph/latest", "name": "latest estate graph",
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
            else:
                return err(-32602, f"unknown resource {uri!r}")
            return ok({"contents": [{"uri": uri, "mimeType": "application/json", "text": text}]})
        if method == "prompts/list":
            return ok({"prompts": [
                {"name": "triage", "description": "Triage the highest-risk findings."},
                {"name": "remediate", "description": "Draft a remediation plan."},
            ]})
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
