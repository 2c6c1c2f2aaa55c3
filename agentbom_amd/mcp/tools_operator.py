"""Operator MCP tools: runtime, identity, shield, tickets, cost, analytics.

Reference surface: src/agent_bom/mcp_server_operator_tools.py (52 tools) +
mcp_tools/{identity,runtime,runtime_evidence,risk_campaigns,ticketing,
posture,cloud,kspm,analysis,graph}.py.  Every write-capable tool is gated by
the shared admin-role + scope + audit-reason contract (``mcp/authz.py``);
reads are unrestricted.

Offline posture: tools that the reference points at live services
(ClickHouse, Jira, cloud APIs) consume exported files here — OTel GenAI
span dumps, proxy audit JSONL, tracker/issue exports, cloud inventory
exports — through the same code paths a live forwarder would use.
"""

from __future__ import annotations

import json
import math
import time
from collections import defaultdict
from pathlib import Path
from typing import Any, Optional

from agentbom_amd.mcp.authz import authorize_write

# Open per-1M-token cost model (USD) used for spend attribution; operators
# can override with AGENT_BOM_COST_MODEL (JSON path).
DEFAULT_PRICES = {
    "claude-sonnet": {"input": 3.0, "output": 15.0},
    "claude-haiku": {"input": 0.8, "output": 4.0},
    "claude-opus": {"input": 15.0, "output": 75.0},
    "gpt-4o": {"input": 2.5, "output": 10.0},
    "default": {"input": 2.0, "output": 8.0},
}

# Canonical role blueprints for runtime policy design: which capability
# classes each production role may exercise.
RUNTIME_BLUEPRINTS = {
    "read-only-analyst": {
        "description": "Dashboards and investigation; never mutates anything.",
        "allowed_capabilities": ["read", "query"],
        "denied_capabilities": ["execute", "write", "delete", "network"],
    },
    "ci-scanner": {
        "description": "Pipeline scanning: reads code and emits findings.",
        "allowed_capabilities": ["read", "query", "write"],
        "denied_capabilities": ["execute", "delete"],
    },
    "remediation-operator": {
        "description": "Applies fixes: writes and network egress, no shell.",
        "allowed_capabilities": ["read", "query", "write", "network"],
        "denied_capabilities": ["execute", "delete"],
    },
    "break-glass-admin": {
        "description": "Incident response; every call must carry an audit reason.",
        "allowed_capabilities": ["read", "query", "write", "network", "execute", "delete"],
        "denied_capabilities": [],
    },
}


def _load_jsonl(path: str, max_rows: int = 200_000) -> list[dict[str, Any]]:
    rows = []
    for line in Path(path).read_text().splitlines()[:max_rows]:
        line = line.strip()
        if not line:
            continue
        try:
            row = json.loads(line)
        except json.JSONDecodeError:
            continue
        if isinstance(row, dict):
            rows.append(row)
    return rows


def _price_for(model: str, prices: dict) -> dict:
    model_l = (model or "").lower()
    for key, p in prices.items():
        if key != "default" and key in model_l:
            return p
    return prices["default"]


def _span_cost(span: dict, prices: dict) -> float:
    p = _price_for(str(span.get("model", "")), prices)
    return (float(span.get("input_tokens", 0)) * p["input"]
            + float(span.get("output_tokens", 0)) * p["output"]) / 1e6


def _audit_tool_calls(rows: list[dict]) -> dict[str, int]:
    counts: dict[str, int] = defaultdict(int)
    for r in rows:
        if r.get("method") == "tools/call" and r.get("tool"):
            counts[str(r["tool"])] += 1
    return counts


def register_operator_tools(server) -> None:  # noqa: C901 — one registrar
    tool = server.tool

    # ── triage / campaigns / diff ─────────────────────────────────────────

    @tool("findings_triage", "Fused triage queue: findings ranked by KEV/EPSS/"
                             "reachability-weighted priority.",
          {"type": "object", "properties": {"limit": {"type": "integer", "default": 25}}})
    def findings_triage(limit: int = 25) -> dict:
        report, _g = server._ensure_scan()

        def prio(f) -> float:
            ev = f.evidence if isinstance(f.evidence, dict) else {}
            return float(ev.get("triage_priority", 0.0))

        findings = sorted(report.to_findings(), key=lambda f: -prio(f))[:limit]
        return {"queue": [{
            "finding_id": f.id, "title": f.title, "severity": f.severity,
            "triage_priority": prio(f), "risk_score": f.risk_score,
            "is_kev": f.is_kev, "epss_score": f.epss_score,
            "reachability": f.reachability, "fixed_version": f.fixed_version,
        } for f in findings]}

    @tool("diff", "Diff the current scan against a saved report JSON "
                  "(new / resolved / persisting findings).",
          {"type": "object", "properties": {"baseline_path": {"type": "string"}},
           "required": ["baseline_path"]})
    def diff(baseline_path: str) -> dict:
        from agentbom_amd.output.json_fmt import to_json
        from agentbom_amd.scan.history import diff_reports

        report, _g = server._ensure_scan()
        baseline = json.loads(Path(baseline_path).read_text())
        return diff_reports(baseline, to_json(report))

    @tool("risk_campaign_workflow",
          "List/create/assign/ticket/verify remediation campaigns. Writes "
          "require admin role + findings:write scope + audit reason.",
          {"type": "object", "properties": {
              "action": {"type": "string",
                         "enum": ["list", "candidates", "create", "assign",
                                  "ticket", "verify"]},
              "dimension": {"type": "string", "default": "package"},
              "campaign_id": {"type": "string"},
              "group_key": {"type": "string"},
              "assignee": {"type": "string"},
              "operator_role": {"type": "string", "default": ""},
              "operator_scopes": {"type": "string", "default": ""},
              "reason": {"type": "string", "default": ""}},
           "required": ["action"]})
    def risk_campaign_workflow(action: str, dimension: str = "package",
                               campaign_id: str = "", group_key: str = "",
                               assignee: str = "", operator_role: str = "",
                               operator_scopes: str = "", reason: str = "") -> dict:
        from agentbom_amd.scan.campaigns import group_findings

        store = server.campaign_store
        if action == "list":
            return {"campaigns": store.list()}
        report, _g = server._ensure_scan()
        if action == "candidates":
            return {"candidates": group_findings(report, dimension=dimension)[:20]}
        ok, blocked = authorize_write(
            action=f"campaign.{action}", operator_role=operator_role,
            operator_scopes=operator_scopes, reason=reason,
            required_scope="findings:write")
        if not ok:
            return blocked
        if action == "create":
            for g in group_findings(report, dimension=dimension):
                if g["group_key"] == group_key:
                    return {"campaign": store.create(g, created_by="mcp-operator")}
            return {"error": f"no candidate group {group_key!r}"}
        if action == "assign":
            c = store.assign(campaign_id, assignee)
            return {"campaign": c} if c else {"error": "unknown campaign"}
        if action == "ticket":
            c = store.get(campaign_id)
            if c is None:
                return {"error": "unknown campaign"}
            t = server.ticket_store.create(
                finding_id=c["finding_ids"][0] if c["finding_ids"] else campaign_id,
                title=f"Remediate {c['group_key']} ({c['finding_count'] if 'finding_count' in c else len(c['finding_ids'])} findings)",
                description=f"Campaign {campaign_id}: fix {c['group_key']}; "
                            f"fix versions {c['fix_versions']}",
                severity="high" if c["max_risk"] >= 7 else "medium")
            store.attach_ticket(campaign_id, t["ticket_id"])
            return {"ticket": t}
        if action == "verify":
            c = store.verify(campaign_id, report)
            return {"campaign": c} if c else {"error": "unknown campaign"}
        return {"error": f"unknown action {action!r}"}

    # ── ticketing ─────────────────────────────────────────────────────────

    @tool("create_ticket", "Create a remediation ticket for a finding "
                           "(renders Jira/GitHub payloads; dedups open tickets). "
                           "Requires admin + findings:write + reason.",
          {"type": "object", "properties": {
              "finding_id": {"type": "string"}, "title": {"type": "string"},
              "description": {"type": "string"},
              "severity": {"type": "string", "default": "medium"},
              "tracker": {"type": "string", "enum": ["jira", "github"],
                          "default": "jira"},
              "operator_role": {"type": "string", "default": ""},
              "operator_scopes": {"type": "string", "default": ""},
              "reason": {"type": "string", "default": ""}},
           "required": ["finding_id", "title", "description"]})
    def create_ticket(finding_id: str, title: str, description: str,
                      severity: str = "medium", tracker: str = "jira",
                      operator_role: str = "", operator_scopes: str = "",
                      reason: str = "") -> dict:
        ok, blocked = authorize_write(
            action="ticket.create", operator_role=operator_role,
            operator_scopes=operator_scopes, reason=reason,
            required_scope="findings:write")
        if not ok:
            return blocked
        return server.ticket_store.create(finding_id, title, description,
                                          severity=severity, tracker=tracker)

    @tool("sync_ticket_status", "Reconcile ticket statuses from a tracker "
                                "export file ({'issues': [{ref, status}]}).",
          {"type": "object", "properties": {"export_path": {"type": "string"}},
           "required": ["export_path"]})
    def sync_ticket_status(export_path: str) -> dict:
        export = json.loads(Path(export_path).read_text())
        return server.ticket_store.sync_from_export(export)

    # ── code / cloud / kspm / marketplace ─────────────────────────────────

    @tool("code_scan", "AST scan a source tree: dangerous sinks + symbol "
                       "reachability index.",
          {"type": "object", "properties": {"path": {"type": "string"}},
           "required": ["path"]})
    def code_scan(path: str) -> dict:
        from agentbom_amd.scan.ast_analysis import analyze_python_source

        findings, symbols = [], set()
        root = Path(path)
        files = [root] if root.is_file() else sorted(root.rglob("*.py"))[:500]
        for f in files:
            try:
                hits, syms = analyze_python_source(f.read_text(), str(f))
            except (OSError, UnicodeDecodeError, SyntaxError):
                continue
            findings.extend(a.to_dict() for a in hits)
            symbols |= syms
        return {"files_scanned": len(files), "findings": findings,
                "symbols_indexed": len(symbols)}

    @tool("cloud_side_scan", "Evaluate an exported cloud inventory (AWS JSON) "
                             "against the CIS / AI-infra check packs.",
          {"type": "object", "properties": {"inventory_path": {"type": "string"}},
           "required": ["inventory_path"]})
    def cloud_side_scan(inventory_path: str) -> dict:
        from agentbom_amd.scan.cloud import scan_cloud_inventory

        results = scan_cloud_inventory(inventory_path)
        return {"checks": [r.to_dict() for r in results],
                "failed": sum(1 for r in results if r.status == "fail")}

    @tool("cloud_inventory", "Summarize an exported cloud inventory file.",
          {"type": "object", "properties": {"inventory_path": {"type": "string"}},
           "required": ["inventory_path"]})
    def cloud_inventory(inventory_path: str) -> dict:
        from agentbom_amd.scan.cloud import load_inventory

        inv = load_inventory(inventory_path)
        return {k: (len(v) if isinstance(v, list) else v) for k, v in inv.items()}

    @tool("cis_benchmark", "CIS-style benchmark over an exported AWS inventory.",
          {"type": "object", "properties": {"inventory_path": {"type": "string"}},
           "required": ["inventory_path"]})
    def cis_benchmark(inventory_path: str) -> dict:
        from agentbom_amd.scan.cloud import evaluate_aws_inventory, load_inventory

        results = evaluate_aws_inventory(load_inventory(inventory_path))
        by_status: dict[str, int] = defaultdict(int)
        for r in results:
            by_status[r.status] += 1
        return {"summary": dict(by_status), "checks": [r.to_dict() for r in results]}

    @tool("kspm_cluster_posture", "Kubernetes posture: exported cluster "
                                  "inventory (runtime pods/RBAC, evidence "
                                  "envelope) or a manifest tree.",
          {"type": "object", "properties": {"path": {"type": "string"}},
           "required": ["path"]})
    def kspm_cluster_posture(path: str) -> dict:
        import os as _os

        if _os.path.isfile(path):
            from agentbom_amd.scan.kspm import scan_cluster_posture

            return scan_cluster_posture(path).to_evidence_dict()
        from agentbom_amd.scan.iac import scan_iac_tree

        findings = [f.to_dict() for f in scan_iac_tree(path)
                    if f.rule_id.startswith("K8S")]
        return {"schema_version": "kspm.manifests.v1",
                "finding_count": len(findings), "findings": findings}

    @tool("marketplace_check", "Pre-install trust check for an MCP server "
                               "package: typosquat, malicious list, advisories.",
          {"type": "object", "properties": {
              "name": {"type": "string"},
              "ecosystem": {"type": "string", "default": "npm"}},
           "required": ["name"]})
    def marketplace_check(name: str, ecosystem: str = "npm") -> dict:
        from agentbom_amd.db.store import load_advisory_windows
        from agentbom_amd.mcp.registry import check_blocklist, lookup_package
        from agentbom_amd.models import Package
        from agentbom_amd.scan.malicious import check_typosquat, flag_malicious_packages
        from agentbom_amd.utils.canonical_ids import normalize_package_name

        pkg = Package(name=name, version="0.0.0", ecosystem=ecosystem)
        flag_malicious_packages([pkg])
        typo = check_typosquat(name, ecosystem)
        norm = normalize_package_name(name, ecosystem)
        advisories = [w.vuln_id for w in load_advisory_windows(offline=True)
                      if w.ecosystem.lower() == ecosystem.lower()
                      and normalize_package_name(w.package_name, w.ecosystem) == norm]
        registry_entry = lookup_package(name)
        blocked = check_blocklist(name)
        high_risk = bool(registry_entry
                         and registry_entry.get("risk_level") in ("high", "critical"))
        verdict = ("block" if pkg.is_malicious or blocked
                   else "warn" if typo or advisories or high_risk else "allow")
        return {"package": name, "ecosystem": ecosystem, "verdict": verdict,
                "is_malicious": pkg.is_malicious,
                "malicious_reason": pkg.malicious_reason,
                "blocklist_hit": blocked,
                "registry": ({k: registry_entry.get(k) for k in
                              ("name", "risk_level", "verified", "category",
                               "credential_env_vars", "risk_justification")}
                             if registry_entry else None),
                "typosquat_of": typo, "known_advisories": sorted(set(advisories))}

    # ── graph ─────────────────────────────────────────────────────────────

    @tool("context_graph", "Bounded neighborhood subgraph around one node.",
          {"type": "object", "properties": {
              "node_id": {"type": "string"},
              "hops": {"type": "integer", "default": 2}},
           "required": ["node_id"]})
    def context_graph(node_id: str, hops: int = 2) -> dict:
        _r, g = server._ensure_scan()
        if node_id not in g.nodes:
            return {"error": f"node {node_id!r} not found"}
        sub = g.traverse_subgraph(node_id, max_depth=min(hops, 4))
        return {"center": node_id, **sub}

    @tool("graph_export", "Export the estate graph (dot/mermaid/graphml/cypher).",
          {"type": "object", "properties": {
              "format": {"type": "string",
                         "enum": ["dot", "mermaid", "graphml", "cypher"],
                         "default": "mermaid"}}})
    def graph_export_tool(format: str = "mermaid") -> dict:
        from agentbom_amd.output import graph_export as ge

        report, _g = server._ensure_scan()
        fn = {"dot": ge.to_dot, "mermaid": ge.to_mermaid,
              "graphml": ge.to_graphml, "cypher": ge.to_cypher}[format]
        return {"format": format, "text": fn(report)}

    @tool("analytics_query", "Aggregate findings: counts and risk by severity/"
                             "ecosystem/agent/impact; optional snapshot trend.",
          {"type": "object", "properties": {
              "group_by": {"type": "string",
                           "enum": ["severity", "ecosystem", "agent", "impact"],
                           "default": "severity"}}})
    def analytics_query(group_by: str = "severity") -> dict:
        report, _g = server._ensure_scan()
        agg: dict[str, dict] = defaultdict(lambda: {"count": 0, "total_risk": 0.0})
        for br in report.blast_radii:
            if group_by == "severity":
                keys = [br.vulnerability.severity.value]
            elif group_by == "ecosystem":
                keys = [br.package.ecosystem]
            elif group_by == "impact":
                keys = [br.impact_category]
            else:
                keys = [a.name for a in br.affected_agents] or ["unattributed"]
            for k in keys:
                agg[k]["count"] += 1
                agg[k]["total_risk"] = round(agg[k]["total_risk"] + br.risk_score, 2)
        return {"group_by": group_by, "groups": dict(agg)}

    # ── runtime evidence / correlation ────────────────────────────────────

    @tool("runtime_correlate", "Join scan findings with a proxy audit log: "
                               "which vulnerable tools were ACTUALLY called.",
          {"type": "object", "properties": {"audit_path": {"type": "string"}},
           "required": ["audit_path"]})
    def runtime_correlate(audit_path: str) -> dict:
        report, _g = server._ensure_scan()
        calls = _audit_tool_calls(_load_jsonl(audit_path))
        confirmed, theoretical = [], []
        for br in report.blast_radii:
            tools = {t.name for t in br.exposed_tools}
            called = {t: calls[t] for t in tools if t in calls}
            row = {"vulnerability_id": br.vulnerability.id,
                   "package": f"{br.package.name}@{br.package.version}",
                   "risk_score": br.risk_score,
                   "called_tools": called,
                   "total_calls": sum(called.values())}
            if called:
                # risk amplification: confirmed attack surface outranks theory
                row["amplified_risk"] = round(min(
                    br.risk_score + min(2.0, math.log10(1 + row["total_calls"])),
                    10.0), 2)
                confirmed.append(row)
            else:
                theoretical.append(row)
        confirmed.sort(key=lambda r: -r["amplified_risk"])
        return {"confirmed_attack_surface": confirmed,
                "theoretical_only": len(theoretical),
                "audited_tool_calls": sum(calls.values())}

    @tool("runtime_production_index", "Metadata-only runtime posture from a "
                                      "proxy audit log (no payloads returned).",
          {"type": "object", "properties": {"audit_path": {"type": "string"}},
           "required": ["audit_path"]})
    def runtime_production_index(audit_path: str) -> dict:
        rows = _load_jsonl(audit_path)
        calls = [r for r in rows if r.get("method") == "tools/call"]
        blocked = [r for r in rows if r.get("action") == "block"]
        sessions = {r.get("session") for r in rows if r.get("session")}
        ts = [float(r["ts"]) for r in rows if isinstance(r.get("ts"), (int, float))]
        return {
            "total_frames": len(rows),
            "tool_calls": len(calls),
            "distinct_tools": len({r.get("tool") for r in calls if r.get("tool")}),
            "block_rate": round(len(blocked) / max(len(rows), 1), 4),
            "active_sessions": len(sessions),
            "freshness_s": round(time.time() - max(ts), 1) if ts else None,
        }

    @tool("runtime_blueprints", "Canonical role/profile blueprints for runtime "
                                "policy design.")
    def runtime_blueprints() -> dict:
        return {"blueprints": RUNTIME_BLUEPRINTS}

    @tool("runtime_blueprint_drift", "Observed runtime traffic outside an "
                                     "approved role blueprint.",
          {"type": "object", "properties": {
              "audit_path": {"type": "string"},
              "blueprint": {"type": "string"}},
           "required": ["audit_path", "blueprint"]})
    def runtime_blueprint_drift(audit_path: str, blueprint: str) -> dict:
        from agentbom_amd.models import MCPTool
        from agentbom_amd.scan.risk import classify_mcp_tool

        bp = RUNTIME_BLUEPRINTS.get(blueprint)
        if bp is None:
            return {"error": f"unknown blueprint {blueprint!r}",
                    "available": sorted(RUNTIME_BLUEPRINTS)}
        denied = set(bp["denied_capabilities"])
        calls = _audit_tool_calls(_load_jsonl(audit_path))
        violations = []
        for tool_name, n in sorted(calls.items(), key=lambda kv: -kv[1]):
            caps = {c.value for c in classify_mcp_tool(
                MCPTool(name=tool_name, description=""))}
            bad = sorted(caps & denied)
            if bad:
                violations.append({"tool": tool_name, "calls": n,
                                   "violating_capabilities": bad})
        drift_score = round(min(
            sum(v["calls"] for v in violations) / max(sum(calls.values()), 1), 1.0), 4)
        return {"blueprint": blueprint, "drift_score": drift_score,
                "violations": violations}

    @tool("drift_incidents", "Open blueprint-drift incidents across all "
                             "blueprints for one audit log.",
          {"type": "object", "properties": {"audit_path": {"type": "string"}},
           "required": ["audit_path"]})
    def drift_incidents(audit_path: str) -> dict:
        incidents = []
        for name in RUNTIME_BLUEPRINTS:
            d = runtime_blueprint_drift(audit_path, name)
            if d.get("violations"):
                incidents.append({"blueprint": name,
                                  "drift_score": d["drift_score"],
                                  "top_violations": d["violations"][:5]})
        return {"open_incidents": incidents}

    # ── cost ──────────────────────────────────────────────────────────────

    def _load_spans(spans_path: str) -> list[dict]:
        return [s for s in _load_jsonl(spans_path)
                if "input_tokens" in s or "output_tokens" in s]

    @tool("cost_report", "LLM spend attribution from exported OTel GenAI spans "
                         "(token counts only; prompts are never read).",
          {"type": "object", "properties": {"spans_path": {"type": "string"}},
           "required": ["spans_path"]})
    def cost_report(spans_path: str) -> dict:
        spans = _load_spans(spans_path)
        by_agent: dict[str, float] = defaultdict(float)
        by_model: dict[str, float] = defaultdict(float)
        for s in spans:
            c = _span_cost(s, DEFAULT_PRICES)
            by_agent[str(s.get("agent", "unknown"))] += c
            by_model[str(s.get("model", "unknown"))] += c
        total = sum(by_agent.values())
        return {"total_usd": round(total, 4),
                "spans": len(spans),
                "by_agent": {k: round(v, 4) for k, v in
                             sorted(by_agent.items(), key=lambda kv: -kv[1])},
                "by_model": {k: round(v, 4) for k, v in
                             sorted(by_model.items(), key=lambda kv: -kv[1])}}

    @tool("cost_forecast", "Project 30-day spend from the span window's "
                           "observed daily run rate.",
          {"type": "object", "properties": {"spans_path": {"type": "string"}},
           "required": ["spans_path"]})
    def cost_forecast(spans_path: str) -> dict:
        spans = _load_spans(spans_path)
        ts = [float(s["ts"]) for s in spans if isinstance(s.get("ts"), (int, float))]
        total = sum(_span_cost(s, DEFAULT_PRICES) for s in spans)
        window_days = max((max(ts) - min(ts)) / 86400, 1 / 24) if len(ts) > 1 else 1.0
        daily = total / window_days
        return {"window_days": round(window_days, 2),
                "observed_usd": round(total, 4),
                "daily_run_rate_usd": round(daily, 4),
                "forecast_30d_usd": round(daily * 30, 2)}

    @tool("cost_allocation", "Spend per (agent, model) pair for chargeback.",
          {"type": "object", "properties": {"spans_path": {"type": "string"}},
           "required": ["spans_path"]})
    def cost_allocation(spans_path: str) -> dict:
        rows: dict[tuple, dict] = {}
        for s in _load_spans(spans_path):
            key = (str(s.get("agent", "unknown")), str(s.get("model", "unknown")))
            r = rows.setdefault(key, {"agent": key[0], "model": key[1],
                                      "spans": 0, "usd": 0.0})
            r["spans"] += 1
            r["usd"] = round(r["usd"] + _span_cost(s, DEFAULT_PRICES), 6)
        return {"allocation": sorted(rows.values(), key=lambda r: -r["usd"])}

    @tool("anomaly_scan", "Statistical outliers: per-agent spend and per-session "
                          "tool-call rates beyond mean+3sigma.",
          {"type": "object", "properties": {
              "spans_path": {"type": "string"},
              "audit_path": {"type": "string"}}})
    def anomaly_scan(spans_path: str = "", audit_path: str = "") -> dict:
        def outliers(values: dict[str, float]) -> list[dict]:
            if len(values) < 3:
                return []
            vals = list(values.values())
            mean = sum(vals) / len(vals)
            var = sum((v - mean) ** 2 for v in vals) / len(vals)
            sigma = math.sqrt(var)
            cut = mean + 3 * sigma
            return [{"key": k, "value": round(v, 4),
                     "threshold": round(cut, 4)}
                    for k, v in values.items() if sigma > 0 and v > cut]

        result: dict[str, Any] = {"spend_outliers": [], "call_rate_outliers": []}
        if spans_path:
            by_agent: dict[str, float] = defaultdict(float)
            for s in _load_spans(spans_path):
                by_agent[str(s.get("agent", "unknown"))] += _span_cost(s, DEFAULT_PRICES)
            result["spend_outliers"] = outliers(by_agent)
        if audit_path:
            by_session: dict[str, float] = defaultdict(float)
            for r in _load_jsonl(audit_path):
                if r.get("method") == "tools/call":
                    by_session[str(r.get("session", "unknown"))] += 1
            result["call_rate_outliers"] = outliers(by_session)
        return result

    # ── proxy / gateway / shield ──────────────────────────────────────────

    @tool("proxy_status", "Verify a proxy audit log's hash chain + frame counts.",
          {"type": "object", "properties": {"audit_path": {"type": "string"}},
           "required": ["audit_path"]})
    def proxy_status(audit_path: str) -> dict:
        from agentbom_amd.runtime.proxy import AuditLog

        intact, n = AuditLog.verify(audit_path)
        rows = _load_jsonl(audit_path)
        return {"chain_intact": intact, "entries": n,
                "blocked": sum(1 for r in rows if r.get("action") == "block"),
                "warned": sum(1 for r in rows if r.get("action") == "warn")}

    @tool("proxy_alerts", "Detector alerts recorded in a proxy audit log.",
          {"type": "object", "properties": {
              "audit_path": {"type": "string"},
              "limit": {"type": "integer", "default": 50}},
           "required": ["audit_path"]})
    def proxy_alerts(audit_path: str, limit: int = 50) -> dict:
        alerts = []
        for r in _load_jsonl(audit_path):
            for a in r.get("alerts") or []:
                alerts.append({"ts": r.get("ts"), "tool": r.get("tool"),
                               "action": r.get("action"), **a})
        return {"alerts": alerts[-limit:], "total": len(alerts)}

    @tool("gateway_status", "Gateway posture: upstreams, breaker states, "
                            "quarantines.")
    def gateway_status() -> dict:
        gw = getattr(server, "_gateway", None)
        if gw is None:
            return {"upstreams": {}, "note": "no gateway running in this session"}
        return {"metrics": dict(gw.metrics),
                "upstreams": {name: {"quarantined": up.quarantined,
                                     "breaker": up.breaker.state()}
                              for name, up in gw.upstreams.items()}}

    @tool("shield_status", "Shield enforcement state: blocked tools, revoked "
                           "credentials, quarantined upstreams.")
    def shield_status() -> dict:
        s = server.shield
        return {"enabled": True,
                "blocked_tools": sorted(s.blocked_tools),
                "revoked_credentials_count": len(s.revoked_credentials),
                "quarantined_upstreams": sorted(s.quarantined_upstreams),
                "actions_applied": len(getattr(s, "action_log", []))}

    def _shield_write(action: str, target: str, operator_role: str,
                      operator_scopes: str, reason: str) -> dict:
        ok, blocked = authorize_write(
            action=f"shield.{action}", operator_role=operator_role,
            operator_scopes=operator_scopes, reason=reason,
            required_scope="shield:write")
        if not ok:
            return blocked
        return server.shield.apply_write_action(action, target, admin=True,
                                                reason=reason)

    _WRITE_SCHEMA = {"type": "object", "properties": {
        "target": {"type": "string"},
        "operator_role": {"type": "string", "default": ""},
        "operator_scopes": {"type": "string", "default": ""},
        "reason": {"type": "string", "default": ""}},
        "required": ["target"]}

    @tool("shield_start", "Block a tool estate-wide (fail-closed). Requires "
                          "admin + shield:write + reason.", _WRITE_SCHEMA)
    def shield_start(target: str, operator_role: str = "",
                     operator_scopes: str = "", reason: str = "") -> dict:
        return _shield_write("block_tool", target, operator_role,
                             operator_scopes, reason)

    @tool("shield_unblock", "Remove a shield tool block. Requires admin + "
                            "shield:write + reason.", _WRITE_SCHEMA)
    def shield_unblock(target: str, operator_role: str = "",
                       operator_scopes: str = "", reason: str = "") -> dict:
        ok, blocked = authorize_write(
            action="shield.unblock", operator_role=operator_role,
            operator_scopes=operator_scopes, reason=reason,
            required_scope="shield:write")
        if not ok:
            return blocked
        server.shield.blocked_tools.discard(target)
        return {"action": "unblock", "target": target, "status": "applied"}

    @tool("shield_break_glass", "Quarantine an upstream NOW (incident "
                                "response). Requires admin + shield:write + reason.",
          _WRITE_SCHEMA)
    def shield_break_glass(target: str, operator_role: str = "",
                           operator_scopes: str = "", reason: str = "") -> dict:
        return _shield_write("quarantine_upstream", target, operator_role,
                             operator_scopes, reason)

    @tool("firewall_check", "Dry-run one JSON-RPC frame through the shield "
                            "decision pipeline (no enforcement).",
          {"type": "object", "properties": {"frame": {"type": "object"}},
           "required": ["frame"]})
    def firewall_check(frame: dict) -> dict:
        return server.shield.decide(frame).to_dict()

    # ── identity lifecycle (write-gated) ──────────────────────────────────

    _ID_WRITE = {"operator_role": {"type": "string", "default": ""},
                 "operator_scopes": {"type": "string", "default": ""},
                 "reason": {"type": "string", "default": ""}}

    @tool("identity_issue", "Issue a scoped agent identity (token returned "
                            "once). Requires admin + identity:write + reason.",
          {"type": "object", "properties": {
              "agent_name": {"type": "string"},
              "scopes": {"type": "string", "default": ""},
              "allowed_tools": {"type": "string", "default": ""},
              "ttl_hours": {"type": "number", "default": 24}, **_ID_WRITE},
           "required": ["agent_name"]})
    def identity_issue(agent_name: str, scopes: str = "", allowed_tools: str = "",
                       ttl_hours: float = 24, operator_role: str = "",
                       operator_scopes: str = "", reason: str = "") -> dict:
        ok, blocked = authorize_write(
            action="identity.issue", operator_role=operator_role,
            operator_scopes=operator_scopes, reason=reason,
            required_scope="identity:write")
        if not ok:
            return blocked
        ident, raw = server.identity_store.issue(
            agent_name,
            scopes=[s for s in scopes.split(",") if s.strip()],
            allowed_tools=[t for t in allowed_tools.split(",") if t.strip()],
            ttl_hours=ttl_hours, actor="mcp-operator", reason=reason)
        return {"identity": ident.to_public_dict(), "token": raw,
                "note": "token is shown exactly once"}

    @tool("identity_rotate", "Rotate an identity with an overlap window. "
                             "Requires admin + identity:write + reason.",
          {"type": "object", "properties": {
              "identity_id": {"type": "string"},
              "overlap_minutes": {"type": "number", "default": 15}, **_ID_WRITE},
           "required": ["identity_id"]})
    def identity_rotate(identity_id: str, overlap_minutes: float = 15,
                        operator_role: str = "", operator_scopes: str = "",
                        reason: str = "") -> dict:
        ok, blocked = authorize_write(
            action="identity.rotate", operator_role=operator_role,
            operator_scopes=operator_scopes, reason=reason,
            required_scope="identity:write")
        if not ok:
            return blocked
        new, raw = server.identity_store.rotate(
            identity_id, overlap_minutes=overlap_minutes,
            actor="mcp-operator", reason=reason)
        if new is None:
            return {"error": "identity not found or not live"}
        return {"identity": new.to_public_dict(), "token": raw}

    @tool("identity_revoke", "Revoke an identity immediately. Requires admin "
                             "+ identity:write + reason.",
          {"type": "object", "properties": {
              "identity_id": {"type": "string"}, **_ID_WRITE},
           "required": ["identity_id"]})
    def identity_revoke(identity_id: str, operator_role: str = "",
                        operator_scopes: str = "", reason: str = "") -> dict:
        ok, blocked = authorize_write(
            action="identity.revoke", operator_role=operator_role,
            operator_scopes=operator_scopes, reason=reason,
            required_scope="identity:write")
        if not ok:
            return blocked
        return {"revoked": server.identity_store.revoke(
            identity_id, actor="mcp-operator", reason=reason)}

    @tool("identity_grant_jit", "Grant a time-boxed JIT scope elevation. "
                                "Requires admin + identity:write + reason.",
          {"type": "object", "properties": {
              "identity_id": {"type": "string"},
              "scopes": {"type": "string"},
              "ttl_minutes": {"type": "number", "default": 60}, **_ID_WRITE},
           "required": ["identity_id", "scopes"]})
    def identity_grant_jit(identity_id: str, scopes: str, ttl_minutes: float = 60,
                           operator_role: str = "", operator_scopes: str = "",
                           reason: str = "") -> dict:
        ok, blocked = authorize_write(
            action="identity.grant_jit", operator_role=operator_role,
            operator_scopes=operator_scopes, reason=reason,
            required_scope="identity:write")
        if not ok:
            return blocked
        g = server.identity_store.grant_jit(
            identity_id, [s for s in scopes.split(",") if s.strip()],
            reason=reason, granted_by="mcp-operator", ttl_minutes=ttl_minutes)
        return {"grant": g.to_public_dict()} if g else {"error": "unknown identity"}

    @tool("identity_revoke_jit", "Revoke a JIT grant early. Requires admin + "
                                 "identity:write + reason.",
          {"type": "object", "properties": {
              "grant_id": {"type": "string"}, **_ID_WRITE},
           "required": ["grant_id"]})
    def identity_revoke_jit(grant_id: str, operator_role: str = "",
                            operator_scopes: str = "", reason: str = "") -> dict:
        ok, blocked = authorize_write(
            action="identity.revoke_jit", operator_role=operator_role,
            operator_scopes=operator_scopes, reason=reason,
            required_scope="identity:write")
        if not ok:
            return blocked
        return {"revoked": server.identity_store.revoke_jit(
            grant_id, actor="mcp-operator", reason=reason)}

    # ── identity reads ────────────────────────────────────────────────────

    @tool("nhi_discover", "Discover non-human identities from Okta/Entra "
                          "export files (offline) or env-configured connectors.",
          {"type": "object", "properties": {
              "okta_export": {"type": "string"},
              "entra_export": {"type": "string"}}})
    def nhi_discover(okta_export: str = "", entra_export: str = "") -> dict:
        from agentbom_amd.identity.nhi import discover_entra_nhis, discover_okta_nhis

        return {
            "okta": discover_okta_nhis(
                export_path=okta_export or None).to_dict(),
            "entra": discover_entra_nhis(
                export_path=entra_export or None).to_dict(),
        }

    @tool("credential_expiry", "Identities whose credentials expire inside "
                               "the window (issued + discovered).",
          {"type": "object", "properties": {
              "within_hours": {"type": "number", "default": 72}}})
    def credential_expiry(within_hours: float = 72) -> dict:
        return {"expiring": server.identity_store.credential_expiry_report(
            within_hours=within_hours)}

    @tool("access_review", "Estate access posture: live identities, wildcard "
                           "scopes, live JIT grants, audit chain state.")
    def access_review() -> dict:
        return server.identity_store.access_review()

    @tool("audit_query", "Query the identity lifecycle audit log.",
          {"type": "object", "properties": {
              "action_prefix": {"type": "string", "default": ""},
              "limit": {"type": "integer", "default": 100}}})
    def audit_query(action_prefix: str = "", limit: int = 100) -> dict:
        entries = server.identity_store.audit_entries()
        if action_prefix:
            entries = [e for e in entries if e["action"].startswith(action_prefix)]
        return {"entries": entries[-limit:], "total": len(entries)}

    @tool("audit_integrity", "Verify the identity audit hash chain end-to-end.")
    def audit_integrity() -> dict:
        return {"chain_valid": server.identity_store.audit_chain_valid(),
                "entries": len(server.identity_store.audit_entries())}

    # ── fleet ─────────────────────────────────────────────────────────────

    @tool("fleet_scan", "Fleet posture: registered members, staleness, "
                        "schedules due.")
    def fleet_scan() -> dict:
        from agentbom_amd.api.fleet import FleetRegistry, ScanScheduler

        fleet = getattr(server, "_fleet", None)
        if fleet is None:
            fleet = (FleetRegistry(), ScanScheduler(run_scan=lambda cfg: None))
            server._fleet = fleet
        registry, scheduler = fleet
        return {"members": registry.list_members(),
                "schedules": [s.to_dict() for s in scheduler.schedules.values()]}

    @tool("effective_permissions", "Per-agent effective tool permissions "
                                   "with grant provenance + least-privilege "
                                   "recommendations.")
    def effective_permissions() -> dict:
        from agentbom_amd.graph.effective_permissions import (
            compute_effective_permissions,
            least_privilege_recommendations,
        )
        from agentbom_amd.graph.nhi_overlay import apply_issued_identity_overlay

        _r, g = server._ensure_scan()
        apply_issued_identity_overlay(g, server.identity_store)
        perms = compute_effective_permissions(g)
        perms["recommendations"] = least_privilege_recommendations(g)
        return perms

    @tool("mcp_auth_posture", "MCP server auth posture: remote no-auth, "
                              "plaintext transport, static tokens vs OAuth.")
    def mcp_auth_posture() -> dict:
        from agentbom_amd.scan.auth_posture import assess_estate

        report, _g = server._ensure_scan()
        return assess_estate(report.agents)

    @tool("a2a_auth_posture", "Inter-agent auth posture: shared static "
                              "credentials, missing mutual auth, over-broad "
                              "delegation, transitive delegation webs.")
    def a2a_auth_posture() -> dict:
        from agentbom_amd.scan.auth_posture import assess_a2a

        report, _g = server._ensure_scan()
        return assess_a2a(report.agents, identity_store=server.identity_store)

    @tool("self_posture", "agent-bom audits its OWN deployment hardening "
                          "(auth, RBAC, tenancy, persistence, signing).")
    def self_posture() -> dict:
        from agentbom_amd.scan.self_posture import evaluate_self_posture

        return evaluate_self_posture()

    @tool("attest_server", "Sign a DSSE scan attestation for one scanned "
                           "MCP server (key from AGENT_BOM_ATTESTATION_KEY).",
          {"type": "object", "properties": {
              "server_name": {"type": "string"},
              "verdict": {"type": "string", "enum": ["pass", "warn", "block"],
                          "default": "pass"}},
           "required": ["server_name"]})
    def attest_server(server_name: str, verdict: str = "pass") -> dict:
        import os

        from agentbom_amd.mcp.server import resolve_mcp_tenant_id
        from agentbom_amd.utils.attestation import attest_scanned_server

        key_hex = os.environ.get("AGENT_BOM_ATTESTATION_KEY", "")
        if not key_hex:
            return {"error": "AGENT_BOM_ATTESTATION_KEY not configured"}
        report, _g = server._ensure_scan()
        target = next((s for a in report.agents for s in a.mcp_servers
                       if s.name == server_name), None)
        if target is None:
            return {"error": f"server {server_name!r} not in latest scan"}
        return attest_scanned_server(
            target, verdict, bytes.fromhex(key_hex),
            key_id=os.environ.get("AGENT_BOM_ATTESTATION_KEY_ID", "operator"),
            tenant_id=resolve_mcp_tenant_id())

    @tool("verify_attestation", "Verify a DSSE scan attestation against the "
                                "operator trust policy (pinned keys only).",
          {"type": "object", "properties": {
              "envelope": {"type": "object"},
              "expected_tenant": {"type": "string", "default": ""}},
           "required": ["envelope"]})
    def verify_attestation_tool(envelope: dict, expected_tenant: str = "") -> dict:
        import os

        from agentbom_amd.utils.attestation import verify_attestation

        key_hex = os.environ.get("AGENT_BOM_ATTESTATION_KEY", "")
        key_id = os.environ.get("AGENT_BOM_ATTESTATION_KEY_ID", "operator")
        policy = {"keys": ({key_id: key_hex} if key_hex else {}),
                  "expected_tenant": expected_tenant or None}
        return verify_attestation(envelope, policy)

    @tool("trust_score", "Supply-chain trust score (0-100 + reasons) for a "
                         "package: malicious/typosquat/registry/advisory/"
                         "name signals.",
          {"type": "object", "properties": {
              "name": {"type": "string"},
              "version": {"type": "string", "default": ""},
              "ecosystem": {"type": "string", "default": "npm"}},
           "required": ["name"]})
    def trust_score_tool(name: str, version: str = "",
                         ecosystem: str = "npm") -> dict:
        from agentbom_amd.db.store import load_advisory_windows
        from agentbom_amd.models import Package
        from agentbom_amd.scan.trust import trust_score

        return trust_score(
            Package(name=name, version=version or "0.0.0", ecosystem=ecosystem),
            advisory_windows=load_advisory_windows(offline=True))

    @tool("effective_reach", "Effective-reach triage bands per finding: "
                             "composite of CVSS/EPSS/KEV/tool-capability/"
                             "credential-tier/agent-breadth.",
          {"type": "object", "properties": {
              "band": {"type": "string",
                       "enum": ["green", "amber", "red", "pulsing-red"]}}})
    def effective_reach(band: str = "") -> dict:
        from agentbom_amd.graph.effective_reach import effective_reach_summary

        report, _g = server._ensure_scan()
        out = effective_reach_summary(report)
        if band:
            out["findings"] = [r for r in out["findings"]
                               if r["band"] == band]
        return out

    @tool("maestro_posture", "MAESTRO KC-layer posture: findings classified "
                             "by agentic-AI layer (KC1 models .. KC6 infra).")
    def maestro_posture() -> dict:
        from agentbom_amd.models.maestro import maestro_summary

        report, _g = server._ensure_scan()
        return maestro_summary([f.to_dict() for f in report.to_findings()])

    @tool("cost_runway", "Budget runway from OTel GenAI span records: "
                         "burn-rate basis, projected period spend, days "
                         "remaining, exhaustion timestamp.",
          {"type": "object", "properties": {
              "spans_path": {"type": "string"},
              "budget_usd": {"type": "number"}},
           "required": ["spans_path"]})
    def cost_runway(spans_path: str, budget_usd: float = 0.0) -> dict:
        from agentbom_amd.api.cost_store import (
            CostBudget,
            LLMCostRecord,
            forecast_spend,
        )

        records = []
        for row in _load_jsonl(spans_path):
            records.append(LLMCostRecord(
                tenant_id="local", agent=str(row.get("agent", "?")),
                cost_usd=float(row.get("cost_usd", 0.0)),
                model=str(row.get("model", "")),
                observed_at=str(row.get("observed_at", ""))))
        budget = CostBudget("local", budget_usd) if budget_usd else None
        return forecast_spend(records, budget)

    @tool("tool_metrics", "Per-tool call/latency/error counters for this "
                          "MCP session.")
    def tool_metrics() -> dict:
        return {name: {**m, "avg_ms": round(m["total_ms"] / max(m["calls"], 1), 2)}
                for name, m in server.tool_metrics.items()}
