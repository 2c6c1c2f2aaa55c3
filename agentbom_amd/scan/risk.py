"""Tool capability classification + MCP server risk scoring.

Reference: src/agent_bom/scanners/risk.py, risk_analyzer.py — semantic
capability classes per tool and the capability-weighted server risk score
(weights from utils/config with identical defaults).
"""

from __future__ import annotations

from enum import Enum

from agentbom_amd.models import MCPServer, MCPTool
from agentbom_amd.utils import config as cfg


class ToolCapability(str, Enum):
    EXECUTE = "execute"
    READ = "read"
    WRITE = "write"
    NETWORK = "network"
    DELETE = "delete"
    QUERY = "query"


_CAP_KEYWORDS: dict[ToolCapability, tuple[str, ...]] = {
    ToolCapability.EXECUTE: ("exec", "shell", "run", "command", "eval", "spawn", "launch"),
    ToolCapability.READ: ("read", "get", "list", "search", "fetch", "view", "cat", "inbox"),
    ToolCapability.WRITE: ("write", "create", "push", "upload", "insert", "upsert", "send",
                           "post", "update", "transform", "load", "export", "reply"),
    ToolCapability.NETWORK: ("http", "request", "url", "download", "webhook", "api", "email"),
    ToolCapability.DELETE: ("delete", "remove", "drop", "destroy", "purge"),
    ToolCapability.QUERY: ("query", "sql", "select", "database"),
}


def classify_mcp_tool(tool: MCPTool) -> set[ToolCapability]:
    """Keyword-over-name/description capability classification."""
    text = f"{tool.name} {tool.description}".lower()
    caps = {cap for cap, kws in _CAP_KEYWORDS.items() if any(k in text for k in kws)}
    for f in tool.schema_findings:
        if "shell-execution" in f:
            caps.add(ToolCapability.EXECUTE)
        if "network-egress" in f:
            caps.add(ToolCapability.NETWORK)
        if "filesystem" in f:
            caps.add(ToolCapability.WRITE)
    return caps


def score_server_risk(server: MCPServer) -> float:
    """Capability-weighted server risk 0-10 (config.SERVER_RISK_* weights)."""
    all_caps: set[ToolCapability] = set()
    for t in server.tools:
        all_caps |= classify_mcp_tool(t)

    tool_factor = min(len(server.tools) * cfg.SERVER_RISK_TOOL_WEIGHT, cfg.SERVER_RISK_TOOL_CAP)
    cred_factor = min(
        len(server.credential_names) * cfg.SERVER_RISK_CRED_WEIGHT, cfg.SERVER_RISK_CRED_CAP
    )
    # toxic capability combos: execute+creds, execute+network, write+network
    combos = 0
    if ToolCapability.EXECUTE in all_caps and server.has_credentials:
        combos += 1
    if ToolCapability.EXECUTE in all_caps and ToolCapability.NETWORK in all_caps:
        combos += 1
    if ToolCapability.WRITE in all_caps and ToolCapability.NETWORK in all_caps:
        combos += 1
    combo_factor = min(combos * cfg.SERVER_RISK_COMBO_WEIGHT, cfg.SERVER_RISK_COMBO_CAP)

    base = min(
        (len(all_caps) / max(len(ToolCapability), 1)) * cfg.SERVER_RISK_BASE_CEILING,
        cfg.SERVER_RISK_BASE_CEILING,
    )
    score = min(base + tool_factor + cred_factor + combo_factor, 10.0)
    if server.registry_verified:
        score = max(score - 1.0, 0.0)
    return round(score, 2)


def server_risk_level(score: float) -> str:
    if score >= cfg.SERVER_RISK_CRITICAL_THRESHOLD:
        return "critical"
    if score >= cfg.SERVER_RISK_HIGH_THRESHOLD:
        return "high"
    if score >= cfg.SERVER_RISK_MEDIUM_THRESHOLD:
        return "medium"
    return "low"


def estate_exec_score(report) -> dict:
    """Executive estate score 0-100 (higher = healthier) with drivers.

    Reference analog: exec_score.py — one number for leadership, derived
    entirely from scan evidence: finding pressure (severity-weighted per
    asset), confirmed-exploit exposure (KEV/malicious), reachability
    discount, and coverage honesty (PARTIAL scans cap the score).
    """
    from agentbom_amd.models import ScanOutcome, active_blast_radii

    brs = active_blast_radii(report.blast_radii)
    assets = max(report.total_agents + report.total_servers, 1)
    sev_w = {"critical": 10.0, "high": 5.0, "medium": 2.0, "low": 0.5}
    pressure = sum(sev_w.get(b.vulnerability.severity.value, 0.0) for b in brs)
    pressure_per_asset = pressure / assets

    score = 100.0
    drivers = []

    deduction = min(pressure_per_asset * 8.0, 50.0)
    if deduction:
        drivers.append({"driver": "finding_pressure", "delta": -round(deduction, 1),
                        "detail": f"{len(brs)} active findings across {assets} assets"})
    score -= deduction

    kev = sum(1 for b in brs if b.vulnerability.is_kev)
    if kev:
        d = min(10.0 + 5.0 * (kev - 1), 25.0)
        drivers.append({"driver": "known_exploited", "delta": -round(d, 1),
                        "detail": f"{kev} KEV finding(s)"})
        score -= d
    mal = sum(1 for b in brs if b.package.is_malicious)
    if mal:
        d = min(15.0 + 5.0 * (mal - 1), 30.0)
        drivers.append({"driver": "malicious_packages", "delta": -round(d, 1),
                        "detail": f"{mal} known-malicious package(s)"})
        score -= d

    unreachable = sum(1 for b in brs if b.reachability == "unreachable")
    if brs and unreachable:
        credit = min(10.0 * unreachable / len(brs), 10.0)
        drivers.append({"driver": "reachability_pruning", "delta": round(credit, 1),
                        "detail": f"{unreachable}/{len(brs)} findings proven unreachable"})
        score += credit

    if report.scan_run and report.scan_run.outcome is not ScanOutcome.COMPLETE:
        drivers.append({"driver": "incomplete_evidence", "delta": "cap 60",
                        "detail": f"scan outcome {report.scan_run.outcome.value}"})
        score = min(score, 60.0)

    score = max(0.0, min(100.0, score))
    grade = ("A" if score >= 90 else "B" if score >= 75 else
             "C" if score >= 60 else "D" if score >= 40 else "F")
    return {"score": round(score, 1), "grade": grade, "drivers": drivers}
