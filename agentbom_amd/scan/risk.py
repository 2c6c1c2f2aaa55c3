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
