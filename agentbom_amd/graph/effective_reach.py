"""Effective-reach triage scoring: incident, not vulnerable-package.

Reference parity: src/agent_bom/effective_reach.py + symbol_reach_triage.py
— a deterministic 0..100 composite per finding that answers "what can
this ACTUALLY do here": a vulnerable package behind a read-only search
tool and the same package behind run_shell + visible AWS_* names are
different incidents.

Coefficients and band thresholds carry the reference's semantics
exactly (scoring parity is the contract):

  (cvss/10)*30 + epss*20 + (40 if KEV) + tool_capability*25
  + cred_visibility*20 + min(agent_breadth,5)*5
  + symbol delta (function_reachable +15 / unreachable -30), clamped 0..100

Bands: >=90 pulsing-red, >70 red, >30 amber, else green.

This build computes the factors directly from BlastRadius rows (the
engine already materialized reachable tools/creds/agents per finding)
instead of re-walking a context graph.
"""

from __future__ import annotations

import re
from dataclasses import dataclass
from typing import Any, Optional

from agentbom_amd.scan.risk import ToolCapability, classify_mcp_tool

# tool capability -> 0..1 weight; multiple tools combine via MAX (one
# shell tool is enough; breadth is not double-counted here)
_CAPABILITY_WEIGHT = {
    ToolCapability.READ: 0.10,
    ToolCapability.QUERY: 0.30,
    ToolCapability.NETWORK: 0.40,
    ToolCapability.WRITE: 0.65,
    ToolCapability.DELETE: 0.75,
    ToolCapability.EXECUTE: 1.00,
}

_CLOUD_CRED = re.compile(
    r"^(AWS_|GCP_|GOOGLE_|AZURE_|OPENAI_|ANTHROPIC_|SNOWFLAKE_|DATABRICKS_|"
    r"K8S_|KUBE)", re.IGNORECASE)
_PROJECT_CRED = re.compile(
    r"(TOKEN|SECRET|KEY|PASSWORD|CREDENTIAL|DATABASE_URL|DSN|WEBHOOK)",
    re.IGNORECASE)
_HOME_CRED = re.compile(r"^(HOME|USER|PWD|SHELL|LANG|TERM|PATH|EDITOR)$",
                        re.IGNORECASE)

_SYMBOL_DELTA = {"function_reachable": 15.0, "package_reachable": 0.0,
                 "unreachable": -30.0}


def credential_tier(env_key: str) -> float:
    """0.10 HOME-scoped / 0.55 project-scoped / 1.0 cloud-API."""
    if _HOME_CRED.match(env_key):
        return 0.10
    if _CLOUD_CRED.match(env_key):
        return 1.00
    if _PROJECT_CRED.search(env_key):
        return 0.55
    return 0.25  # unknown name: above HOME, below a proven token


def tool_weight(tool) -> float:
    caps = classify_mcp_tool(tool)
    return max((_CAPABILITY_WEIGHT.get(c, 0.0) for c in caps), default=0.10)


@dataclass
class ReachScore:
    cvss: float
    epss: float
    is_kev: bool
    tool_capability: float
    cred_visibility: float
    agent_breadth: int
    reachable_tools: tuple = ()
    reachable_creds: tuple = ()
    reachable_agents: tuple = ()
    symbol_reachability: Optional[str] = None

    @property
    def composite(self) -> float:
        score = (
            (max(0.0, min(self.cvss, 10.0)) / 10.0) * 30.0
            + max(0.0, min(self.epss, 1.0)) * 20.0
            + (40.0 if self.is_kev else 0.0)
            + max(0.0, min(self.tool_capability, 1.0)) * 25.0
            + max(0.0, min(self.cred_visibility, 1.0)) * 20.0
            + max(0, min(self.agent_breadth, 5)) * 5.0
        )
        score += _SYMBOL_DELTA.get(self.symbol_reachability or "", 0.0)
        return round(max(0.0, min(score, 100.0)), 2)

    @property
    def band(self) -> str:
        c = self.composite
        if c >= 90.0:
            return "pulsing-red"
        if c > 70.0:
            return "red"
        if c > 30.0:
            return "amber"
        return "green"

    def as_breakdown(self) -> dict[str, Any]:
        return {
            "cvss": round(self.cvss, 2),
            "epss": round(self.epss, 4),
            "is_kev": self.is_kev,
            "tool_capability": round(self.tool_capability, 3),
            "cred_visibility": round(self.cred_visibility, 3),
            "agent_breadth": self.agent_breadth,
            "reachable_tools": list(self.reachable_tools),
            "reachable_creds": list(self.reachable_creds),
            "reachable_agents": list(self.reachable_agents),
            "symbol_reachability": self.symbol_reachability,
            "composite": self.composite,
            "band": self.band,
        }


def reach_score_for_blast(br) -> ReachScore:
    """Factor extraction from one BlastRadius row."""
    v = br.vulnerability
    tools = list(br.exposed_tools)
    creds = list(br.exposed_credentials)
    return ReachScore(
        cvss=float(v.cvss_score or 0.0),
        epss=float(v.epss_score or 0.0),
        is_kev=bool(v.is_kev),
        tool_capability=max((tool_weight(t) for t in tools), default=0.0),
        cred_visibility=max((credential_tier(c) for c in creds), default=0.0),
        agent_breadth=len(br.affected_agents),
        reachable_tools=tuple(t.name for t in tools[:8]),
        reachable_creds=tuple(creds[:8]),
        reachable_agents=tuple(a.name for a in br.affected_agents[:8]),
        symbol_reachability=br.symbol_reachability,
    )


def effective_reach_summary(report) -> dict[str, Any]:
    """Per-band counts + the scored findings list for the report/API."""
    rows = []
    bands = {"green": 0, "amber": 0, "red": 0, "pulsing-red": 0}
    for br in report.blast_radii:
        if br.suppressed:
            continue
        rs = reach_score_for_blast(br)
        bands[rs.band] += 1
        rows.append({"vulnerability_id": br.vulnerability.id,
                     "package": f"{br.package.name}@{br.package.version}",
                     **rs.as_breakdown()})
    rows.sort(key=lambda r: (-r["composite"], r["vulnerability_id"]))
    return {"schema_version": "1", "bands": bands, "findings": rows}
