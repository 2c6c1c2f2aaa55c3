"""BlastRadius: per-(package, vulnerability) reach analysis + risk scoring.

Reference: src/agent_bom/models.py:906-1135 (model + calculate_risk_score +
reachability/is_actionable), src/agent_bom/scanners/blast_radius.py (multi-hop
delegation expansion), src/agent_bom/vex.py:537 (VEX suppression gate).

Risk-score math uses the config weights (identical defaults) and is also the
specification for the GPU scoring kernel (ops/csrc/blast.hip scores findings
with the same formula on u32 reach counts; parity-tested in tests/).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Optional

from agentbom_amd.models.core import Agent, MCPServer, MCPTool, Package, Severity, Vulnerability
from agentbom_amd.utils import config as cfg

_VEX_SUPPRESSED_STATUSES = {"not_affected", "fixed"}


def is_vex_suppressed(vuln: Any) -> bool:
    """True if VEX marks a vulnerability not_affected or fixed."""
    return str(getattr(vuln, "vex_status", "") or "").lower() in _VEX_SUPPRESSED_STATUSES


HOP_RISK_FACTORS: dict[int, float] = {1: 1.0, 2: 0.7, 3: 0.5, 4: 0.35, 5: 0.25}


def risk_score_from_counts(
    *,
    severity: Severity,
    n_agents: int,
    n_creds: int,
    n_tools: int,
    has_ai_context: bool,
    is_kev: bool,
    epss_score: Optional[float],
    scorecard_score: Optional[float],
    graph_reachable: Optional[bool] = None,
    symbol_reachability: Optional[str] = None,
) -> float:
    """The scalar risk formula — single source of truth shared by the CPU
    model and the GPU scoring kernel (which implements exactly this in f32)."""
    base = {
        Severity.CRITICAL: cfg.RISK_BASE_CRITICAL,
        Severity.HIGH: cfg.RISK_BASE_HIGH,
        Severity.MEDIUM: cfg.RISK_BASE_MEDIUM,
        Severity.LOW: cfg.RISK_BASE_LOW,
    }.get(severity, 0.0)

    agent_factor = min(n_agents * cfg.RISK_AGENT_WEIGHT, cfg.RISK_AGENT_CAP)
    cred_factor = min(n_creds * cfg.RISK_CRED_WEIGHT, cfg.RISK_CRED_CAP)
    tool_factor = min(n_tools * cfg.RISK_TOOL_WEIGHT, cfg.RISK_TOOL_CAP)

    ai_signals = sum([bool(has_ai_context), n_creds > 0, n_tools > 0])
    ai_boost = cfg.RISK_AI_BOOST if ai_signals >= 2 else 0.0

    kev_boost = cfg.RISK_KEV_BOOST if is_kev else 0.0
    epss_boost = cfg.RISK_EPSS_BOOST if (epss_score or 0) >= cfg.EPSS_CRITICAL_THRESHOLD else 0.0

    scorecard_boost = 0.0
    if scorecard_score is not None:
        if scorecard_score < cfg.RISK_SCORECARD_TIER1_THRESHOLD:
            scorecard_boost = cfg.RISK_SCORECARD_TIER1_BOOST
        elif scorecard_score < cfg.RISK_SCORECARD_TIER2_THRESHOLD:
            scorecard_boost = cfg.RISK_SCORECARD_TIER2_BOOST
        elif scorecard_score < cfg.RISK_SCORECARD_TIER3_THRESHOLD:
            scorecard_boost = cfg.RISK_SCORECARD_TIER3_BOOST

    reach_adjustment = 0.0
    if graph_reachable is True:
        reach_adjustment = cfg.RISK_REACHABLE_BOOST
    elif graph_reachable is False:
        reach_adjustment = -cfg.RISK_UNREACHABLE_PENALTY
    if symbol_reachability == "function_reachable":
        reach_adjustment = max(reach_adjustment, cfg.RISK_REACHABLE_BOOST)
    elif symbol_reachability == "unreachable":
        reach_adjustment = min(reach_adjustment, -cfg.RISK_UNREACHABLE_PENALTY)

    total = (base + agent_factor + cred_factor + tool_factor + ai_boost
             + kev_boost + epss_boost + scorecard_boost + reach_adjustment)
    return max(0.0, min(total, 10.0))


@dataclass
class BlastRadius:
    """Blast-radius analysis for one (package, vulnerability) pair."""

    vulnerability: Vulnerability
    package: Package
    affected_servers: list[MCPServer]
    affected_agents: list[Agent]
    exposed_credentials: list[str]
    exposed_tools: list[MCPTool]
    phantom_tools: list[MCPTool] = field(default_factory=list)
    risk_score: float = 0.0
    ai_risk_context: Optional[str] = None
    # Per-framework compliance tag fields (16 frameworks)
    owasp_tags: list[str] = field(default_factory=list)
    atlas_tags: list[str] = field(default_factory=list)
    attack_tags: list[str] = field(default_factory=list)
    nist_ai_rmf_tags: list[str] = field(default_factory=list)
    owasp_mcp_tags: list[str] = field(default_factory=list)
    owasp_agentic_tags: list[str] = field(default_factory=list)
    eu_ai_act_tags: list[str] = field(default_factory=list)
    nist_csf_tags: list[str] = field(default_factory=list)
    iso_27001_tags: list[str] = field(default_factory=list)
    soc2_tags: list[str] = field(default_factory=list)
    cis_tags: list[str] = field(default_factory=list)
    cmmc_tags: list[str] = field(default_factory=list)
    nist_800_53_tags: list[str] = field(default_factory=list)
    fedramp_tags: list[str] = field(default_factory=list)
    pci_dss_tags: list[str] = field(default_factory=list)
    ai_summary: Optional[str] = None
    suppressed: bool = False
    suppression_id: Optional[str] = None
    suppression_state: Optional[str] = None
    suppression_reason: Optional[str] = None
    unsuppressed_risk_score: Optional[float] = None
    # CWE-aware impact context
    impact_category: str = "code-execution"
    all_server_credentials: list[str] = field(default_factory=list)
    all_server_tools: list[MCPTool] = field(default_factory=list)
    attack_vector_summary: Optional[str] = None
    # Multi-hop delegation
    hop_depth: int = 1
    delegation_chain: list[str] = field(default_factory=list)
    transitive_agents: list[dict[str, Any]] = field(default_factory=list)
    transitive_credentials: list[str] = field(default_factory=list)
    transitive_risk_score: float = 0.0
    # Evidence-backed graph reachability (attack-path proof)
    graph_reachable: Optional[bool] = None
    graph_min_hop_distance: Optional[int] = None
    graph_reachable_from_agents: list[str] = field(default_factory=list)
    # Function-level symbol reachability
    symbol_reachability: Optional[str] = None
    reachable_affected_symbols: list[str] = field(default_factory=list)
    # Structural dependency closure
    dependency_reachable: Optional[bool] = None
    dependency_min_hop_distance: Optional[int] = None
    dependency_reachable_from_agents: list[str] = field(default_factory=list)

    def calculate_risk_score(self) -> float:
        if self.suppressed or is_vex_suppressed(self.vulnerability):
            self.risk_score = 0.0
            self.transitive_risk_score = 0.0
            return self.risk_score
        self.risk_score = risk_score_from_counts(
            severity=self.vulnerability.severity,
            n_agents=len(self.affected_agents),
            n_creds=len(self.exposed_credentials),
            n_tools=len(self.exposed_tools),
            has_ai_context=bool(self.ai_risk_context),
            is_kev=self.vulnerability.is_kev,
            epss_score=self.vulnerability.epss_score,
            scorecard_score=self.package.scorecard_score,
            graph_reachable=self.graph_reachable,
            symbol_reachability=self.symbol_reachability,
        )
        return self.risk_score

    @property
    def reachability(self) -> str:
        """confirmed / likely / unlikely / unknown."""
        has_creds = bool(self.exposed_credentials)
        has_tools = bool(self.exposed_tools)
        is_direct = self.package.is_direct
        is_high = self.vulnerability.severity in (Severity.CRITICAL, Severity.HIGH)
        has_agents = bool(self.affected_agents)
        declaration_only = self.package.reachability_evidence == "declaration_only"

        if (has_creds or has_tools) and is_direct:
            return "confirmed"
        if declaration_only and not has_creds and not has_tools:
            return "unknown"
        if has_creds or has_tools or (is_direct and has_agents) or is_high:
            return "likely"
        if not is_direct and not has_creds and not has_tools:
            return "unlikely"
        return "unknown"

    @property
    def is_actionable(self) -> bool:
        if self.suppressed or is_vex_suppressed(self.vulnerability):
            return False
        if self.vulnerability.is_kev:
            return True
        if self.vulnerability.severity in (Severity.CRITICAL, Severity.HIGH):
            return True
        if self.exposed_credentials or self.exposed_tools:
            return True
        if self.package.is_direct:
            return True
        if self.package.is_malicious:
            return True
        return False

    @property
    def layer_attribution(self):
        return sorted(self.package.occurrences, key=lambda o: (o.layer_index, o.layer_id, o.package_path or ""))


def expand_blast_radius_hops(
    blast_radii: list[BlastRadius],
    agents: list[Agent],
    max_depth: int = 1,
) -> None:
    """Multi-hop delegation BFS over agent<->server sharing, hop-decayed risk.

    CPU reference semantics; at estate scale the same expansion runs as the
    hop-limited GPU BFS with per-hop decay (ops/csrc/bfs.hip).
    """
    max_depth = max(1, min(max_depth, 5))
    if max_depth <= 1:
        return

    server_to_agents: dict[str, list[Agent]] = {}
    agent_to_servers: dict[str, list[str]] = {}
    for agent in agents:
        agent_to_servers[agent.name] = [s.name for s in agent.mcp_servers]
        for server in agent.mcp_servers:
            server_to_agents.setdefault(server.name, []).append(agent)

    for br in blast_radii:
        direct_agents = {a.name for a in br.affected_agents}
        direct_servers = {s.name for s in br.affected_servers}
        visited_agents = set(direct_agents)
        visited_servers = set(direct_servers)
        transitive_agents: list[dict] = []
        transitive_creds: list[str] = []
        chains: list[str] = []

        queue: list[tuple[str, int, list[str]]] = []
        for agent in br.affected_agents:
            for sname in agent_to_servers.get(agent.name, []):
                if sname not in direct_servers:
                    queue.append((agent.name, 1, [agent.name, sname]))
                    visited_servers.add(sname)

        max_hop = 1
        while queue:
            _aname, hop, chain = queue.pop(0)
            if hop >= max_depth:
                continue
            current_server = chain[-1]
            for nxt in server_to_agents.get(current_server, []):
                if nxt.name in visited_agents:
                    continue
                visited_agents.add(nxt.name)
                next_hop = hop + 1
                max_hop = max(max_hop, next_hop)
                new_chain = chain + [nxt.name]
                chain_str = "→".join(new_chain)
                chains.append(chain_str)
                creds: list[str] = []
                for server in nxt.mcp_servers:
                    creds.extend(server.credential_names)
                transitive_agents.append(
                    {"name": nxt.name, "type": nxt.agent_type.value, "hop": next_hop, "chain": chain_str}
                )
                transitive_creds.extend(set(creds))
                if next_hop < max_depth:
                    for sname in agent_to_servers.get(nxt.name, []):
                        if sname not in visited_servers:
                            visited_servers.add(sname)
                            queue.append((nxt.name, next_hop, new_chain + [sname]))

        if transitive_agents:
            br.hop_depth = max_hop
            br.delegation_chain = chains
            br.transitive_agents = transitive_agents
            br.transitive_credentials = list(set(transitive_creds))
            factor = HOP_RISK_FACTORS.get(max_hop, 0.25)
            br.transitive_risk_score = round(br.risk_score * factor, 2)


def active_blast_radii(blast_radii: list[BlastRadius]) -> list[BlastRadius]:
    return [br for br in blast_radii if not is_vex_suppressed(br.vulnerability)]
