"""Scan execution-quality contract + the AIBOMReport aggregate.

Reference: src/agent_bom/evidence/scan_run.py (ScanOutcome/ScanScope/
ScanIssue/ScanRun) and src/agent_bom/models.py:1163-1520 (AIBOMReport).
Execution quality is independent of the policy/finding verdict: a complete
scan can still exit non-zero, and a partial scan is never rendered clean.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from datetime import datetime, timezone
from enum import Enum
from typing import Any, Optional

from agentbom_amd.models.blast import BlastRadius, active_blast_radii
from agentbom_amd.models.core import Agent, AgentType, Severity
from agentbom_amd.models.finding import Finding, blast_radius_to_finding


def _sanitize_text(value: Any, max_len: int = 500) -> str:
    text = str(value or "").strip()
    text = "".join(ch for ch in text if ch.isprintable())
    return text[:max_len]


class ScanOutcome(str, Enum):
    COMPLETE = "complete"
    PARTIAL = "partial"
    FAILED = "failed"


class ScanScopeStatus(str, Enum):
    COMPLETE = "complete"
    PARTIAL = "partial"
    UNSUPPORTED = "unsupported"
    UNAVAILABLE = "unavailable"
    PERMISSION_DENIED = "permission_denied"
    SKIPPED = "skipped"


@dataclass(frozen=True)
class ScanScope:
    name: str
    status: ScanScopeStatus
    requested: bool = True
    item_count: Optional[int] = None
    message: str = ""

    def __post_init__(self) -> None:
        object.__setattr__(self, "name", _sanitize_text(self.name, 100) or "unknown")
        if isinstance(self.status, str):
            object.__setattr__(self, "status", ScanScopeStatus(self.status))
        if self.item_count is not None:
            object.__setattr__(self, "item_count", max(0, int(self.item_count)))
        object.__setattr__(self, "message", _sanitize_text(self.message, 500))

    def to_dict(self) -> dict[str, Any]:
        return {
            "name": self.name,
            "status": self.status.value,
            "requested": self.requested,
            "item_count": self.item_count,
            "message": self.message,
        }


@dataclass(frozen=True)
class ScanIssue:
    code: str
    stage: str
    source: str
    message: str
    severity: str = "warning"
    affects_coverage: bool = True

    def __post_init__(self) -> None:
        object.__setattr__(self, "code", _sanitize_text(self.code, 100) or "scan_issue")
        object.__setattr__(self, "stage", _sanitize_text(self.stage, 100) or "scan")
        object.__setattr__(self, "source", _sanitize_text(self.source, 200) or "agent-bom")
        object.__setattr__(self, "message", _sanitize_text(self.message, 1000) or "Scan execution issue")
        if self.severity not in ("warning", "error"):
            object.__setattr__(self, "severity", "warning")

    def to_dict(self) -> dict[str, Any]:
        return {
            "code": self.code,
            "stage": self.stage,
            "source": self.source,
            "message": self.message,
            "severity": self.severity,
            "affects_coverage": self.affects_coverage,
        }


@dataclass
class ScanRun:
    """Canonical scan outcome plus bounded, structured execution issues."""

    outcome: ScanOutcome = ScanOutcome.COMPLETE
    issues: list[ScanIssue] = field(default_factory=list)
    scopes: list[ScanScope] = field(default_factory=list)

    def __post_init__(self) -> None:
        if isinstance(self.outcome, str):
            self.outcome = ScanOutcome(self.outcome)
        seen: set[tuple] = set()
        deduped: list[ScanIssue] = []
        for issue in self.issues:
            key = (issue.code, issue.stage, issue.source, issue.message)
            if key not in seen:
                seen.add(key)
                deduped.append(issue)
        self.issues = deduped
        seen_scopes: set[str] = set()
        dedup_scopes: list[ScanScope] = []
        for scope in self.scopes:
            if scope.name not in seen_scopes:
                seen_scopes.add(scope.name)
                dedup_scopes.append(scope)
        self.scopes = dedup_scopes
        self._derive_partial()

    def _derive_partial(self) -> None:
        if self.outcome == ScanOutcome.FAILED:
            return
        degraded_issue = any(i.affects_coverage for i in self.issues)
        degraded_scope = any(
            s.requested and s.status not in (ScanScopeStatus.COMPLETE, ScanScopeStatus.SKIPPED)
            for s in self.scopes
        )
        if degraded_issue or degraded_scope:
            self.outcome = ScanOutcome.PARTIAL

    def add_issue(self, issue: ScanIssue) -> None:
        self.issues.append(issue)
        self.__post_init__()

    def to_dict(self) -> dict[str, Any]:
        return {
            "outcome": self.outcome.value,
            "issues": [i.to_dict() for i in self.issues],
            "scopes": [s.to_dict() for s in self.scopes],
        }


def classify_agent_kind(agent: Agent) -> str:
    """client / background / synthetic display classification."""
    if agent.agent_type != AgentType.CUSTOM:
        return "client"
    if (agent.name or "").startswith(("sbom:", "image:")):
        return "synthetic"
    return "background"


@dataclass
class AIBOMReport:
    """Complete AI-BOM report: agents + blast radii + unified findings."""

    agents: list[Agent] = field(default_factory=list)
    blast_radii: list[BlastRadius] = field(default_factory=list)
    generated_at: datetime = field(default_factory=lambda: datetime.now(timezone.utc))
    scan_id: str = ""
    tool_version: str = ""
    executive_summary: Optional[str] = None
    findings: list[Finding] = field(default_factory=list)
    scan_sources: list[str] = field(default_factory=list)
    scan_run: ScanRun = field(default_factory=ScanRun)
    coverage_warnings: list[dict[str, Any]] = field(default_factory=list)
    warnings: list[str] = field(default_factory=list)
    scan_performance_data: Optional[dict[str, Any]] = None
    intel_matches: Optional[list[dict[str, Any]]] = None
    vuln_data_freshness: Optional[dict[str, Any]] = None
    # Optional side blocks (subset of the reference's; extend as surfaces land)
    iac_findings_data: Optional[dict[str, Any]] = None
    ai_inventory_data: Optional[dict[str, Any]] = None
    project_inventory_data: Optional[dict[str, Any]] = None
    license_report: Optional[dict[str, Any]] = None
    vex_data: Optional[dict[str, Any]] = None
    context_graph_data: Optional[dict[str, Any]] = None
    toxic_combination_findings_data: Optional[list[Any]] = None
    enforcement_data: Optional[dict[str, Any]] = None
    delta_data: Optional[dict[str, Any]] = None
    # Remaining reference side blocks travel in this bucket keyed by their
    # reference field name, so JSON emission can stay key-compatible.
    extra_data: dict[str, Any] = field(default_factory=dict)

    def __post_init__(self) -> None:
        if not self.tool_version:
            from agentbom_amd import __version__

            self.tool_version = __version__

    # ── inventory summaries ────────────────────────────────────────────────

    @property
    def has_mcp_context(self) -> bool:
        return any(s.is_mcp_surface for a in self.agents for s in a.mcp_servers)

    @property
    def has_agent_context(self) -> bool:
        return any(a.agent_type != AgentType.CUSTOM for a in self.agents)

    @property
    def agent_class_counts(self) -> dict[str, int]:
        counts = {"client": 0, "background": 0}
        for agent in self.agents:
            kind = classify_agent_kind(agent)
            if kind in counts:
                counts[kind] += 1
        return counts

    @property
    def total_agents(self) -> int:
        return len(self.agents)

    @property
    def total_servers(self) -> int:
        # Synthetic dependency-bearing surfaces must not inflate MCP counts.
        return sum(1 for a in self.agents for s in a.mcp_servers if s.is_mcp_surface)

    @property
    def total_packages(self) -> int:
        return sum(a.total_packages for a in self.agents)

    @property
    def total_vulnerabilities(self) -> int:
        return sum(a.total_vulnerabilities for a in self.agents)

    @property
    def critical_vulns(self) -> list[BlastRadius]:
        return [br for br in active_blast_radii(self.blast_radii) if br.vulnerability.severity == Severity.CRITICAL]

    def severity_counts(self) -> dict[str, int]:
        counts = {s.value: 0 for s in Severity}
        for br in active_blast_radii(self.blast_radii):
            counts[br.vulnerability.severity.value] += 1
        return counts

    def to_findings(self) -> list[Finding]:
        """Unified findings list; auto-populated from blast_radii when empty."""
        base = list(self.findings)
        existing = {f.canonical_id for f in base}
        for br in self.blast_radii:
            finding = blast_radius_to_finding(br)
            if finding.canonical_id not in existing:
                base.append(finding)
                existing.add(finding.canonical_id)
        return base
