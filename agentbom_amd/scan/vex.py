"""VEX (Vulnerability Exploitability eXchange) generation and suppression.

Reference: src/agent_bom/vex.py — OpenVEX-shaped documents; statements with
status affected / not_affected / fixed / under_investigation; applying a
document stamps vex_status/vex_justification onto matching vulnerabilities
and ``not_affected``/``fixed`` suppress scoring, counts and exit gates
(models/blast.is_vex_suppressed).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Any, Optional

from agentbom_amd.models import AIBOMReport

VEX_STATUSES = ("affected", "not_affected", "fixed", "under_investigation")
VEX_JUSTIFICATIONS = (
    "component_not_present",
    "vulnerable_code_not_present",
    "vulnerable_code_not_in_execute_path",
    "vulnerable_code_cannot_be_controlled_by_adversary",
    "inline_mitigations_already_exist",
)


@dataclass
class VexStatement:
    vulnerability_id: str
    status: str = "affected"
    justification: Optional[str] = None
    products: list[str] = field(default_factory=list)  # purls; empty = all
    impact_statement: Optional[str] = None

    def __post_init__(self) -> None:
        if self.status not in VEX_STATUSES:
            raise ValueError(f"invalid VEX status {self.status!r}")
        if self.justification and self.justification not in VEX_JUSTIFICATIONS:
            raise ValueError(f"invalid VEX justification {self.justification!r}")

    def to_dict(self) -> dict[str, Any]:
        d: dict[str, Any] = {
            "vulnerability": {"name": self.vulnerability_id},
            "status": self.status,
        }
        if self.justification:
            d["justification"] = self.justification
        if self.products:
            d["products"] = [{"@id": p} for p in self.products]
        if self.impact_statement:
            d["impact_statement"] = self.impact_statement
        return d


@dataclass
class VexDocument:
    statements: list[VexStatement] = field(default_factory=list)
    author: str = "agent-bom"
    version: int = 1

    def to_dict(self) -> dict[str, Any]:
        return {
            "@context": "https://openvex.dev/ns/v0.2.0",
            "author": self.author,
            "timestamp": datetime.now(timezone.utc).isoformat(),
            "version": self.version,
            "statements": [s.to_dict() for s in self.statements],
        }

    @classmethod
    def from_dict(cls, data: dict[str, Any]) -> "VexDocument":
        stmts = []
        for raw in data.get("statements", []):
            vuln = raw.get("vulnerability")
            vid = vuln.get("name") if isinstance(vuln, dict) else str(vuln or "")
            stmts.append(
                VexStatement(
                    vulnerability_id=vid,
                    status=raw.get("status", "affected"),
                    justification=raw.get("justification"),
                    products=[p.get("@id", "") for p in raw.get("products", []) if isinstance(p, dict)],
                    impact_statement=raw.get("impact_statement"),
                )
            )
        return cls(statements=stmts, author=data.get("author", "agent-bom"),
                   version=data.get("version", 1))


def apply_vex(report: AIBOMReport, vex: VexDocument) -> int:
    """Stamp vex_status/justification onto matching vulnerabilities.

    Returns count updated.  Suppressed rows stay in the data model for audit
    but are excluded from counts/exit gates via is_vex_suppressed."""
    by_id = {s.vulnerability_id: s for s in vex.statements}
    updated = 0

    def stamp(vuln, pkg_purl: Optional[str]) -> None:
        nonlocal updated
        stmt = by_id.get(vuln.id) or next(
            (by_id[a] for a in vuln.aliases if a in by_id), None
        )
        if stmt is None:
            return
        if stmt.products and pkg_purl and pkg_purl not in stmt.products:
            return
        vuln.vex_status = stmt.status
        vuln.vex_justification = stmt.justification
        updated += 1

    for agent in report.agents:
        for server in agent.mcp_servers:
            for pkg in server.packages:
                purl = pkg.purl or f"pkg:{pkg.ecosystem}/{pkg.name}@{pkg.version}"
                for v in pkg.vulnerabilities:
                    stamp(v, purl)
    for br in report.blast_radii:
        stamp(br.vulnerability, None)
        br.calculate_risk_score()
    return updated


def generate_vex(report: AIBOMReport, default_status: str = "affected") -> VexDocument:
    """Generate a VEX document covering every finding in the report."""
    stmts = []
    seen: set[str] = set()
    for br in report.blast_radii:
        if br.vulnerability.id in seen:
            continue
        seen.add(br.vulnerability.id)
        purl = f"pkg:{br.package.ecosystem}/{br.package.name}@{br.package.version}"
        stmts.append(
            VexStatement(
                vulnerability_id=br.vulnerability.id,
                status=br.vulnerability.vex_status or default_status,
                justification=br.vulnerability.vex_justification,
                products=[purl],
            )
        )
    return VexDocument(statements=stmts)
