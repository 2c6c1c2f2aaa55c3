"""Remediation campaigns: group findings into trackable fix waves.

Reference surface: src/agent_bom/mcp_tools/risk_campaigns.py +
mcp_server_operator_tools.py risk_campaign_workflow (list / assign /
ticket / verify against the same store REST uses).

A campaign groups open findings by a shared fix dimension (same package,
same vulnerability, or same agent), assigns an owner, spawns tickets via
the ticket store, and is *verified closed* only when a fresh scan no longer
reports any of its member findings — closure is evidence-driven, never
declared.
"""

from __future__ import annotations

import json
import secrets
import sqlite3
import threading
import time
from collections import defaultdict
from typing import Any, Optional

from agentbom_amd.models import AIBOMReport, blast_radius_to_finding


def _finding_key(br: Any) -> str:
    """Stable across scans: the unified finding's canonical UUID."""
    return blast_radius_to_finding(br).canonical_id

_SCHEMA = """
CREATE TABLE IF NOT EXISTS campaigns (
    campaign_id TEXT PRIMARY KEY,
    status TEXT NOT NULL,
    doc TEXT NOT NULL
);
"""

_GROUP_DIMENSIONS = ("package", "vulnerability", "agent")


def group_findings(report: AIBOMReport, dimension: str = "package",
                   min_risk: float = 0.0) -> list[dict[str, Any]]:
    """Candidate campaigns: findings sharing one fix dimension, risk-ranked."""
    if dimension not in _GROUP_DIMENSIONS:
        raise ValueError(f"dimension must be one of {_GROUP_DIMENSIONS}")
    groups: dict[str, list] = defaultdict(list)
    for br in report.blast_radii:
        if br.risk_score < min_risk:
            continue
        if dimension == "package":
            key = f"{br.package.ecosystem}:{br.package.name}"
        elif dimension == "vulnerability":
            key = br.vulnerability.id
        else:
            key = br.affected_agents[0].name if br.affected_agents else "unassigned"
        groups[key].append(br)
    out = []
    for key, brs in groups.items():
        out.append({
            "group_key": key,
            "dimension": dimension,
            "finding_count": len(brs),
            "max_risk": max(b.risk_score for b in brs),
            "total_risk": round(sum(b.risk_score for b in brs), 2),
            "finding_ids": [_finding_key(b) for b in brs],
            "vulnerability_ids": sorted({b.vulnerability.id for b in brs}),
            "fix_versions": sorted({b.vulnerability.fixed_version for b in brs
                                    if b.vulnerability.fixed_version}),
        })
    return sorted(out, key=lambda g: -g["total_risk"])


class CampaignStore:
    """SQLite-backed remediation campaigns; thread-safe."""

    def __init__(self, path: str = ":memory:"):
        self._lock = threading.Lock()
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.executescript(_SCHEMA)

    def create(self, group: dict[str, Any], created_by: str = "operator") -> dict[str, Any]:
        with self._lock:
            campaign = {
                "campaign_id": f"cmp-{secrets.token_hex(5)}",
                "group_key": group["group_key"],
                "dimension": group["dimension"],
                "finding_ids": list(group["finding_ids"]),
                "vulnerability_ids": list(group.get("vulnerability_ids", [])),
                "fix_versions": list(group.get("fix_versions", [])),
                "max_risk": group.get("max_risk", 0.0),
                "status": "open",
                "assignee": None,
                "ticket_ids": [],
                "created_by": created_by,
                "created_at": time.time(),
                "verified_at": None,
                "residual_findings": None,
            }
            self._save(campaign)
            return campaign

    def _save(self, c: dict[str, Any]) -> None:
        self._db.execute(
            "INSERT INTO campaigns (campaign_id, status, doc) VALUES (?,?,?)"
            " ON CONFLICT(campaign_id) DO UPDATE SET status=excluded.status,"
            " doc=excluded.doc",
            (c["campaign_id"], c["status"], json.dumps(c)))
        self._db.commit()

    def get(self, campaign_id: str) -> Optional[dict[str, Any]]:
        row = self._db.execute("SELECT doc FROM campaigns WHERE campaign_id=?",
                               (campaign_id,)).fetchone()
        return json.loads(row[0]) if row else None

    def list(self, status: Optional[str] = None) -> list[dict[str, Any]]:
        if status:
            rows = self._db.execute(
                "SELECT doc FROM campaigns WHERE status=? ORDER BY rowid", (status,))
        else:
            rows = self._db.execute("SELECT doc FROM campaigns ORDER BY rowid")
        return [json.loads(r[0]) for r in rows.fetchall()]

    def assign(self, campaign_id: str, assignee: str) -> Optional[dict[str, Any]]:
        with self._lock:
            c = self.get(campaign_id)
            if c is None:
                return None
            c["assignee"] = assignee
            if c["status"] == "open":
                c["status"] = "in_progress"
            self._save(c)
            return c

    def attach_ticket(self, campaign_id: str, ticket_id: str) -> Optional[dict[str, Any]]:
        with self._lock:
            c = self.get(campaign_id)
            if c is None:
                return None
            if ticket_id not in c["ticket_ids"]:
                c["ticket_ids"].append(ticket_id)
            self._save(c)
            return c

    def verify(self, campaign_id: str, fresh_report: AIBOMReport) -> Optional[dict[str, Any]]:
        """Evidence-driven closure: resolved only if no member finding recurs."""
        with self._lock:
            c = self.get(campaign_id)
            if c is None:
                return None
            current = {_finding_key(br) for br in fresh_report.blast_radii}
            residual = sorted(set(c["finding_ids"]) & current)
            c["residual_findings"] = residual
            if residual:
                c["status"] = "in_progress"
            else:
                c["status"] = "resolved"
                c["verified_at"] = time.time()
            self._save(c)
            return c

    def close(self) -> None:
        self._db.close()
