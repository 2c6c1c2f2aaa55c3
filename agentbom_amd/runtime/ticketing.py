"""Remediation ticketing: local ticket store + tracker payload rendering.

Reference surface: src/agent_bom/mcp_server_ticketing_tools.py
(create_ticket / sync_ticket_status).  There is no egress here, so the
store is the source of truth and the Jira/GitHub payloads are rendered for
an out-of-band forwarder; ``sync_ticket_status`` reconciles statuses from
an exported tracker dump the same way the cloud/NHI modules consume
exported inventories.
"""

from __future__ import annotations

import json
import secrets
import sqlite3
import threading
import time
from typing import Any, Optional

_VALID_STATUSES = ("open", "in_progress", "resolved", "wont_fix", "closed")

_SCHEMA = """
CREATE TABLE IF NOT EXISTS tickets (
    ticket_id TEXT PRIMARY KEY,
    finding_id TEXT NOT NULL,
    status TEXT NOT NULL,
    doc TEXT NOT NULL
);
"""


def render_jira_payload(ticket: dict[str, Any], project_key: str = "SEC") -> dict[str, Any]:
    sev = str(ticket.get("severity", "medium")).lower()
    priority = {"critical": "Highest", "high": "High", "medium": "Medium",
                "low": "Low"}.get(sev, "Medium")
    return {
        "fields": {
            "project": {"key": project_key},
            "issuetype": {"name": "Bug"},
            "summary": ticket["title"],
            "description": ticket["description"],
            "priority": {"name": priority},
            "labels": ["agent-bom", f"severity-{sev}"]
            + ([f"cve-{ticket['vulnerability_id']}"] if ticket.get("vulnerability_id") else []),
        }
    }


def render_github_payload(ticket: dict[str, Any]) -> dict[str, Any]:
    sev = str(ticket.get("severity", "medium")).lower()
    return {
        "title": ticket["title"],
        "body": ticket["description"],
        "labels": ["security", "agent-bom", f"severity:{sev}"],
    }


class TicketStore:
    """SQLite-backed remediation tickets keyed to finding ids; thread-safe."""

    def __init__(self, path: str = ":memory:"):
        self._lock = threading.Lock()
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.executescript(_SCHEMA)

    def create(self, finding_id: str, title: str, description: str,
               severity: str = "medium", vulnerability_id: Optional[str] = None,
               assignee: Optional[str] = None, tracker: str = "jira",
               created_by: str = "mcp-operator") -> dict[str, Any]:
        with self._lock:
            existing = self.get_by_finding(finding_id)
            if existing and existing["status"] not in ("resolved", "closed", "wont_fix"):
                return {**existing, "deduplicated": True}
            ticket = {
                "ticket_id": f"abt-{secrets.token_hex(5)}",
                "finding_id": finding_id,
                "title": title,
                "description": description,
                "severity": severity,
                "vulnerability_id": vulnerability_id,
                "assignee": assignee,
                "tracker": tracker,
                "status": "open",
                "created_by": created_by,
                "created_at": time.time(),
                "updated_at": time.time(),
                "external_ref": None,
            }
            ticket["tracker_payload"] = (
                render_jira_payload(ticket) if tracker == "jira"
                else render_github_payload(ticket))
            self._db.execute(
                "INSERT INTO tickets (ticket_id, finding_id, status, doc)"
                " VALUES (?,?,?,?)",
                (ticket["ticket_id"], finding_id, "open", json.dumps(ticket)))
            self._db.commit()
            return ticket

    def get(self, ticket_id: str) -> Optional[dict[str, Any]]:
        row = self._db.execute("SELECT doc FROM tickets WHERE ticket_id=?",
                               (ticket_id,)).fetchone()
        return json.loads(row[0]) if row else None

    def get_by_finding(self, finding_id: str) -> Optional[dict[str, Any]]:
        row = self._db.execute(
            "SELECT doc FROM tickets WHERE finding_id=? ORDER BY rowid DESC LIMIT 1",
            (finding_id,)).fetchone()
        return json.loads(row[0]) if row else None

    def list(self, status: Optional[str] = None) -> list[dict[str, Any]]:
        if status:
            rows = self._db.execute(
                "SELECT doc FROM tickets WHERE status=? ORDER BY rowid", (status,))
        else:
            rows = self._db.execute("SELECT doc FROM tickets ORDER BY rowid")
        return [json.loads(r[0]) for r in rows.fetchall()]

    def update_status(self, ticket_id: str, status: str,
                      external_ref: Optional[str] = None) -> Optional[dict[str, Any]]:
        if status not in _VALID_STATUSES:
            raise ValueError(f"invalid status {status!r}; one of {_VALID_STATUSES}")
        with self._lock:
            ticket = self.get(ticket_id)
            if ticket is None:
                return None
            ticket["status"] = status
            ticket["updated_at"] = time.time()
            if external_ref:
                ticket["external_ref"] = external_ref
            self._db.execute("UPDATE tickets SET status=?, doc=? WHERE ticket_id=?",
                             (status, json.dumps(ticket), ticket_id))
            self._db.commit()
            return ticket

    def sync_from_export(self, export: dict[str, Any]) -> dict[str, Any]:
        """Reconcile statuses from a tracker dump: {"issues": [{ref, status}]}.

        Tracker statuses map onto the locked vocabulary; unknown refs are
        reported, never created.
        """
        status_map = {"done": "resolved", "closed": "closed", "resolved": "resolved",
                      "in progress": "in_progress", "in_progress": "in_progress",
                      "open": "open", "to do": "open", "wontfix": "wont_fix",
                      "won't fix": "wont_fix"}
        by_ref = {t.get("external_ref"): t for t in self.list() if t.get("external_ref")}
        updated, unknown = [], []
        for issue in export.get("issues", []):
            ref = str(issue.get("ref", ""))
            mapped = status_map.get(str(issue.get("status", "")).strip().lower())
            ticket = by_ref.get(ref)
            if ticket is None:
                unknown.append(ref)
                continue
            if mapped and ticket["status"] != mapped:
                self.update_status(ticket["ticket_id"], mapped)
                updated.append({"ticket_id": ticket["ticket_id"], "status": mapped})
        return {"updated": updated, "unknown_refs": unknown,
                "open_tickets": len(self.list("open"))}

    def close(self) -> None:
        self._db.close()
