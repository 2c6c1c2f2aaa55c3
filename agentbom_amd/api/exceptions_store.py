"""Vulnerability exception / waiver workflow.

Reference parity: src/agent_bom/api/exception_store.py — security teams
grant TIME-BOXED exceptions for specific (CVE, package, server) scopes
with an approval workflow:

    PENDING → APPROVED → EXPIRED
    PENDING → REJECTED
    APPROVED → REVOKED

Wildcards: ``vuln_id="*"`` waives every CVE on a package;
``package_name="*"`` waives one CVE everywhere; an empty server scope
matches all servers.  Only APPROVED, unexpired exceptions suppress —
``apply_exceptions_to_report`` stamps matching blast radii
``suppressed`` with ``suppression_state="exception"`` so scoring, exit
gates and counts all honor the waiver through the existing suppression
contract (models/blast.py:117-121).  Every lifecycle transition appends
to a hash-chained audit log (same scheme as identity/lifecycle.py).
"""

from __future__ import annotations

import hashlib
import json
import sqlite3
import threading
from dataclasses import dataclass, field
from datetime import datetime, timedelta, timezone
from typing import Any, Optional
from uuid import uuid4

PENDING = "pending"
APPROVED = "approved"
REJECTED = "rejected"
EXPIRED = "expired"
REVOKED = "revoked"

_TRANSITIONS = {
    (PENDING, APPROVED), (PENDING, REJECTED), (APPROVED, REVOKED),
}


def _now() -> datetime:
    return datetime.now(timezone.utc)


@dataclass
class VulnException:
    """One waiver request; covers a (vuln, package, server) scope."""

    vuln_id: str
    package_name: str
    reason: str
    requested_by: str
    server_name: str = ""            # "" or "*" = all servers
    tenant_id: str = "default"
    exception_id: str = ""
    status: str = PENDING
    approved_by: str = ""
    created_at: str = ""
    expires_at: str = ""
    approved_at: str = ""
    revoked_at: str = ""

    def __post_init__(self) -> None:
        if not self.exception_id:
            self.exception_id = f"exc-{uuid4().hex[:12]}"
        if not self.created_at:
            self.created_at = _now().isoformat()

    def is_expired(self, at: Optional[datetime] = None) -> bool:
        if not self.expires_at:
            return False
        return (at or _now()).isoformat() > self.expires_at

    def matches(self, vuln_id: str, package_name: str,
                server_name: str = "", at: Optional[datetime] = None) -> bool:
        """True when this waiver currently covers the finding."""
        if self.status != APPROVED or self.is_expired(at):
            return False
        return ((self.vuln_id == "*" or self.vuln_id == vuln_id)
                and (self.package_name == "*"
                     or self.package_name == package_name)
                and (self.server_name in ("", "*")
                     or self.server_name == server_name))

    def to_dict(self) -> dict[str, Any]:
        return {k: getattr(self, k) for k in (
            "exception_id", "vuln_id", "package_name", "server_name",
            "reason", "requested_by", "approved_by", "status", "created_at",
            "expires_at", "approved_at", "revoked_at", "tenant_id")}


_SCHEMA = """
CREATE TABLE IF NOT EXISTS vuln_exceptions (
    exception_id TEXT PRIMARY KEY,
    tenant_id TEXT NOT NULL,
    status TEXT NOT NULL,
    doc TEXT NOT NULL
);
CREATE TABLE IF NOT EXISTS exception_audit (
    seq INTEGER PRIMARY KEY AUTOINCREMENT,
    ts TEXT NOT NULL,
    action TEXT NOT NULL,
    actor TEXT NOT NULL,
    subject TEXT NOT NULL,
    reason TEXT NOT NULL,
    prev_hash TEXT NOT NULL,
    entry_hash TEXT NOT NULL
);
"""


class ExceptionStore:
    """SQLite-backed waiver store (":memory:" default) with chained audit."""

    def __init__(self, path: str = ":memory:"):
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.executescript(_SCHEMA)
        self._lock = threading.Lock()

    # ── audit chain ───────────────────────────────────────────────────────

    def _audit(self, action: str, actor: str, subject: str, reason: str) -> None:
        row = self._db.execute(
            "SELECT entry_hash FROM exception_audit ORDER BY seq DESC LIMIT 1"
        ).fetchone()
        prev = row[0] if row else "genesis"
        ts = _now().isoformat()
        entry = hashlib.sha256(
            f"{prev}|{ts}|{action}|{actor}|{subject}|{reason}".encode()
        ).hexdigest()
        self._db.execute(
            "INSERT INTO exception_audit (ts, action, actor, subject, reason,"
            " prev_hash, entry_hash) VALUES (?,?,?,?,?,?,?)",
            (ts, action, actor, subject, reason, prev, entry))

    def audit_chain_valid(self) -> bool:
        prev = "genesis"
        for ts, action, actor, subject, reason, prev_hash, entry_hash in \
                self._db.execute("SELECT ts, action, actor, subject, reason,"
                                 " prev_hash, entry_hash FROM exception_audit"
                                 " ORDER BY seq"):
            if prev_hash != prev:
                return False
            want = hashlib.sha256(
                f"{prev}|{ts}|{action}|{actor}|{subject}|{reason}".encode()
            ).hexdigest()
            if want != entry_hash:
                return False
            prev = entry_hash
        return True

    # ── lifecycle ─────────────────────────────────────────────────────────

    def request(self, exc: VulnException) -> VulnException:
        with self._lock:
            self._db.execute(
                "INSERT INTO vuln_exceptions (exception_id, tenant_id, status,"
                " doc) VALUES (?,?,?,?)",
                (exc.exception_id, exc.tenant_id, exc.status,
                 json.dumps(exc.to_dict())))
            self._audit("exception.request", exc.requested_by,
                        exc.exception_id, exc.reason)
            self._db.commit()
        return exc

    def _transition(self, exception_id: str, new_status: str, actor: str,
                    reason: str, tenant_id: Optional[str],
                    ttl_days: Optional[float] = None) -> Optional[VulnException]:
        with self._lock:
            exc = self.get(exception_id, tenant_id)
            if exc is None or (exc.status, new_status) not in _TRANSITIONS:
                return None
            exc.status = new_status
            now = _now()
            if new_status == APPROVED:
                exc.approved_by = actor
                exc.approved_at = now.isoformat()
                if ttl_days is not None and not exc.expires_at:
                    exc.expires_at = (now + timedelta(days=ttl_days)).isoformat()
            elif new_status == REVOKED:
                exc.revoked_at = now.isoformat()
            self._db.execute(
                "UPDATE vuln_exceptions SET status=?, doc=? WHERE exception_id=?",
                (exc.status, json.dumps(exc.to_dict()), exception_id))
            self._audit(f"exception.{new_status}", actor, exception_id, reason)
            self._db.commit()
            return exc

    def approve(self, exception_id: str, actor: str, reason: str = "",
                tenant_id: Optional[str] = None,
                ttl_days: float = 90.0) -> Optional[VulnException]:
        """Approve with a DEFAULT 90-day expiry — waivers are time-boxed
        unless the request carried an explicit expires_at."""
        return self._transition(exception_id, APPROVED, actor, reason,
                                tenant_id, ttl_days=ttl_days)

    def reject(self, exception_id: str, actor: str, reason: str = "",
               tenant_id: Optional[str] = None) -> Optional[VulnException]:
        return self._transition(exception_id, REJECTED, actor, reason, tenant_id)

    def revoke(self, exception_id: str, actor: str, reason: str = "",
               tenant_id: Optional[str] = None) -> Optional[VulnException]:
        return self._transition(exception_id, REVOKED, actor, reason, tenant_id)

    # ── reads ─────────────────────────────────────────────────────────────

    def get(self, exception_id: str,
            tenant_id: Optional[str] = None) -> Optional[VulnException]:
        q = "SELECT doc FROM vuln_exceptions WHERE exception_id=?"
        args: list[Any] = [exception_id]
        if tenant_id is not None:
            q += " AND tenant_id=?"
            args.append(tenant_id)
        row = self._db.execute(q, args).fetchone()
        return VulnException(**json.loads(row[0])) if row else None

    def list(self, tenant_id: str = "default",
             status: Optional[str] = None) -> list[VulnException]:
        q = "SELECT doc FROM vuln_exceptions WHERE tenant_id=?"
        args: list[Any] = [tenant_id]
        if status:
            q += " AND status=?"
            args.append(status)
        out = [VulnException(**json.loads(doc))
               for (doc,) in self._db.execute(q + " ORDER BY exception_id", args)]
        # surface expiry without mutating stored state
        for exc in out:
            if exc.status == APPROVED and exc.is_expired():
                exc.status = EXPIRED
        return out

    def active_for(self, tenant_id: str, vuln_id: str, package_name: str,
                   server_name: str = "") -> Optional[VulnException]:
        for exc in self.list(tenant_id, status=APPROVED):
            if exc.matches(vuln_id, package_name, server_name):
                return exc
        return None


def apply_exceptions_to_report(report, store: ExceptionStore,
                               tenant_id: str = "default") -> int:
    """Suppress blast radii covered by an APPROVED, unexpired waiver.

    Stamps the existing suppression contract fields so scoring/exit
    gates/counts honor the exception exactly like an ignore rule; the
    pre-waiver risk survives in ``unsuppressed_risk_score``."""
    approved = [e for e in store.list(tenant_id, status=APPROVED)
                if not e.is_expired()]
    if not approved:
        return 0
    n = 0
    for br in report.blast_radii:
        if br.suppressed:
            continue
        servers = [s.name for s in br.affected_servers] or [""]
        hit = next((e for e in approved for srv in servers
                    if e.matches(br.vulnerability.id, br.package.name, srv)),
                   None)
        if hit is not None:
            br.unsuppressed_risk_score = br.risk_score
            br.suppressed = True
            br.suppression_id = hit.exception_id
            br.suppression_state = "exception"
            br.suppression_reason = hit.reason
            n += 1
    return n
