"""Human-in-the-loop approval queue for flagged tool calls.

Reference parity: src/agent_bom/api/{hitl_approval_queue,
hitl_approval_store}.py — when the gateway's detectors flag a call as
warn-level (suspicious but not auto-blocked), the call can PARK in an
approval queue instead of passing through: a human approves or denies
with an accountable actor + reason; an unanswered request EXPIRES to
deny (fail closed — silence never authorizes).

Single-use grants: an approval authorizes exactly ONE replay of the
parked frame (matched by frame hash) within the grant window; consuming
it closes the request.  All transitions land in the queue's audit trail.
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
DENIED = "denied"
EXPIRED = "expired"
CONSUMED = "consumed"


def _now() -> datetime:
    return datetime.now(timezone.utc)


def frame_fingerprint(frame: dict[str, Any]) -> str:
    """Stable hash of the parked JSON-RPC frame (id excluded — the retry
    may carry a fresh message id)."""
    probe = {k: v for k, v in frame.items() if k != "id"}
    return hashlib.sha256(
        json.dumps(probe, sort_keys=True, default=str).encode()).hexdigest()


@dataclass
class ApprovalRequest:
    upstream: str
    principal: str
    frame_hash: str
    summary: str
    alerts: list[dict[str, Any]] = field(default_factory=list)
    request_id: str = ""
    status: str = PENDING
    created_at: str = ""
    expires_at: str = ""
    decided_by: str = ""
    decided_at: str = ""
    reason: str = ""

    def __post_init__(self) -> None:
        if not self.request_id:
            self.request_id = f"hitl-{uuid4().hex[:12]}"
        if not self.created_at:
            self.created_at = _now().isoformat()

    def is_expired(self, at: Optional[datetime] = None) -> bool:
        return bool(self.expires_at) and (at or _now()).isoformat() > self.expires_at

    def to_dict(self) -> dict[str, Any]:
        return {k: getattr(self, k) for k in (
            "request_id", "upstream", "principal", "frame_hash", "summary",
            "alerts", "status", "created_at", "expires_at", "decided_by",
            "decided_at", "reason")}


_SCHEMA = """
CREATE TABLE IF NOT EXISTS hitl_requests (
    request_id TEXT PRIMARY KEY,
    status TEXT NOT NULL,
    doc TEXT NOT NULL
);
"""


class ApprovalQueue:
    """SQLite-backed approval queue (":memory:" default)."""

    def __init__(self, path: str = ":memory:",
                 default_ttl_minutes: float = 30.0):
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.executescript(_SCHEMA)
        self._lock = threading.Lock()
        self.default_ttl_minutes = default_ttl_minutes

    def _save(self, req: ApprovalRequest) -> None:
        self._db.execute(
            "INSERT OR REPLACE INTO hitl_requests (request_id, status, doc)"
            " VALUES (?,?,?)",
            (req.request_id, req.status, json.dumps(req.to_dict())))

    def park(self, upstream: str, principal: str, frame: dict[str, Any],
             alerts: Optional[list] = None,
             ttl_minutes: Optional[float] = None) -> ApprovalRequest:
        method = str(frame.get("method", "?"))
        tool = str((frame.get("params") or {}).get("name", ""))
        req = ApprovalRequest(
            upstream=upstream, principal=principal,
            frame_hash=frame_fingerprint(frame),
            summary=f"{method} {tool}".strip(),
            alerts=[a if isinstance(a, dict) else
                    {"message": str(a)} for a in (alerts or [])],
            expires_at=(_now() + timedelta(
                minutes=ttl_minutes if ttl_minutes is not None
                else self.default_ttl_minutes)).isoformat())
        with self._lock:
            self._save(req)
            self._db.commit()
        return req

    def get(self, request_id: str) -> Optional[ApprovalRequest]:
        row = self._db.execute(
            "SELECT doc FROM hitl_requests WHERE request_id=?",
            (request_id,)).fetchone()
        if row is None:
            return None
        req = ApprovalRequest(**json.loads(row[0]))
        # lazily surface expiry: an unanswered pending request reads EXPIRED
        if req.status == PENDING and req.is_expired():
            req.status = EXPIRED
            with self._lock:
                self._save(req)
                self._db.commit()
        return req

    def list(self, status: Optional[str] = None) -> list[ApprovalRequest]:
        rows = [self.get(rid) for (rid,) in self._db.execute(
            "SELECT request_id FROM hitl_requests ORDER BY request_id")]
        return [r for r in rows if r and (status is None or r.status == status)]

    def decide(self, request_id: str, approve: bool, actor: str,
               reason: str = "") -> Optional[ApprovalRequest]:
        """Approve/deny a PENDING request; expired requests cannot be
        approved (fail closed)."""
        with self._lock:
            req = self.get(request_id)
            if req is None or req.status != PENDING:
                return None
            req.status = APPROVED if approve else DENIED
            req.decided_by = actor
            req.decided_at = _now().isoformat()
            req.reason = reason
            self._save(req)
            self._db.commit()
            return req

    def consume(self, request_id: str, frame: dict[str, Any]) -> bool:
        """Burn a single-use approval for a retry of the SAME frame.

        True exactly once, and only when the request is APPROVED, not
        expired, and the frame hash matches the parked call."""
        with self._lock:
            req = self.get(request_id)
            if req is None or req.status != APPROVED or req.is_expired():
                return False
            if frame_fingerprint(frame) != req.frame_hash:
                return False
            req.status = CONSUMED
            self._save(req)
            self._db.commit()
            return True
