"""Compliance Hub: multi-source external-finding ingest + current state.

Reference parity: src/agent_bom/api/{hub_ingest,hub_current_payload,
hub_overview_cache,compliance_hub_store}.py — connectors and external
scanners (Wiz/Prisma/SIEM exports, other agent-bom sites) push finding
batches into ONE hub; the hub keeps:

- an append-only LEDGER of ingest events,
- the CURRENT state per (tenant, source, finding_key) with
  first_seen/last_seen,
- ABSENT reconciliation: a finding the source stops reporting is
  RESOLVED (never silently dropped),
- per-ingest DELTAS (new / changed / resolved) for downstream sinks,
- a generation-stamped OVERVIEW cache (invalidated by writes, so the
  dashboard read path never rescans the table per request).
"""

from __future__ import annotations

import hashlib
import json
import sqlite3
import threading
import time
from typing import Any, Optional

_SCHEMA = """
CREATE TABLE IF NOT EXISTS hub_current (
    tenant_id TEXT NOT NULL,
    source TEXT NOT NULL,
    finding_key TEXT NOT NULL,
    payload TEXT NOT NULL,
    severity TEXT NOT NULL,
    status TEXT NOT NULL,
    first_seen REAL NOT NULL,
    last_seen REAL NOT NULL,
    resolved_at REAL,
    PRIMARY KEY (tenant_id, source, finding_key)
);
CREATE TABLE IF NOT EXISTS hub_ledger (
    seq INTEGER PRIMARY KEY AUTOINCREMENT,
    ts REAL NOT NULL,
    tenant_id TEXT NOT NULL,
    source TEXT NOT NULL,
    event TEXT NOT NULL
);
"""


def finding_key(row: dict[str, Any]) -> str:
    """Stable identity for an external finding: explicit id wins, else a
    hash of the (vuln, asset, title) triple."""
    for key in ("finding_id", "id"):
        if row.get(key):
            return str(row[key])
    probe = json.dumps([row.get("vulnerability_id") or row.get("cve_id"),
                        row.get("asset") or row.get("resource"),
                        row.get("title")], sort_keys=True, default=str)
    return "hub-" + hashlib.sha256(probe.encode()).hexdigest()[:16]


class ComplianceHub:
    """SQLite-backed hub (":memory:" default)."""

    def __init__(self, path: str = ":memory:"):
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.executescript(_SCHEMA)
        self._lock = threading.Lock()
        self._generation = 0
        self._overview_cache: dict[str, tuple[int, dict]] = {}

    # ── ingest (the single shared write path) ─────────────────────────────

    def ingest(self, tenant_id: str, source: str,
               findings: list[dict[str, Any]],
               reconcile_absent: bool = True) -> dict[str, Any]:
        """Upsert a source's batch; reconcile what the source stopped
        reporting; return the delta for downstream sinks."""
        now = time.time()
        new: list[str] = []
        changed: list[str] = []
        seen: set[str] = set()
        with self._lock:
            for row in findings:
                if not isinstance(row, dict):
                    continue
                key = finding_key(row)
                seen.add(key)
                payload = json.dumps(row, sort_keys=True, default=str)
                sev = str(row.get("severity", "unknown")).lower()
                cur = self._db.execute(
                    "SELECT payload, status FROM hub_current WHERE"
                    " tenant_id=? AND source=? AND finding_key=?",
                    (tenant_id, source, key)).fetchone()
                if cur is None:
                    self._db.execute(
                        "INSERT INTO hub_current (tenant_id, source,"
                        " finding_key, payload, severity, status,"
                        " first_seen, last_seen) VALUES (?,?,?,?,?,?,?,?)",
                        (tenant_id, source, key, payload, sev, "open",
                         now, now))
                    new.append(key)
                else:
                    reopened = cur[1] == "resolved"
                    if cur[0] != payload or reopened:
                        changed.append(key)
                    self._db.execute(
                        "UPDATE hub_current SET payload=?, severity=?,"
                        " status='open', last_seen=?, resolved_at=NULL WHERE"
                        " tenant_id=? AND source=? AND finding_key=?",
                        (payload, sev, now, tenant_id, source, key))
            resolved: list[str] = []
            if reconcile_absent:
                rows = self._db.execute(
                    "SELECT finding_key FROM hub_current WHERE tenant_id=?"
                    " AND source=? AND status='open'",
                    (tenant_id, source)).fetchall()
                for (key,) in rows:
                    if key not in seen:
                        self._db.execute(
                            "UPDATE hub_current SET status='resolved',"
                            " resolved_at=? WHERE tenant_id=? AND source=?"
                            " AND finding_key=?",
                            (now, tenant_id, source, key))
                        resolved.append(key)
            delta = {"source": source, "ingested": len(seen),
                     "new": sorted(new), "changed": sorted(changed),
                     "resolved": sorted(resolved), "at": now}
            self._db.execute(
                "INSERT INTO hub_ledger (ts, tenant_id, source, event)"
                " VALUES (?,?,?,?)",
                (now, tenant_id, source, json.dumps(
                    {k: (len(v) if isinstance(v, list) else v)
                     for k, v in delta.items()})))
            self._db.commit()
            self._generation += 1  # invalidates overview caches
        return delta

    # ── reads ─────────────────────────────────────────────────────────────

    def findings(self, tenant_id: str, source: Optional[str] = None,
                 status: str = "open",
                 limit: int = 1000) -> list[dict[str, Any]]:
        q = ("SELECT source, finding_key, payload, severity, status,"
             " first_seen, last_seen FROM hub_current WHERE tenant_id=?")
        args: list[Any] = [tenant_id]
        if source:
            q += " AND source=?"
            args.append(source)
        if status:
            q += " AND status=?"
            args.append(status)
        q += " ORDER BY source, finding_key LIMIT ?"
        args.append(limit)
        out = []
        for src, key, payload, sev, st, first, last in \
                self._db.execute(q, args):
            out.append({"source": src, "finding_key": key,
                        "severity": sev, "status": st,
                        "first_seen": first, "last_seen": last,
                        "payload": json.loads(payload)})
        return out

    def overview(self, tenant_id: str) -> dict[str, Any]:
        """Per-source and per-severity aggregates — generation-cached."""
        cached = self._overview_cache.get(tenant_id)
        if cached is not None and cached[0] == self._generation:
            return cached[1]
        by_source: dict[str, dict[str, int]] = {}
        by_sev: dict[str, int] = {}
        open_total = resolved_total = 0
        for src, sev, status, n in self._db.execute(
                "SELECT source, severity, status, COUNT(*) FROM hub_current"
                " WHERE tenant_id=? GROUP BY source, severity, status",
                (tenant_id,)):
            row = by_source.setdefault(src, {"open": 0, "resolved": 0})
            row[status] = row.get(status, 0) + n
            if status == "open":
                open_total += n
                by_sev[sev] = by_sev.get(sev, 0) + n
            else:
                resolved_total += n
        out = {"generation": self._generation,
               "sources": dict(sorted(by_source.items())),
               "open_by_severity": dict(sorted(by_sev.items())),
               "open_total": open_total, "resolved_total": resolved_total}
        self._overview_cache[tenant_id] = (self._generation, out)
        return out

    def ledger(self, tenant_id: str, limit: int = 100) -> list[dict[str, Any]]:
        return [{"ts": ts, "source": src, **json.loads(event)}
                for ts, src, event in self._db.execute(
                    "SELECT ts, source, event FROM hub_ledger WHERE"
                    " tenant_id=? ORDER BY seq DESC LIMIT ?",
                    (tenant_id, limit))]
