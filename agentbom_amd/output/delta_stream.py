"""Finding delta stream: new / resolved / changed events with a watermark.

Reference parity: src/agent_bom/delta_stream.py — SIEM/data-lake sinks get
per-finding delta events instead of full snapshot re-lists.  State lives in
a small SQLite watermark store keyed by the finding canonical id with a
content hash; each emit produces only what changed since the last one, with
a monotonically increasing sequence number so consumers can resume.

Formats: ``ndjson`` (one JSON object per event) and ``ocsf`` (wrapped as
OCSF 1.1 Vulnerability Finding events, same mapping as output/ocsf.py).
"""

from __future__ import annotations

import hashlib
import json
import sqlite3
import threading
import time
from typing import Any, Callable, Optional

from agentbom_amd.models import AIBOMReport, blast_radius_to_finding

_SCHEMA = """
CREATE TABLE IF NOT EXISTS finding_state (
    canonical_id TEXT PRIMARY KEY,
    content_hash TEXT NOT NULL,
    first_seen REAL NOT NULL,
    last_seen REAL NOT NULL,
    doc TEXT NOT NULL
);
CREATE TABLE IF NOT EXISTS watermark (
    id INTEGER PRIMARY KEY CHECK (id = 1),
    seq INTEGER NOT NULL,
    ts REAL NOT NULL
);
"""

# fields whose change constitutes a "changed" event (not cosmetic noise)
_TRACKED_FIELDS = (
    "risk_score", "severity", "is_kev", "epss_score", "fixed_version",
    "reachability", "vex_suppressed", "suppressed",
)


def _finding_row(br) -> dict[str, Any]:
    f = blast_radius_to_finding(br)
    return {
        "canonical_id": f.canonical_id,
        "vulnerability_id": f.vulnerability_id,
        "package": f"{br.package.name}@{br.package.version}",
        "ecosystem": br.package.ecosystem,
        "risk_score": round(float(br.risk_score), 4),
        "severity": br.vulnerability.severity.value,
        "is_kev": bool(br.vulnerability.is_kev),
        "epss_score": br.vulnerability.epss_score,
        "fixed_version": br.vulnerability.fixed_version,
        "reachability": br.reachability,
        "vex_suppressed": bool(getattr(br, "vex_suppressed", False)),
        "suppressed": bool(br.suppressed),
        "affected_agents": sorted(a.name for a in br.affected_agents),
    }


def _content_hash(row: dict[str, Any]) -> str:
    basis = {k: row.get(k) for k in _TRACKED_FIELDS}
    return hashlib.sha256(
        json.dumps(basis, sort_keys=True, default=str).encode()).hexdigest()


def _to_ocsf_event(event: dict[str, Any]) -> dict[str, Any]:
    row = event.get("finding") or {}
    activity = {"new": 1, "changed": 2, "resolved": 3}[event["kind"]]
    return {
        "class_uid": 2002,  # Vulnerability Finding
        "class_name": "Vulnerability Finding",
        "activity_id": activity,
        "time": int(event["ts"] * 1000),
        "metadata": {"product": {"name": "agent-bom"}, "version": "1.1.0",
                     "sequence": event["seq"]},
        "finding_info": {"uid": event["canonical_id"],
                         "title": row.get("vulnerability_id")},
        "vulnerabilities": [{"cve": {"uid": row.get("vulnerability_id")},
                             "severity": row.get("severity"),
                             "is_exploit_available": row.get("is_kev")}],
        "severity": row.get("severity"),
        "unmapped": row,
    }


class DeltaStreamer:
    """Stateful delta emitter. One instance per sink stream; thread-safe."""

    def __init__(self, state_path: str = ":memory:",
                 fmt: str = "ndjson",
                 reconcile_absent: bool = True):
        if fmt not in ("ndjson", "ocsf"):
            raise ValueError(f"unknown delta format {fmt!r}")
        self.fmt = fmt
        self.reconcile_absent = reconcile_absent
        self._lock = threading.Lock()
        self._db = sqlite3.connect(state_path, check_same_thread=False)
        self._db.executescript(_SCHEMA)

    # ── watermark ─────────────────────────────────────────────────────────

    @property
    def watermark(self) -> dict[str, Any]:
        row = self._db.execute("SELECT seq, ts FROM watermark WHERE id=1").fetchone()
        return {"seq": row[0], "ts": row[1]} if row else {"seq": 0, "ts": None}

    def _advance(self, n_events: int, now: float) -> int:
        wm = self.watermark
        seq = wm["seq"]
        self._db.execute(
            "INSERT INTO watermark (id, seq, ts) VALUES (1, ?, ?)"
            " ON CONFLICT(id) DO UPDATE SET seq=excluded.seq, ts=excluded.ts",
            (seq + n_events, now))
        return seq

    # ── emit ──────────────────────────────────────────────────────────────

    def emit(self, report: AIBOMReport,
             sink: Optional[Callable[[str], None]] = None) -> list[dict[str, Any]]:
        """Diff the report against stored state; return (and sink) events."""
        with self._lock:
            now = time.time()
            current = {}
            for br in report.blast_radii:
                row = _finding_row(br)
                current[row["canonical_id"]] = row

            stored = {cid: (chash, doc) for cid, chash, doc in self._db.execute(
                "SELECT canonical_id, content_hash, doc FROM finding_state")}

            events: list[dict[str, Any]] = []
            for cid in sorted(current):
                row = current[cid]
                chash = _content_hash(row)
                if cid not in stored:
                    events.append({"kind": "new", "canonical_id": cid,
                                   "finding": row})
                elif stored[cid][0] != chash:
                    prev = json.loads(stored[cid][1])
                    changed = {k: {"from": prev.get(k), "to": row.get(k)}
                               for k in _TRACKED_FIELDS
                               if prev.get(k) != row.get(k)}
                    events.append({"kind": "changed", "canonical_id": cid,
                                   "finding": row, "changes": changed})
                self._db.execute(
                    "INSERT INTO finding_state (canonical_id, content_hash,"
                    " first_seen, last_seen, doc) VALUES (?,?,?,?,?)"
                    " ON CONFLICT(canonical_id) DO UPDATE SET"
                    " content_hash=excluded.content_hash,"
                    " last_seen=excluded.last_seen, doc=excluded.doc",
                    (cid, chash, now, now, json.dumps(row)))

            if self.reconcile_absent:
                for cid in sorted(set(stored) - set(current)):
                    events.append({"kind": "resolved", "canonical_id": cid,
                                   "finding": json.loads(stored[cid][1])})
                    self._db.execute(
                        "DELETE FROM finding_state WHERE canonical_id=?", (cid,))

            base_seq = self._advance(len(events), now)
            self._db.commit()

            for i, ev in enumerate(events):
                ev["seq"] = base_seq + i + 1
                ev["ts"] = now
            if sink is not None:
                for ev in events:
                    payload = _to_ocsf_event(ev) if self.fmt == "ocsf" else ev
                    sink(json.dumps(payload, default=str))
            return events

    def close(self) -> None:
        self._db.close()
