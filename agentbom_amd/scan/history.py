"""Scan history, report diffing, and asset lifecycle tracking.

Reference: src/agent_bom/history.py (snapshots in ~/.agent-bom/history +
diff_reports), asset_tracker.py (first_seen/last_seen/resolved/MTTR),
scan_cache.py (TTL'd OSV response cache).
"""

from __future__ import annotations

import json
import os
import sqlite3
import time
from datetime import datetime, timezone
from pathlib import Path
from typing import Any, Optional


def _history_dir() -> Path:
    root = Path(os.environ.get("AGENT_BOM_HOME", str(Path.home() / ".agent-bom")))
    d = root / "history"
    d.mkdir(parents=True, exist_ok=True)
    return d


def save_report_snapshot(report_json: dict[str, Any], history_dir: Optional[Path] = None) -> Path:
    d = history_dir or _history_dir()
    ts = datetime.now(timezone.utc).strftime("%Y%m%dT%H%M%SZ")
    path = d / f"scan-{ts}-{report_json.get('scan_id', 'local')[:8]}.json"
    path.write_text(json.dumps(report_json, default=str))
    return path


def list_snapshots(history_dir: Optional[Path] = None) -> list[Path]:
    d = history_dir or _history_dir()
    return sorted(d.glob("scan-*.json"))


def diff_reports(old: dict[str, Any], new: dict[str, Any]) -> dict[str, Any]:
    """Finding-level diff: new / resolved / unchanged, package inventory delta."""

    def finding_keys(doc: dict) -> dict[str, dict]:
        out = {}
        for row in doc.get("blast_radius", []):
            key = f"{row.get('vulnerability_id')}|{row.get('package_name')}"
            out[key] = row
        return out

    old_f = finding_keys(old)
    new_f = finding_keys(new)
    new_keys = sorted(set(new_f) - set(old_f))
    resolved_keys = sorted(set(old_f) - set(new_f))

    def pkg_set(doc: dict) -> set[str]:
        return {
            f"{p.get('ecosystem')}:{p.get('name')}@{p.get('version')}"
            for p in doc.get("packages", [])
        }

    old_p, new_p = pkg_set(old), pkg_set(new)
    return {
        "schema_version": "1",
        "new_findings": [new_f[k] for k in new_keys],
        "resolved_findings": [
            {"vulnerability_id": old_f[k].get("vulnerability_id"),
             "package": old_f[k].get("package")} for k in resolved_keys
        ],
        "unchanged_count": len(set(new_f) & set(old_f)),
        "packages_added": sorted(new_p - old_p),
        "packages_removed": sorted(old_p - new_p),
        "summary": {
            "new": len(new_keys),
            "resolved": len(resolved_keys),
            "unchanged": len(set(new_f) & set(old_f)),
        },
    }


class AssetTracker:
    """first_seen / last_seen / resolved lifecycle + MTTR per finding."""

    def __init__(self, path: Optional[str | Path] = None):
        db = Path(path) if path else _history_dir().parent / "assets.db"
        if str(db) != ":memory:":
            db.parent.mkdir(parents=True, exist_ok=True)
        # API threads share the tracker; sqlite guards are relaxed and the
        # caller (server/state) serializes writes per request
        self.conn = sqlite3.connect(str(db), check_same_thread=False)
        self.conn.execute(
            """CREATE TABLE IF NOT EXISTS finding_lifecycle (
                finding_key TEXT PRIMARY KEY,
                vulnerability_id TEXT,
                package TEXT,
                first_seen REAL NOT NULL,
                last_seen REAL NOT NULL,
                resolved_at REAL
            )"""
        )
        self.conn.commit()

    def observe_scan(self, report_json: dict[str, Any], now: Optional[float] = None) -> dict:
        now = now or time.time()
        current = {
            f"{r.get('vulnerability_id')}|{r.get('package_name')}": r
            for r in report_json.get("blast_radius", [])
        }
        new = reopened = 0
        for key, row in current.items():
            cur = self.conn.execute(
                "SELECT resolved_at FROM finding_lifecycle WHERE finding_key=?", (key,)
            ).fetchone()
            if cur is None:
                self.conn.execute(
                    "INSERT INTO finding_lifecycle(finding_key, vulnerability_id, package,"
                    " first_seen, last_seen) VALUES (?,?,?,?,?)",
                    (key, row.get("vulnerability_id"), row.get("package"), now, now),
                )
                new += 1
            else:
                if cur[0] is not None:
                    reopened += 1
                self.conn.execute(
                    "UPDATE finding_lifecycle SET last_seen=?, resolved_at=NULL"
                    " WHERE finding_key=?", (now, key),
                )
        # resolve everything not in this scan
        resolved = 0
        for (key,) in self.conn.execute(
            "SELECT finding_key FROM finding_lifecycle WHERE resolved_at IS NULL"
        ).fetchall():
            if key not in current:
                self.conn.execute(
                    "UPDATE finding_lifecycle SET resolved_at=? WHERE finding_key=?", (now, key)
                )
                resolved += 1
        self.conn.commit()
        return {"new": new, "resolved": resolved, "reopened": reopened,
                "active": len(current)}

    def mttr_seconds(self) -> Optional[float]:
        rows = self.conn.execute(
            "SELECT resolved_at - first_seen FROM finding_lifecycle"
            " WHERE resolved_at IS NOT NULL"
        ).fetchall()
        if not rows:
            return None
        return sum(r[0] for r in rows) / len(rows)


class ScanCache:
    """TTL'd advisory-response cache (reference scan_cache.py)."""

    def __init__(self, path: Optional[str | Path] = None, ttl_s: float = 6 * 3600):
        db = Path(path) if path else _history_dir().parent / "scan_cache.db"
        db.parent.mkdir(parents=True, exist_ok=True)
        self.ttl_s = ttl_s
        self.conn = sqlite3.connect(str(db))
        self.conn.execute(
            "CREATE TABLE IF NOT EXISTS cache (key TEXT PRIMARY KEY, value TEXT,"
            " stored_at REAL)"
        )
        self.conn.commit()

    def get(self, key: str) -> Optional[Any]:
        row = self.conn.execute(
            "SELECT value, stored_at FROM cache WHERE key=?", (key,)
        ).fetchone()
        if row is None:
            return None
        value, stored_at = row
        if time.time() - stored_at > self.ttl_s:
            self.conn.execute("DELETE FROM cache WHERE key=?", (key,))
            self.conn.commit()
            return None
        return json.loads(value)

    def put(self, key: str, value: Any) -> None:
        self.conn.execute(
            "INSERT OR REPLACE INTO cache(key, value, stored_at) VALUES (?,?,?)",
            (key, json.dumps(value, default=str), time.time()),
        )
        self.conn.commit()
