"""Local advisory store (SQLite) + arena export.

Schema-compatible with the reference local vuln DB (reference:
src/agent_bom/db/schema.py:106-186 — vulns / affected / epss_scores /
kev_entries / sync_meta; one row per affected WINDOW, PK over all three
bounds so multi-branch advisories never collapse).

Beyond the reference, ``export_windows()`` feeds the GPU advisory arena
(db/arena.py): the whole store becomes sorted columnar arrays resident in
HBM, refreshed on sync.
"""

from __future__ import annotations

import json
import os
import sqlite3
from datetime import datetime, timezone
from pathlib import Path
from typing import Iterable, Optional, Sequence

from agentbom_amd.db.arena import AdvisoryArena, AdvisoryWindow, build_arena
from agentbom_amd.models.core import Severity
from agentbom_amd.utils.canonical_ids import normalize_package_ecosystem, normalize_package_name

SCHEMA_VERSION = 5

_DDL = """
PRAGMA journal_mode = WAL;
PRAGMA synchronous = NORMAL;
PRAGMA foreign_keys = ON;
PRAGMA busy_timeout = 30000;

CREATE TABLE IF NOT EXISTS schema_version (version INTEGER NOT NULL);

CREATE TABLE IF NOT EXISTS vulns (
    id              TEXT PRIMARY KEY,
    summary         TEXT NOT NULL,
    severity        TEXT NOT NULL,
    cvss_score      REAL,
    cvss_vector     TEXT,
    fixed_version   TEXT,
    cwe_ids         TEXT DEFAULT '',
    aliases         TEXT DEFAULT '',
    published       TEXT,
    modified        TEXT,
    source          TEXT NOT NULL
);

CREATE TABLE IF NOT EXISTS affected (
    vuln_id         TEXT NOT NULL REFERENCES vulns(id) ON DELETE CASCADE,
    ecosystem       TEXT NOT NULL,
    package_name    TEXT NOT NULL,
    introduced      TEXT,
    fixed           TEXT,
    last_affected   TEXT,
    PRIMARY KEY (vuln_id, ecosystem, package_name, introduced, fixed, last_affected)
);
CREATE INDEX IF NOT EXISTS idx_affected_pkg ON affected(ecosystem, package_name);

CREATE TABLE IF NOT EXISTS epss_scores (
    cve_id          TEXT PRIMARY KEY,
    probability     REAL NOT NULL,
    percentile      REAL,
    updated_at      TEXT NOT NULL
);

CREATE TABLE IF NOT EXISTS kev_entries (
    cve_id          TEXT PRIMARY KEY,
    date_added      TEXT,
    due_date        TEXT,
    product         TEXT,
    vendor_project  TEXT
);

CREATE TABLE IF NOT EXISTS sync_meta (
    source          TEXT PRIMARY KEY,
    last_synced     TEXT,
    record_count    INTEGER DEFAULT 0,
    metadata_json   TEXT DEFAULT ''
);
"""


def default_db_path() -> Path:
    root = Path(os.environ.get("AGENT_BOM_HOME", str(Path.home() / ".agent-bom")))
    root.mkdir(parents=True, exist_ok=True)
    return root / "vuln.db"


class AdvisoryStore:
    """SQLite advisory store with columnar arena export."""

    def __init__(self, path: str | Path):
        self.path = Path(path)
        self.path.parent.mkdir(parents=True, exist_ok=True)
        self.conn = sqlite3.connect(str(self.path))
        self.conn.executescript(_DDL)
        cur = self.conn.execute("SELECT COUNT(*) FROM schema_version")
        if cur.fetchone()[0] == 0:
            self.conn.execute("INSERT INTO schema_version(version) VALUES (?)", (SCHEMA_VERSION,))
        self.conn.commit()

    def close(self) -> None:
        self.conn.close()

    # ── ingest ─────────────────────────────────────────────────────────────

    def ingest_windows(self, windows: Sequence[AdvisoryWindow], source: str = "osv") -> int:
        n = 0
        for w in windows:
            # NOTE: never INSERT OR REPLACE into vulns — REPLACE is
            # DELETE+INSERT and the affected.vuln_id ON DELETE CASCADE would
            # silently wipe sibling windows of a multi-branch advisory
            # (exactly the collapse the schema-v5 PK exists to prevent).
            self.conn.execute(
                "INSERT INTO vulns(id, summary, severity, cvss_score, fixed_version,"
                " cwe_ids, aliases, source) VALUES (?,?,?,?,?,?,?,?)"
                " ON CONFLICT(id) DO UPDATE SET summary=excluded.summary,"
                " severity=excluded.severity, cvss_score=excluded.cvss_score,"
                " fixed_version=excluded.fixed_version, cwe_ids=excluded.cwe_ids,"
                " aliases=excluded.aliases, source=excluded.source",
                (
                    w.vuln_id, w.summary, w.severity.value, w.cvss_score,
                    w.fixed_version or w.fixed, ",".join(w.cwe_ids), ",".join(w.aliases), source,
                ),
            )
            self.conn.execute(
                "INSERT OR IGNORE INTO affected(vuln_id, ecosystem, package_name, introduced,"
                " fixed, last_affected) VALUES (?,?,?,?,?,?)",
                (
                    w.vuln_id,
                    normalize_package_ecosystem(w.ecosystem),
                    normalize_package_name(w.package_name, w.ecosystem),
                    w.introduced or "",
                    w.fixed or "",
                    w.last_affected or "",
                ),
            )
            if w.epss_score is not None:
                self.conn.execute(
                    "INSERT OR REPLACE INTO epss_scores(cve_id, probability, percentile,"
                    " updated_at) VALUES (?,?,?,?)",
                    (w.vuln_id, w.epss_score, None, _now()),
                )
            if w.is_kev:
                self.conn.execute(
                    "INSERT OR REPLACE INTO kev_entries(cve_id, date_added, due_date, product,"
                    " vendor_project) VALUES (?,?,?,?,?)",
                    (w.vuln_id, "2023-09-27", "2023-10-04", w.package_name, w.ecosystem),
                )
            n += 1
        self.conn.execute(
            "INSERT OR REPLACE INTO sync_meta(source, last_synced, record_count) VALUES (?,?,?)",
            (source, _now(), n),
        )
        self.conn.commit()
        return n

    # ── export ─────────────────────────────────────────────────────────────

    def export_windows(self) -> list[AdvisoryWindow]:
        """All affected windows joined to vuln metadata + EPSS/KEV."""
        rows = self.conn.execute(
            """
            SELECT a.ecosystem, a.package_name, a.introduced, a.fixed, a.last_affected,
                   v.id, v.summary, v.severity, v.cvss_score, v.cwe_ids, v.aliases,
                   v.fixed_version,
                   e.probability, (k.cve_id IS NOT NULL) AS is_kev
            FROM affected a
            JOIN vulns v ON v.id = a.vuln_id
            LEFT JOIN epss_scores e ON e.cve_id = v.id
            LEFT JOIN kev_entries k ON k.cve_id = v.id
            ORDER BY a.ecosystem, a.package_name, v.id, a.introduced, a.fixed
            """
        ).fetchall()
        out = []
        for (eco, name, intro, fixed, last, vid, summary, sev, cvss, cwes, aliases,
             fixed_version, epss, kev) in rows:
            try:
                severity = Severity(sev)
            except ValueError:
                severity = Severity.UNKNOWN
            out.append(
                AdvisoryWindow(
                    ecosystem=eco,
                    package_name=name,
                    vuln_id=vid,
                    introduced=intro or None,
                    fixed=fixed or None,
                    last_affected=last or None,
                    severity=severity,
                    cvss_score=cvss,
                    epss_score=epss,
                    is_kev=bool(kev),
                    unfixed=not fixed and not last,
                    summary=summary,
                    cwe_ids=tuple(c for c in (cwes or "").split(",") if c),
                    aliases=tuple(a for a in (aliases or "").split(",") if a),
                    fixed_version=fixed_version,
                )
            )
        return out

    def build_arena(self, include_unfixed: bool = False) -> AdvisoryArena:
        return build_arena(self.export_windows(), include_unfixed=include_unfixed)

    def status(self) -> dict:
        counts = {
            "vulns": self.conn.execute("SELECT COUNT(*) FROM vulns").fetchone()[0],
            "affected_windows": self.conn.execute("SELECT COUNT(*) FROM affected").fetchone()[0],
            "epss_scores": self.conn.execute("SELECT COUNT(*) FROM epss_scores").fetchone()[0],
            "kev_entries": self.conn.execute("SELECT COUNT(*) FROM kev_entries").fetchone()[0],
        }
        sync = {
            row[0]: {"last_synced": row[1], "record_count": row[2]}
            for row in self.conn.execute("SELECT source, last_synced, record_count FROM sync_meta")
        }
        return {"path": str(self.path), "schema_version": SCHEMA_VERSION,
                "counts": counts, "sync": sync,
                "freshness": grade_freshness(sync)}


def grade_freshness(sync: dict, now=None) -> dict:
    """Advisory staleness grading (reference: vuln_freshness.py).

    fresh < 3 days <= stale < 30 days <= expired; never-synced sources are
    "unknown".  Scanning against an expired DB is surfaced as a scan
    warning — decisions made on month-old advisories mislead.
    """
    from datetime import datetime, timezone

    now = now or datetime.now(timezone.utc)
    per_source = {}
    worst = "fresh"
    order = {"fresh": 0, "stale": 1, "expired": 2, "unknown": 3}
    for source, meta in sync.items():
        last = meta.get("last_synced")
        try:
            age_days = (now - datetime.fromisoformat(str(last))).total_seconds() / 86400
        except (TypeError, ValueError):
            per_source[source] = {"grade": "unknown", "age_days": None}
            worst = max(worst, "unknown", key=lambda g: order[g])
            continue
        grade = ("fresh" if age_days < 3 else
                 "stale" if age_days < 30 else "expired")
        per_source[source] = {"grade": grade, "age_days": round(age_days, 2)}
        worst = max(worst, grade, key=lambda g: order[g])
    if not per_source:
        worst = "unknown"
    return {"overall": worst, "sources": per_source}


def _now() -> str:
    return datetime.now(timezone.utc).isoformat()


def load_advisory_windows(offline: bool = False, path: Optional[Path] = None) -> list[AdvisoryWindow]:
    """Windows for a real scan: the local DB if present, else the bundled
    demo advisories (offline-image fallback so scans always have coverage)."""
    db_path = path or default_db_path()
    if db_path.exists():
        store = AdvisoryStore(db_path)
        try:
            windows = store.export_windows()
            if windows:
                return windows
        finally:
            store.close()
    from agentbom_amd.scan.demo import demo_advisory_windows

    return demo_advisory_windows()
