"""SQLite graph store: snapshots, pagination, diffs, evidence manifests.

Reference: src/agent_bom/api/graph_store.py:490 (SQLiteGraphStore:
snapshot save/load, cursor pagination, search rows),
src/agent_bom/db/graph_store.py (retention policy, snapshot digests,
graph_evidence_manifest :1721, graph_history :1763) and the
GraphStoreProtocol (:154).

One store serves both roles here: durable snapshots on disk while the hot
query path stays in-process (UnifiedGraph / GPU CSR) — the Postgres
server-side-walk tier of the reference is replaced by the engine.
"""

from __future__ import annotations

import hashlib
import json
import sqlite3
import time
from datetime import datetime, timezone
from pathlib import Path
from typing import Any, Optional

from agentbom_amd.graph.container import UnifiedGraph

SCHEMA = """
CREATE TABLE IF NOT EXISTS graph_snapshots (
    snapshot_id   TEXT PRIMARY KEY,
    tenant_id     TEXT NOT NULL DEFAULT 'default',
    scan_id       TEXT,
    created_at    REAL NOT NULL,
    node_count    INTEGER NOT NULL,
    edge_count    INTEGER NOT NULL,
    digest        TEXT NOT NULL,
    complete      INTEGER NOT NULL DEFAULT 1
);
CREATE TABLE IF NOT EXISTS graph_nodes (
    snapshot_id   TEXT NOT NULL,
    node_id       TEXT NOT NULL,
    entity_type   TEXT NOT NULL,
    label         TEXT NOT NULL,
    layer         TEXT,
    status        TEXT NOT NULL,
    properties    TEXT NOT NULL DEFAULT '{}',
    tags          TEXT NOT NULL DEFAULT '[]',
    PRIMARY KEY (snapshot_id, node_id)
);
CREATE INDEX IF NOT EXISTS idx_nodes_type ON graph_nodes(snapshot_id, entity_type);
CREATE INDEX IF NOT EXISTS idx_nodes_label ON graph_nodes(snapshot_id, label);
CREATE TABLE IF NOT EXISTS graph_edges (
    snapshot_id   TEXT NOT NULL,
    source        TEXT NOT NULL,
    target        TEXT NOT NULL,
    relationship  TEXT NOT NULL,
    weight        REAL NOT NULL DEFAULT 1.0,
    bidirectional INTEGER NOT NULL DEFAULT 0,
    properties    TEXT NOT NULL DEFAULT '{}',
    PRIMARY KEY (snapshot_id, source, target, relationship)
);
CREATE INDEX IF NOT EXISTS idx_edges_src ON graph_edges(snapshot_id, source);
"""


class SQLiteGraphStore:
    """Durable snapshot store with retention + evidence manifests."""

    def __init__(self, path: str | Path, retention: int = 10):
        self.path = Path(path)
        self.path.parent.mkdir(parents=True, exist_ok=True)
        self.retention = retention
        self.conn = sqlite3.connect(str(self.path))
        self.conn.executescript(SCHEMA)
        self.conn.commit()

    def close(self) -> None:
        self.conn.close()

    # ── save / load ───────────────────────────────────────────────────────

    def save_snapshot(self, graph: UnifiedGraph, scan_id: str = "",
                      tenant_id: str = "default") -> str:
        doc = graph.to_dict()
        digest = hashlib.sha256(
            json.dumps(doc, sort_keys=True, default=str).encode()
        ).hexdigest()
        snapshot_id = f"snap-{int(time.time() * 1000)}-{digest[:8]}"
        cur = self.conn
        cur.execute(
            "INSERT INTO graph_snapshots(snapshot_id, tenant_id, scan_id, created_at,"
            " node_count, edge_count, digest, complete) VALUES (?,?,?,?,?,?,?,?)",
            (snapshot_id, tenant_id, scan_id, time.time(), graph.node_count,
             graph.edge_count, digest, int(not graph.truncated)),
        )
        cur.executemany(
            "INSERT INTO graph_nodes(snapshot_id, node_id, entity_type, label, layer,"
            " status, properties, tags) VALUES (?,?,?,?,?,?,?,?)",
            [
                (snapshot_id, n["id"], n["entity_type"], n["label"], n["layer"],
                 n["status"], json.dumps(n["properties"], default=str),
                 json.dumps(n["tags"]))
                for n in doc["nodes"]
            ],
        )
        cur.executemany(
            "INSERT OR IGNORE INTO graph_edges(snapshot_id, source, target, relationship,"
            " weight, bidirectional, properties) VALUES (?,?,?,?,?,?,?)",
            [
                (snapshot_id, e["source"], e["target"], e["relationship"], e["weight"],
                 int(e["bidirectional"]), json.dumps(e["properties"], default=str))
                for e in doc["edges"]
            ],
        )
        self.conn.commit()
        self._apply_retention(tenant_id)
        return snapshot_id

    def load_snapshot(self, snapshot_id: str) -> Optional[UnifiedGraph]:
        row = self.conn.execute(
            "SELECT 1 FROM graph_snapshots WHERE snapshot_id=?", (snapshot_id,)
        ).fetchone()
        if row is None:
            return None
        nodes = [
            {
                "id": r[0], "entity_type": r[1], "label": r[2], "layer": r[3],
                "status": r[4], "properties": json.loads(r[5]), "tags": json.loads(r[6]),
            }
            for r in self.conn.execute(
                "SELECT node_id, entity_type, label, layer, status, properties, tags"
                " FROM graph_nodes WHERE snapshot_id=? ORDER BY node_id", (snapshot_id,)
            )
        ]
        edges = [
            {
                "source": r[0], "target": r[1], "relationship": r[2], "weight": r[3],
                "bidirectional": bool(r[4]), "properties": json.loads(r[5]),
            }
            for r in self.conn.execute(
                "SELECT source, target, relationship, weight, bidirectional, properties"
                " FROM graph_edges WHERE snapshot_id=?", (snapshot_id,)
            )
        ]
        return UnifiedGraph.from_dict({"nodes": nodes, "edges": edges})

    def list_snapshots(self, tenant_id: str = "default") -> list[dict[str, Any]]:
        return [
            {"snapshot_id": r[0], "scan_id": r[1], "created_at": r[2],
             "node_count": r[3], "edge_count": r[4], "digest": r[5],
             "complete": bool(r[6])}
            for r in self.conn.execute(
                "SELECT snapshot_id, scan_id, created_at, node_count, edge_count,"
                " digest, complete FROM graph_snapshots WHERE tenant_id=?"
                " ORDER BY created_at DESC", (tenant_id,)
            )
        ]

    def latest_snapshot_id(self, tenant_id: str = "default") -> Optional[str]:
        row = self.conn.execute(
            "SELECT snapshot_id FROM graph_snapshots WHERE tenant_id=?"
            " ORDER BY created_at DESC LIMIT 1", (tenant_id,)
        ).fetchone()
        return row[0] if row else None

    def _apply_retention(self, tenant_id: str) -> None:
        ids = [r[0] for r in self.conn.execute(
            "SELECT snapshot_id FROM graph_snapshots WHERE tenant_id=?"
            " ORDER BY created_at DESC", (tenant_id,)
        )]
        for old in ids[self.retention:]:
            self.conn.execute("DELETE FROM graph_snapshots WHERE snapshot_id=?", (old,))
            self.conn.execute("DELETE FROM graph_nodes WHERE snapshot_id=?", (old,))
            self.conn.execute("DELETE FROM graph_edges WHERE snapshot_id=?", (old,))
        self.conn.commit()

    # ── paginated reads ───────────────────────────────────────────────────

    def nodes_page(self, snapshot_id: str, cursor: Optional[str] = None,
                   limit: int = 100, entity_type: Optional[str] = None) -> dict[str, Any]:
        """Cursor pagination over node_id ordering."""
        q = "SELECT node_id, entity_type, label, status FROM graph_nodes WHERE snapshot_id=?"
        params: list[Any] = [snapshot_id]
        if cursor:
            q += " AND node_id > ?"
            params.append(cursor)
        if entity_type:
            q += " AND entity_type = ?"
            params.append(entity_type)
        q += " ORDER BY node_id LIMIT ?"
        params.append(limit + 1)
        rows = self.conn.execute(q, params).fetchall()
        has_more = len(rows) > limit
        rows = rows[:limit]
        return {
            "nodes": [
                {"id": r[0], "entity_type": r[1], "label": r[2], "status": r[3]}
                for r in rows
            ],
            "next_cursor": rows[-1][0] if has_more and rows else None,
        }

    def search_nodes(self, snapshot_id: str, query: str, limit: int = 100) -> list[dict]:
        rows = self.conn.execute(
            "SELECT node_id, entity_type, label, status FROM graph_nodes"
            " WHERE snapshot_id=? AND (label LIKE ? OR node_id LIKE ?)"
            " ORDER BY node_id LIMIT ?",
            (snapshot_id, f"%{query}%", f"%{query}%", limit),
        ).fetchall()
        return [{"id": r[0], "entity_type": r[1], "label": r[2], "status": r[3]}
                for r in rows]

    # ── diff + evidence ───────────────────────────────────────────────────

    def diff_snapshots(self, old_id: str, new_id: str) -> dict[str, Any]:
        def node_set(sid: str) -> set[str]:
            return {r[0] for r in self.conn.execute(
                "SELECT node_id FROM graph_nodes WHERE snapshot_id=?", (sid,))}

        def edge_set(sid: str) -> set[tuple]:
            return {tuple(r) for r in self.conn.execute(
                "SELECT source, target, relationship FROM graph_edges"
                " WHERE snapshot_id=?", (sid,))}

        on, nn = node_set(old_id), node_set(new_id)
        oe, ne = edge_set(old_id), edge_set(new_id)
        return {
            "old": old_id,
            "new": new_id,
            "nodes_added": sorted(nn - on),
            "nodes_removed": sorted(on - nn),
            "edges_added": sorted(ne - oe),
            "edges_removed": sorted(oe - ne),
            "summary": {
                "nodes_added": len(nn - on), "nodes_removed": len(on - nn),
                "edges_added": len(ne - oe), "edges_removed": len(oe - ne),
            },
        }

    def evidence_manifest(self, tenant_id: str = "default",
                          scan_id: Optional[str] = None) -> dict[str, Any]:
        """agent-bom.graph_evidence_manifest/v1 (db/graph_store.py:1721)."""
        snaps = self.list_snapshots(tenant_id)
        latest = snaps[0] if snaps else None
        diff_summary = None
        baseline = snaps[1]["snapshot_id"] if len(snaps) > 1 else None
        if latest and baseline:
            diff_summary = self.diff_snapshots(baseline, latest["snapshot_id"])["summary"]
        return {
            "schema_version": "agent-bom.graph_evidence_manifest/v1",
            "tenant_id": tenant_id,
            "scan_id": scan_id or (latest["scan_id"] if latest else None),
            "generated_at": datetime.now(timezone.utc).isoformat(),
            "scan_created_at": (
                datetime.fromtimestamp(latest["created_at"], timezone.utc).isoformat()
                if latest else None
            ),
            "graph_digest": latest["digest"] if latest else None,
            "findings_digest": None,
            "diff_baseline_scan_id": baseline,
            "diff_summary": diff_summary,
            "counts": {
                "nodes": latest["node_count"] if latest else 0,
                "edges": latest["edge_count"] if latest else 0,
                "snapshots": len(snaps),
            },
            "included_tables": ["graph_snapshots", "graph_nodes", "graph_edges"],
            "excluded_private_fields": ["env_values", "credential_values"],
            "retention_policy": {"snapshots": self.retention},
        }

    def graph_history(self, tenant_id: str = "default") -> dict[str, Any]:
        """Retained snapshots + adjacent diff summaries (db/graph_store.py:1763)."""
        snaps = self.list_snapshots(tenant_id)
        entries = []
        for i, snap in enumerate(snaps):
            entry = dict(snap)
            if i + 1 < len(snaps):
                entry["diff_vs_previous"] = self.diff_snapshots(
                    snaps[i + 1]["snapshot_id"], snap["snapshot_id"]
                )["summary"]
            entries.append(entry)
        return {"schema_version": "agent-bom.graph_history/v1",
                "tenant_id": tenant_id, "snapshots": entries}
