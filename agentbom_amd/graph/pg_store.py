"""Postgres graph store: the shared-deployment mirror of graph/store.py.

Reference parity target: src/agent_bom/api/postgres_graph.py:250
PostgresGraphStore (tenant-scoped tables + RLS policies).  In this build
the HOT graph path is the GPU engine; Postgres is the durable multi-
replica control-plane tier, so this store mirrors SQLiteGraphStore's
protocol (save/load/list/diff/evidence) over any DB-API connection:

- ``conninfo`` uses psycopg when installed (optional dependency — the
  ROCm image ships without it; the store raises a clear error otherwise);
- ``connection_factory`` injects any DB-API connection, which is how the
  tests run the FULL store logic against an in-process shim
  (tests/test_pg_store.py) without a server;
- tenant isolation is structural (every query filters tenant_id) AND
  declarative: ``rls_ddl()`` emits the row-level-security policies a real
  Postgres deployment applies (CREATE POLICY ... USING tenant_id =
  current_setting('abom.tenant')), and ``set_tenant()`` pins the session
  GUC the policies read.
"""

from __future__ import annotations

import json
import time
import uuid
from typing import Any, Callable, Optional

from agentbom_amd.graph.container import UnifiedGraph

_DDL = [
    """CREATE TABLE IF NOT EXISTS graph_snapshots (
        snapshot_id TEXT PRIMARY KEY,
        tenant_id   TEXT NOT NULL DEFAULT 'default',
        scan_id     TEXT NOT NULL DEFAULT '',
        created_at  DOUBLE PRECISION NOT NULL,
        node_count  INTEGER NOT NULL,
        edge_count  INTEGER NOT NULL,
        digest      TEXT NOT NULL DEFAULT '',
        complete    BOOLEAN NOT NULL DEFAULT TRUE
    )""",
    """CREATE TABLE IF NOT EXISTS graph_nodes (
        snapshot_id TEXT NOT NULL,
        tenant_id   TEXT NOT NULL DEFAULT 'default',
        node_id     TEXT NOT NULL,
        doc         TEXT NOT NULL,
        PRIMARY KEY (snapshot_id, node_id)
    )""",
    """CREATE TABLE IF NOT EXISTS graph_edges (
        snapshot_id TEXT NOT NULL,
        tenant_id   TEXT NOT NULL DEFAULT 'default',
        ord         INTEGER NOT NULL,
        doc         TEXT NOT NULL,
        PRIMARY KEY (snapshot_id, ord)
    )""",
]

_RLS_DDL = [
    "ALTER TABLE {t} ENABLE ROW LEVEL SECURITY",
    "DROP POLICY IF EXISTS abom_tenant_isolation ON {t}",
    ("CREATE POLICY abom_tenant_isolation ON {t} USING "
     "(tenant_id = current_setting('abom.tenant', true))"),
]


def rls_ddl() -> list[str]:
    """Row-level-security statements for a real Postgres deployment."""
    out = []
    for table in ("graph_snapshots", "graph_nodes", "graph_edges"):
        out.extend(stmt.format(t=table) for stmt in _RLS_DDL)
    return out


class PostgresGraphStore:
    """SQLiteGraphStore-protocol store over a DB-API connection."""

    def __init__(self, conninfo: Optional[str] = None,
                 connection_factory: Optional[Callable[[], Any]] = None,
                 retention: int = 10, apply_rls: bool = False):
        if connection_factory is not None:
            self.conn = connection_factory()
        else:
            try:
                import psycopg
            except ImportError as exc:  # pragma: no cover - env-dependent
                raise RuntimeError(
                    "PostgresGraphStore needs psycopg (pip install psycopg) "
                    "or an injected connection_factory") from exc
            self.conn = psycopg.connect(conninfo or "")
        self.retention = retention
        cur = self.conn.cursor()
        for ddl in _DDL:
            cur.execute(ddl)
        if apply_rls:
            for stmt in rls_ddl():
                cur.execute(stmt)
        self.conn.commit()

    def close(self) -> None:
        self.conn.close()

    def set_tenant(self, tenant_id: str) -> None:
        """Pin the session tenant GUC the RLS policies read."""
        cur = self.conn.cursor()
        cur.execute("SELECT set_config('abom.tenant', %s, false)", (tenant_id,))
        self.conn.commit()

    # ── protocol (mirrors graph/store.SQLiteGraphStore) ────────────────────

    def save_snapshot(self, graph: UnifiedGraph, scan_id: str = "",
                      tenant_id: str = "default") -> str:
        import hashlib

        digest = hashlib.sha256(
            json.dumps(graph.to_dict(), sort_keys=True, default=str).encode()
        ).hexdigest()
        snapshot_id = str(uuid.uuid4())
        cur = self.conn.cursor()
        cur.execute(
            "INSERT INTO graph_snapshots(snapshot_id, tenant_id, scan_id,"
            " created_at, node_count, edge_count, digest, complete)"
            " VALUES (%s,%s,%s,%s,%s,%s,%s,%s)",
            (snapshot_id, tenant_id, scan_id, time.time(), graph.node_count,
             graph.edge_count, digest, True))
        cur.executemany(
            "INSERT INTO graph_nodes(snapshot_id, tenant_id, node_id, doc)"
            " VALUES (%s,%s,%s,%s)",
            [(snapshot_id, tenant_id, nid, json.dumps(node.to_dict()))
             for nid, node in sorted(graph.nodes.items())])
        cur.executemany(
            "INSERT INTO graph_edges(snapshot_id, tenant_id, ord, doc)"
            " VALUES (%s,%s,%s,%s)",
            [(snapshot_id, tenant_id, i, json.dumps(e.to_dict()))
             for i, e in enumerate(graph.edges)])
        self.conn.commit()
        self._apply_retention(tenant_id)
        return snapshot_id

    def load_snapshot(self, snapshot_id: str) -> Optional[UnifiedGraph]:
        cur = self.conn.cursor()
        cur.execute("SELECT 1 FROM graph_snapshots WHERE snapshot_id=%s",
                    (snapshot_id,))
        if cur.fetchone() is None:
            return None
        cur.execute("SELECT doc FROM graph_nodes WHERE snapshot_id=%s"
                    " ORDER BY node_id", (snapshot_id,))
        nodes = [json.loads(r[0]) for r in cur.fetchall()]
        cur.execute("SELECT doc FROM graph_edges WHERE snapshot_id=%s"
                    " ORDER BY ord", (snapshot_id,))
        edges = [json.loads(r[0]) for r in cur.fetchall()]
        return UnifiedGraph.from_dict({"nodes": nodes, "edges": edges})

    def list_snapshots(self, tenant_id: str = "default") -> list[dict[str, Any]]:
        cur = self.conn.cursor()
        cur.execute(
            "SELECT snapshot_id, scan_id, created_at, node_count, edge_count,"
            " digest, complete FROM graph_snapshots WHERE tenant_id=%s"
            " ORDER BY created_at DESC", (tenant_id,))
        cols = ("snapshot_id", "scan_id", "created_at", "node_count",
                "edge_count", "digest", "complete")
        return [dict(zip(cols, row)) for row in cur.fetchall()]

    def latest_snapshot_id(self, tenant_id: str = "default") -> Optional[str]:
        cur = self.conn.cursor()
        cur.execute(
            "SELECT snapshot_id FROM graph_snapshots WHERE tenant_id=%s"
            " ORDER BY created_at DESC LIMIT 1", (tenant_id,))
        row = cur.fetchone()
        return row[0] if row else None

    def diff_snapshots(self, old_id: str, new_id: str) -> dict[str, Any]:
        def node_ids(sid: str) -> set[str]:
            cur = self.conn.cursor()
            cur.execute("SELECT node_id FROM graph_nodes WHERE snapshot_id=%s",
                        (sid,))
            return {r[0] for r in cur.fetchall()}

        old_nodes, new_nodes = node_ids(old_id), node_ids(new_id)
        return {
            "nodes_added": sorted(new_nodes - old_nodes),
            "nodes_removed": sorted(old_nodes - new_nodes),
            "node_count_delta": len(new_nodes) - len(old_nodes),
        }

    def _apply_retention(self, tenant_id: str) -> None:
        cur = self.conn.cursor()
        cur.execute(
            "SELECT snapshot_id FROM graph_snapshots WHERE tenant_id=%s"
            " ORDER BY created_at DESC", (tenant_id,))
        stale = [r[0] for r in cur.fetchall()][self.retention:]
        for sid in stale:
            cur.execute("DELETE FROM graph_nodes WHERE snapshot_id=%s", (sid,))
            cur.execute("DELETE FROM graph_edges WHERE snapshot_id=%s", (sid,))
            cur.execute("DELETE FROM graph_snapshots WHERE snapshot_id=%s", (sid,))
        self.conn.commit()
