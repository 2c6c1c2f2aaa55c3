"""PostgresGraphStore logic over an in-process DB-API shim.

psycopg is not in this image, so the full store logic (DDL, snapshot
round-trips, retention, tenant filtering, diff) runs against a sqlite
connection wrapped with a %s->? paramstyle adapter; Postgres-only
surfaces (set_config GUC, RLS DDL) are intercepted and recorded so the
emitted statements are still asserted.
"""

from __future__ import annotations

import re
import sqlite3

import pytest

from agentbom_amd.graph.container import UnifiedEdge, UnifiedGraph, UnifiedNode
from agentbom_amd.graph.pg_store import PostgresGraphStore, rls_ddl
from agentbom_amd.graph.types import EntityType, RelationshipType


class _Cursor:
    def __init__(self, real, recorded):
        self._c = real
        self._rec = recorded

    def _xlate(self, sql: str) -> str:
        sql = sql.replace("DOUBLE PRECISION", "REAL")
        return sql.replace("%s", "?")

    def execute(self, sql, params=()):
        if "set_config" in sql or "POLICY" in sql or "ROW LEVEL SECURITY" in sql:
            self._rec.append((sql, tuple(params)))
            return self
        self._c.execute(self._xlate(sql), params)
        return self

    def executemany(self, sql, rows):
        self._c.executemany(self._xlate(sql), rows)
        return self

    def fetchone(self):
        return self._c.fetchone()

    def fetchall(self):
        return self._c.fetchall()


class _Conn:
    def __init__(self):
        self._c = sqlite3.connect(":memory:")
        self.recorded: list = []

    def cursor(self):
        return _Cursor(self._c.cursor(), self.recorded)

    def commit(self):
        self._c.commit()

    def close(self):
        self._c.close()


def _graph(n: int = 4, tag: str = "a") -> UnifiedGraph:
    g = UnifiedGraph()
    for i in range(n):
        g.add_node(UnifiedNode(id=f"{tag}{i}", entity_type=EntityType.AGENT,
                               label=f"{tag}{i}"))
    for i in range(n - 1):
        g.add_edge(UnifiedEdge(source=f"{tag}{i}", target=f"{tag}{i+1}",
                               relationship=RelationshipType.USES))
    return g


@pytest.fixture
def store():
    return PostgresGraphStore(connection_factory=_Conn, retention=3)


def test_snapshot_roundtrip(store):
    g = _graph(5)
    sid = store.save_snapshot(g, scan_id="s1", tenant_id="acme")
    loaded = store.load_snapshot(sid)
    assert loaded is not None
    assert set(loaded.nodes) == set(g.nodes)
    assert loaded.edge_count == g.edge_count
    assert store.load_snapshot("missing") is None


def test_tenant_filtered_listing(store):
    store.save_snapshot(_graph(2, "x"), tenant_id="acme")
    store.save_snapshot(_graph(3, "y"), tenant_id="globex")
    acme = store.list_snapshots("acme")
    globex = store.list_snapshots("globex")
    assert len(acme) == 1 and len(globex) == 1
    assert acme[0]["node_count"] == 2 and globex[0]["node_count"] == 3
    assert store.latest_snapshot_id("acme") == acme[0]["snapshot_id"]
    assert store.latest_snapshot_id("nobody") is None


def test_retention(store):
    for i in range(5):
        store.save_snapshot(_graph(2, f"t{i}"), tenant_id="acme")
    assert len(store.list_snapshots("acme")) == 3  # retention=3


def test_diff(store):
    a = store.save_snapshot(_graph(3, "n"), tenant_id="d")
    b = store.save_snapshot(_graph(4, "n"), tenant_id="d")
    d = store.diff_snapshots(a, b)
    assert d["nodes_added"] == ["n3"] and d["node_count_delta"] == 1


def test_set_tenant_emits_guc(store):
    store.set_tenant("acme")
    assert any("set_config" in sql and params == ("acme",)
               for sql, params in store.conn.recorded)


def test_rls_policy_statements():
    stmts = rls_ddl()
    assert len(stmts) == 9  # 3 tables x (enable, drop, create policy)
    for table in ("graph_snapshots", "graph_nodes", "graph_edges"):
        assert any(f"ON {table} USING" in s and "abom.tenant" in s for s in stmts)
        assert any(re.search(rf"ALTER TABLE {table} ENABLE ROW LEVEL", s) for s in stmts)


def test_rls_applied_at_init():
    conn = _Conn()
    PostgresGraphStore(connection_factory=lambda: conn, apply_rls=True)
    assert sum("POLICY" in sql for sql, _ in conn.recorded) >= 3
