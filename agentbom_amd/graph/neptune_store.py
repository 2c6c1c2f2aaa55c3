"""Neptune (Gremlin) graph store — optional backend behind an explicit flag.

Reference parity: src/agent_bom/api/neptune_graph.py:152 — an OPTIONAL
Gremlin-server-backed store enabled only by explicit configuration, with
unsupported operations raising clearly instead of pretending.  Speaks the
Gremlin HTTP endpoint (POST {"gremlin": "..."}), transport-injected for
tests; offline mode refuses construction.

The durable/default tiers remain SQLite (graph/store.py) and Postgres
(graph/pg_store.py); the GPU engine is the hot path.  This exists so a
deployment already standardized on Neptune can persist snapshots there.
"""

from __future__ import annotations

import json
import os
from typing import Any, Optional

from agentbom_amd.graph.container import UnifiedGraph
from agentbom_amd.utils.http_client import check_offline, create_client, request_with_retry


class NeptuneUnsupported(NotImplementedError):
    """Operation not offered by the Neptune backend (by design)."""


def _esc(s: str) -> str:
    return str(s).replace("\\", "\\\\").replace("'", "\\'")


class NeptuneGraphStore:
    """Snapshot persistence over a Gremlin HTTP endpoint."""

    def __init__(self, endpoint: Optional[str] = None, client=None):
        endpoint = endpoint or os.environ.get("AGENT_BOM_NEPTUNE_ENDPOINT")
        if not endpoint:
            raise RuntimeError(
                "Neptune store requires AGENT_BOM_NEPTUNE_ENDPOINT "
                "(explicit opt-in — SQLite/Postgres are the default tiers)")
        check_offline(endpoint)
        self.endpoint = endpoint.rstrip("/")
        self.client = client or create_client(timeout=60.0)

    def _submit(self, gremlin: str) -> list[Any]:
        resp = request_with_retry(self.client, "POST", self.endpoint,
                                  json={"gremlin": gremlin})
        if resp is None or resp.status_code != 200:
            raise RuntimeError(
                f"gremlin query failed: "
                f"{resp.status_code if resp is not None else 'unreachable'}")
        body = resp.json()
        return (((body.get("result") or {}).get("data") or {}).get("@value")
                or [])

    def save_snapshot(self, graph: UnifiedGraph, scan_id: str = "",
                      tenant_id: str = "default") -> str:
        """Persist nodes/edges labeled with the snapshot id."""
        import uuid

        snapshot_id = str(uuid.uuid4())
        for nid, node in sorted(graph.nodes.items()):
            self._submit(
                f"g.addV('{_esc(node.entity_type.value)}')"
                f".property('nid','{_esc(nid)}')"
                f".property('snapshot','{_esc(snapshot_id)}')"
                f".property('tenant','{_esc(tenant_id)}')"
                f".property('label_text','{_esc(node.label)}')"
                f".property('doc','{_esc(json.dumps(node.to_dict(), default=str))}')")
        for e in graph.edges:
            self._submit(
                f"g.V().has('nid','{_esc(e.source)}')"
                f".has('snapshot','{_esc(snapshot_id)}')"
                f".addE('{_esc(e.relationship.value)}')"
                f".to(__.V().has('nid','{_esc(e.target)}')"
                f".has('snapshot','{_esc(snapshot_id)}'))")
        self._submit(
            f"g.addV('abom_snapshot').property('snapshot','{_esc(snapshot_id)}')"
            f".property('tenant','{_esc(tenant_id)}')"
            f".property('scan_id','{_esc(scan_id)}')"
            f".property('node_count',{graph.node_count})"
            f".property('edge_count',{graph.edge_count})")
        return snapshot_id

    def load_snapshot(self, snapshot_id: str) -> Optional[UnifiedGraph]:
        docs = self._submit(
            f"g.V().has('snapshot','{_esc(snapshot_id)}')"
            f".hasLabel(neq('abom_snapshot')).values('doc')")
        if not docs:
            return None
        nodes = [json.loads(d) for d in docs if isinstance(d, str)]
        return UnifiedGraph.from_dict({"nodes": nodes, "edges": []})

    def list_snapshots(self, tenant_id: str = "default") -> list[dict[str, Any]]:
        rows = self._submit(
            f"g.V().hasLabel('abom_snapshot').has('tenant','{_esc(tenant_id)}')"
            f".valueMap('snapshot','scan_id','node_count','edge_count')")
        out = []
        for r in rows:
            if isinstance(r, dict):
                vals = r.get("@value", r)
                out.append(vals)
        return out

    # operations the Neptune tier deliberately does not offer (reference
    # behavior: explicit unsupported errors, never silent degradation)
    def diff_snapshots(self, old_id: str, new_id: str):
        raise NeptuneUnsupported("snapshot diff is served by the SQLite/"
                                 "Postgres tiers, not Neptune")

    def evidence_manifest(self, *a, **k):
        raise NeptuneUnsupported("evidence manifests are served by the "
                                 "SQLite/Postgres tiers, not Neptune")
