"""Store-backed UnifiedGraph: same interface, bounded RSS.

Reference parity: src/agent_bom/graph/store_backed.py (StoreBackedUnifiedGraph
— LRU node cache + dirty write-back over a SQLite workspace, auto-wired for
>= 5,000-entity builds so estate-scale graph construction doesn't hold every
node payload in RAM).

MI355X-first design: the ADJACENCY SKELETON (node ids, edge list, forward/
reverse index) stays in RAM — it is what traversals and the GPU CSR build
touch, and it is compact (ints + ids).  Only the node PAYLOADS (properties,
tags, labels) spill to SQLite, fetched on demand through an LRU with dirty
write-back.  Every UnifiedGraph method works unchanged because ``nodes``
keeps its Mapping contract.
"""

from __future__ import annotations

import json
import sqlite3
import tempfile
from collections import OrderedDict
from collections.abc import MutableMapping
from pathlib import Path
from typing import Any, Iterator, Optional

from agentbom_amd.graph.container import UnifiedGraph, UnifiedNode
from agentbom_amd.graph.types import EntityType, GraphSemanticLayer, NodeStatus

AUTO_SPILL_THRESHOLD = 5_000

_SCHEMA = """
CREATE TABLE IF NOT EXISTS live_nodes (
    node_id TEXT PRIMARY KEY,
    payload TEXT NOT NULL
);
"""


def _serialize(node: UnifiedNode) -> str:
    return json.dumps({
        "id": node.id, "entity_type": node.entity_type.value,
        "label": node.label,
        "layer": node.layer.value if node.layer else None,
        "status": node.status.value, "properties": node.properties,
        "tags": node.tags, "canonical_id": node.canonical_id,
        "first_seen": node.first_seen, "last_seen": node.last_seen,
    }, default=str)


def _deserialize(payload: str) -> UnifiedNode:
    d = json.loads(payload)
    return UnifiedNode(
        id=d["id"], entity_type=EntityType(d["entity_type"]), label=d["label"],
        layer=GraphSemanticLayer(d["layer"]) if d.get("layer") else None,
        status=NodeStatus(d.get("status", "active")),
        properties=d.get("properties", {}), tags=d.get("tags", []),
        canonical_id=d.get("canonical_id"),
        first_seen=d.get("first_seen"), last_seen=d.get("last_seen"),
    )


class _SpillingNodeMap(MutableMapping):
    """id → UnifiedNode backed by SQLite with an LRU of live objects.

    Nodes returned to callers may be MUTATED in place (builder overlays do),
    so every cached entry counts as dirty and is written back on eviction
    and on flush() — correctness over write-avoidance.
    """

    def __init__(self, db_path: str | Path, capacity: int = 4096):
        self._conn = sqlite3.connect(str(db_path))
        self._conn.executescript(_SCHEMA)
        self._cache: OrderedDict[str, UnifiedNode] = OrderedDict()
        self._capacity = max(16, capacity)
        self._ids: set[str] = set(
            r[0] for r in self._conn.execute("SELECT node_id FROM live_nodes"))

    # ── Mapping interface ──────────────────────────────────────────────
    def __getitem__(self, nid: str) -> UnifiedNode:
        node = self._cache.get(nid)
        if node is not None:
            self._cache.move_to_end(nid)
            return node
        if nid not in self._ids:
            raise KeyError(nid)
        row = self._conn.execute(
            "SELECT payload FROM live_nodes WHERE node_id = ?", (nid,)).fetchone()
        if row is None:  # pragma: no cover — ids/table desync guard
            raise KeyError(nid)
        node = _deserialize(row[0])
        self._admit(nid, node)
        return node

    def __setitem__(self, nid: str, node: UnifiedNode) -> None:
        self._ids.add(nid)
        self._admit(nid, node)

    def __delitem__(self, nid: str) -> None:
        self._ids.discard(nid)
        self._cache.pop(nid, None)
        self._conn.execute("DELETE FROM live_nodes WHERE node_id = ?", (nid,))

    def __contains__(self, nid: object) -> bool:
        return nid in self._ids

    def __iter__(self) -> Iterator[str]:
        return iter(sorted(self._ids))

    def __len__(self) -> int:
        return len(self._ids)

    def values(self):
        for nid in self:
            yield self[nid]

    def items(self):
        for nid in self:
            yield nid, self[nid]

    # ── cache mechanics ────────────────────────────────────────────────
    def _admit(self, nid: str, node: UnifiedNode) -> None:
        self._cache[nid] = node
        self._cache.move_to_end(nid)
        while len(self._cache) > self._capacity:
            old_id, old_node = self._cache.popitem(last=False)
            self._write(old_id, old_node)

    def _write(self, nid: str, node: UnifiedNode) -> None:
        self._conn.execute(
            "INSERT OR REPLACE INTO live_nodes (node_id, payload) VALUES (?, ?)",
            (nid, _serialize(node)))

    def flush(self) -> None:
        for nid, node in self._cache.items():
            self._write(nid, node)
        self._conn.commit()

    def close(self) -> None:
        self.flush()
        self._conn.close()


class StoreBackedUnifiedGraph(UnifiedGraph):
    """UnifiedGraph whose node payloads live in SQLite behind an LRU."""

    def __init__(self, node_budget: Optional[int] = None,
                 db_path: Optional[str | Path] = None,
                 cache_capacity: int = 4096):
        super().__init__(**({"node_budget": node_budget} if node_budget else {}))
        if db_path is None:
            self._workspace = tempfile.NamedTemporaryFile(
                prefix="abom-graph-", suffix=".sqlite", delete=False)
            db_path = self._workspace.name
        self.db_path = str(db_path)
        self.nodes = _SpillingNodeMap(db_path, capacity=cache_capacity)

    def flush(self) -> None:
        self.nodes.flush()

    def close(self) -> None:
        self.nodes.close()

    def to_memory(self) -> UnifiedGraph:
        """Materialize back into a plain in-RAM UnifiedGraph."""
        g = UnifiedGraph(node_budget=self.node_budget)
        for nid in self.nodes:
            g.add_node(self.nodes[nid])
        for e in self.edges:
            g.add_edge(e)
        g.truncated, g.dropped_nodes = self.truncated, self.dropped_nodes
        return g


def maybe_store_backed(expected_entities: int,
                       node_budget: Optional[int] = None) -> UnifiedGraph:
    """The reference's auto-wire rule: estate builds >= 5,000 entities get
    the spilling container, small builds stay fully in RAM."""
    if expected_entities >= AUTO_SPILL_THRESHOLD:
        return StoreBackedUnifiedGraph(node_budget=node_budget)
    return UnifiedGraph(**({"node_budget": node_budget} if node_budget else {}))
