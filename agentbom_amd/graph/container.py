"""UnifiedGraph — the in-RAM estate graph container.

Reference: src/agent_bom/graph/{node.py,edge.py,container.py} —
merge-union add_node, (src,tgt,rel)-deduped add_edge, forward+reverse
adjacency (bidirectional edges materialized both ways), BFS /
shortest_path / reachable_from / bounded traverse_subgraph / impact_of
(≤4 hops) / search / degree centrality / approx-betweenness bottlenecks /
typed views / node-budget truncation with completeness metadata.

At estate scale (≥ AGENT_BOM_GPU_GRAPH_THRESHOLD entities) ``to_csr()``
exports the container to the device CSR consumed by the HIP kernels
(graph/gpu_engine.py); this container remains the small-scale path and
the correctness oracle.
"""

from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field
from typing import Any, Callable, Iterable, Optional

from agentbom_amd.graph.types import EntityType, GraphSemanticLayer, NodeStatus, RelationshipType

DEFAULT_NODE_BUDGET = 250_000
IMPACT_MAX_HOPS = 4


@dataclass
class UnifiedNode:
    id: str
    entity_type: EntityType
    label: str
    layer: Optional[GraphSemanticLayer] = None
    status: NodeStatus = NodeStatus.ACTIVE
    properties: dict[str, Any] = field(default_factory=dict)
    tags: list[str] = field(default_factory=list)
    canonical_id: Optional[str] = None
    first_seen: Optional[str] = None
    last_seen: Optional[str] = None

    def merge(self, other: "UnifiedNode") -> None:
        """Union-merge: keep existing values, absorb new properties/tags."""
        for k, v in other.properties.items():
            self.properties.setdefault(k, v)
        for t in other.tags:
            if t not in self.tags:
                self.tags.append(t)
        if other.status == NodeStatus.VULNERABLE:
            self.status = NodeStatus.VULNERABLE
        self.last_seen = other.last_seen or self.last_seen

    def to_dict(self) -> dict[str, Any]:
        return {
            "id": self.id,
            "entity_type": self.entity_type.value,
            "label": self.label,
            "layer": self.layer.value if self.layer else None,
            "status": self.status.value,
            "properties": self.properties,
            "tags": self.tags,
            "canonical_id": self.canonical_id,
            "first_seen": self.first_seen,
            "last_seen": self.last_seen,
        }


@dataclass
class UnifiedEdge:
    source: str
    target: str
    relationship: RelationshipType
    weight: float = 1.0  # 0-10
    bidirectional: bool = False
    traversable: bool = True
    confidence: float = 1.0
    evidence: Optional[str] = None
    properties: dict[str, Any] = field(default_factory=dict)

    def to_dict(self) -> dict[str, Any]:
        return {
            "source": self.source,
            "target": self.target,
            "relationship": self.relationship.value,
            "weight": self.weight,
            "bidirectional": self.bidirectional,
            "traversable": self.traversable,
            "confidence": self.confidence,
            "evidence": self.evidence,
            "properties": self.properties,
        }


@dataclass
class AttackPath:
    """A scored entry→target path (attack-path fusion output)."""

    id: str
    nodes: list[str]
    relationships: list[str]
    score: float
    entry: str
    target: str
    techniques: list[str] = field(default_factory=list)
    narrative: str = ""

    def to_dict(self) -> dict[str, Any]:
        return {
            "id": self.id, "nodes": self.nodes, "relationships": self.relationships,
            "score": self.score, "entry": self.entry, "target": self.target,
            "techniques": self.techniques, "narrative": self.narrative,
        }


class UnifiedGraph:
    """Adjacency-indexed graph with bounded traversals + completeness."""

    def __init__(self, node_budget: int = DEFAULT_NODE_BUDGET):
        self.nodes: dict[str, UnifiedNode] = {}
        self.edges: list[UnifiedEdge] = []
        self._adj: dict[str, list[int]] = {}
        self._radj: dict[str, list[int]] = {}
        self._edge_keys: set[tuple[str, str, str]] = set()
        self.node_budget = node_budget
        self.truncated = False
        self.dropped_nodes = 0

    # ── construction ──────────────────────────────────────────────────────

    def add_node(self, node: UnifiedNode) -> bool:
        existing = self.nodes.get(node.id)
        if existing is not None:
            existing.merge(node)
            return True
        if len(self.nodes) >= self.node_budget:
            self.truncated = True
            self.dropped_nodes += 1
            return False
        self.nodes[node.id] = node
        self._adj.setdefault(node.id, [])
        self._radj.setdefault(node.id, [])
        return True

    def add_edge(self, edge: UnifiedEdge) -> bool:
        if edge.source not in self.nodes or edge.target not in self.nodes:
            return False
        key = (edge.source, edge.target, edge.relationship.value)
        if key in self._edge_keys:
            return False
        self._edge_keys.add(key)
        idx = len(self.edges)
        self.edges.append(edge)
        self._adj[edge.source].append(idx)
        self._radj[edge.target].append(idx)
        if edge.bidirectional:
            self._adj[edge.target].append(idx)
            self._radj[edge.source].append(idx)
        return True

    @property
    def node_count(self) -> int:
        return len(self.nodes)

    @property
    def edge_count(self) -> int:
        return len(self.edges)

    def completeness(self) -> dict[str, Any]:
        return {
            "complete": not self.truncated,
            "node_budget": self.node_budget,
            "node_count": self.node_count,
            "dropped_nodes": self.dropped_nodes,
        }

    # ── neighbors ─────────────────────────────────────────────────────────

    def _out(self, nid: str, allowed: Optional[set[RelationshipType]] = None,
             reverse: bool = False) -> Iterable[tuple[str, UnifiedEdge]]:
        table = self._radj if reverse else self._adj
        for ei in table.get(nid, ()):
            e = self.edges[ei]
            if not e.traversable:
                continue
            if allowed is not None and e.relationship not in allowed:
                continue
            if reverse:
                other = e.source if e.target == nid else e.target
            else:
                other = e.target if e.source == nid else e.source
            yield other, e

    def neighbors(self, nid: str, direction: str = "out") -> list[str]:
        if direction == "out":
            return [n for n, _ in self._out(nid)]
        if direction == "in":
            return [n for n, _ in self._out(nid, reverse=True)]
        return list({n for n, _ in self._out(nid)} | {n for n, _ in self._out(nid, reverse=True)})

    # ── traversals (reference container.py:613-720) ───────────────────────

    def bfs(self, start: str, max_depth: int = 10,
            allowed: Optional[set[RelationshipType]] = None,
            reverse: bool = False) -> dict[str, int]:
        """Hop distances from ``start`` (inclusive, start=0)."""
        if start not in self.nodes:
            return {}
        dist = {start: 0}
        q = deque([start])
        while q:
            u = q.popleft()
            if dist[u] >= max_depth:
                continue
            for v, _e in self._out(u, allowed, reverse):
                if v not in dist:
                    dist[v] = dist[u] + 1
                    q.append(v)
        return dist

    def shortest_path(self, source: str, target: str,
                      allowed: Optional[set[RelationshipType]] = None) -> Optional[list[str]]:
        if source not in self.nodes or target not in self.nodes:
            return None
        parent: dict[str, Optional[str]] = {source: None}
        q = deque([source])
        while q:
            u = q.popleft()
            if u == target:
                path = []
                cur: Optional[str] = u
                while cur is not None:
                    path.append(cur)
                    cur = parent[cur]
                return path[::-1]
            for v, _e in self._out(u, allowed):
                if v not in parent:
                    parent[v] = u
                    q.append(v)
        return None

    def reachable_from(self, start: str, max_depth: int = 10,
                       allowed: Optional[set[RelationshipType]] = None) -> set[str]:
        return set(self.bfs(start, max_depth, allowed)) - {start}

    def impact_of(self, nid: str, max_hops: int = IMPACT_MAX_HOPS) -> dict[str, Any]:
        """Bounded blast-radius neighborhood of a node (both directions)."""
        fwd = self.bfs(nid, max_hops)
        rev = self.bfs(nid, max_hops, reverse=True)
        combined: dict[str, int] = dict(fwd)
        for k, v in rev.items():
            combined[k] = min(combined.get(k, v), v)
        by_type: dict[str, list[str]] = {}
        for n, _hop in combined.items():
            if n == nid:
                continue
            node = self.nodes[n]
            by_type.setdefault(node.entity_type.value, []).append(n)
        return {
            "node_id": nid,
            "total_impacted": len(combined) - 1,
            "max_hops": max_hops,
            "hops": combined,
            "by_entity_type": {k: sorted(v) for k, v in sorted(by_type.items())},
        }

    def traverse_subgraph(self, start: str, max_depth: int = 3, max_nodes: int = 500,
                          allowed: Optional[set[RelationshipType]] = None) -> dict[str, Any]:
        """Bounded traversal returning nodes+edges with truncation metadata."""
        if start not in self.nodes:
            return {"nodes": [], "edges": [], "truncated": False, "complete": True}
        seen = {start}
        order = [start]
        edges_out: list[UnifiedEdge] = []
        q = deque([(start, 0)])
        truncated = False
        while q:
            u, d = q.popleft()
            if d >= max_depth:
                continue
            for v, e in self._out(u, allowed):
                if e not in edges_out and (v in seen or len(seen) < max_nodes):
                    edges_out.append(e)
                if v in seen:
                    continue
                if len(seen) >= max_nodes:
                    truncated = True
                    continue
                seen.add(v)
                order.append(v)
                q.append((v, d + 1))
        return {
            "nodes": [self.nodes[n].to_dict() for n in order],
            "edges": [e.to_dict() for e in edges_out],
            "truncated": truncated,
            "complete": not truncated,
        }

    # ── search / analytics (reference container.py:527-900) ───────────────

    def search(self, query: str = "", entity_types: Optional[list[EntityType]] = None,
               tags: Optional[list[str]] = None, status: Optional[NodeStatus] = None,
               limit: int = 100) -> list[UnifiedNode]:
        q = (query or "").lower()
        out = []
        for node in self.nodes.values():
            if entity_types and node.entity_type not in entity_types:
                continue
            if status and node.status != status:
                continue
            if tags and not set(tags) & set(node.tags):
                continue
            if q and q not in node.label.lower() and q not in node.id.lower():
                continue
            out.append(node)
            if len(out) >= limit:
                break
        return out

    def degree_centrality(self, top_n: int = 20) -> list[tuple[str, int]]:
        degrees = {
            nid: len(self._adj.get(nid, ())) + len(self._radj.get(nid, ()))
            for nid in self.nodes
        }
        return sorted(degrees.items(), key=lambda kv: (-kv[1], kv[0]))[:top_n]

    def pagerank(self, damping: float = 0.85, top_n: Optional[int] = None,
                 backend: Optional[str] = None) -> dict[str, float]:
        """Influence ranking via the pluggable analytics backend
        (graph/backend.py: native numpy by default, networkx when asked
        for and installed — reference graph_backend.py:25)."""
        from agentbom_amd.graph.backend import get_backend

        ranks = get_backend(backend).pagerank(self, damping=damping)
        if top_n is not None:
            top = sorted(ranks.items(), key=lambda kv: (-kv[1], kv[0]))[:top_n]
            return dict(top)
        return ranks

    def betweenness(self, sample: int = 64,
                    backend: Optional[str] = None) -> dict[str, float]:
        """Brandes betweenness through the analytics backend (exact when
        sample >= |V|); `bottlenecks()` remains the cheap approximation."""
        from agentbom_amd.graph.backend import get_backend

        return get_backend(backend).betweenness(self, sample=sample)

    def bottlenecks(self, top_n: int = 10, sample: int = 50) -> list[tuple[str, float]]:
        """Approximate betweenness via BFS from a deterministic node sample."""
        import itertools

        counts: dict[str, int] = {}
        sources = list(itertools.islice(sorted(self.nodes), sample))
        for s in sources:
            parent: dict[str, Optional[str]] = {s: None}
            q = deque([s])
            while q:
                u = q.popleft()
                for v, _e in self._out(u):
                    if v not in parent:
                        parent[v] = u
                        q.append(v)
            for leaf in parent:
                cur = parent[leaf]
                while cur is not None and parent[cur] is not None:
                    counts[cur] = counts.get(cur, 0) + 1
                    cur = parent[cur]
        total = sum(counts.values()) or 1
        ranked = sorted(counts.items(), key=lambda kv: (-kv[1], kv[0]))[:top_n]
        return [(n, c / total) for n, c in ranked]

    # ── typed views (reference container.py:956-1015) ─────────────────────

    _VIEW_FILTERS: dict[str, Callable] = {}

    def view(self, name: str) -> dict[str, Any]:
        """inventory / attack-path / lateral / compliance / runtime views."""
        if name == "inventory":
            keep_rel = {RelationshipType.HOSTS, RelationshipType.USES, RelationshipType.DEPENDS_ON,
                        RelationshipType.PROVIDES_TOOL, RelationshipType.CONTAINS,
                        RelationshipType.EXPOSES_CRED, RelationshipType.PART_OF}
        elif name == "attack-path":
            keep_rel = {RelationshipType.VULNERABLE_TO, RelationshipType.AFFECTS,
                        RelationshipType.EXPLOITABLE_VIA, RelationshipType.EXPOSES_CRED,
                        RelationshipType.REACHES_TOOL, RelationshipType.EXPOSED_TO,
                        RelationshipType.CAN_ACCESS}
        elif name == "lateral":
            keep_rel = {RelationshipType.SHARES_SERVER, RelationshipType.SHARES_CRED,
                        RelationshipType.LATERAL_PATH, RelationshipType.DELEGATED_TO}
        elif name == "runtime":
            keep_rel = {RelationshipType.INVOKED, RelationshipType.CALLED,
                        RelationshipType.USED_CREDENTIAL, RelationshipType.ACCESSED,
                        RelationshipType.ACTED_AS, RelationshipType.DELEGATED_TO}
        elif name == "compliance":
            keep_rel = {RelationshipType.VULNERABLE_TO, RelationshipType.AFFECTS,
                        RelationshipType.BELONGS_TO}
        else:
            raise ValueError(f"unknown view {name!r}")
        edges = [e for e in self.edges if e.relationship in keep_rel]
        node_ids = {e.source for e in edges} | {e.target for e in edges}
        return {
            "view": name,
            "nodes": [self.nodes[n].to_dict() for n in sorted(node_ids)],
            "edges": [e.to_dict() for e in edges],
        }

    # ── serialization / export ────────────────────────────────────────────

    def to_dict(self) -> dict[str, Any]:
        return {
            "schema_version": "1",
            "nodes": [self.nodes[n].to_dict() for n in sorted(self.nodes)],
            "edges": [e.to_dict() for e in self.edges],
            "completeness": self.completeness(),
        }

    @classmethod
    def from_dict(cls, data: dict[str, Any]) -> "UnifiedGraph":
        g = cls()
        for n in data.get("nodes", []):
            g.add_node(
                UnifiedNode(
                    id=n["id"], entity_type=EntityType(n["entity_type"]), label=n["label"],
                    layer=GraphSemanticLayer(n["layer"]) if n.get("layer") else None,
                    status=NodeStatus(n.get("status", "active")),
                    properties=n.get("properties", {}), tags=n.get("tags", []),
                    canonical_id=n.get("canonical_id"),
                )
            )
        for e in data.get("edges", []):
            g.add_edge(
                UnifiedEdge(
                    source=e["source"], target=e["target"],
                    relationship=RelationshipType(e["relationship"]),
                    weight=e.get("weight", 1.0), bidirectional=e.get("bidirectional", False),
                    traversable=e.get("traversable", True),
                    confidence=e.get("confidence", 1.0), evidence=e.get("evidence"),
                )
            )
        return g

    def export(self, fmt: str) -> str:
        if fmt == "dot":
            lines = ["digraph estate {", "  rankdir=LR;"]
            for n in sorted(self.nodes):
                node = self.nodes[n]
                lines.append(f'  "{n}" [label="{node.entity_type.value}: {node.label}"];')
            for e in self.edges:
                lines.append(f'  "{e.source}" -> "{e.target}" [label="{e.relationship.value}"];')
            lines.append("}")
            return "\n".join(lines) + "\n"
        if fmt == "mermaid":
            ids = {nid: f"n{i}" for i, nid in enumerate(sorted(self.nodes))}
            lines = ["graph LR"]
            for nid, short in ids.items():
                label = self.nodes[nid].label.replace('"', "'")
                lines.append(f'  {short}["{label}"]')
            for e in self.edges:
                lines.append(f"  {ids[e.source]} -->|{e.relationship.value}| {ids[e.target]}")
            return "\n".join(lines) + "\n"
        if fmt == "graphml":
            from xml.sax.saxutils import escape

            out = ['<?xml version="1.0"?>',
                   '<graphml xmlns="http://graphml.graphdrawing.org/xmlns">',
                   '  <graph edgedefault="directed">']
            for nid in sorted(self.nodes):
                out.append(f'    <node id="{escape(nid)}"/>')
            for i, e in enumerate(self.edges):
                out.append(f'    <edge id="e{i}" source="{escape(e.source)}" target="{escape(e.target)}"/>')
            out += ["  </graph>", "</graphml>"]
            return "\n".join(out) + "\n"
        if fmt == "cypher":
            lines = []
            for nid in sorted(self.nodes):
                node = self.nodes[nid]
                label = node.entity_type.value.replace("-", "_").title().replace("_", "")
                lines.append(f"MERGE (:{label} {{id: '{nid}'}});")
            for e in self.edges:
                lines.append(
                    f"MATCH (a {{id: '{e.source}'}}), (b {{id: '{e.target}'}}) "
                    f"MERGE (a)-[:{e.relationship.value.upper()}]->(b);"
                )
            return "\n".join(lines) + "\n"
        raise ValueError(f"unknown export format {fmt!r}")

    # ── GPU bridge ────────────────────────────────────────────────────────

    def to_csr(self):
        """(node_index, row_off, col, etype) numpy arrays for the HIP engine.

        Node order is sorted id (deterministic); edge types use
        types.REL_CODE.  Bidirectional edges emit both directions."""
        import numpy as np

        from agentbom_amd.graph.types import REL_CODE

        order = sorted(self.nodes)
        index = {nid: i for i, nid in enumerate(order)}
        src, dst, et = [], [], []
        for e in self.edges:
            code = REL_CODE[e.relationship]
            src.append(index[e.source])
            dst.append(index[e.target])
            et.append(min(code, 255))
            if e.bidirectional:
                src.append(index[e.target])
                dst.append(index[e.source])
                et.append(min(code, 255))
        n = len(order)
        src_a = np.asarray(src, dtype=np.int64)
        dst_a = np.asarray(dst, dtype=np.int64)
        et_a = np.asarray(et, dtype=np.uint8)
        o = np.argsort(src_a, kind="stable")
        counts = np.bincount(src_a, minlength=n)
        row_off = np.zeros(n + 1, dtype=np.int64)
        np.cumsum(counts, out=row_off[1:])
        return order, row_off, dst_a[o].astype(np.uint32), et_a[o]
