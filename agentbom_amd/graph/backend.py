"""Graph-analytics backend protocol: PageRank + betweenness centrality.

Reference parity: src/agent_bom/graph_backend.py:25 — an optional analytics
backend behind a Protocol, NetworkX when installed, with the container's
built-in approximations otherwise.  MI355X build ships a NATIVE numpy
backend as the default (power-iteration PageRank, Brandes betweenness on a
bounded sample) so the analytics work in the air-gapped image where
networkx is absent; the NetworkX adapter remains for environments that
have it (``AGENT_BOM_GRAPH_BACKEND=networkx``).
"""

from __future__ import annotations

import os
from typing import Protocol, runtime_checkable

import numpy as np


@runtime_checkable
class GraphBackendProtocol(Protocol):
    """What an analytics backend must provide over a UnifiedGraph."""

    name: str

    def pagerank(self, graph, damping: float = 0.85,
                 iterations: int = 50) -> dict[str, float]: ...

    def betweenness(self, graph, sample: int = 64) -> dict[str, float]: ...


class NativeBackend:
    """Numpy analytics over the in-RAM adjacency — no third-party deps."""

    name = "native"

    @staticmethod
    def _index(graph) -> tuple[list[str], dict[str, int]]:
        ids = sorted(graph.nodes)
        return ids, {nid: i for i, nid in enumerate(ids)}

    def pagerank(self, graph, damping: float = 0.85,
                 iterations: int = 50) -> dict[str, float]:
        ids, pos = self._index(graph)
        n = len(ids)
        if n == 0:
            return {}
        # column-stochastic sparse transition: follow traversable edges
        src_idx, dst_idx = [], []
        for e in graph.edges:
            if getattr(e, "traversable", True):
                src_idx.append(pos[e.source])
                dst_idx.append(pos[e.target])
                if getattr(e, "bidirectional", False):
                    src_idx.append(pos[e.target])
                    dst_idx.append(pos[e.source])
        src = np.asarray(src_idx, dtype=np.int64)
        dst = np.asarray(dst_idx, dtype=np.int64)
        out_deg = np.bincount(src, minlength=n).astype(np.float64)
        rank = np.full(n, 1.0 / n)
        teleport = (1.0 - damping) / n
        for _ in range(iterations):
            contrib = np.where(out_deg > 0, rank / np.maximum(out_deg, 1), 0.0)
            new = np.full(n, teleport)
            if len(src):
                np.add.at(new, dst, damping * contrib[src])
            # dangling mass redistributed uniformly
            new += damping * rank[out_deg == 0].sum() / n
            if np.abs(new - rank).sum() < 1e-10:
                rank = new
                break
            rank = new
        return {nid: float(rank[i]) for i, nid in enumerate(ids)}

    def betweenness(self, graph, sample: int = 64) -> dict[str, float]:
        """Brandes' algorithm from a deterministic source sample (exact when
        sample >= |V|), unweighted."""
        ids, pos = self._index(graph)
        n = len(ids)
        scores = dict.fromkeys(ids, 0.0)
        if n < 3:
            return scores
        adj: list[list[int]] = [[] for _ in range(n)]
        for e in graph.edges:
            if getattr(e, "traversable", True):
                adj[pos[e.source]].append(pos[e.target])
                if getattr(e, "bidirectional", False):
                    adj[pos[e.target]].append(pos[e.source])
        sources = range(n) if sample >= n else range(0, n, max(1, n // sample))
        for s in sources:
            # single-source shortest-path counts
            dist = [-1] * n
            sigma = [0.0] * n
            preds: list[list[int]] = [[] for _ in range(n)]
            dist[s], sigma[s] = 0, 1.0
            order: list[int] = []
            frontier = [s]
            while frontier:
                nxt = []
                for v in frontier:
                    order.append(v)
                    for w in adj[v]:
                        if dist[w] < 0:
                            dist[w] = dist[v] + 1
                            nxt.append(w)
                        if dist[w] == dist[v] + 1:
                            sigma[w] += sigma[v]
                            preds[w].append(v)
                frontier = nxt
            delta = [0.0] * n
            for w in reversed(order):
                for v in preds[w]:
                    delta[v] += sigma[v] / sigma[w] * (1.0 + delta[w])
                if w != s:
                    scores[ids[w]] += delta[w]
        norm = (n - 1) * (n - 2)
        if norm > 0:
            for k in scores:
                scores[k] /= norm
        return scores


class NetworkXBackend:
    """Adapter for environments that have networkx installed."""

    name = "networkx"

    def __init__(self) -> None:
        import networkx  # raises ImportError where absent — caller gates

        self._nx = networkx

    def _to_nx(self, graph):
        g = self._nx.DiGraph()
        g.add_nodes_from(graph.nodes)
        for e in graph.edges:
            if getattr(e, "traversable", True):
                g.add_edge(e.source, e.target)
                if getattr(e, "bidirectional", False):
                    g.add_edge(e.target, e.source)
        return g

    def pagerank(self, graph, damping: float = 0.85,
                 iterations: int = 50) -> dict[str, float]:
        return dict(self._nx.pagerank(self._to_nx(graph), alpha=damping,
                                      max_iter=iterations))

    def betweenness(self, graph, sample: int = 64) -> dict[str, float]:
        g = self._to_nx(graph)
        k = min(sample, g.number_of_nodes()) or None
        return dict(self._nx.betweenness_centrality(g, k=k, seed=7))


def get_backend(name: str | None = None) -> GraphBackendProtocol:
    """Resolve the analytics backend: arg > $AGENT_BOM_GRAPH_BACKEND > native.

    Asking for networkx where it isn't installed falls back to native with
    the same results contract (never an ImportError at call time).
    """
    name = (name or os.environ.get("AGENT_BOM_GRAPH_BACKEND") or "native").lower()
    if name == "networkx":
        try:
            return NetworkXBackend()
        except ImportError:
            return NativeBackend()
    return NativeBackend()
