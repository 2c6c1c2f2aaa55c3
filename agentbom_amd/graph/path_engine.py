"""Hop-bounded max-score path DP driver (attack / exposure paths).

Runs the label-correcting relaxation (ops/csrc/paths.hip on GPU,
ops/cpu_ref.path_relax on CPU — bit-identical semantics) and reconstructs
the top-k gated paths.  This is the estate-scale replacement for the
reference's DFS walk (src/agent_bom/graph/attack_path_fusion.py:194-377):

- label(v, d, g) = best f32 score of any d-hop path from an entry to v
  with gate state g (g=1 once the path used a vuln-class edge or touched a
  gated node), packed ((ordered_f32(score) << 32) | winner_edge) so one
  atomicMax ranks by score with deterministic edge-index tie-breaks;
- per-hop label arrays are kept (N*2 u64 per hop, HBM-resident at estate
  scale) so every reported path is reconstructed EXACTLY, score re-derived
  along the walk;
- targets = gated labels on target-class nodes, best over all hop depths,
  ranked by (score desc, node id asc) — deterministic.

Semantics vs the DFS: the DFS enumerates per (entry, target) and keeps the
best; the DP merges entries and keeps the best per target (with the entry
recovered from the reconstruction).  On acyclic estates the best-per-
target results coincide (tests/test_attack_paths.py asserts it); on cyclic
graphs the DP may report a non-simple walk, which reconstruction rejects
(dropped from the top-k, the next candidate takes its place).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import numpy as np

NO_EDGE = 0xFFFFFFFF
_SEED_PACK = (0x80000000 << 32) | NO_EDGE  # ordered_f32(0.0) | no-edge


@dataclass
class PathHit:
    """One reconstructed attack/exposure path over numeric node ids."""

    nodes: list  # entry ... target (global node ids)
    edges: list  # winner edge index per hop
    etypes: list  # edge type per hop
    score: float
    target: int
    entry: int


def _unpack_score(packed_u64: np.ndarray) -> np.ndarray:
    hi = (packed_u64 >> np.uint64(32)) & np.uint64(0x7FFFFFFF)
    return hi.astype(np.uint32).view(np.float32)


def _np(a, dtype=None):
    """numpy view of a numpy array or (possibly device) torch tensor."""
    if a is None:
        return None
    if hasattr(a, "cpu"):
        a = a.cpu().numpy()
    return np.asarray(a, dtype=dtype) if dtype else np.asarray(a)


def run_path_dp(
    edge_src, col, etype, edge_weight,
    num_nodes: int,
    entries,
    node_boost, etype_boost, etype_trav, etype_gate, node_gate,
    target_mask,
    max_depth: int = 6,
    k: int = 100,
    device=None,
) -> list[PathHit]:
    """Full DP + reconstruction.  numpy in, PathHit list out.

    On a CUDA ``device`` the relaxation runs the HIP kernel with all label
    levels resident in HBM; otherwise the numpy oracle.
    """
    use_gpu = device is not None and getattr(device, "type", str(device)).startswith("cuda")
    N = num_nodes

    seed = np.zeros(N * 2, dtype=np.uint64)
    entries = _np(entries, np.int64)
    seed[entries * 2] = np.uint64(_SEED_PACK)

    if use_gpu:
        levels = _run_gpu(edge_src, col, etype, edge_weight, seed, node_boost,
                          etype_boost, etype_trav, etype_gate, node_gate,
                          max_depth, device)
    else:
        from agentbom_amd.ops import cpu_ref

        edge_src, col, etype = _np(edge_src), _np(col), _np(etype)
        edge_weight = _np(edge_weight)
        node_boost, etype_boost = _np(node_boost), _np(etype_boost)
        etype_trav, etype_gate = _np(etype_trav), _np(etype_gate)
        node_gate = _np(node_gate)
        levels = [seed]
        for _ in range(max_depth):
            levels.append(cpu_ref.path_relax(
                edge_src, col, etype, edge_weight, levels[-1], node_boost,
                etype_boost, etype_trav, etype_gate, node_gate))

    # best gated label per target node over all depths >= 1
    tgt_nodes = np.nonzero(_np(target_mask))[0]
    if not len(tgt_nodes):
        return []
    stacked = np.stack([lv[tgt_nodes * 2 + 1] for lv in levels[1:]])  # [D, T]
    best_d = np.argmax(stacked, axis=0)
    best = stacked[best_d, np.arange(len(tgt_nodes))]
    hit = best != 0
    if not hit.any():
        return []
    cand_nodes = tgt_nodes[hit]
    cand_depth = best_d[hit] + 1
    cand_packed = best[hit]
    cand_score = _unpack_score(cand_packed)
    # rank: score desc, node id asc (deterministic); take extra candidates
    # because cyclic reconstructions may be rejected
    order = np.lexsort((cand_nodes, -cand_score))
    order = order[: max(k * 2, k + 16)]

    out: list[PathHit] = []
    esrc, ecol, etyp = _np(edge_src), _np(col), _np(etype)
    ew_np, nb_np, eb_np = _np(edge_weight), _np(node_boost), _np(etype_boost)
    eg_np, ng_np = _np(etype_gate), _np(node_gate)
    for i in order:
        if len(out) >= k:
            break
        hitp = _reconstruct(levels, int(cand_nodes[i]), int(cand_depth[i]),
                            esrc, ecol, etyp, ew_np, nb_np, eb_np, eg_np, ng_np)
        if hitp is not None:
            out.append(hitp)
    return out


def _run_gpu(edge_src, col, etype, edge_weight, seed, node_boost, etype_boost,
             etype_trav, etype_gate, node_gate, max_depth, device):
    import torch

    from agentbom_amd.ops import native

    def dev_t(a, dtype):
        if a is None:
            return None
        if torch.is_tensor(a):  # numpy 2.x arrays also expose .device
            return a.to(device=device, dtype=dtype)
        return torch.from_numpy(np.ascontiguousarray(a)).to(device=device, dtype=dtype)

    src_t = dev_t(edge_src, torch.int32)
    col_t = dev_t(col, torch.int32)
    et_t = dev_t(etype, torch.uint8)
    ew_t = dev_t(edge_weight, torch.float32)
    nb_t = dev_t(node_boost, torch.float32)
    eb_t = dev_t(etype_boost, torch.float32)
    tv_t = dev_t(etype_trav, torch.uint8)
    eg_t = dev_t(etype_gate, torch.uint8)
    ng_t = dev_t(node_gate, torch.uint8)

    cur = dev_t(seed.view(np.int64), torch.int64)
    levels_dev = [cur]
    for _ in range(max_depth):
        nxt = torch.zeros_like(cur)
        native.path_relax(src_t, col_t, et_t, ew_t, levels_dev[-1], nxt,
                          nb_t, eb_t, tv_t, eg_t, ng_t)
        levels_dev.append(nxt)
    return [lv.cpu().numpy().view(np.uint64) for lv in levels_dev]


def _reconstruct(levels, node: int, depth: int, esrc, ecol, etyp, edge_weight,
                 node_boost, etype_boost, etype_gate, node_gate) -> Optional[PathHit]:
    """Walk winner edges back to the entry; None for non-simple walks.

    Gate-source disambiguation mirrors the kernel's atomicMax: the stored
    score equals max over feasible predecessor labels + step; ties prefer
    the gated predecessor (fixed rule, documented in the module docstring).
    """
    f32 = np.float32
    g = 1
    v, d = node, depth
    nodes = [node]
    edges: list[int] = []
    etypes: list[int] = []
    score = float(_unpack_score(np.array([levels[d][v * 2 + g]], dtype=np.uint64))[0])
    while d > 0:
        packed = int(levels[d][v * 2 + g])
        if packed == 0:
            return None
        e = packed & NO_EDGE
        if e == NO_EDGE:
            break  # reached a seed label
        u = int(esrc[e])
        et = int(etyp[e])
        step = f32(etype_boost[et]) + f32(node_boost[v])
        if edge_weight is not None:
            step = f32(step + f32(edge_weight[e]) * f32(0.3))
        my_score = _unpack_score(np.array([packed], dtype=np.uint64))[0]
        gate_edge = bool(etype_gate[et]) or (node_gate is not None and bool(node_gate[v]))
        prev = levels[d - 1]
        g_next = None
        if g == 1:
            lu1 = int(prev[u * 2 + 1])
            if lu1 and f32(_unpack_score(np.array([lu1], dtype=np.uint64))[0] + step) == my_score:
                g_next = 1
            if g_next is None and gate_edge:
                lu0 = int(prev[u * 2])
                if lu0 and f32(_unpack_score(np.array([lu0], dtype=np.uint64))[0] + step) == my_score:
                    g_next = 0
        else:
            lu0 = int(prev[u * 2])
            if lu0 and f32(_unpack_score(np.array([lu0], dtype=np.uint64))[0] + step) == my_score:
                g_next = 0
        if g_next is None:
            return None  # inconsistent label chain (shouldn't happen)
        edges.append(int(e))
        etypes.append(et)
        nodes.append(u)
        v, g, d = u, g_next, d - 1
    nodes.reverse()
    edges.reverse()
    etypes.reverse()
    if len(set(nodes)) != len(nodes):
        return None  # non-simple walk (cycle) — rejected
    return PathHit(nodes=nodes, edges=edges, etypes=etypes,
                   score=round(min(score, 100.0), 2),
                   target=node, entry=nodes[0])
