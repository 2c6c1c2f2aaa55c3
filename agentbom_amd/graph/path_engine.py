"""Hop-bounded max-score path DP driver (attack / exposure paths).

Runs the label-correcting relaxation (ops/csrc/paths.hip on GPU,
ops/cpu_ref.path_relax on CPU — bit-identical semantics) and reconstructs
the top-k gated paths.  This is the estate-scale replacement for the
reference's DFS walk (src/agent_bom/graph/attack_path_fusion.py:194-377):

- label(v, d, g) = best f32 score of any d-hop path from an entry to v
  with gate state g (g=1 once the path used a vuln-class edge or touched a
  gated node), packed ((ordered_f32(score) << 32) | winner_edge) so one
  atomicMax ranks by score with deterministic edge-index tie-breaks;
- the relaxation never walks straight back to a label's predecessor
  (2-cycle guard — on lateral agent<->server edges the unguarded max-score
  walk always ping-pongs the best-boosted server);
- candidates are ALL (target, depth) gated labels ranked by score; each is
  reconstructed exactly (score re-derived along the walk) and non-simple
  walks are rejected — depth-2 labels (agent -> gated server -> jewel) are
  simple by construction, so every reachable target eventually yields a
  valid path; the first (best) accepted candidate per target wins;
- on GPU the label levels stay in HBM: selection is a device topk and
  reconstruction pulls only the touched labels (small gathers), not the
  level arrays.

Ranking is (score desc, packed-label tie-break, node asc) — deterministic
(graph contract: determinism given the same inventory).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import numpy as np

NO_EDGE = 0xFFFFFFFF
_SEED_PACK = (0x80000000 << 32) | NO_EDGE  # ordered_f32(0.0) | no-edge
_INT64_MIN = -(1 << 63)


@dataclass
class PathHit:
    """One reconstructed attack/exposure path over numeric node ids."""

    nodes: list  # entry ... target (global node ids)
    edges: list  # winner edge index per hop
    etypes: list  # edge type per hop
    score: float
    target: int
    entry: int


def _unpack_score(packed_u64) -> np.ndarray:
    arr = np.asarray(packed_u64, dtype=np.uint64)
    hi = (arr >> np.uint64(32)) & np.uint64(0x7FFFFFFF)
    return hi.astype(np.uint32).view(np.float32)


def _np(a, dtype=None):
    """numpy view of a numpy array or (possibly device) torch tensor."""
    if a is None:
        return None
    if hasattr(a, "cpu") and not isinstance(a, np.ndarray):
        a = a.cpu().numpy()
    return np.asarray(a, dtype=dtype) if dtype else np.asarray(a)


class _NumpyFetch:
    """Label/edge/boost accessors over host arrays (CPU oracle path)."""

    def __init__(self, levels, esrc, ecol, etyp, eweight, nboost):
        self.levels = levels
        self.esrc, self.etyp = esrc, etyp
        self.eweight, self.nboost = eweight, nboost

    def labels(self, d, nodes, g):
        return self.levels[d][np.asarray(nodes, dtype=np.int64) * 2 + g]

    def labels_batch(self, d_arr, v_arr, g_arr):
        out = np.zeros(len(v_arr), dtype=np.uint64)
        d_arr = np.asarray(d_arr)
        idx = np.asarray(v_arr, dtype=np.int64) * 2 + np.asarray(g_arr)
        for dd in np.unique(d_arr):
            m = d_arr == dd
            out[m] = self.levels[int(dd)][idx[m]]
        return out

    def edge_src(self, e):
        return self.esrc[np.asarray(e, dtype=np.int64)]

    def edge_type(self, e):
        return self.etyp[np.asarray(e, dtype=np.int64)]

    def edge_w(self, e):
        if self.eweight is None:
            return None
        return self.eweight[np.asarray(e, dtype=np.int64)]

    def node_boost(self, v):
        return self.nboost[np.asarray(v, dtype=np.int64)]


class _DeviceFetch:
    """Same accessors via small device gathers (labels stay in HBM)."""

    def __init__(self, torch, levels_dev, src_t, et_t, ew_t, nb_t):
        self.torch = torch
        self.L = levels_dev  # list of int64 [N*2] device tensors
        self.src_t, self.et_t = src_t, et_t
        self.ew_t, self.nb_t = ew_t, nb_t
        self.dev = src_t.device

    def _idx(self, a):
        return self.torch.from_numpy(np.asarray(a, dtype=np.int64)).to(self.dev)

    def labels(self, d, nodes, g):
        out = self.L[d][self._idx(np.asarray(nodes, dtype=np.int64) * 2 + g)]
        return out.cpu().numpy().view(np.uint64)

    def labels_batch(self, d_arr, v_arr, g_arr):
        out = np.zeros(len(v_arr), dtype=np.uint64)
        d_arr = np.asarray(d_arr)
        idx = np.asarray(v_arr, dtype=np.int64) * 2 + np.asarray(g_arr)
        for dd in np.unique(d_arr):
            m = d_arr == dd
            got = self.L[int(dd)][self._idx(idx[m])].cpu().numpy()
            out[m] = got.view(np.uint64)
        return out

    def edge_src(self, e):
        return self.src_t[self._idx(e)].cpu().numpy().astype(np.int64)

    def edge_type(self, e):
        return self.et_t[self._idx(e)].cpu().numpy()

    def edge_w(self, e):
        if self.ew_t is None:
            return None
        return self.ew_t[self._idx(e)].cpu().numpy()

    def node_boost(self, v):
        return self.nb_t[self._idx(v)].cpu().numpy()


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
    """Full DP + reconstruction.  Returns ranked PathHit list."""
    use_gpu = device is not None and getattr(device, "type", str(device)).startswith("cuda")
    N = num_nodes

    seed = np.zeros(N * 2, dtype=np.uint64)
    entries = _np(entries, np.int64)
    seed[entries * 2] = np.uint64(_SEED_PACK)

    eb_np = _np(etype_boost, np.float32)
    eg_np = _np(etype_gate)
    ng_np = _np(node_gate)

    if use_gpu:
        import torch

        levels_dev, src_t, et_t, ew_t, nb_t = _run_gpu(
            edge_src, col, etype, edge_weight, seed, node_boost,
            etype_boost, etype_trav, etype_gate, node_gate, max_depth, device)
        fetch = _DeviceFetch(torch, levels_dev, src_t, et_t, ew_t, nb_t)
        cand_nodes, cand_depth, cand_packed = _select_device(
            torch, levels_dev, target_mask, k, device)
    else:
        from agentbom_amd.ops import cpu_ref

        esrc_np, col_np, et_np = _np(edge_src), _np(col), _np(etype)
        ew_np = _np(edge_weight)
        nb_np = _np(node_boost, np.float32)
        tv_np = _np(etype_trav)
        levels = [seed]
        for _ in range(max_depth):
            levels.append(cpu_ref.path_relax(
                esrc_np, col_np, et_np, ew_np, levels[-1], nb_np,
                eb_np, tv_np, eg_np, ng_np))
        fetch = _NumpyFetch(levels, esrc_np, col_np, et_np, ew_np, nb_np)
        cand_nodes, cand_depth, cand_packed = _select_numpy(levels, _np(target_mask), k)

    # reconstruct in rank order; first accepted candidate per target wins.
    # Batched lockstep walk: all candidates step back one hop together, so
    # the device path does a handful of gathers per hop instead of ~30
    # synchronized reads per path (p50 178 ms -> single-digit ms at 10M).
    out: list[PathHit] = []
    done_targets: set[int] = set()
    pos = 0
    max_attempts = max(64, 16 * k)
    attempts = 0
    while len(out) < k and pos < len(cand_nodes) and attempts < max_attempts:
        batch_idx = []
        batch_seen = set()
        while pos < len(cand_nodes) and len(batch_idx) < max(k * 2, 32):
            t = int(cand_nodes[pos])
            if t not in done_targets and t not in batch_seen:
                batch_idx.append(pos)
                batch_seen.add(t)
            pos += 1
        if not batch_idx:
            break
        attempts += len(batch_idx)
        hits = _reconstruct_batch(
            fetch,
            [int(cand_nodes[i]) for i in batch_idx],
            [int(cand_depth[i]) for i in batch_idx],
            eb_np, eg_np, ng_np)
        for hit in hits:
            if hit is not None and hit.target not in done_targets and len(out) < k:
                out.append(hit)
                done_targets.add(hit.target)
    return out


def _select_numpy(levels, target_mask, k):
    """All (target, depth) gated candidates ranked by score (numpy)."""
    tgt = np.nonzero(target_mask)[0]
    if not len(tgt):
        return [], [], []
    stacked = np.stack([lv[tgt * 2 + 1] for lv in levels[1:]])  # [D, T]
    d_idx, t_idx = np.nonzero(stacked)
    if not len(d_idx):
        return [], [], []
    packed = stacked[d_idx, t_idx]
    nodes = tgt[t_idx]
    # exact u64 ordering: ~packed ascending == packed (score) descending
    order = np.lexsort((nodes, ~packed))
    return nodes[order], d_idx[order] + 1, packed[order]


def _select_device(torch, levels_dev, target_mask, k, device):
    """Device-side candidate selection: no full-level host copies."""
    tm = target_mask
    if not torch.is_tensor(tm):
        tm = torch.from_numpy(np.ascontiguousarray(tm)).to(device)
    tgt = torch.nonzero(tm.to(torch.bool)).flatten()
    if not tgt.numel():
        return [], [], []
    gidx = tgt * 2 + 1
    rows = [lv[gidx] for lv in levels_dev[1:]]
    stacked = torch.stack(rows)  # [D, T] int64 (valid labels are negative)
    flat = stacked.flatten()
    valid = flat != 0
    ranked = torch.where(valid, flat, torch.full_like(flat, _INT64_MIN))
    n_take = min(int(valid.sum().item()), max(64, 16 * k))
    if n_take == 0:
        return [], [], []
    topv, topi = torch.topk(ranked, n_take)
    keep = topv != _INT64_MIN
    topv, topi = topv[keep], topi[keep]
    T = tgt.numel()
    d_idx = (topi // T).cpu().numpy()
    nodes = tgt[(topi % T)].cpu().numpy()
    packed = topv.cpu().numpy().view(np.uint64)
    return nodes, d_idx + 1, packed


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
    return levels_dev, src_t, et_t, ew_t, nb_t


def _reconstruct_batch(fetch, nodes, depths, eb_np, eg_np, ng_np):
    """Lockstep batched reconstruction of many candidates.

    Same semantics as _reconstruct (kept as the single-path reference and
    used by its tests), vectorized: each hop gathers labels/edges for the
    whole batch at once."""
    f32 = np.float32
    M = len(nodes)
    v = np.asarray(nodes, dtype=np.int64)
    g = np.ones(M, dtype=np.int64)
    d = np.asarray(depths, dtype=np.int64)
    alive = np.ones(M, dtype=bool)
    done = np.zeros(M, dtype=bool)
    nodes_acc: list[list[int]] = [[int(x)] for x in v]
    edges_acc: list[list[int]] = [[] for _ in range(M)]
    etypes_acc: list[list[int]] = [[] for _ in range(M)]
    first = fetch.labels_batch(d, v, g)
    score0 = _unpack_score(first)
    alive &= first != 0

    max_steps = int(d.max()) if M else 0
    for _ in range(max_steps):
        act = alive & ~done & (d > 0)
        if not act.any():
            break
        ai = np.nonzero(act)[0]
        packed = fetch.labels_batch(d[ai], v[ai], g[ai])
        dead = packed == 0
        alive[ai[dead]] = False
        live = ~dead
        ai = ai[live]
        packed = packed[live]
        if not len(ai):
            continue
        e = (packed & np.uint64(NO_EDGE)).astype(np.int64)
        is_seed = e == NO_EDGE
        done[ai[is_seed]] = True
        ai = ai[~is_seed]
        if not len(ai):
            continue
        e = e[~is_seed]
        packed = packed[~is_seed]
        u = fetch.edge_src(e).astype(np.int64)
        et = fetch.edge_type(e).astype(np.int64)
        step = eb_np[et].astype(f32) + fetch.node_boost(v[ai]).astype(f32)
        w = fetch.edge_w(e)
        if w is not None:
            step = (step + w.astype(f32) * f32(0.3)).astype(f32)
        my_score = _unpack_score(packed)
        gate_edge = eg_np[et].astype(bool)
        if ng_np is not None:
            gate_edge = gate_edge | ng_np[v[ai]].astype(bool)

        lu0 = fetch.labels_batch(d[ai] - 1, u, np.zeros(len(ai), dtype=np.int64))
        lu1 = fetch.labels_batch(d[ai] - 1, u, np.ones(len(ai), dtype=np.int64))

        def feasible(lab):
            ok = lab != 0
            score_match = np.zeros(len(lab), dtype=bool)
            if ok.any():
                cand = (_unpack_score(lab) + step).astype(f32)
                score_match = cand == my_score
            pe = (lab & np.uint64(NO_EDGE)).astype(np.int64)
            has_pe = ok & (pe != NO_EDGE)
            pred_ok = np.ones(len(lab), dtype=bool)
            if has_pe.any():
                preds = fetch.edge_src(np.where(has_pe, pe, 0)).astype(np.int64)
                pred_ok = ~(has_pe & (preds == v[ai]))
            return ok & score_match & pred_ok

        f1 = feasible(lu1)
        f0 = feasible(lu0)
        gcur = g[ai]
        g_next = np.full(len(ai), -1, dtype=np.int64)
        g_next[(gcur == 1) & f1] = 1
        sel0 = (gcur == 1) & ~f1 & gate_edge & f0
        g_next[sel0] = 0
        g_next[(gcur == 0) & f0] = 0
        bad = g_next < 0
        alive[ai[bad]] = False
        keep = ~bad
        ai = ai[keep]
        if not len(ai):
            continue
        for idx_pos, i in enumerate(np.nonzero(keep)[0]):
            j = ai[idx_pos]
            edges_acc[j].append(int(e[i]))
            etypes_acc[j].append(int(et[i]))
            nodes_acc[j].append(int(u[i]))
        v[ai] = u[keep]
        g[ai] = g_next[keep]
        d[ai] -= 1
        done[ai[d[ai] == 0]] = True  # walked all hops back to a seed level

    hits: list[Optional[PathHit]] = []
    for j in range(M):
        if not alive[j] or not done[j]:
            hits.append(None)
            continue
        path_nodes = list(reversed(nodes_acc[j]))
        if len(set(path_nodes)) != len(path_nodes):
            hits.append(None)  # non-simple walk — rejected
            continue
        hits.append(PathHit(
            nodes=path_nodes,
            edges=list(reversed(edges_acc[j])),
            etypes=list(reversed(etypes_acc[j])),
            score=round(min(float(score0[j]), 100.0), 2),
            target=path_nodes[-1], entry=path_nodes[0]))
    return hits


def _reconstruct(fetch, node: int, depth: int, eb_np, eg_np, ng_np) -> Optional[PathHit]:
    """Walk winner edges back to the entry; None for non-simple walks.

    Gate-source disambiguation mirrors the kernel: the stored score equals
    the feasible predecessor label + step; a predecessor label whose own
    predecessor is this hop's target was blocked by the 2-cycle guard and
    is not feasible.  Ties prefer the gated predecessor (fixed rule)."""
    f32 = np.float32
    g = 1
    v, d = node, depth
    nodes = [node]
    edges: list[int] = []
    etypes: list[int] = []
    packed0 = int(fetch.labels(d, [v], g)[0])
    if packed0 == 0:
        return None
    score = float(_unpack_score([packed0])[0])
    while d > 0:
        packed = int(fetch.labels(d, [v], g)[0])
        if packed == 0:
            return None
        e = packed & NO_EDGE
        if e == NO_EDGE:
            break  # seed label
        u = int(fetch.edge_src([e])[0])
        et = int(fetch.edge_type([e])[0])
        step = f32(eb_np[et]) + f32(fetch.node_boost([v])[0])
        w = fetch.edge_w([e])
        if w is not None:
            step = f32(step + f32(w[0]) * f32(0.3))
        my_score = _unpack_score([packed])[0]
        gate_edge = bool(eg_np[et]) or (ng_np is not None and bool(ng_np[v]))
        prev = fetch.labels(d - 1, [u, u], [0, 1])
        lu0, lu1 = int(prev[0]), int(prev[1])

        def feasible(lab):
            if not lab:
                return False
            if f32(_unpack_score([lab])[0] + step) != my_score:
                return False
            pe = lab & NO_EDGE
            if pe != NO_EDGE and int(fetch.edge_src([pe])[0]) == v:
                return False  # blocked by the 2-cycle guard in relax
            return True

        g_next = None
        if g == 1:
            if feasible(lu1):
                g_next = 1
            elif gate_edge and feasible(lu0):
                g_next = 0
        else:
            if feasible(lu0):
                g_next = 0
        if g_next is None:
            return None  # inconsistent label chain
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
