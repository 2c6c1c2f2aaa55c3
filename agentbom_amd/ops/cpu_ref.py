"""CPU (numpy) reference implementations of the engine ops.

These are the numerics oracle for the HIP kernels (tests/test_ops_gpu.py
compares device results bit-for-bit / set-for-set against these) and the
engine used on GPU-less development machines.  Semantics mirror
ops/csrc/{match,graph}.hip exactly.
"""

from __future__ import annotations

from typing import Optional

import numpy as np

UNVISITED = np.uint32(0xFFFFFFFF)

# Window flag bits (contract with csrc/abom_common.h and db/arena.py)
WF_HAS_INTRO = 1
WF_HAS_FIXED = 2
WF_HAS_LAST = 4
WF_CPU_FALLBACK = 8
WF_UNFIXED_SUPPRESSED = 16
PF_ENCODABLE = 1


def _key_lt(ahi, alo, bhi, blo):
    return (ahi < bhi) | ((ahi == bhi) & (alo < blo))


def match(pkg_group_key, pkg_key_hi, pkg_key_lo, pkg_flags,
          group_keys, group_off, windows: dict):
    """Vectorized reference of the match kernel.

    Returns (pkg_idx, win_idx) int64 arrays sorted by (pkg, window).
    """
    P = len(pkg_group_key)
    enc = (pkg_flags & PF_ENCODABLE) != 0
    gidx = np.searchsorted(group_keys, pkg_group_key)
    in_range = gidx < len(group_keys)
    gidx_c = np.clip(gidx, 0, max(len(group_keys) - 1, 0))
    hit = enc & in_range & (group_keys[gidx_c] == pkg_group_key) if len(group_keys) else np.zeros(P, bool)

    pkg_out = []
    win_out = []
    wf = windows["flags"]
    ihi, ilo = windows["intro_hi"], windows["intro_lo"]
    fhi, flo = windows["fixed_hi"], windows["fixed_lo"]
    lhi, llo = windows["last_hi"], windows["last_lo"]
    for p in np.nonzero(hit)[0]:
        g = gidx[p]
        khi, klo = pkg_key_hi[p], pkg_key_lo[p]
        for w in range(group_off[g], group_off[g + 1]):
            f = wf[w]
            if f & (WF_CPU_FALLBACK | WF_UNFIXED_SUPPRESSED):
                continue
            if (f & WF_HAS_INTRO) and _key_lt(khi, klo, ihi[w], ilo[w]):
                continue
            if (f & WF_HAS_FIXED) and not _key_lt(khi, klo, fhi[w], flo[w]):
                continue
            if (f & WF_HAS_LAST) and _key_lt(lhi[w], llo[w], khi, klo):
                continue
            pkg_out.append(p)
            win_out.append(w)
    pkg_arr = np.asarray(pkg_out, dtype=np.int64)
    win_arr = np.asarray(win_out, dtype=np.int64)
    order = np.lexsort((win_arr, pkg_arr)) if len(pkg_arr) else np.array([], dtype=np.int64)
    return pkg_arr[order], win_arr[order]


def bfs(row_off, col, sources, num_nodes: int,
        etype: Optional[np.ndarray] = None, allowed_mask: int = 0xFFFFFFFF,
        max_levels: int = 64):
    """Level-synchronous multi-source BFS; u32 dist, UNVISITED sentinel."""
    dist = np.full(num_nodes, UNVISITED, dtype=np.uint32)
    frontier = np.unique(np.asarray(sources, dtype=np.int64))
    dist[frontier] = 0
    level = 0
    while len(frontier) and level < max_levels:
        level += 1
        nxt = []
        for u in frontier:
            beg, end = row_off[u], row_off[u + 1]
            for e in range(beg, end):
                if etype is not None and not ((allowed_mask >> int(etype[e])) & 1):
                    continue
                v = col[e]
                if dist[v] == UNVISITED:
                    dist[v] = level
                    nxt.append(v)
        frontier = np.asarray(nxt, dtype=np.int64)
    return dist


def bfs_level(row_off, col, frontier, dist, level: int,
              etype: Optional[np.ndarray] = None, allowed_mask: int = 0xFFFFFFFF):
    """Expand ONE BFS level (claim semantics identical to the HIP kernel).

    ``dist`` may exceed the local node count (distributed sent-markers)."""
    nxt = []
    for u in frontier:
        for e in range(row_off[u], row_off[u + 1]):
            if etype is not None and not ((allowed_mask >> int(etype[e])) & 1):
                continue
            v = int(col[e])
            if dist[v] == UNVISITED:
                dist[v] = level
                nxt.append(v)
    return np.asarray(nxt, dtype=np.int64)


def impact_query(row_off, col, query_sources, etype=None, allowed_mask: int = 0xFFFFFFFF,
                 max_hops: int = 4, max_nodes: int = 4096):
    """Reference bounded blast-radius query.  Returns list of {node: hop}."""
    results = []
    for s in query_sources:
        hops = {int(s): 0}
        frontier = [int(s)]
        truncated = False
        for hop in range(1, max_hops + 1):
            nxt = []
            for u in frontier:
                for e in range(row_off[u], row_off[u + 1]):
                    if etype is not None and not ((allowed_mask >> int(etype[e])) & 1):
                        continue
                    v = int(col[e])
                    if v not in hops:
                        if len(hops) >= max_nodes:
                            truncated = True
                            continue
                        hops[v] = hop
                        nxt.append(v)
            frontier = nxt
        results.append((hops, truncated))
    return results


def risk_score(severity, n_agents, n_creds, n_tools, flags, epss, scorecard, reach,
               weights: Optional[np.ndarray] = None):
    """Vectorized reference of the risk-score kernel (f32 arithmetic)."""
    if weights is None:
        from agentbom_amd.ops.native import risk_weights_array

        weights = risk_weights_array()
    w = weights.astype(np.float32)
    sev = np.asarray(severity)
    base = np.zeros(len(sev), dtype=np.float32)
    base[sev == 5] = w[0]
    base[sev == 4] = w[1]
    base[sev == 3] = w[2]
    base[sev == 2] = w[3]
    af = np.minimum(n_agents.astype(np.float32) * w[4], w[5])
    cf = np.minimum(n_creds.astype(np.float32) * w[6], w[7])
    tf = np.minimum(n_tools.astype(np.float32) * w[8], w[9])
    ai_signals = (flags & 1).astype(np.int32) + (n_creds > 0) + (n_tools > 0)
    ai = np.where(ai_signals >= 2, w[10], 0.0).astype(np.float32)
    kev = np.where((flags & 2) != 0, w[11], 0.0).astype(np.float32)
    ep = np.where(epss >= w[13], w[12], 0.0).astype(np.float32)
    sc = np.zeros(len(sev), dtype=np.float32)
    has_sc = scorecard >= 0
    sc = np.where(has_sc & (scorecard < w[14]), w[15], sc)
    sc = np.where(has_sc & (scorecard >= w[14]) & (scorecard < w[16]), w[17], sc)
    sc = np.where(has_sc & (scorecard >= w[16]) & (scorecard < w[18]), w[19], sc)
    ra = np.where(reach == 1, w[20], np.where(reach == 0, -w[21], 0.0)).astype(np.float32)
    total = base + af + cf + tf + ai + kev + ep + sc + ra
    out = np.clip(total, 0.0, 10.0).astype(np.float32)
    out[(flags & 4) != 0] = 0.0
    return out


def severity_histogram(owner, severity, num_containers: int):
    hist = np.zeros((num_containers, 6), dtype=np.uint32)
    np.add.at(hist, (np.asarray(owner, dtype=np.int64), np.asarray(severity, dtype=np.int64)), 1)
    return hist


# ── attack/exposure path DP (oracle for csrc/paths.hip) ─────────────────────

PATH_NO_EDGE = np.uint64(0xFFFFFFFF)


def _ordered_f32(f):
    """Order-preserving u32 image of non-negative float32 scores."""
    u = np.asarray(f, dtype=np.float32).view(np.uint32)
    return (u | np.uint32(0x80000000)).astype(np.uint64)


def path_unordered_f32(u):
    """Inverse of _ordered_f32 for valid (non-negative-score) labels."""
    raw = (np.asarray(u, dtype=np.uint64) & np.uint64(0x7FFFFFFF)).astype(np.uint32)
    return raw.view(np.float32)


def path_pack(score, edge):
    return (_ordered_f32(score) << np.uint64(32)) | np.asarray(edge, dtype=np.uint64)


def path_relax(edge_src, col, etype, edge_weight, cur, node_boost,
               etype_boost, etype_trav, etype_gate, node_gate):
    """One hop of the max-score path DP; exact mirror of abom_path_relax.

    ``cur``/returned ``nxt`` are u64 [N*2] packed labels
    ((ordered_f32(score) << 32) | winner_edge); slot 0 = ungated path,
    slot 1 = path that used a gate edge or touched a gated node.  The
    kernel's atomicMax == np.maximum.at (both order-independent).
    """
    nxt = np.zeros_like(cur)
    et = np.asarray(etype)
    keep = etype_trav[et].astype(bool)
    e_idx = np.nonzero(keep)[0]
    if not len(e_idx):
        return nxt
    u = np.asarray(edge_src, dtype=np.int64)[e_idx]
    v = np.asarray(col, dtype=np.int64)[e_idx]
    step = etype_boost[et[e_idx]].astype(np.float32) + node_boost[v].astype(np.float32)
    if edge_weight is not None:
        step = step + edge_weight[e_idx].astype(np.float32) * np.float32(0.3)
    gate = etype_gate[et[e_idx]].astype(bool)
    if node_gate is not None:
        gate = gate | node_gate[v].astype(bool)

    esrc_all = np.asarray(edge_src, dtype=np.int64)

    def not_backtrack(labels):
        # predecessor-avoidance: block relaxing straight back to the node
        # the label came from (matches the kernel's 2-cycle guard)
        pe = (labels & np.uint64(0xFFFFFFFF)).astype(np.int64)
        no_edge = pe == 0xFFFFFFFF
        pred = np.where(no_edge, -1, esrc_all[np.where(no_edge, 0, pe)])
        return no_edge | (pred != v)

    lu0 = cur[u * 2]
    m = (lu0 != 0) & not_backtrack(lu0)
    if m.any():
        s = path_unordered_f32(lu0[m] >> np.uint64(32)) + step[m]
        cand = path_pack(s, e_idx[m])
        np.maximum.at(nxt, v[m] * 2 + gate[m], cand)
    lu1 = cur[u * 2 + 1]
    m = (lu1 != 0) & not_backtrack(lu1)
    if m.any():
        s = path_unordered_f32(lu1[m] >> np.uint64(32)) + step[m]
        cand = path_pack(s, e_idx[m])
        np.maximum.at(nxt, v[m] * 2 + 1, cand)
    return nxt
