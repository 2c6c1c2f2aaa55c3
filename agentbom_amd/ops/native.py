"""ctypes bindings for the gfx950 HIP engine (_abom_gpu.so).

Tensors are allocated and owned by PyTorch (the caching allocator manages
HBM); raw device pointers + the current torch HIP stream are handed to the
C ABI.  On a machine with a visible GPU the native library is REQUIRED —
a missing .so raises instead of silently falling back to a slow path
(AGENT_BOM_GPU_REQUIRED=0 opts out for CPU-only dev boxes).
"""

from __future__ import annotations

import ctypes
import os
from pathlib import Path
from typing import Optional

import numpy as np

_SO_PATH = Path(__file__).resolve().parent / "_abom_gpu.so"
_lib: Optional[ctypes.CDLL] = None

_c = ctypes.c_void_p
_i64 = ctypes.c_longlong
_u32 = ctypes.c_uint
_i32 = ctypes.c_int

_SIGNATURES = {
    "abom_abi_version": ([], _i32),
    "abom_device_count": ([], _i32),
    "abom_synchronize": ([_c], _i32),
    "abom_error_string": ([_i32], ctypes.c_char_p),
    "abom_match": ([_c] * 4 + [_i64] + [_c, _c, _i64] + [_c] * 7 + [_c] + [_c, _c] + [_c] + [_c, _c, _i64, _c], _i32),
    "abom_bfs_init": ([_c, _i64, _c], _i32),
    "abom_bfs_seed": ([_c, _i64, _c, _c, _c, _c, _c], _i32),
    "abom_bfs_expand": ([_c, _c, _c, _u32, _c, _i64, _c, _u32, _c, _c, _c, _c, _c, _i64, _c], _i32),
    "abom_bfs_expand_heavy": ([_c, _c, _c, _u32, _c, _c, _c, _u32, _c, _c, _c, _i64, _c], _i32),
    "abom_bfs_run": ([_c, _c, _c, _u32, _c, _i64, _c, _i64, _c, _c, _c, _c, _i32, _c, _i64, ctypes.c_double, _c, _c, _c, _c, _c], _i32),
    "abom_bfs_expand_edges": ([_c, _c, _c, _u32, _i64, _c, _c, _u32, _c, _c, _c, _i32, _i64, _c], _i32),
    "abom_impact_query": ([_c, _c, _c, _u32, _c, _i32, _i32, _i32, _c, _c, _c, _c, _c], _i32),
    "abom_risk_score": ([_c] * 8 + [_c, _i64, ctypes.POINTER(ctypes.c_float), _c], _i32),
    "abom_blast_counts": ([_c, _i64] + [_c] * 6 + [_i32] * 4 + [_c] * 4 + [_c], _i32),
    "abom_severity_histogram": ([_c, _c, _c, _i64, _c], _i32),
    "abom_score_gather": ([_c] * 11 + [_c] * 4 + [_i64, ctypes.POINTER(ctypes.c_float), _c], _i32),
    "abom_path_relax": ([_c] * 11 + [_i64, _c], _i32),
    "abom_rank_order": ([_c, _i64, _c, _c, _i32, _c, _c], _i32),
}


class NativeUnavailable(RuntimeError):
    pass


def load(required: Optional[bool] = None) -> ctypes.CDLL:
    """Load the engine .so, binding signatures once."""
    global _lib
    if _lib is not None:
        return _lib
    if not _SO_PATH.exists():
        if required is None:
            from agentbom_amd.utils import config as cfg

            required = cfg.GPU_REQUIRED and _torch_has_gpu()
        if required:
            raise NativeUnavailable(
                f"HIP engine missing: {_SO_PATH} not built. Run "
                "`python -m agentbom_amd.ops.build` (requires hipcc). The GPU "
                "path never silently falls back — set AGENT_BOM_GPU_REQUIRED=0 "
                "only on CPU-only development machines."
            )
        raise NativeUnavailable(f"{_SO_PATH} not built")
    lib = ctypes.CDLL(str(_SO_PATH))
    for name, (argtypes, restype) in _SIGNATURES.items():
        fn = getattr(lib, name)
        fn.argtypes = argtypes
        fn.restype = restype
    _lib = lib
    return lib


def _torch_has_gpu() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


def available() -> bool:
    try:
        load(required=False)
        return True
    except NativeUnavailable:
        return False


def _check(rc: int, what: str) -> None:
    if rc != 0:
        lib = load()
        msg = lib.abom_error_string(abs(rc)).decode() if rc else ""
        raise RuntimeError(f"{what} failed: hip error {rc} ({msg})")


def _ptr(t) -> ctypes.c_void_p:
    return ctypes.c_void_p(t.data_ptr())


def _stream() -> ctypes.c_void_p:
    import torch

    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


# ── High-level ops (torch tensors on cuda device) ──────────────────────────


def match_launch(pkg_group_key, pkg_key_hi, pkg_key_lo, pkg_flags, group_keys,
                 group_off, windows: dict, capacity: Optional[int] = None,
                 pkg_win_range=None, order=None) -> dict:
    """Launch the match kernel on the CURRENT stream without syncing.

    Returns a pending handle for :func:`match_finalize`.  Used by the engine
    to overlap the match with the reach BFS on a second HIP stream.
    ``pkg_win_range``: optional (wbeg u32[P], wend u32[P]) precomputed by the
    engine for a resident estate — the kernel then skips the group search.
    """
    import torch

    lib = load()
    P = pkg_group_key.numel()
    if P >= 1 << 32:
        raise ValueError(f"match: {P} packages exceeds the u32 pair encoding; shard the batch")
    G = group_keys.numel()
    cap = capacity or max(1024, P // 4)
    dev = pkg_group_key.device
    out_pairs = torch.empty(cap, dtype=torch.int64, device=dev)
    out_count = torch.zeros(1, dtype=torch.int32, device=dev)
    wbeg_p = wend_p = None
    if pkg_win_range is not None:
        wbeg_p, wend_p = _ptr(pkg_win_range[0]), _ptr(pkg_win_range[1])
    order_p = _ptr(order) if order is not None else None
    rc = lib.abom_match(
        _ptr(pkg_group_key), _ptr(pkg_key_hi), _ptr(pkg_key_lo), _ptr(pkg_flags), P,
        _ptr(group_keys), _ptr(group_off), G,
        _ptr(windows["intro_hi"]), _ptr(windows["intro_lo"]),
        _ptr(windows["fixed_hi"]), _ptr(windows["fixed_lo"]),
        _ptr(windows["last_hi"]), _ptr(windows["last_lo"]),
        _ptr(windows["flags"]),
        _ptr(windows["packed"]) if "packed" in windows else None,
        wbeg_p, wend_p, order_p,
        _ptr(out_pairs), _ptr(out_count), cap, _stream(),
    )
    _check(rc, "abom_match")
    return {"out_pairs": out_pairs, "out_count": out_count, "cap": cap,
            "args": (pkg_group_key, pkg_key_hi, pkg_key_lo, pkg_flags,
                     group_keys, group_off, windows),
            "kw": {"pkg_win_range": pkg_win_range, "order": order}}


def match_finalize(pending: dict, sort: bool = True):
    """Read the pending match count and produce (pkg_idx, win_idx).

    On capacity overflow (count > cap) the match is relaunched synchronously
    with a grown buffer — rare, the launch heuristic is P//4.

    ``sort=False`` skips the canonical (pkg<<32|win) sort — for callers that
    impose their own order downstream (the dedup-expand path re-sorts after
    fanning out to package rows, so sorting the distinct-row pairs here is
    pure waste).
    """
    import torch

    n = int(pending["out_count"].item())
    if n < 0:
        # int32 device counter wrapped: >2^31 candidate matches. Fail loudly
        # rather than slicing with a wrapped count.
        raise RuntimeError("match: device match count overflowed int32; shard the package batch")
    if n > pending["cap"]:
        relaunched = match_launch(*pending["args"], capacity=int(n * 1.2) + 1024,
                                  **pending.get("kw", {}))
        return match_finalize(relaunched, sort=sort)
    pairs = pending["out_pairs"][:n]
    if sort:
        pairs, _ = torch.sort(pairs)
    return (pairs >> 32).to(torch.int64), (pairs & 0xFFFFFFFF).to(torch.int64)


def match(pkg_group_key, pkg_key_hi, pkg_key_lo, pkg_flags, group_keys, group_off,
          windows: dict, capacity: Optional[int] = None, pkg_win_range=None):
    """Bulk version-range match.  Returns sorted (pkg_idx u32, window_idx u32).

    All tensors device-resident; ``windows`` carries intro/fixed/last hi+lo
    (int64 bit-patterns) and flags (uint8).  Retries with a larger buffer if
    capacity overflows.
    """
    pending = match_launch(pkg_group_key, pkg_key_hi, pkg_key_lo, pkg_flags,
                           group_keys, group_off, windows, capacity=capacity,
                           pkg_win_range=pkg_win_range)
    return match_finalize(pending)


def bfs(row_off, col, sources, num_nodes: int, etype=None, allowed_mask: int = 0xFFFFFFFF,
        max_levels: int = 64, workspace: Optional[dict] = None, edge_src=None,
        rev=None):
    """Multi-source BFS over CSR; returns u32 dist (UNVISITED = 0xFFFFFFFF).

    ``workspace`` may carry preallocated buffers (dist/frontier_a/frontier_b/
    heavy/counters) to avoid reallocation in steady-state serving.
    """
    import torch

    lib = load()
    dev = row_off.device
    ws = workspace or {}
    dist = ws.get("dist")
    if dist is None or dist.numel() < num_nodes:
        dist = torch.empty(num_nodes, dtype=torch.int32, device=dev)
    fa = ws.get("frontier_a")
    if fa is None or fa.numel() < num_nodes:
        fa = torch.empty(num_nodes, dtype=torch.int32, device=dev)
    fb = ws.get("frontier_b")
    if fb is None or fb.numel() < num_nodes:
        fb = torch.empty(num_nodes, dtype=torch.int32, device=dev)
    hq = ws.get("heavy")
    if hq is None or hq.numel() < num_nodes:
        hq = torch.empty(num_nodes, dtype=torch.int32, device=dev)
    ctr = ws.get("counters")
    if ctr is None or ctr.numel() < 4:
        ctr = torch.zeros(4, dtype=torch.int32, device=dev)
    # dense-mode bitmaps: 3 x ceil(N/32) u32 words, L2/LLC-resident
    bit_words = (num_nodes + 31) // 32
    bits = ws.get("bits")
    if bits is None or bits.numel() < 3 * bit_words:
        bits = torch.empty(3 * bit_words, dtype=torch.int32, device=dev)
    if workspace is not None:
        workspace.update(dist=dist, frontier_a=fa, frontier_b=fb, heavy=hq,
                         counters=ctr, bits=bits)

    et = _ptr(etype) if etype is not None else None
    if os.environ.get("AGENT_BOM_BFS_DENSE", "1") == "0":
        edge_src = None  # A/B toggle: disable edge-centric dense-frontier mode
    # Bitmap dense mode measured 2x SLOWER than dist-probe mode on MI355X:
    # atomicOr packs 32 nodes per cache line and the 8 XCDs' private L2s
    # ping-pong those shared lines through the coherence point, while the
    # scattered dist[] writes touch each line once.  Kept behind an opt-in
    # flag for future single-XCD / small-graph experiments.
    bits_ptr = None
    if os.environ.get("AGENT_BOM_BFS_BITS", "0") == "1":
        bits_ptr = _ptr(bits)
    es = _ptr(edge_src) if edge_src is not None else None
    num_edges = col.numel()
    avg_degree = num_edges / max(num_nodes, 1)
    # direction-optimized dense levels: pass the reverse CSR for bottom-up
    rev_off_p = rev_col_p = rev_et_p = None
    if rev is not None and os.environ.get("AGENT_BOM_BFS_BOTTOMUP", "1") != "0":
        rev_off_p = _ptr(rev["row_off"])
        rev_col_p = _ptr(rev["col"])
        rev_et_p = _ptr(rev["etype"]) if rev.get("etype") is not None else None
    rc = lib.abom_bfs_run(
        _ptr(row_off), _ptr(col), et, allowed_mask,
        _ptr(sources), sources.numel(), _ptr(dist), num_nodes,
        _ptr(fa), _ptr(fb), _ptr(hq), _ptr(ctr), max_levels,
        es, num_edges, float(avg_degree), bits_ptr,
        rev_off_p, rev_col_p, rev_et_p, _stream(),
    )
    if rc < 0:
        _check(-rc, "abom_bfs_run")
    return dist[:num_nodes]


def bfs_level(row_off, col, frontier, dist, level: int, etype=None,
              allowed_mask: int = 0xFFFFFFFF, workspace: Optional[dict] = None):
    """Expand ONE BFS level; returns the next frontier (claimed vertices).

    ``dist`` may be larger than the local graph (distributed mode: the tail
    indexes remote nodes and a claim there marks "queued for send").  Used by
    parallel/dist_bfs.py between RCCL frontier exchanges.
    """
    import torch

    lib = load()
    dev = row_off.device
    ws = workspace if workspace is not None else {}
    cap = dist.numel()
    nxt = ws.get("next")
    if nxt is None or nxt.numel() < cap:
        nxt = torch.empty(cap, dtype=torch.int32, device=dev)
        ws["next"] = nxt
    hq = ws.get("heavy")
    if hq is None or hq.numel() < max(frontier.numel(), 1):
        hq = torch.empty(max(frontier.numel(), 1024), dtype=torch.int32, device=dev)
        ws["heavy"] = hq
    ctr = ws.get("counters")
    if ctr is None or ctr.numel() < 4:
        ctr = torch.zeros(4, dtype=torch.int32, device=dev)
        ws["counters"] = ctr
    ctr.zero_()
    et = _ptr(etype) if etype is not None else None
    deg_ptr = ctypes.c_void_p(ctr.data_ptr() + 8)
    rc = lib.abom_bfs_expand(
        _ptr(row_off), _ptr(col), et, allowed_mask, _ptr(frontier), frontier.numel(),
        _ptr(dist), level, _ptr(nxt), _ptr(ctr), _ptr(hq),
        ctypes.c_void_p(ctr.data_ptr() + 4), deg_ptr, cap, _stream(),
    )
    _check(rc, "abom_bfs_expand")
    rc = lib.abom_bfs_expand_heavy(
        _ptr(row_off), _ptr(col), et, allowed_mask, _ptr(hq),
        ctypes.c_void_p(ctr.data_ptr() + 4), _ptr(dist), level, _ptr(nxt), _ptr(ctr),
        deg_ptr, cap, _stream(),
    )
    _check(rc, "abom_bfs_expand_heavy")
    n = int(ctr[0].item())
    return nxt[:n].clone()


def bfs_level_edges(row_off, col, edge_src, frontier_dist_level: int, dist,
                    etype=None, allowed_mask: int = 0xFFFFFFFF,
                    workspace: Optional[dict] = None):
    """Edge-centric ONE-level expansion (dense frontiers, distributed mode).

    Claims every unvisited neighbor of vertices whose dist equals
    ``frontier_dist_level`` and RETURNS the claimed list (local + remote ids
    alike) — the materialized frontier the RCCL exchange needs.  Used by
    parallel/dist_bfs.py when the frontier's degree sum is a large share of
    the shard's edges (same threshold as the single-GPU dense mode).
    """
    import torch

    lib = load()
    dev = row_off.device
    ws = workspace if workspace is not None else {}
    cap = dist.numel()
    nxt = ws.get("next")
    if nxt is None or nxt.numel() < cap:
        nxt = torch.empty(cap, dtype=torch.int32, device=dev)
        ws["next"] = nxt
    ctr = ws.get("counters")
    if ctr is None or ctr.numel() < 4:
        ctr = torch.zeros(4, dtype=torch.int32, device=dev)
        ws["counters"] = ctr
    ctr.zero_()
    et = _ptr(etype) if etype is not None else None
    rc = lib.abom_bfs_expand_edges(
        _ptr(edge_src), _ptr(col), et, allowed_mask, col.numel(),
        _ptr(row_off), _ptr(dist), frontier_dist_level, _ptr(nxt), _ptr(ctr),
        ctypes.c_void_p(ctr.data_ptr() + 8), 1, cap, _stream(),
    )
    _check(rc, "abom_bfs_expand_edges")
    n = int(ctr[0].item())
    return nxt[:n].clone()


def impact_query(row_off, col, query_sources, etype=None, allowed_mask: int = 0xFFFFFFFF,
                 max_hops: int = 4, max_nodes: int = 4096):
    """Batched bounded blast-radius queries (one block per query).

    Returns (nodes u32 [Q, max_nodes], hops u8 [Q, max_nodes], counts u32 [Q],
    truncated u8 [Q]).
    """
    import torch

    lib = load()
    dev = row_off.device
    Q = query_sources.numel()
    out_nodes = torch.empty((Q, max_nodes), dtype=torch.int32, device=dev)
    out_hops = torch.empty((Q, max_nodes), dtype=torch.uint8, device=dev)
    out_counts = torch.zeros(Q, dtype=torch.int32, device=dev)
    out_trunc = torch.zeros(Q, dtype=torch.uint8, device=dev)
    et = _ptr(etype) if etype is not None else None
    rc = lib.abom_impact_query(
        _ptr(row_off), _ptr(col), et, allowed_mask, _ptr(query_sources), Q,
        max_hops, max_nodes, _ptr(out_nodes), _ptr(out_hops), _ptr(out_counts),
        _ptr(out_trunc), _stream(),
    )
    _check(rc, "abom_impact_query")
    return out_nodes, out_hops, out_counts, out_trunc


def blast_counts(pkgs, rev, fwd, et_codes: tuple[int, int, int, int],
                 node_is_db_cred, node_is_db_tool):
    """Fused per-package reach counts.  Returns (counts [n,6] int32,
    overflow u8 [n]) — overflowed rows must be recomputed by the caller's
    sort-based path."""
    import torch

    lib = load()
    n = pkgs.numel()
    dev = pkgs.device
    out_counts = torch.zeros(n * 6, dtype=torch.int32, device=dev)
    out_overflow = torch.zeros(n, dtype=torch.uint8, device=dev)
    rc = lib.abom_blast_counts(
        _ptr(pkgs), n,
        _ptr(rev["row_off"]), _ptr(rev["col"]), _ptr(rev["etype"]),
        _ptr(fwd["row_off"]), _ptr(fwd["col"]), _ptr(fwd["etype"]),
        et_codes[0], et_codes[1], et_codes[2], et_codes[3],
        _ptr(node_is_db_cred), _ptr(node_is_db_tool),
        _ptr(out_counts), _ptr(out_overflow), _stream(),
    )
    _check(rc, "abom_blast_counts")
    return out_counts.view(n, 6), out_overflow


def risk_weights_array() -> np.ndarray:
    """The 22-float weight vector, in kernel order, from live config."""
    from agentbom_amd.utils import config as cfg

    return np.array(
        [
            cfg.RISK_BASE_CRITICAL, cfg.RISK_BASE_HIGH, cfg.RISK_BASE_MEDIUM, cfg.RISK_BASE_LOW,
            cfg.RISK_AGENT_WEIGHT, cfg.RISK_AGENT_CAP, cfg.RISK_CRED_WEIGHT, cfg.RISK_CRED_CAP,
            cfg.RISK_TOOL_WEIGHT, cfg.RISK_TOOL_CAP, cfg.RISK_AI_BOOST, cfg.RISK_KEV_BOOST,
            cfg.RISK_EPSS_BOOST, cfg.EPSS_CRITICAL_THRESHOLD,
            cfg.RISK_SCORECARD_TIER1_THRESHOLD, cfg.RISK_SCORECARD_TIER1_BOOST,
            cfg.RISK_SCORECARD_TIER2_THRESHOLD, cfg.RISK_SCORECARD_TIER2_BOOST,
            cfg.RISK_SCORECARD_TIER3_THRESHOLD, cfg.RISK_SCORECARD_TIER3_BOOST,
            cfg.RISK_REACHABLE_BOOST, cfg.RISK_UNREACHABLE_PENALTY,
        ],
        dtype=np.float32,
    )


def risk_score(severity, n_agents, n_creds, n_tools, flags, epss, scorecard, reach):
    """GPU risk scoring — same formula as models/blast.risk_score_from_counts."""
    import torch

    lib = load()
    n = severity.numel()
    out = torch.empty(n, dtype=torch.float32, device=severity.device)
    w = risk_weights_array()
    rc = lib.abom_risk_score(
        _ptr(severity), _ptr(n_agents), _ptr(n_creds), _ptr(n_tools), _ptr(flags),
        _ptr(epss), _ptr(scorecard), _ptr(reach), _ptr(out), n,
        w.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), _stream(),
    )
    _check(rc, "abom_risk_score")
    return out


def score_gather(win_idx, pkg_nodes, pos, arena_sev, arena_kev, arena_epss,
                 arena_impact, cred_lut, tool_lut, counts2d, dist):
    """Fused per-finding gather + risk score (one kernel for the whole
    post-counts chain).  Returns (scores f32, n_agents i32, n_creds i32,
    n_tools i32)."""
    import torch

    lib = load()
    n = win_idx.numel()
    dev = win_idx.device
    out_scores = torch.empty(n, dtype=torch.float32, device=dev)
    out_agents = torch.empty(n, dtype=torch.int32, device=dev)
    out_creds = torch.empty(n, dtype=torch.int32, device=dev)
    out_tools = torch.empty(n, dtype=torch.int32, device=dev)
    w = risk_weights_array()
    rc = lib.abom_score_gather(
        _ptr(win_idx), _ptr(pkg_nodes), _ptr(pos), _ptr(arena_sev),
        _ptr(arena_kev), _ptr(arena_epss), _ptr(arena_impact), _ptr(cred_lut),
        _ptr(tool_lut), _ptr(counts2d), _ptr(dist), _ptr(out_scores),
        _ptr(out_agents), _ptr(out_creds), _ptr(out_tools), n,
        w.ctypes.data_as(ctypes.POINTER(ctypes.c_float)), _stream(),
    )
    _check(rc, "abom_score_gather")
    return out_scores, out_agents, out_creds, out_tools


def severity_histogram(owner, severity, num_containers: int):
    import torch

    lib = load()
    hist = torch.zeros(num_containers * 6, dtype=torch.int32, device=owner.device)
    rc = lib.abom_severity_histogram(_ptr(owner), _ptr(severity), _ptr(hist),
                                     owner.numel(), _stream())
    _check(rc, "abom_severity_histogram")
    return hist.view(num_containers, 6)


def path_relax(edge_src, col, etype, edge_weight, cur, nxt, node_boost,
               etype_boost, etype_trav, etype_gate, node_gate):
    """One hop of the max-score path DP (ops/csrc/paths.hip).

    ``cur``/``nxt``: int64 [N*2] tensors holding the packed u64 labels
    ((ordered_f32(score) << 32) | winner_edge); ``nxt`` must be pre-zeroed.
    ``edge_weight`` and ``node_gate`` may be None.
    """
    lib = load()
    rc = lib.abom_path_relax(
        _ptr(edge_src), _ptr(col), _ptr(etype),
        _ptr(edge_weight) if edge_weight is not None else None,
        _ptr(cur), _ptr(nxt), _ptr(node_boost), _ptr(etype_boost),
        _ptr(etype_trav), _ptr(etype_gate),
        _ptr(node_gate) if node_gate is not None else None,
        etype.numel(), _stream(),
    )
    _check(rc, "abom_path_relax")


_RANK_BINS = 1024
_RANK_CHUNK = 256


def rank_bucket_order(scores, workspace: Optional[dict] = None):
    """Deterministic finding rank via the stable counting-sort kernel.

    Bit-equal to gpu_engine.rank_order's quantized key sort (bucket asc,
    index asc) — four small launches, no rocprim (ops/csrc/rank.hip)."""
    import torch

    lib = load()
    n = scores.numel()
    dev = scores.device
    if n == 0:
        return torch.empty(0, dtype=torch.int64, device=dev)
    nblocks = (n + _RANK_CHUNK - 1) // _RANK_CHUNK
    ws = workspace if workspace is not None else {}
    hist = ws.get("rank_hist")
    if hist is None or hist.numel() < nblocks * _RANK_BINS:
        hist = torch.empty(nblocks * _RANK_BINS, dtype=torch.int32, device=dev)
        ws["rank_hist"] = hist
    scratch = ws.get("rank_scratch")
    if scratch is None:
        scratch = torch.empty(2 * _RANK_BINS, dtype=torch.int32, device=dev)
        ws["rank_scratch"] = scratch
    order = torch.empty(n, dtype=torch.int64, device=dev)
    rc = lib.abom_rank_order(_ptr(scores), n, _ptr(hist), _ptr(scratch),
                             nblocks, _ptr(order), _stream())
    _check(rc, "abom_rank_order")
    return order
