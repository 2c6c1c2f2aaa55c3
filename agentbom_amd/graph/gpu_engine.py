"""GPU estate engine: CSR residency in HBM + the blast-radius pipeline.

The estate graph (agents, servers, creds, tools, packages) is built as
forward and reverse CSR on-device (torch sort/bincount/cumsum — GPU CSR
construction) and stays resident in HBM3E.  On top of it:

- ``match()``            bulk advisory matching (ops/csrc/match.hip)
- ``dependency_reach()`` multi-source BFS from all agents (graph.hip)
- ``blast_counts()``     segmented joins: per-finding distinct agent/cred/
                         tool reach with CWE-impact filtering
- ``score()``            GPU risk scoring (reference formula)
- ``step()``             full findings pipeline: match -> reach -> join ->
                         score -> rank (the bench.py timed region)
- ``blast_radius_query()`` batched bounded BFS for p50 query latency /
                         the /v1/graph read path

Replaces the reference's scan_packages hot loop + graph dict walks
(reference: src/agent_bom/scanners/package_scan.py:1307-2028,
src/agent_bom/graph/{container,dependency_reach}.py) — see SURVEY.md §2.11.
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from agentbom_amd.models.cwe_impact import (
    IMPACT_CODE,
    IMPACT_CREDENTIAL_ACCESS,
    IMPACT_CODE_EXECUTION,
    IMPACT_DATA_LEAK,
    IMPACT_FILE_ACCESS,
    IMPACT_INJECTION,
    IMPACT_SSRF,
)
from agentbom_amd.scan.synth import ET_CONTAINS, ET_HAS_CRED, ET_PROVIDES_TOOL, ET_USES, SyntheticEstate

# Impact-code lookup tables (u8 code -> filter class), device-ready.
_FULL_CRED = {IMPACT_CODE[c] for c in (IMPACT_CODE_EXECUTION, IMPACT_CREDENTIAL_ACCESS,
                                       IMPACT_DATA_LEAK, IMPACT_FILE_ACCESS, IMPACT_SSRF)}
_DB_CRED = {IMPACT_CODE[IMPACT_INJECTION]}
_FULL_TOOL = _FULL_CRED
_DB_TOOL = {IMPACT_CODE[IMPACT_INJECTION]}


def _impact_lut(full: set, db: set):
    """9-entry u8 LUT: 2 = full reach, 1 = db-only, 0 = none."""
    lut = np.zeros(9, dtype=np.uint8)
    for c in full:
        lut[c] = 2
    for c in db:
        lut[c] = 1
    return lut


_SIGN64 = -0x8000000000000000  # xor flips int64 order into u64 order


def _precompute_win_ranges(torch, arena: dict, pkg_group_key_sorted):
    """(wbeg u32[P], wend u32[P]) advisory window range per package row.

    Keys are u64 hashes stored as int64 bit patterns; xor with the sign bit
    makes torch.searchsorted order match the kernel's unsigned compare."""
    gk = arena["group_keys"]
    G = gk.numel()
    P = pkg_group_key_sorted.numel()
    dev = pkg_group_key_sorted.device
    if G == 0:
        z = torch.zeros(P, dtype=torch.int32, device=dev)
        return z, z.clone()
    g = torch.searchsorted(gk ^ _SIGN64, pkg_group_key_sorted ^ _SIGN64)
    gc = g.clamp(max=G - 1)
    valid = (g < G) & (gk[gc] == pkg_group_key_sorted)
    off = arena["group_off"].to(torch.int32)
    zero = torch.zeros(P, dtype=torch.int32, device=dev)
    wbeg = torch.where(valid, off[gc], zero)
    wend = torch.where(valid, off[gc + 1], zero)
    return wbeg.contiguous(), wend.contiguous()


def build_dedup_match_layout(torch, arena: dict, gk, hi, lo, flags):
    """Dedup (group, version, flags) rows for the resident-estate match.

    Real estates install the same (name, version) on many servers; the
    reference dedups before matching (package_scan.py:1794-1816).  The
    engine mirrors that: match runs once per DISTINCT row, results fan back
    out to package rows.  Returns None when dedup would not pay
    (>90% distinct).  Layout:
      u_*       unique row columns (lexicographically sorted)
      run_off   int64 [U+1]: unique row -> its span in ``perm2``
      perm2     int64 [P]: sorted-row -> original package index
      u_ranges  precomputed window ranges for the unique rows
    """
    P = gk.numel()
    # lexicographic stable sort by (gk, hi, lo, flags)
    idx = torch.argsort(flags.to(torch.int64), stable=True)
    for keycol in (lo, hi, gk):
        idx = idx[torch.argsort(keycol[idx], stable=True)]
    g2, h2, l2, f2 = gk[idx], hi[idx], lo[idx], flags[idx]
    neq = (g2[1:] != g2[:-1]) | (h2[1:] != h2[:-1]) | (l2[1:] != l2[:-1]) \
        | (f2[1:] != f2[:-1])
    starts = torch.cat([
        torch.zeros(1, dtype=torch.int64, device=gk.device),
        torch.nonzero(neq).flatten() + 1,
    ])
    U = starts.numel()
    if U > int(P * 0.9):
        return None
    run_off = torch.cat([starts, torch.tensor([P], dtype=torch.int64, device=gk.device)])
    layout = {
        "u_gk": g2[starts].contiguous(),
        "u_hi": h2[starts].contiguous(),
        "u_lo": l2[starts].contiguous(),
        "u_flags": f2[starts].contiguous(),
        "run_off": run_off,
        "perm2": idx,
    }
    layout["u_ranges"] = _precompute_win_ranges(torch, arena, layout["u_gk"])
    # Match schedule experiments (AGENT_BOM_MATCH_SCHEDULE):
    #   row   — rows in window-count-descending order.  Measured SLOWER
    #           (2.44 vs 2.27 ms/step): equal-wcnt rows from DIFFERENT
    #           groups interleave, so wave lanes stop sharing a window
    #           range and the cooperative walk (64x traffic saving on
    #           zipf heads) stops firing.
    #   group — GROUPS ordered by descending total work (rows x windows),
    #           rows within a group stay adjacent and in original order:
    #           cooperative walk intact, heavy groups launch first.
    #   off   — natural lexicographic order (default).
    import os

    mode = os.environ.get("AGENT_BOM_MATCH_SCHEDULE", "off")
    if mode in ("1", "row"):
        wcnt = (layout["u_ranges"][1] - layout["u_ranges"][0]).to(torch.int64)
        layout["heavy_order"] = torch.argsort(
            -wcnt, stable=True).to(torch.int32).contiguous()
    elif mode == "group":
        wbeg, wend = layout["u_ranges"]
        wcnt = (wend - wbeg).to(torch.int64)
        # group id = first row index of the run of equal wbeg (rows are
        # lexicographically sorted, so equal-group rows are adjacent)
        wb = wbeg.to(torch.int64)
        new_grp = torch.cat([torch.ones(1, dtype=torch.bool, device=wb.device),
                             wb[1:] != wb[:-1]])
        grp = torch.cumsum(new_grp.to(torch.int64), 0) - 1
        n_grp = int(grp[-1].item()) + 1 if grp.numel() else 0
        rows_per = torch.bincount(grp, minlength=n_grp)
        gw = torch.zeros(n_grp, dtype=torch.int64, device=wb.device)
        gw.scatter_add_(0, grp, wcnt)  # total windows walked per group
        # stable argsort by descending group work -> group visit order
        grp_rank = torch.empty(n_grp, dtype=torch.int64, device=wb.device)
        grp_rank[torch.argsort(-gw, stable=True)] = torch.arange(
            n_grp, device=wb.device)
        # row key = (group rank, original row index) — one stable sort
        U = wb.numel()
        bits = max(1, (U - 1).bit_length())
        key = (grp_rank[grp] << bits) | torch.arange(U, device=wb.device)
        layout["heavy_order"] = torch.argsort(key).to(torch.int32).contiguous()
    else:
        layout["heavy_order"] = None
    return layout


def expand_dedup_matches(torch, layout: dict, su, sw):
    """Fan unique-row matches back out to per-package (orig_idx, win) pairs,
    sorted by (pkg << 32 | win) — identical to the non-dedup output.

    Accepts UNSORTED (su, sw) — the one F-sized value sort at the end
    canonicalizes, so match_finalize can skip its pair sort (sort=False).
    A cross-product construction replacing the int64 value sort with a
    stable int32 argsort was A/B-measured at BOTH scales and rejected
    (2.50 vs 2.38 ms at 10M pkgs; 18.96 vs 18.69 ms at 100M): argsort
    internally sorts (key, index) pairs — the same 8 bytes/element the
    packed sort moves — so nothing is saved at any size; see
    profiles/r02_rank_kernel.md."""
    dev = su.device
    if su.numel() == 0:
        e = torch.empty(0, dtype=torch.int64, device=dev)
        return e, e.clone()
    run_off = layout["run_off"]
    cnt = run_off[su + 1] - run_off[su]
    rep = torch.repeat_interleave(
        torch.arange(su.numel(), device=dev), cnt)
    ends = torch.cumsum(cnt, 0)
    base = torch.cat([torch.zeros(1, dtype=torch.int64, device=dev), ends[:-1]])
    within = torch.arange(int(ends[-1].item()) if ends.numel() else 0,
                          device=dev) - base[rep]
    rows2 = run_off[su][rep] + within
    orig = layout["perm2"][rows2]
    packed = (orig << 32) | sw[rep]
    packed, _ = torch.sort(packed)
    return (packed >> 32), (packed & 0xFFFFFFFF)



_RANK_WS: dict = {}


def rank_order(torch, scores):
    """Deterministic finding rank: score desc (0.01 quantization), index asc.

    Scores are clipped to [0, 10] and serialized rounded, so 0.01 buckets
    preserve every visible distinction.  On GPU the rank is the stable
    1024-bin counting-sort kernel (ops/csrc/rank.hip — four small launches,
    no rocprim); the CPU path packs (bucket,
    index) into one integer key and sorts.  Both produce the identical
    (bucket asc, index asc) order -> bit-equal CPU/GPU.
    """
    n = scores.numel()
    if n == 0:
        return torch.empty(0, dtype=torch.int64, device=scores.device)
    if scores.is_cuda:
        from agentbom_amd.ops import native

        return native.rank_bucket_order(scores.contiguous(), _RANK_WS)
    q = (scores * 100).round().clamp(0, 1000).to(torch.int64)
    bits = max(1, (n - 1).bit_length())
    key = ((1000 - q) << bits) | torch.arange(n, device=scores.device)
    if (1010 << bits) < (1 << 31):
        key = key.to(torch.int32)
    skey, _ = torch.sort(key)
    return (skey.to(torch.int64)) & ((1 << bits) - 1)


class EstateEngine:
    """Device-resident estate + advisory arena + the findings pipeline."""

    def __init__(self, estate: SyntheticEstate, device: str = "cuda"):
        import torch

        self.torch = torch
        self.estate = estate
        self.device = torch.device(device)
        self.N = estate.num_nodes
        dev = self.device

        src = torch.from_numpy(estate.edge_src).to(dev)
        dst = torch.from_numpy(estate.edge_dst).to(dev)
        et = torch.from_numpy(estate.edge_type).to(dev)

        self.fwd = self._build_csr(src, dst, et)
        self.rev = self._build_csr(dst, src, et)

        self.pkg_group_key = torch.from_numpy(estate.pkg_name_id.view(np.int64)).to(dev)
        self.pkg_key_hi = torch.from_numpy(estate.pkg_key_hi.view(np.int64)).to(dev)
        self.pkg_key_lo = torch.from_numpy(estate.pkg_key_lo.view(np.int64)).to(dev)
        self.pkg_flags = torch.from_numpy(estate.pkg_flags).to(dev)
        # Match-locality layout: packages sorted by group key so adjacent
        # lanes binary-search identical tree paths (L1/L2-resident) and walk
        # the same window runs.  ``pkg_perm`` maps sorted row -> original
        # package index for output reporting.
        self.pkg_perm = torch.argsort(self.pkg_group_key, stable=True)
        self.pkg_group_key_sorted = self.pkg_group_key[self.pkg_perm].contiguous()
        self.pkg_key_hi_sorted = self.pkg_key_hi[self.pkg_perm].contiguous()
        self.pkg_key_lo_sorted = self.pkg_key_lo[self.pkg_perm].contiguous()
        self.pkg_flags_sorted = self.pkg_flags[self.pkg_perm].contiguous()

        self.arena = estate.arena.to_torch(dev)
        # resident-estate match index: per-package advisory window range,
        # precomputed once (estate and arena are static between steps) so
        # the match kernel skips the per-package group binary search
        self.pkg_win_range = _precompute_win_ranges(
            torch, self.arena, self.pkg_group_key_sorted)
        # dedup-match layout (None when the estate is >90% distinct rows)
        self.match_dedup = build_dedup_match_layout(
            torch, self.arena, self.pkg_group_key, self.pkg_key_hi,
            self.pkg_key_lo, self.pkg_flags) if self.use_gpu else None
        self.cred_is_db = torch.from_numpy(estate.cred_is_db).to(dev)
        self.tool_is_db = torch.from_numpy(estate.tool_is_db).to(dev)
        self.cred_lut = torch.from_numpy(_impact_lut(_FULL_CRED, _DB_CRED)).to(dev)
        self.tool_lut = torch.from_numpy(_impact_lut(_FULL_TOOL, _DB_TOOL)).to(dev)

        self.agent_ids = torch.arange(estate.n_agents, dtype=torch.int32, device=dev)
        self._bfs_ws: dict = {}
        self._match_stream = None  # side stream: match ∥ reach BFS overlap

        # per-node db-credential / db-tool indicator (for impact filtering)
        self.node_is_db_cred = torch.zeros(self.N, dtype=torch.uint8, device=dev)
        self.node_is_db_cred[
            estate.cred_base + torch.arange(estate.n_creds, device=dev)
        ] = self.cred_is_db
        self.node_is_db_tool = torch.zeros(self.N, dtype=torch.uint8, device=dev)
        self.node_is_db_tool[
            estate.tool_base + torch.arange(estate.n_tools, device=dev)
        ] = self.tool_is_db

    def _build_csr(self, src, dst, et):
        """GPU CSR construction: stable sort by src, bincount offsets."""
        torch = self.torch
        order = torch.argsort(src, stable=True)
        row_counts = torch.bincount(src, minlength=self.N)
        row_off = torch.zeros(self.N + 1, dtype=torch.int64, device=src.device)
        torch.cumsum(row_counts, 0, out=row_off[1:])
        return {
            "row_off": row_off,
            "col": dst[order].to(torch.int32),
            "etype": et[order].contiguous(),
            # col-aligned edge sources: enables edge-centric dense-frontier BFS
            "src": src[order].to(torch.int32),
        }

    # ── pipeline stages ────────────────────────────────────────────────────

    @property
    def use_gpu(self) -> bool:
        return self.device.type == "cuda"

    def match(self):
        """(finding_pkg_idx, finding_win_idx) int64 device tensors."""
        if self.use_gpu:
            from agentbom_amd.ops import native

            torch = self.torch
            sp, sw = native.match(
                self.pkg_group_key_sorted, self.pkg_key_hi_sorted,
                self.pkg_key_lo_sorted, self.pkg_flags_sorted,
                self.arena["group_keys"], self.arena["group_off"], self.arena["windows"],
                pkg_win_range=self.pkg_win_range,
            )
            # map sorted rows back to original package indices, re-sort for
            # the deterministic (pkg, window) output order
            orig = self.pkg_perm[sp]
            packed = (orig << 32) | sw
            packed, _ = torch.sort(packed)
            return (packed >> 32), (packed & 0xFFFFFFFF)
        # CPU oracle path (tests / dev boxes)
        from agentbom_amd.ops import cpu_ref

        torch = self.torch
        win = {k: v.numpy().view(np.uint64) if v.dtype == torch.int64 else v.numpy()
               for k, v in self.arena["windows"].items()}
        pkg_i, win_i = cpu_ref.match(
            self.pkg_group_key.numpy().view(np.uint64),
            self.pkg_key_hi.numpy().view(np.uint64),
            self.pkg_key_lo.numpy().view(np.uint64),
            self.pkg_flags.numpy(),
            self.arena["group_keys"].numpy().view(np.uint64),
            self.arena["group_off"].numpy().view(np.uint32),
            win,
        )
        return torch.from_numpy(pkg_i), torch.from_numpy(win_i)

    def dependency_reach(self):
        """u32 hop distance from the nearest agent, per node (multi-source BFS)."""
        mask = (1 << ET_USES) | (1 << ET_CONTAINS) | (1 << ET_HAS_CRED) | (1 << ET_PROVIDES_TOOL)
        if self.use_gpu:
            from agentbom_amd.ops import native

            return native.bfs(
                self.fwd["row_off"], self.fwd["col"], self.agent_ids, self.N,
                etype=self.fwd["etype"], allowed_mask=mask, workspace=self._bfs_ws,
                edge_src=self.fwd["src"], rev=self.rev,
            )
        from agentbom_amd.ops import cpu_ref

        dist = cpu_ref.bfs(
            self.fwd["row_off"].numpy(), self.fwd["col"].numpy(),
            self.agent_ids.numpy(), self.N,
            etype=self.fwd["etype"].numpy(), allowed_mask=mask,
        )
        return self.torch.from_numpy(dist.view(np.int32)).to(self.torch.int32)

    def _expand(self, csr, rows, carry, want_type: int):
        """Segmented gather: neighbors of ``rows`` with ``carry`` repeated.

        Returns (neighbor_node, carried_value) filtered to edge type
        ``want_type``.
        """
        torch = self.torch
        row_off = csr["row_off"]
        beg = row_off[rows]
        cnt = row_off[rows + 1] - beg
        total = int(cnt.sum().item())
        if total == 0:
            empty = torch.empty(0, dtype=torch.int64, device=self.device)
            return empty, empty
        rep = torch.repeat_interleave(torch.arange(rows.numel(), device=self.device), cnt)
        cum = torch.zeros(rows.numel(), dtype=torch.int64, device=self.device)
        torch.cumsum(cnt, 0, out=cum)
        base = torch.cat([torch.zeros(1, dtype=torch.int64, device=self.device), cum[:-1]])
        eidx = torch.arange(total, device=self.device) - base[rep] + beg[rep]
        keep = csr["etype"][eidx] == want_type
        nbr = csr["col"][eidx][keep].to(torch.int64)
        return nbr, carry[rep[keep]]

    def _distinct_counts(self, key_a, key_b, n_bins: int, weights=None):
        """Count distinct (a, b) pairs per a; optional 0/1 weight per b kind."""
        torch = self.torch
        if key_a.numel() == 0:
            z = torch.zeros(n_bins, dtype=torch.int32, device=self.device)
            return z if weights is None else (z, z.clone())
        pair = key_a * self.N + key_b
        uniq = torch.unique(pair)
        ua = uniq // self.N
        ub = uniq % self.N
        cnt = torch.bincount(ua, minlength=n_bins).to(torch.int32)
        if weights is None:
            return cnt
        w = weights[ub].to(torch.int64)
        cnt_w = torch.bincount(ua, weights=w.to(torch.float64), minlength=n_bins)
        return cnt, cnt_w.to(torch.int32)

    def blast_counts(self, finding_pkgs):
        """Per unique finding-package reach counts.

        GPU path: ONE fused wave-per-package HIP kernel (ops/csrc/blast.hip);
        zipf-head packages that overflow its LDS caps fall back to the exact
        sort-based join below.  CPU path: the sort-based join throughout.
        Returns dense tensors indexed by position in ``uniq_pkgs``.
        """
        torch = self.torch
        # finding_pkgs arrives sorted (the match-finalize pair sort), so
        # unique_consecutive avoids torch.unique's internal re-sort
        uniq_pkgs = torch.unique_consecutive(finding_pkgs)
        if self.use_gpu:
            return self._blast_counts_fused(uniq_pkgs)
        return self._blast_counts_join(uniq_pkgs)

    def _blast_counts_fused(self, uniq_pkgs):
        from agentbom_amd.ops import native

        torch = self.torch
        counts, overflow = native.blast_counts(
            uniq_pkgs.to(torch.int32).contiguous(), self.rev, self.fwd,
            (ET_CONTAINS, ET_USES, ET_HAS_CRED, ET_PROVIDES_TOOL),
            self.node_is_db_cred, self.node_is_db_tool,
        )
        result = {
            "uniq_pkgs": uniq_pkgs,
            "n_servers": counts[:, 0].to(torch.int32),
            "n_agents": counts[:, 1].to(torch.int32),
            "n_creds_all": counts[:, 2].to(torch.int32),
            "n_creds_db": counts[:, 3].to(torch.int32),
            "n_tools_all": counts[:, 4].to(torch.int32),
            "n_tools_db": counts[:, 5].to(torch.int32),
        }
        ov_idx = torch.nonzero(overflow).flatten()
        if ov_idx.numel():
            heavy = self._blast_counts_join(uniq_pkgs[ov_idx])
            counts = counts.clone()
            for k, key in enumerate(("n_servers", "n_agents", "n_creds_all",
                                     "n_creds_db", "n_tools_all", "n_tools_db")):
                result[key] = result[key].clone()
                result[key][ov_idx] = heavy[key]
                counts[ov_idx, k] = heavy[key]
        result["counts2d"] = counts  # [U,6] — feeds the fused score_gather
        return result

    def _blast_counts_join(self, uniq_pkgs):
        """Sort-based exact join (torch ops) — CPU path + overflow fallback."""
        torch = self.torch
        # position index for dense bincounts
        n_up = uniq_pkgs.numel()
        pos = torch.arange(n_up, device=self.device)

        # package -> servers (reverse CONTAINS)
        srv, carry = self._expand(self.rev, uniq_pkgs, pos, ET_CONTAINS)
        n_servers = self._distinct_counts(carry, srv, n_up)

        # servers -> agents (reverse USES), carry package position
        ag, carry2 = self._expand(self.rev, srv, carry, ET_USES)
        n_agents = self._distinct_counts(carry2, ag, n_up)

        # servers -> creds / tools (forward)
        cr, carry3 = self._expand(self.fwd, srv, carry, ET_HAS_CRED)
        n_creds_all, n_creds_db = self._distinct_counts(
            carry3, cr, n_up, weights=self.node_is_db_cred
        )
        tl, carry4 = self._expand(self.fwd, srv, carry, ET_PROVIDES_TOOL)
        n_tools_all, n_tools_db = self._distinct_counts(
            carry4, tl, n_up, weights=self.node_is_db_tool
        )
        return {
            "uniq_pkgs": uniq_pkgs,
            "n_servers": n_servers,
            "n_agents": n_agents,
            "n_creds_all": n_creds_all,
            "n_creds_db": n_creds_db,
            "n_tools_all": n_tools_all,
            "n_tools_db": n_tools_db,
        }

    def step(self, blast_depth: int = 1, reach_dist=None):
        """One full findings pass.  Returns summary dict.

        ``reach_dist`` overrides the locally-computed dependency-reach
        distances (multi-GPU mode passes the shard's slice of the
        distributed BFS result)."""
        torch = self.torch
        if self.use_gpu:
            # Overlap: the match kernel has no dependency on the reach BFS,
            # so it runs on a side HIP stream while the BFS host loop (which
            # syncs its own stream every level) drives the default stream.
            from agentbom_amd.ops import native

            if self._match_stream is None:
                self._match_stream = torch.cuda.Stream(device=self.device)
            side = self._match_stream
            side.wait_stream(torch.cuda.current_stream())
            dd = self.match_dedup
            with torch.cuda.stream(side):
                if dd is not None:
                    # dedup-match: one kernel row per DISTINCT (name,
                    # version); results fan back out below
                    pending = native.match_launch(
                        dd["u_gk"], dd["u_hi"], dd["u_lo"], dd["u_flags"],
                        self.arena["group_keys"], self.arena["group_off"],
                        self.arena["windows"], pkg_win_range=dd["u_ranges"],
                        order=dd["heavy_order"])
                else:
                    pending = native.match_launch(
                        self.pkg_group_key_sorted, self.pkg_key_hi_sorted,
                        self.pkg_key_lo_sorted, self.pkg_flags_sorted,
                        self.arena["group_keys"], self.arena["group_off"],
                        self.arena["windows"], pkg_win_range=self.pkg_win_range)
            dist = reach_dist if reach_dist is not None else self.dependency_reach()
            torch.cuda.current_stream().wait_stream(side)
            # downstream always re-canonicalizes; skip the pair sort
            sp, sw = native.match_finalize(pending, sort=False)
            if dd is not None:
                pkg_idx, win_idx = expand_dedup_matches(torch, dd, sp, sw)
            else:
                orig = self.pkg_perm[sp]
                packed = (orig << 32) | sw
                packed, _ = torch.sort(packed)
                pkg_idx, win_idx = (packed >> 32), (packed & 0xFFFFFFFF)
        else:
            pkg_idx, win_idx = self.match()
            dist = reach_dist if reach_dist is not None else self.dependency_reach()
        n_findings = pkg_idx.numel()

        pkg_nodes = pkg_idx + self.estate.pkg_base
        counts = self.blast_counts(pkg_nodes)

        # map each finding to its unique-package position
        pos = torch.searchsorted(counts["uniq_pkgs"], pkg_nodes)

        if self.use_gpu:
            # fused gather + score: arena gathers, CWE-impact LUT count
            # selection, reachability and the risk formula in ONE kernel
            from agentbom_amd.ops import native

            scores, n_agents, n_creds, n_tools = native.score_gather(
                win_idx.contiguous(), pkg_nodes.contiguous(), pos.contiguous(),
                self.arena["severity"], self.arena["kev"], self.arena["epss"],
                self.arena["impact"], self.cred_lut, self.tool_lut,
                counts["counts2d"].contiguous(), dist)
            return {
                "n_findings": int(n_findings),
                "scores": scores,
                "order": rank_order(torch, scores),
                "pkg_idx": pkg_idx,
                "win_idx": win_idx,
                "n_agents": n_agents,
                "n_creds": n_creds,
                "n_tools": n_tools,
                "reach_dist": dist,
            }

        # CWE-impact filtering via LUTs: 2=full, 1=db-only, 0=none
        impact = self.arena["impact"].to(torch.int64)[win_idx]
        cred_cls = self.cred_lut.to(torch.int64)[impact]
        tool_cls = self.tool_lut.to(torch.int64)[impact]
        n_creds = torch.where(
            cred_cls == 2, counts["n_creds_all"][pos],
            torch.where(cred_cls == 1, counts["n_creds_db"][pos],
                        torch.zeros_like(pos, dtype=torch.int32)),
        )
        n_tools = torch.where(
            tool_cls == 2, counts["n_tools_all"][pos],
            torch.where(tool_cls == 1, counts["n_tools_db"][pos],
                        torch.zeros_like(pos, dtype=torch.int32)),
        )
        n_agents = counts["n_agents"][pos]

        sev = self.arena["severity"][win_idx]
        kev = self.arena["kev"][win_idx]
        epss = self.arena["epss"][win_idx]
        flags = (kev.to(torch.uint8) << 1)  # bit1 kev; no ai_ctx/suppressed in synth
        scorecard = torch.full((n_findings,), -1.0, dtype=torch.float32, device=self.device)
        # dist is int32 with UNVISITED as the -1 bit pattern (comparing the
        # u32 literal would promote to int64 and never match)
        reach_known = dist[pkg_nodes] != -1
        reach = reach_known.to(torch.int8)

        from agentbom_amd.ops import cpu_ref

        scores = torch.from_numpy(
            cpu_ref.risk_score(
                sev.numpy(), n_agents.numpy().astype(np.uint32),
                n_creds.numpy().astype(np.uint32), n_tools.numpy().astype(np.uint32),
                flags.numpy(), epss.numpy(), scorecard.numpy(), reach.numpy(),
            )
        )

        # deterministic rank: quantized score desc, finding index asc
        order = rank_order(torch, scores)
        return {
            "n_findings": int(n_findings),
            "scores": scores,
            "order": order,
            "pkg_idx": pkg_idx,
            "win_idx": win_idx,
            "n_agents": n_agents,
            "n_creds": n_creds,
            "n_tools": n_tools,
            "reach_dist": dist,
        }

    def rollup(self, step_res=None):
        """Estate CONTAINS-tree collapse on device.

        The estate tree is two levels (agent -> server -> package), so the
        collapse is two segmented reductions kept on the GPU: per-server
        severity histograms over contained findings (severity_histogram
        HIP kernel on the GPU path), then index_add-merged up the USES
        edges into per-agent aggregates.  Returns device tensors:
        histograms [n,6] (codes 5..0 mapped to columns 0..5), worst code,
        finding counts — the estate-scale analog of graph/rollup.py's
        Python CONTAINS walk (reference graph/rollup.py:313-580).
        """
        torch = self.torch
        est = self.estate
        if step_res is None:
            step_res = self.step()
        win_idx = step_res["win_idx"].to(torch.int64)
        sev_code = self.arena["severity"].to(torch.int64)[win_idx]  # 5..1
        sev_col = (5 - sev_code).clamp(0, 5)  # 0=critical .. 5=unknown
        pkg_nodes = (step_res["pkg_idx"] + est.pkg_base).to(torch.int64)

        # finding -> containing servers (reverse CONTAINS), severity carried
        uniq = torch.unique(pkg_nodes)
        pos_of_uniq = torch.arange(uniq.numel(), device=self.device)
        srv, carry = self._expand(self.rev, uniq, pos_of_uniq, ET_CONTAINS)
        # distinct (package, server): duplicate CONTAINS edges count once
        pair = torch.unique((carry << 32) | srv)
        carry = pair >> 32
        srv = pair & 0xFFFFFFFF
        # map each (finding) to each server containing its package
        fpos = torch.searchsorted(uniq, pkg_nodes)
        # (uniq_pos -> servers) join back onto findings: build per-uniq run
        order = torch.argsort(carry, stable=True)
        srv_sorted = srv[order]
        carry_sorted = carry[order]
        run_counts = torch.bincount(carry_sorted, minlength=uniq.numel())
        run_off = torch.zeros(uniq.numel() + 1, dtype=torch.int64, device=self.device)
        torch.cumsum(run_counts, 0, out=run_off[1:])
        cnt_f = run_counts[fpos]
        rep = torch.repeat_interleave(
            torch.arange(pkg_nodes.numel(), device=self.device), cnt_f)
        ends = torch.cumsum(cnt_f, 0)
        base = torch.cat([torch.zeros(1, dtype=torch.int64, device=self.device),
                          ends[:-1]])
        within = torch.arange(int(ends[-1].item()) if ends.numel() else 0,
                              device=self.device) - base[rep]
        srv_per = srv_sorted[run_off[fpos[rep]] + within]
        sev_per = sev_col[rep]

        srv_local = (srv_per - est.server_base).to(torch.int64)
        if self.use_gpu:
            from agentbom_amd.ops import native

            # the histogram kernel reads u32 owner indices
            srv_hist = native.severity_histogram(
                srv_local.to(torch.int32).contiguous(),
                sev_per.to(torch.uint8).contiguous(), est.n_servers)
            srv_hist = srv_hist.to(torch.int64)
        else:
            flat = torch.bincount(srv_local * 6 + sev_per,
                                  minlength=est.n_servers * 6)
            srv_hist = flat.view(est.n_servers, 6)

        # server -> agent merge (reverse USES rows of servers)
        srv_ids = torch.arange(est.n_servers, device=self.device) + est.server_base
        ag, srv_carry = self._expand(self.rev, srv_ids,
                                     torch.arange(est.n_servers, device=self.device),
                                     ET_USES)
        agent_hist = torch.zeros(est.n_agents, 6, dtype=torch.int64,
                                 device=self.device)
        agent_hist.index_add_(0, ag, srv_hist[srv_carry])

        def worst_of(hist):
            has = hist > 0
            first = torch.argmax(has.to(torch.int8), dim=1)
            any_f = has.any(dim=1)
            return torch.where(any_f, first, torch.full_like(first, 6))

        return {
            "server_hist": srv_hist, "agent_hist": agent_hist,
            "server_worst": worst_of(srv_hist), "agent_worst": worst_of(agent_hist),
            "server_findings": srv_hist.sum(dim=1),
            "agent_findings": agent_hist.sum(dim=1),
        }

    # ── attack / exposure paths (GPU path DP) ──────────────────────────────

    ET_LATERAL = 4  # reversed USES edge (server -> agent): lateral movement

    def _path_edges(self):
        """Edge arrays for the path DP: forward edges + reversed USES edges
        (lateral movement via shared servers).  Cached device tensors."""
        cached = getattr(self, "_path_edges_cache", None)
        if cached is not None:
            return cached
        torch = self.torch
        est = self.estate
        uses = est.edge_type == ET_USES
        src = np.concatenate([est.edge_src, est.edge_dst[uses]])
        dst = np.concatenate([est.edge_dst, est.edge_src[uses]])
        et = np.concatenate([est.edge_type,
                             np.full(int(uses.sum()), self.ET_LATERAL, dtype=np.uint8)])
        dev = self.device
        cached = (
            torch.from_numpy(src).to(device=dev, dtype=torch.int32),
            torch.from_numpy(dst).to(device=dev, dtype=torch.int32),
            torch.from_numpy(et).to(device=dev, dtype=torch.uint8),
        )
        self._path_edges_cache = cached
        return cached

    def attack_paths(self, step_res=None, k: int = 100, max_depth: int = 6):
        """Top-k attack/exposure paths over the estate (HIP path DP).

        Entry class: agents.  Crown jewels: credentials and tools.  Gate:
        the path must traverse a server that contains a matched-vulnerable
        package (the estate analog of the vulnerable_to/exploitable_via
        edge gate in graph/attack_paths.py).  Node boosts mirror
        _node_boost: vulnerable +1.5, severity map, KEV +2.0.
        Replaces the reference's DFS at estate scale
        (src/agent_bom/graph/attack_path_fusion.py:194).
        """
        from agentbom_amd.graph.path_engine import run_path_dp

        torch = self.torch
        est = self.estate
        if step_res is None:
            step_res = self.step()

        pkg_nodes = (step_res["pkg_idx"] + est.pkg_base).to(torch.int64)
        win_idx = step_res["win_idx"].to(torch.int64)
        sev = self.arena["severity"].to(torch.int64)[win_idx]
        kev = self.arena["kev"].to(torch.int64)[win_idx]
        sev_boost = torch.zeros_like(sev, dtype=torch.float32)
        sev_boost[sev == 5] = 2.0   # critical
        sev_boost[sev == 4] = 1.2   # high
        sev_boost[sev == 3] = 0.6   # medium
        boost = 1.5 + sev_boost + kev.to(torch.float32) * 2.0

        node_boost = torch.zeros(self.N, dtype=torch.float32, device=self.device)
        node_boost.scatter_reduce_(0, pkg_nodes, boost, reduce="amax")

        # gated servers: contain >= 1 matched package; server boost = the
        # best boost among its matched packages (compromise value)
        uniq = torch.unique(pkg_nodes)
        pos = torch.arange(uniq.numel(), device=self.device)
        srv, carry = self._expand(self.rev, uniq, pos, ET_CONTAINS)
        node_gate = torch.zeros(self.N, dtype=torch.uint8, device=self.device)
        if srv.numel():
            node_gate[srv] = 1
            pkg_best = torch.zeros(uniq.numel(), dtype=torch.float32, device=self.device)
            pkg_best.scatter_reduce_(0, torch.searchsorted(uniq, pkg_nodes),
                                     boost, reduce="amax")
            node_boost.scatter_reduce_(0, srv, pkg_best[carry], reduce="amax")

        etype_boost = np.zeros(256, dtype=np.float32)
        etype_boost[ET_USES] = 0.3
        etype_boost[ET_CONTAINS] = 0.3
        etype_boost[ET_HAS_CRED] = 1.5       # exposes_cred class
        etype_boost[ET_PROVIDES_TOOL] = 0.3
        etype_boost[self.ET_LATERAL] = 1.0   # shares_server class
        etype_trav = np.zeros(256, dtype=np.uint8)
        etype_trav[[ET_USES, ET_CONTAINS, ET_HAS_CRED, ET_PROVIDES_TOOL,
                    self.ET_LATERAL]] = 1
        etype_gate = np.zeros(256, dtype=np.uint8)

        target_mask = np.zeros(self.N, dtype=np.uint8)
        target_mask[est.cred_base:est.cred_base + est.n_creds] = 1
        target_mask[est.tool_base:est.tool_base + est.n_tools] = 1

        src, dst, et = self._path_edges()
        return run_path_dp(
            src, dst, et, None, self.N, self.agent_ids.to(torch.int64),
            node_boost, etype_boost, etype_trav, etype_gate, node_gate,
            target_mask, max_depth=max_depth, k=k,
            device=self.device if self.use_gpu else None,
        )

    def enable_query_graph(self, batch: int = 1, max_hops: int = 4,
                           max_nodes: int = 4096) -> None:
        """Capture the blast-radius query into a hipGraph for serving.

        One capture per (batch, max_hops, max_nodes) shape: the kernel launch
        sequence is recorded once against static in/out buffers; each query
        then copies ids into the static input and REPLAYS the graph — no
        per-query launch construction on the hot path (the serving analog of
        capturing launch-bound inner loops in hipGraphs).
        """
        torch = self.torch
        from agentbom_amd.ops import native

        dev = self.device
        q_in = torch.zeros(batch, dtype=torch.int32, device=dev)
        out_nodes = torch.empty((batch, max_nodes), dtype=torch.int32, device=dev)
        out_hops = torch.empty((batch, max_nodes), dtype=torch.uint8, device=dev)
        out_counts = torch.zeros(batch, dtype=torch.int32, device=dev)
        out_trunc = torch.zeros(batch, dtype=torch.uint8, device=dev)
        lib = native.load()
        import ctypes

        def _launch():
            rc = lib.abom_impact_query(
                native._ptr(self.rev["row_off"]), native._ptr(self.rev["col"]),
                None, ctypes.c_uint(0xFFFFFFFF), native._ptr(q_in), batch,
                max_hops, max_nodes, native._ptr(out_nodes),
                native._ptr(out_hops), native._ptr(out_counts),
                native._ptr(out_trunc), native._stream())
            native._check(rc, "abom_impact_query")

        # warmup on a side stream, then capture
        s = torch.cuda.Stream(device=dev)
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            _launch()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            _launch()
        self._query_graph = {
            "graph": graph, "batch": batch, "max_hops": max_hops,
            "max_nodes": max_nodes, "q_in": q_in, "out_nodes": out_nodes,
            "out_hops": out_hops, "out_counts": out_counts,
            "out_trunc": out_trunc,
        }

    def blast_radius_query(self, node_ids, max_hops: int = 4, max_nodes: int = 4096):
        """Bounded blast-radius neighborhoods (reverse direction: who reaches
        this package / what this vuln exposes)."""
        q = node_ids.to(self.torch.int32)
        if self.use_gpu:
            g = getattr(self, "_query_graph", None)
            if (g is not None and q.numel() == g["batch"]
                    and max_hops == g["max_hops"]
                    and max_nodes == g["max_nodes"]):
                # the kernel fully writes counts/truncated per query, so
                # replay needs only the input copy
                g["q_in"].copy_(q.to(self.device))
                g["graph"].replay()
                return (g["out_nodes"], g["out_hops"], g["out_counts"],
                        g["out_trunc"])
            from agentbom_amd.ops import native

            return native.impact_query(
                self.rev["row_off"], self.rev["col"], q, etype=None,
                allowed_mask=0xFFFFFFFF, max_hops=max_hops, max_nodes=max_nodes,
            )
        from agentbom_amd.ops import cpu_ref

        torch = self.torch
        results = cpu_ref.impact_query(
            self.rev["row_off"].numpy(), self.rev["col"].numpy(), q.numpy(),
            max_hops=max_hops, max_nodes=max_nodes,
        )
        Q = q.numel()
        out_nodes = torch.zeros((Q, max_nodes), dtype=torch.int32)
        out_hops = torch.zeros((Q, max_nodes), dtype=torch.uint8)
        out_counts = torch.zeros(Q, dtype=torch.int32)
        out_trunc = torch.zeros(Q, dtype=torch.uint8)
        for qi, (hops, trunc) in enumerate(results):
            items = list(hops.items())[:max_nodes]
            for j, (node, hop) in enumerate(items):
                out_nodes[qi, j] = node
                out_hops[qi, j] = hop
            out_counts[qi] = len(items)
            out_trunc[qi] = 1 if trunc else 0
        return out_nodes, out_hops, out_counts, out_trunc
