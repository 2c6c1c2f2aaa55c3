"""Deterministic synthetic estate generator (10M-node scale, numpy).

Rebuilds the reference's seeded benchmark-estate generator idea
(reference: scripts/generate_graph_benchmark_estate.py — skewed fan-out,
~1% of agents own 18-32 servers) at GPU scale: node-id ranges per entity
class, typed edge lists, per-package encoded version keys, and a synthetic
advisory arena with controllable match density.

Node id space (contiguous ranges):
  agents [0, A) | servers [A, A+S) | creds | tools | packages (last)

Edge types (compact engine codes; the full RelationshipType enum of the
graph layer maps onto these for traversal classes):
  0 USES (agent->server)        1 CONTAINS (server->package)
  2 HAS_CREDENTIAL (server->cred)  3 PROVIDES_TOOL (server->tool)
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import numpy as np

from agentbom_amd.db.arena import AdvisoryArena, build_arena_from_columns
from agentbom_amd.ops.cpu_ref import PF_ENCODABLE, WF_HAS_FIXED, WF_HAS_INTRO, WF_HAS_LAST
from agentbom_amd.utils.version_keys import RELEASE_RANK, pack_batch

ET_USES = 0
ET_CONTAINS = 1
ET_HAS_CRED = 2
ET_PROVIDES_TOOL = 3


@dataclass
class SyntheticEstate:
    n_agents: int
    n_servers: int
    n_creds: int
    n_tools: int
    n_packages: int
    # node-id range starts
    agent_base: int
    server_base: int
    cred_base: int
    tool_base: int
    pkg_base: int
    num_nodes: int
    # typed edges (forward direction)
    edge_src: np.ndarray  # int64
    edge_dst: np.ndarray
    edge_type: np.ndarray  # uint8
    # package columns
    pkg_name_id: np.ndarray  # uint64 group keys
    pkg_key_hi: np.ndarray  # uint64
    pkg_key_lo: np.ndarray
    pkg_flags: np.ndarray  # uint8
    # credential classification (for CWE impact filtering)
    cred_is_db: np.ndarray  # uint8 [n_creds]
    tool_is_db: np.ndarray  # uint8 [n_tools]
    arena: AdvisoryArena
    seed: int

    @property
    def num_edges(self) -> int:
        return len(self.edge_src)


def _osv_shaped_arena(target_windows: int, name_catalog: int, seed: int) -> AdvisoryArena:
    """OSV-dump-shaped arena: exactly ``target_windows`` windows.

    Realism properties the legacy uniform arena lacked:
    - zipf-correlated branch counts: popular names (low name id — package
      name popularity is pareto over ids) carry MORE advisory windows,
      so the kernel's window walk meets long runs exactly where package
      density is highest (the adversarial case for the zipf-head groups);
    - multi-window branches per advisory (1..64 windows per name);
    - mixed window forms: ~10% last_affected (inclusive upper bound),
      ~1% open-ended introduced-only rows (unfixed app advisories — kept
      matched by default per the round-1 advisor fix), remainder
      introduced/fixed pairs.
    """
    rng = np.random.default_rng([seed, 0xA5A5])
    mean_w = 4.0
    n_est = max(1, int(target_windows / mean_w) + 1024)
    counts = 1 + np.minimum((rng.pareto(1.5, n_est) * 2.2).astype(np.int64), 63)
    n_est = min(n_est, name_catalog)  # at most one branch-group per name
    counts = counts[:n_est]
    cum = np.cumsum(counts)
    cut = int(np.searchsorted(cum, target_windows))
    if cut >= len(counts):
        counts[-1] += target_windows - cum[-1]
    else:
        counts = counts[: cut + 1]
        counts[-1] -= int(cum[cut]) - target_windows
    n_names = len(counts)
    # distinct names, branch-heavy groups on the popular (low-id) names
    vuln_names = np.sort(rng.choice(name_catalog, n_names, replace=False)).astype(np.uint64)
    counts = np.sort(counts)[::-1]
    w_names = np.repeat(vuln_names, counts)
    W = len(w_names)

    # windows span MINOR versions within one major branch (the OSV norm:
    # an advisory's affected window covers a few releases of one branch) —
    # keeps per-window match probability realistic (~0.5%) so the finding
    # rate stays a few percent of the estate even at millions of windows
    i1 = rng.integers(0, 20, W, dtype=np.int64)
    i2 = rng.integers(0, 28, W, dtype=np.int64)
    i3 = rng.integers(0, 60, W, dtype=np.int64)
    span = 1 + np.minimum(rng.pareto(2.0, W).astype(np.int64), 5)
    f1 = i1
    f2 = np.minimum(i2 + span, 30)
    f3 = rng.integers(0, 60, W, dtype=np.int64)
    zeros = np.zeros(W, dtype=np.int64)
    rank = np.full(W, RELEASE_RANK, dtype=np.int64)
    ihi, ilo = pack_batch(zeros, i1, i2, i3, zeros, rank, zeros)
    uhi, ulo = pack_batch(zeros, f1, f2, f3, zeros, rank, zeros)

    form = rng.random(W)
    last_mask = form < 0.10           # introduced + last_affected (inclusive)
    open_mask = (form >= 0.10) & (form < 0.11)  # introduced-only (no fix yet)
    fhi = np.where(last_mask | open_mask, 0, uhi).astype(np.uint64)
    flo = np.where(last_mask | open_mask, 0, ulo).astype(np.uint64)
    lhi = np.where(last_mask, uhi, 0).astype(np.uint64)
    llo = np.where(last_mask, ulo, 0).astype(np.uint64)
    wflags = np.full(W, WF_HAS_INTRO | WF_HAS_FIXED, dtype=np.uint8)
    wflags[last_mask] = WF_HAS_INTRO | WF_HAS_LAST
    wflags[open_mask] = WF_HAS_INTRO
    # open-ended rows: bias introduced into the top of the version space so
    # each unfixed advisory bounds its own match set (~1-2% of versions)
    n_open = int(open_mask.sum())
    oh, ol = pack_batch(
        np.zeros(n_open, dtype=np.int64),
        rng.integers(18, 20, n_open, dtype=np.int64),
        rng.integers(24, 30, n_open, dtype=np.int64),
        i3[open_mask],
        np.zeros(n_open, dtype=np.int64),
        np.full(n_open, RELEASE_RANK, dtype=np.int64),
        np.zeros(n_open, dtype=np.int64))
    ihi[open_mask] = oh
    ilo[open_mask] = ol

    sev = rng.choice([5, 4, 3, 2], size=W, p=[0.1, 0.3, 0.4, 0.2]).astype(np.uint8)
    cvss = (rng.random(W) * 10).astype(np.float32)
    epss = np.where(rng.random(W) < 0.7, rng.random(W) ** 2, -1.0).astype(np.float32)
    kev = (rng.random(W) < 0.02).astype(np.uint8)
    impact = rng.choice(
        np.arange(9, dtype=np.uint8), size=W,
        p=[0.35, 0.08, 0.10, 0.04, 0.08, 0.12, 0.10, 0.05, 0.08],
    ).astype(np.uint8)

    return build_arena_from_columns(
        w_names, ihi.astype(np.uint64), ilo.astype(np.uint64), fhi, flo,
        lhi, llo, wflags, sev, cvss, epss, kev, impact=impact,
    )


def _splitmix64(x: np.ndarray) -> np.ndarray:
    """Vectorized splitmix64 over uint64 (wrapping arithmetic)."""
    with np.errstate(over="ignore"):
        x = (x + np.uint64(0x9E3779B97F4A7C15))
        x = (x ^ (x >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
        x = (x ^ (x >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
        return x ^ (x >> np.uint64(31))


def _pool_versions(rng, name_id: np.ndarray, name_catalog: int, seed: int):
    """Per-name finite release pools -> (key_hi, key_lo) per package.

    Release counts are zipf-distributed (8..2048); a package's version is a
    uniform draw from its name's pool, derived by hashing (name, rel_idx)
    into the same (major<20, minor<30, patch<60) space the arena windows
    cover."""
    n_names_used = name_catalog
    with np.errstate(over="ignore"):
        nrel = 8 + np.minimum(
            (np.random.default_rng([seed, 0x9E1E]).pareto(1.2, n_names_used) * 16)
            .astype(np.int64), 2040)
    rel_idx = (rng.random(len(name_id)) * nrel[name_id.astype(np.int64)]).astype(np.uint64)
    h = _splitmix64(name_id * np.uint64(0x9E3779B97F4A7C15) ^ _splitmix64(rel_idx))
    n1 = (h % np.uint64(20)).astype(np.int64)
    n2 = ((h >> np.uint64(8)) % np.uint64(30)).astype(np.int64)
    n3 = ((h >> np.uint64(16)) % np.uint64(60)).astype(np.int64)
    zeros = np.zeros(len(name_id), dtype=np.int64)
    rank = np.full(len(name_id), RELEASE_RANK, dtype=np.int64)
    return pack_batch(zeros, n1, n2, n3, zeros, rank, zeros)


def _triple_keys(rng, n, lo=(0, 0, 0), hi=(20, 30, 60)):
    n1 = rng.integers(lo[0], hi[0], n, dtype=np.int64)
    n2 = rng.integers(lo[1], hi[1], n, dtype=np.int64)
    n3 = rng.integers(lo[2], hi[2], n, dtype=np.int64)
    zeros = np.zeros(n, dtype=np.int64)
    rank = np.full(n, RELEASE_RANK, dtype=np.int64)
    hi_k, lo_k = pack_batch(zeros, n1, n2, n3, zeros, rank, zeros)
    return (n1, n2, n3), (hi_k, lo_k)


def generate_estate(
    n_agents: int = 1000,
    n_servers: int = 5000,
    n_packages: int = 100_000,
    name_catalog: int = 20_000,
    vulnerable_name_fraction: float = 0.05,
    windows_per_name: int = 2,
    extra_pkg_share: float = 0.5,
    creds_per_server: float = 2.0,
    tools_per_server: float = 4.0,
    seed: int = 1234,
    arena_windows: Optional[int] = None,
) -> SyntheticEstate:
    """Generate a seeded estate.  Sizes are node counts; edges follow.

    ``arena_windows``: when set, build an OSV-dump-shaped arena of exactly
    that many windows (zipf-correlated branch counts per name, mixed
    fixed/last_affected/open window forms) instead of the legacy uniform
    2-windows-per-name arena — VERDICT r1 item 3 (the real OSV corpus is
    millions of windows with multi-window branches, reference
    src/agent_bom/db/schema.py:88-142)."""
    rng = np.random.default_rng(seed)

    n_creds = max(64, n_servers // 2)
    n_tools = max(64, n_servers)
    agent_base = 0
    server_base = n_agents
    cred_base = server_base + n_servers
    tool_base = cred_base + n_creds
    pkg_base = tool_base + n_tools
    num_nodes = pkg_base + n_packages

    # ── agent -> server (skewed: ~1% heavy agents own 18-32x the servers) ──
    heavy = rng.random(n_agents) < 0.01
    weights = np.where(heavy, 25.0, 1.0)
    weights /= weights.sum()
    owner = rng.choice(n_agents, size=n_servers, p=weights)
    # ~10% of servers are shared with a second agent (lateral movement paths)
    shared_mask = rng.random(n_servers) < 0.10
    second = rng.integers(0, n_agents, n_servers)
    shared_idx = np.nonzero(shared_mask & (second != owner))[0]
    us_src = np.concatenate([owner, second[shared_idx]])
    us_dst = np.concatenate([np.arange(n_servers), shared_idx]) + server_base

    # ── server -> package ──────────────────────────────────────────────────
    home_server = rng.integers(0, n_servers, n_packages)
    sp_src = home_server + server_base
    sp_dst = np.arange(n_packages, dtype=np.int64) + pkg_base
    n_extra = int(n_packages * extra_pkg_share)
    if n_extra:
        # popularity-skewed extra containment (popular packages shared widely)
        pop = (rng.pareto(1.5, n_extra) * n_packages / 50).astype(np.int64) % n_packages
        ex_src = rng.integers(0, n_servers, n_extra) + server_base
        ex_dst = pop + pkg_base
        sp_src = np.concatenate([sp_src, ex_src])
        sp_dst = np.concatenate([sp_dst, ex_dst])

    # ── server -> cred / tool ─────────────────────────────────────────────
    def attach(n_items, base, mean):
        counts = rng.poisson(mean, n_servers)
        total = int(counts.sum())
        src = np.repeat(np.arange(n_servers), counts) + server_base
        dst = rng.integers(0, n_items, total) + base
        return src, dst

    sc_src, sc_dst = attach(n_creds, cred_base, creds_per_server)
    st_src, st_dst = attach(n_tools, tool_base, tools_per_server)

    edge_src = np.concatenate([us_src, sp_src, sc_src, st_src]).astype(np.int64)
    edge_dst = np.concatenate([us_dst, sp_dst, sc_dst, st_dst]).astype(np.int64)
    edge_type = np.concatenate(
        [
            np.full(len(us_src), ET_USES, dtype=np.uint8),
            np.full(len(sp_src), ET_CONTAINS, dtype=np.uint8),
            np.full(len(sc_src), ET_HAS_CRED, dtype=np.uint8),
            np.full(len(st_src), ET_PROVIDES_TOOL, dtype=np.uint8),
        ]
    )

    # ── package identities + versions ─────────────────────────────────────
    # zipf-ish name popularity over the catalog
    name_id = (rng.pareto(1.2, n_packages) * name_catalog / 20).astype(np.uint64) % name_catalog
    if arena_windows is not None:
        # realism mode: each name has a FINITE release pool (real packages
        # ship tens-to-hundreds of releases; an estate with 40k installs of
        # one library covers few distinct versions).  Version = seeded hash
        # of (name, release_idx), so duplicate (name, version) rows occur at
        # real-estate rates and the engine's dedup-match layout has the
        # same leverage the reference's dedup pass has
        # (package_scan.py:1794-1816).
        key_hi, key_lo = _pool_versions(rng, name_id, name_catalog, seed)
    else:
        _, (key_hi, key_lo) = _triple_keys(rng, n_packages)
    pkg_flags = np.full(n_packages, PF_ENCODABLE, dtype=np.uint8)

    # ── synthetic advisory arena ──────────────────────────────────────────
    if arena_windows is not None:
        arena = _osv_shaped_arena(arena_windows, name_catalog, seed)
        return SyntheticEstate(
            n_agents=n_agents, n_servers=n_servers, n_creds=n_creds,
            n_tools=n_tools, n_packages=n_packages,
            agent_base=agent_base, server_base=server_base, cred_base=cred_base,
            tool_base=tool_base, pkg_base=pkg_base, num_nodes=num_nodes,
            edge_src=edge_src, edge_dst=edge_dst, edge_type=edge_type,
            pkg_name_id=name_id.astype(np.uint64), pkg_key_hi=key_hi,
            pkg_key_lo=key_lo, pkg_flags=pkg_flags,
            cred_is_db=(rng.random(n_creds) < 0.3).astype(np.uint8),
            tool_is_db=(rng.random(n_tools) < 0.2).astype(np.uint8),
            arena=arena,
            seed=seed,
        )
    n_vuln_names = max(1, int(name_catalog * vulnerable_name_fraction))
    vuln_names = rng.choice(name_catalog, n_vuln_names, replace=False).astype(np.uint64)
    W = n_vuln_names * windows_per_name
    w_names = np.repeat(vuln_names, windows_per_name)
    # introduced in the lower half of the version space, fixed above it: a
    # package in a window iff introduced <= key < fixed
    i1 = rng.integers(0, 10, W, dtype=np.int64)
    i2 = rng.integers(0, 30, W, dtype=np.int64)
    i3 = rng.integers(0, 60, W, dtype=np.int64)
    span = rng.integers(1, 8, W, dtype=np.int64)
    f1 = i1 + span
    f2 = rng.integers(0, 30, W, dtype=np.int64)
    f3 = rng.integers(0, 60, W, dtype=np.int64)
    zeros = np.zeros(W, dtype=np.int64)
    rank = np.full(W, RELEASE_RANK, dtype=np.int64)
    ihi, ilo = pack_batch(zeros, i1, i2, i3, zeros, rank, zeros)
    fhi, flo = pack_batch(zeros, f1, f2, f3, zeros, rank, zeros)
    wflags = np.full(W, WF_HAS_INTRO | WF_HAS_FIXED, dtype=np.uint8)
    sev = rng.choice([5, 4, 3, 2], size=W, p=[0.1, 0.3, 0.4, 0.2]).astype(np.uint8)
    cvss = (rng.random(W) * 10).astype(np.float32)
    epss = np.where(rng.random(W) < 0.7, rng.random(W) ** 2, -1.0).astype(np.float32)
    kev = (rng.random(W) < 0.02).astype(np.uint8)
    # CWE-impact mix (codes: 0 code-exec .. 7 client-side, 8 unknown)
    impact = rng.choice(
        np.arange(9, dtype=np.uint8), size=W,
        p=[0.35, 0.08, 0.10, 0.04, 0.08, 0.12, 0.10, 0.05, 0.08],
    ).astype(np.uint8)

    arena = build_arena_from_columns(
        w_names.astype(np.uint64), ihi, ilo, fhi, flo,
        np.zeros(W, dtype=np.uint64), np.zeros(W, dtype=np.uint64),
        wflags, sev, cvss, epss, kev, impact=impact,
    )

    return SyntheticEstate(
        n_agents=n_agents, n_servers=n_servers, n_creds=n_creds, n_tools=n_tools,
        n_packages=n_packages,
        agent_base=agent_base, server_base=server_base, cred_base=cred_base,
        tool_base=tool_base, pkg_base=pkg_base, num_nodes=num_nodes,
        edge_src=edge_src, edge_dst=edge_dst, edge_type=edge_type,
        pkg_name_id=name_id.astype(np.uint64), pkg_key_hi=key_hi, pkg_key_lo=key_lo,
        pkg_flags=pkg_flags,
        cred_is_db=(rng.random(n_creds) < 0.3).astype(np.uint8),
        tool_is_db=(rng.random(n_tools) < 0.2).astype(np.uint8),
        arena=arena,
        seed=seed,
    )
