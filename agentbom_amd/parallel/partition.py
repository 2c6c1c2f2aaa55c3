"""Estate partitioning across ranks (one shard per GPU).

Each rank generates/owns one shard of the global estate: global node ids
are ``rank * stride + local_id`` (stride = identical per-rank node count),
and a deterministic fraction of agent->server USES edges is rewired to the
next rank's shard so traversals genuinely cross xGMI (lateral movement
between sub-estates).  All ranks can compute every remote id locally, so
partition construction needs no communication.
"""

from __future__ import annotations

import numpy as np

from agentbom_amd.scan.synth import ET_USES, SyntheticEstate, generate_estate


def generate_shard(rank: int, world: int, cross_fraction: float = 0.05,
                   seed: int = 1234, **estate_kw) -> tuple[SyntheticEstate, dict]:
    """Generate this rank's shard + its globalized edge arrays.

    Returns (local_estate, global_edges) where global_edges carries
    ``src``/``dst``/``etype`` with global ids (cross edges rewired).
    """
    est = generate_estate(seed=seed + rank, **estate_kw)
    stride = est.num_nodes
    base = rank * stride
    src = est.edge_src + base
    dst = est.edge_dst + base
    et = est.edge_type.copy()

    if world > 1 and cross_fraction > 0:
        rng = np.random.default_rng(seed * 7919 + rank)
        uses = np.nonzero(et == ET_USES)[0]
        n_cross = int(len(uses) * cross_fraction)
        if n_cross:
            pick = rng.choice(uses, n_cross, replace=False)
            next_base = ((rank + 1) % world) * stride
            # same server position on the neighbour shard
            dst[pick] = (est.edge_dst[pick] - 0) + next_base
    return est, {"src": src, "dst": dst, "etype": et, "stride": stride,
                 "num_global": stride * world}


def build_global_csr(edges: dict, device) -> dict:
    """CSR over the GLOBAL row space for this rank's edges (torch)."""
    import torch

    num_global = edges["num_global"]
    src = torch.from_numpy(edges["src"]).to(device)
    dst = torch.from_numpy(edges["dst"]).to(device)
    et = torch.from_numpy(edges["etype"]).to(device)
    order = torch.argsort(src, stable=True)
    counts = torch.bincount(src, minlength=num_global)
    row_off = torch.zeros(num_global + 1, dtype=torch.int64, device=device)
    torch.cumsum(counts, 0, out=row_off[1:])
    return {
        "row_off": row_off,
        "col": dst[order].to(torch.int32),
        "etype": et[order].contiguous(),
        # col-aligned edge sources: enables the edge-centric dense-frontier
        # level expansion in dist_bfs (same layout as the single-GPU engine)
        "src": src[order].to(torch.int32),
    }
