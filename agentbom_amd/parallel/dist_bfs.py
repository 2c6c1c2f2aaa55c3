"""Hash-partitioned estate BFS with padded all-to-all frontier exchange.

Multi-GPU (one process per GPU, ``torch.distributed`` backend "nccl" = RCCL
on ROCm) traversal of ONE estate CSR hash-partitioned by node id across
ranks (owner = id % world, parallel/partition.py).  xGMI is point-to-point
(7 links x ~153 GB/s per GPU), so the per-level frontier exchange uses
direct per-peer sends via ONE padded equal-split ``all_to_all_single``
(parallel/exchange.py) — counts ride in the buffer, so there is no
per-level count round-trip through the host (the old
path did two D2H syncs per level before the payload exchange).

Per level:
  local expand (HIP kernel on GPU / cpu_ref on CPU; claims local AND
  remote ids — a remote claim is the "queued for send" de-dup marker)
  -> padded all-to-all of remote claims routed by owner
  -> ONE all_reduce of [global_frontier_size, overflow_flag]
     (termination and exchange-overflow decided together, one host sync)
  -> receiver re-claims against its authoritative dist
  -> next frontier = local claims + newly-claimed received ids.

Level-synchronous, so hop counts are exact regardless of arrival order.
On (rare) cap overflow every rank retries the level's exchange with a
collectively grown cap — the claim markers make the retry idempotent.

The same code path runs under gloo at world 2-4 in the CPU tests
(tests/test_dist_engine.py) and under RCCL on the 8-GPU node.
"""

from __future__ import annotations

from agentbom_amd.parallel.exchange import all_to_all_padded

UNVISITED = 0xFFFFFFFF


def _expand_level(csr, frontier, dist, level, etype, allowed_mask, use_gpu, ws):
    if use_gpu:
        from agentbom_amd.ops import native

        frontier = frontier.to(dist.device, dtype=frontier.dtype)
        # dense-frontier handoff, same threshold as the single-GPU engine:
        # when the frontier's outgoing degree is a big share of the shard's
        # edges, the one-thread-per-edge pass with coalesced src/col streams
        # beats per-vertex neighbor loops.  Claims stay materialized (the
        # RCCL exchange needs the list).
        if csr.get("src") is not None and frontier.numel():
            import torch

            f64 = frontier.to(torch.int64)
            deg = (csr["row_off"][f64 + 1] - csr["row_off"][f64]).sum()
            if int(deg) > csr["col"].numel() // 8:
                return native.bfs_level_edges(
                    csr["row_off"], csr["col"], csr["src"], level - 1, dist,
                    etype=etype, allowed_mask=allowed_mask, workspace=ws,
                )
        return native.bfs_level(
            csr["row_off"], csr["col"], frontier, dist, level,
            etype=etype, allowed_mask=allowed_mask, workspace=ws,
        )
    import torch

    from agentbom_amd.ops import cpu_ref

    nxt = cpu_ref.bfs_level(
        csr["row_off"].numpy(), csr["col"].numpy(), frontier.numpy(),
        dist.numpy().view("uint32"), level,
        etype=etype.numpy() if etype is not None else None, allowed_mask=allowed_mask,
    )
    return torch.from_numpy(nxt).to(torch.int32)


def distributed_reach(
    csr: dict,
    sources,
    num_global: int,
    world: int,
    rank: int,
    etype=None,
    allowed_mask: int = 0xFFFFFFFF,
    max_levels: int = 64,
    group=None,
    cap: int = 0,
    workspace=None,
):
    """Multi-source BFS over the hash-partitioned estate.

    ``csr`` has global-indexed ``row_off`` (num_global+1) and global
    ``col`` holding only this rank's owned-source edges; ``sources`` are
    owned global ids.  Returns the global-size u32 dist tensor
    (authoritative for owned ids; remote entries are send markers).
    ``cap`` seeds the exchange block size (auto-grown collectively).
    """
    import torch
    import torch.distributed as dist_mod

    from agentbom_amd.utils import config as _cfg

    device = csr["row_off"].device
    use_gpu = device.type == "cuda"
    if cap <= 0 and _cfg.DIST_EXCHANGE_CAP > 0:
        cap = _cfg.DIST_EXCHANGE_CAP
    if max_levels == 64:
        max_levels = _cfg.DIST_MAX_BFS_LEVELS

    ws = workspace if workspace is not None else {}
    dist = ws.get("ddist")
    if dist is None or dist.numel() < num_global:
        dist = torch.empty(num_global, dtype=torch.int32, device=device)
        ws["ddist"] = dist
    dist.fill_(-1)  # -1 int32 bit pattern == UNVISITED u32
    frontier = sources.to(device=device, dtype=torch.int32)
    dist[frontier.to(torch.int64)] = 0

    if cap <= 0:
        cap = max(4096, num_global // max(world * world, 1) + 2)

    stat = torch.zeros(2, dtype=torch.int64, device=device if use_gpu else "cpu")
    level = 0
    while level < max_levels:
        level += 1
        nxt = _expand_level(csr, frontier, dist, level, etype, allowed_mask, use_gpu, ws)

        if world > 1:
            ids64 = nxt.to(torch.int64)
            owner = ids64 % world
            local_next = nxt[owner == rank]
            remote = nxt[owner != rank]
            rowner = owner[owner != rank]
            while True:
                recv, overflow = all_to_all_padded(
                    remote, rowner, world, cap, group=group, ws=ws)
                # fold overflow into the termination all_reduce: one host
                # sync decides both (and keeps the retry collective)
                stat[0] = local_next.numel() + remote.numel()
                stat[1] = overflow.to(torch.int64)
                dist_mod.all_reduce(stat, group=group)
                g_frontier, g_overflow = int(stat[0].item()), int(stat[1].item())
                if not g_overflow:
                    break
                cap *= 4  # identical growth on every rank
            if recv.numel():
                r64 = torch.unique(recv.to(torch.int64))
                fresh = r64[dist[r64] == -1]
                dist[fresh] = level
                frontier = torch.cat([local_next, fresh.to(torch.int32)])
            else:
                frontier = local_next
            # g_frontier counts claims BEFORE receiver-side rejection, so a
            # level whose claims were all rejected runs one extra (empty,
            # cheap) level before the zero shows up — still exact.
            if g_frontier == 0:
                break
        else:
            frontier = nxt
            if frontier.numel() == 0:
                break
    return dist
