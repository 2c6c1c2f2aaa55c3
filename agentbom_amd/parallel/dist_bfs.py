"""Hash-partitioned estate BFS with all-to-all frontier exchange.

Multi-GPU (one process per GPU, ``torch.distributed`` backend "nccl" = RCCL
on ROCm) traversal of an estate CSR partitioned by node-id range across
ranks.  xGMI is point-to-point (7 links x ~153 GB/s per GPU), so the
per-level frontier exchange uses ``all_to_all_single`` — direct per-peer
sends that can drive all links concurrently — never a ring pattern
(SURVEY.md §5 'Distributed communication backend').

Partitioning model:
- world of W ranks, each owning ``stride`` consecutive global node ids
  (rank r owns [r*stride, (r+1)*stride));
- each rank holds the CSR of edges whose SOURCE it owns, with rows indexed
  by GLOBAL id (empty rows for remote ids) and cols holding global ids;
- ``dist`` is global-size per rank: authoritative for owned nodes; for
  remote nodes a claim is the "already queued for send" marker, so each
  remote vertex is sent at most once per rank.

Per level: local expand (HIP kernel on GPU / cpu_ref on CPU) -> split the
claimed frontier by owner -> exchange counts then ids (all_to_all_single)
-> receiver re-claims against its authoritative dist -> next frontier =
local claims + newly-claimed received ids.  Level-synchronous, so hop
counts are exact regardless of arrival order.  Termination: all_reduce of
the global frontier size.

CPU tests run this with the gloo backend at world_size 2
(tests/test_dist_bfs.py); the driver's 8-GPU scaling bench runs it over
RCCL.
"""

from __future__ import annotations

from typing import Optional

UNVISITED = 0xFFFFFFFF


def _expand_level(csr, frontier, dist, level, etype, allowed_mask, use_gpu, ws):
    if use_gpu:
        from agentbom_amd.ops import native

        frontier = frontier.to(dist.device, dtype=frontier.dtype)
        # dense-frontier handoff, same threshold as the single-GPU engine:
        # when the frontier's outgoing degree is a big share of the shard's
        # edges, the one-thread-per-edge pass with coalesced src/col streams
        # beats per-vertex neighbor loops.  Claims stay materialized (the
        # RCCL exchange needs the list), so this is the build_frontier=1
        # edge kernel, not the dist-driven one.
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


def _exchange_frontier(nxt_sorted, owner_sorted, send_counts, rank, world, use_gpu, group):
    """Per-peer frontier exchange.

    On RCCL: ``all_to_all_single`` (direct per-peer sends over xGMI links).
    On gloo (CPU tests): emulated via all_gather, since gloo lacks
    all_to_all — semantics identical, only for the test path.
    """
    import torch
    import torch.distributed as dist_mod

    # zero the self slot: our own claims are already in local_next
    send_counts = send_counts.clone()
    self_slot = torch.nonzero(owner_sorted == rank).flatten()
    keep = owner_sorted != rank
    payload = nxt_sorted[keep].contiguous()
    counts = send_counts
    counts[rank] = 0

    backend = dist_mod.get_backend(group)
    if backend == "nccl":
        counts_dev = counts.to(nxt_sorted.device)
        recv_counts = torch.zeros_like(counts_dev)
        dist_mod.all_to_all_single(recv_counts, counts_dev, group=group)
        recv_cpu = recv_counts.to("cpu")
        send_cpu = counts.to("cpu")
        recv_buf = torch.empty(int(recv_cpu.sum().item()), dtype=payload.dtype,
                               device=payload.device)
        dist_mod.all_to_all_single(
            recv_buf, payload,
            output_split_sizes=recv_cpu.tolist(),
            input_split_sizes=send_cpu.tolist(),
            group=group,
        )
        return recv_buf
    # gloo emulation: all_gather of (owner, id) pairs, filter to mine
    sizes = [torch.zeros(world, dtype=torch.int64) for _ in range(world)]
    dist_mod.all_gather(sizes, counts.to(torch.int64).cpu(), group=group)
    maxlen = max(int(s.sum().item()) for s in sizes) or 1
    padded = torch.full((maxlen,), -1, dtype=torch.int64)
    padded[: payload.numel()] = payload.to(torch.int64).cpu()
    owners_padded = torch.full((maxlen,), -1, dtype=torch.int64)
    owners_padded[: payload.numel()] = owner_sorted[keep].to(torch.int64).cpu()
    gathered_ids = [torch.empty(maxlen, dtype=torch.int64) for _ in range(world)]
    gathered_own = [torch.empty(maxlen, dtype=torch.int64) for _ in range(world)]
    dist_mod.all_gather(gathered_ids, padded, group=group)
    dist_mod.all_gather(gathered_own, owners_padded, group=group)
    mine = []
    for r in range(world):
        if r == rank:
            continue
        ids = gathered_ids[r]
        own = gathered_own[r]
        sel = ids[own == rank]
        if sel.numel():
            mine.append(sel)
    if not mine:
        return torch.empty(0, dtype=nxt_sorted.dtype)
    return torch.cat(mine).to(nxt_sorted.dtype)


def distributed_reach(
    csr: dict,
    sources,
    num_global: int,
    stride: int,
    etype=None,
    allowed_mask: int = 0xFFFFFFFF,
    max_levels: int = 64,
    group=None,
):
    """Multi-source BFS over the partitioned estate.

    ``csr`` has global-indexed ``row_off`` (num_global+1) and global ``col``;
    ``sources`` are locally-owned global ids.  Returns the global-size dist
    tensor (authoritative for owned ids).
    """
    import torch
    import torch.distributed as dist_mod

    world = dist_mod.get_world_size(group)
    rank = dist_mod.get_rank(group)
    device = csr["row_off"].device
    use_gpu = device.type == "cuda"

    dist = torch.full((num_global,), -1, dtype=torch.int32, device=device)
    # -1 as int32 bit pattern == UNVISITED u32
    frontier = sources.to(device=device, dtype=torch.int32)
    dist[frontier.to(torch.int64)] = 0

    ws: dict = {}
    level = 0
    while level < max_levels:
        level += 1
        nxt = _expand_level(csr, frontier, dist, level, etype, allowed_mask, use_gpu, ws)

        ids64 = nxt.to(torch.int64)
        owner = torch.div(ids64, stride, rounding_mode="floor")
        order = torch.argsort(owner, stable=True)
        nxt_sorted = nxt[order]
        owner_sorted = owner[order]
        send_counts = torch.bincount(owner_sorted, minlength=world)

        # local portion of this rank's own expansion stays local; the rest is
        # exchanged peer-to-peer (self slot is sent empty).
        local_mask = owner_sorted == rank
        local_next = nxt_sorted[local_mask]
        recv_remote = _exchange_frontier(
            nxt_sorted, owner_sorted, send_counts, rank, world, use_gpu, group
        )

        if recv_remote.numel():
            r64 = torch.unique(recv_remote.to(torch.int64))
            fresh = r64[dist[r64] == -1]
            dist[fresh] = level
            frontier = torch.cat([local_next, fresh.to(torch.int32)])
        else:
            frontier = local_next

        # global termination
        sz = torch.tensor([frontier.numel()], dtype=torch.int64)
        if use_gpu:
            sz = sz.to(device)
        dist_mod.all_reduce(sz, group=group)
        if int(sz.item()) == 0:
            break
    return dist
