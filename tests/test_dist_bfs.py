"""Distributed frontier-exchange BFS over ONE hash-partitioned estate.

Validates parallel/dist_bfs.distributed_reach (padded equal-split
all-to-all, counts piggybacked — the identical code path RCCL runs on the
8-GPU node) against a single-process BFS over the unpartitioned estate:
hop distances must be identical for every owned node on every rank.  Also
unit-tests the exchange overflow/retry protocol.
"""

from __future__ import annotations

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from agentbom_amd.ops import cpu_ref
from agentbom_amd.scan.synth import generate_estate

WORLD = 2
ESTATE_KW = dict(n_agents=40, n_servers=150, n_packages=1500, name_catalog=400,
                 seed=99)


def _worker(rank: int, world: int, port: int, result_dir: str, cap: int):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist_mod

    dist_mod.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from agentbom_amd.parallel.dist_bfs import distributed_reach
        from agentbom_amd.parallel.partition import build_csr, partition_estate

        est = generate_estate(**ESTATE_KW)
        part = partition_estate(est, rank, world)
        csr = build_csr(part.fwd_src, part.fwd_dst, part.fwd_type,
                        est.num_nodes, torch.device("cpu"))
        sources = torch.from_numpy(part.own_agents).to(torch.int32)
        dist = distributed_reach(
            csr, sources, est.num_nodes, world, rank, etype=csr["etype"],
            cap=cap,
        )
        own = np.arange(est.num_nodes) % world == rank
        np.save(os.path.join(result_dir, f"dist_{rank}.npy"),
                dist.numpy()[own])
    finally:
        dist_mod.destroy_process_group()


@pytest.mark.timeout(180)
@pytest.mark.parametrize("cap", [0, 8], ids=["auto_cap", "tiny_cap_overflow_retry"])
def test_distributed_reach_matches_single(tmp_path, cap):
    """cap=8 forces the padded exchange to overflow and retry collectively
    (grown cap), exercising the recovery path end-to-end."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]

    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, WORLD, port, str(tmp_path), cap))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=150)
        assert p.exitcode == 0

    est = generate_estate(**ESTATE_KW)
    row_counts = np.bincount(est.edge_src, minlength=est.num_nodes)
    row_off = np.zeros(est.num_nodes + 1, dtype=np.int64)
    np.cumsum(row_counts, out=row_off[1:])
    order = np.argsort(est.edge_src, kind="stable")
    expected = cpu_ref.bfs(
        row_off, est.edge_dst[order], np.arange(est.n_agents), est.num_nodes,
        etype=est.edge_type[order],
    )

    for r in range(WORLD):
        got = np.load(tmp_path / f"dist_{r}.npy").view(np.uint32)
        own = np.arange(est.num_nodes) % WORLD == r
        assert np.array_equal(got, expected[own]), f"rank {r} dist mismatch"


def test_padded_exchange_world1_passthrough():
    from agentbom_amd.parallel.exchange import exchange_sized

    v = torch.arange(10, dtype=torch.int64)
    local, recv = exchange_sized(v, torch.zeros(10, dtype=torch.int64), 0, 1)
    assert torch.equal(local, v) and recv.numel() == 0
