"""Distributed frontier-exchange BFS: 2-process gloo test (CPU).

Validates the partitioned traversal (parallel/dist_bfs.py) against a
single-process BFS over the union graph — hop distances must be identical
for every owned node on every rank.  The same code path runs over RCCL on
the 8-GPU node (backend "nccl"); only the exchange primitive differs.
"""

from __future__ import annotations

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from agentbom_amd.ops import cpu_ref
from agentbom_amd.parallel.partition import build_global_csr, generate_shard

WORLD = 2
ESTATE_KW = dict(n_agents=40, n_servers=150, n_packages=1500, name_catalog=400)


def _union_graph(world: int, seed: int = 99):
    srcs, dsts, ets = [], [], []
    stride = num_global = None
    agents = []
    for r in range(world):
        est, edges = generate_shard(r, world, cross_fraction=0.1, seed=seed, **ESTATE_KW)
        srcs.append(edges["src"])
        dsts.append(edges["dst"])
        ets.append(edges["etype"])
        stride = edges["stride"]
        num_global = edges["num_global"]
        agents.append(np.arange(est.n_agents) + r * stride)
    src = np.concatenate(srcs)
    dst = np.concatenate(dsts)
    et = np.concatenate(ets)
    order = np.argsort(src, kind="stable")
    counts = np.bincount(src, minlength=num_global)
    row_off = np.zeros(num_global + 1, dtype=np.int64)
    np.cumsum(counts, out=row_off[1:])
    return row_off, dst[order], et[order], np.concatenate(agents), stride, num_global


def _worker(rank: int, world: int, port: int, result_dir: str):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist_mod

    dist_mod.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from agentbom_amd.parallel.dist_bfs import distributed_reach

        est, edges = generate_shard(rank, world, cross_fraction=0.1, seed=99, **ESTATE_KW)
        csr = build_global_csr(edges, torch.device("cpu"))
        sources = torch.arange(est.n_agents, dtype=torch.int32) + rank * edges["stride"]
        dist = distributed_reach(
            csr, sources, edges["num_global"], edges["stride"], etype=csr["etype"],
        )
        own = dist[rank * edges["stride"]: (rank + 1) * edges["stride"]]
        np.save(os.path.join(result_dir, f"dist_{rank}.npy"), own.numpy())
    finally:
        dist_mod.destroy_process_group()


@pytest.mark.timeout(180)
def test_distributed_reach_matches_union(tmp_path):
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]

    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, WORLD, port, str(tmp_path))) for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=150)
        assert p.exitcode == 0

    row_off, col, et, agents, stride, num_global = _union_graph(WORLD)
    expected = cpu_ref.bfs(row_off, col, agents, num_global, etype=et)

    for r in range(WORLD):
        got = np.load(tmp_path / f"dist_{r}.npy").view(np.uint32)
        exp = expected[r * stride: (r + 1) * stride]
        assert np.array_equal(got, exp), f"rank {r} dist mismatch"
