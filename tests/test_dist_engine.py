"""Distributed pipeline == single engine, on ONE hash-partitioned estate.

The round-2 deliverable (VERDICT r1 'Next round' #1): the multi-process
run partitions a single seeded estate (no per-rank independent estates)
and every stage — match, reach BFS, blast joins, scoring — must produce
results provably equal to a single-engine run on the unpartitioned estate.

Runs the REAL exchange path (padded equal-split all_to_all_single) over
gloo at world 2 and 4; the identical code runs over RCCL on the 8-GPU
node.  Merged per-rank findings must match the single engine bit-for-bit:
same (pkg, window) set, same distinct agent/cred/tool counts per finding,
same risk scores.
"""

from __future__ import annotations

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from agentbom_amd.graph.gpu_engine import EstateEngine
from agentbom_amd.scan.synth import generate_estate

ESTATE_KW = dict(n_agents=60, n_servers=240, n_packages=4000, name_catalog=800,
                 seed=424, arena_windows=1500)


def _single_reference():
    est = generate_estate(**ESTATE_KW)
    eng = EstateEngine(est, device="cpu")
    res = eng.step()
    return est, res


def _worker(rank: int, world: int, port: int, result_dir: str):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist_mod

    dist_mod.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from agentbom_amd.parallel.dist_engine import DistEstateEngine

        est = generate_estate(**ESTATE_KW)
        eng = DistEstateEngine(est, rank, world, device="cpu")
        res = eng.step()
        np.savez(
            os.path.join(result_dir, f"r{rank}.npz"),
            pkg_idx=res["pkg_idx"].numpy(),
            win_idx=res["win_idx"].numpy(),
            scores=res["scores"].numpy(),
            n_agents=res["n_agents"].numpy(),
            n_creds=res["n_creds"].numpy(),
            n_tools=res["n_tools"].numpy(),
        )
        # second step must be identical (workspace reuse is clean)
        res2 = eng.step()
        assert torch.equal(res["pkg_idx"], res2["pkg_idx"])
        assert torch.equal(res["scores"], res2["scores"])
        # distributed bounded queries (collective: every rank issues the
        # same query sequence; each saves its own authoritative slice)
        for qi in range(world):
            q = eng.blast_radius_query(est.pkg_base + qi, max_hops=4)
            np.save(os.path.join(result_dir, f"q{qi}_r{rank}.npy"), q.numpy())
    finally:
        dist_mod.destroy_process_group()


def _run_world(world: int, tmp_path):
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, world, port, str(tmp_path)))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0


@pytest.mark.timeout(360)
@pytest.mark.parametrize("world", [2, 3, 4])
def test_dist_pipeline_equals_single_engine(world, tmp_path):
    _run_world(world, tmp_path)
    est, ref = _single_reference()

    # merge per-rank findings
    parts = [np.load(tmp_path / f"r{r}.npz") for r in range(world)]
    pkg = np.concatenate([p["pkg_idx"] for p in parts])
    win = np.concatenate([p["win_idx"] for p in parts])
    scores = np.concatenate([p["scores"] for p in parts])
    n_agents = np.concatenate([p["n_agents"] for p in parts])
    n_creds = np.concatenate([p["n_creds"] for p in parts])
    n_tools = np.concatenate([p["n_tools"] for p in parts])

    order = np.lexsort((win, pkg))
    pkg, win = pkg[order], win[order]
    scores, n_agents = scores[order], n_agents[order]
    n_creds, n_tools = n_creds[order], n_tools[order]

    rp = ref["pkg_idx"].numpy()
    rw = ref["win_idx"].numpy()
    rorder = np.lexsort((rw, rp))

    assert np.array_equal(pkg, rp[rorder]), "finding (pkg) sets differ"
    assert np.array_equal(win, rw[rorder]), "finding (window) sets differ"
    assert np.array_equal(n_agents, ref["n_agents"].numpy()[rorder]), \
        "distinct agent counts differ (cross-shard reach lost?)"
    assert np.array_equal(n_creds, ref["n_creds"].numpy()[rorder])
    assert np.array_equal(n_tools, ref["n_tools"].numpy()[rorder])
    assert np.array_equal(scores, ref["scores"].numpy()[rorder]), "risk scores differ"

    # distributed bounded query equals a single-engine reverse BFS
    from agentbom_amd.ops import cpu_ref

    eng = EstateEngine(est, device="cpu")
    for qi in range(world):
        exp = cpu_ref.bfs(
            eng.rev["row_off"].numpy(), eng.rev["col"].numpy(),
            np.array([est.pkg_base + qi], dtype=np.int64), est.num_nodes,
            max_levels=4,
        )
        for r in range(world):
            got = np.load(tmp_path / f"q{qi}_r{r}.npy").view(np.uint32)
            own = np.arange(est.num_nodes) % world == r
            assert np.array_equal(got[own], exp[own]), \
                f"query {qi} dist mismatch on rank {r}"


def test_dist_engine_world1_equals_single():
    """Degenerate world=1: DistEstateEngine must equal EstateEngine exactly
    (no process group needed — also the single-GPU entry for -m gpu runs)."""
    from agentbom_amd.parallel.dist_engine import DistEstateEngine

    est = generate_estate(**ESTATE_KW)
    ref = EstateEngine(est, device="cpu").step()
    got = DistEstateEngine(est, 0, 1, device="cpu").step()
    assert torch.equal(got["pkg_idx"], ref["pkg_idx"])
    assert torch.equal(got["win_idx"], ref["win_idx"])
    assert torch.equal(got["n_agents"], ref["n_agents"])
    assert torch.equal(got["n_creds"], ref["n_creds"])
    assert torch.equal(got["n_tools"], ref["n_tools"])
    assert torch.equal(got["scores"], ref["scores"])
