#!/usr/bin/env python3
"""Flagship benchmark: findings/sec on the 10M-pkg-per-GPU synthetic estate.

One step = one full findings pass over the estate:
  bulk version-range match (HIP kernel, packages vs advisory arena)
  -> dependency-reach BFS (HIP frontier kernels; multi-GPU: ONE global
     estate hash-partitioned by node id with padded RCCL all-to-all
     frontier exchange over xGMI)
  -> blast-radius joins (GPU joins: distinct agents/creds/tools per
     finding, CWE-impact filtered; multi-GPU: owner-computes two-shuffle
     distributed joins — cross-shard reach included)
  -> risk scoring (HIP kernel, reference formula)
  -> ranking (sort by score desc).

Multi-GPU is weak scaling of ONE estate: the global estate is N x the
per-GPU size, every rank owns the node ids congruent to its rank
(provably equal to a single-engine run on the same estate —
tests/test_dist_engine.py).  value = aggregate findings/sec over all N
GPUs.  After the timed region, p50 blast-radius query latency (the second
BASELINE metric) is measured: single-GPU via hipGraph replay; multi-GPU
via collective distributed bounded BFS queries.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--packages P]
For N>1 the driver launches via torch.distributed.run (one rank per GPU).
"""

from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import time


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--packages", type=int, default=10_000_000,
                    help="packages per GPU shard (default: the 10M headline)")
    ap.add_argument("--agents", type=int, default=100_000)
    ap.add_argument("--servers", type=int, default=500_000)
    ap.add_argument("--name-catalog", type=int, default=1_000_000)
    ap.add_argument("--arena-windows", type=int, default=2_000_000,
                    help="advisory windows per GPU-worth of catalog (OSV-dump-"
                         "shaped: zipf branch counts, mixed window forms)")
    ap.add_argument("--queries", type=int, default=201,
                    help="blast-radius latency queries after the timed region")
    ap.add_argument("--seed", type=int, default=1234)
    ap.add_argument("--device", default="cuda")
    return ap.parse_args()


def main():
    args = parse_args()
    import torch

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(args.gpus, world)

    use_gpu = args.device.startswith("cuda")
    if use_gpu and not torch.cuda.is_available():
        print("ERROR: no GPU visible; bench requires an MI355X "
              "(use --device cpu --packages 100000 for a dev dry-run)", file=sys.stderr)
        sys.exit(2)

    distributed = world > 1
    if distributed:
        import torch.distributed as dist_mod

        backend = "nccl" if use_gpu else "gloo"
        if use_gpu:
            torch.cuda.set_device(local_rank)
        dist_mod.init_process_group(backend)
    device = f"cuda:{local_rank}" if use_gpu else "cpu"

    from agentbom_amd.scan.synth import generate_estate

    # ONE global estate, weak scaling: global sizes = per-GPU sizes x world
    estate_kw = dict(
        n_agents=args.agents * n_gpus, n_servers=args.servers * n_gpus,
        n_packages=args.packages * n_gpus, name_catalog=args.name_catalog * n_gpus,
        arena_windows=args.arena_windows * n_gpus if args.arena_windows else None,
    )

    t_gen = time.perf_counter()
    est = generate_estate(seed=args.seed, **estate_kw)
    gen_s = time.perf_counter() - t_gen

    t_build = time.perf_counter()
    if distributed:
        from agentbom_amd.parallel.dist_engine import DistEstateEngine

        engine = DistEstateEngine(est, rank, world, device=device)
    else:
        from agentbom_amd.graph.gpu_engine import EstateEngine

        engine = EstateEngine(est, device=device)
    if use_gpu:
        torch.cuda.synchronize()
    build_s = time.perf_counter() - t_build

    def one_step():
        return engine.step()

    # warmup
    for _ in range(args.warmup):
        res = one_step()
    n_findings_local = res["n_findings"] if args.warmup else None

    if distributed:
        import torch.distributed as dist_mod

        dist_mod.barrier()
    if use_gpu:
        torch.cuda.synchronize()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        res = one_step()
    if use_gpu:
        torch.cuda.synchronize()
    if distributed:
        dist_mod.barrier()
    elapsed = time.perf_counter() - t0

    n_findings_local = res["n_findings"]

    # aggregate: MAX elapsed over ranks, SUM findings over ranks
    if distributed:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_gpu else "cpu")
        dist_mod.all_reduce(t, op=dist_mod.ReduceOp.MAX)
        elapsed = float(t.item())
        f = torch.tensor([n_findings_local], dtype=torch.float64,
                         device=device if use_gpu else "cpu")
        dist_mod.all_reduce(f, op=dist_mod.ReduceOp.SUM)
        total_findings_per_step = int(f.item())
    else:
        total_findings_per_step = n_findings_local

    ms_per_step = elapsed / args.steps * 1000.0
    findings_per_sec = total_findings_per_step * args.steps / elapsed

    # p50 blast-radius query latency
    p50_ms = None
    if distributed and args.queries > 0:
        # collective distributed bounded-BFS queries: every rank participates,
        # rank 0 reports the median wall time per query
        qidx = torch.arange(args.queries, dtype=torch.int64) * 997 % est.n_packages
        qnodes = (qidx + est.pkg_base).tolist()
        engine.blast_radius_query(qnodes[0], max_hops=4)  # warm
        if use_gpu:
            torch.cuda.synchronize()
        dist_mod.barrier()
        lat = []
        for qn in qnodes:
            tq = time.perf_counter()
            engine.blast_radius_query(qn, max_hops=4)
            if use_gpu:
                torch.cuda.synchronize()
            lat.append((time.perf_counter() - tq) * 1000.0)
        p50_ms = statistics.median(lat)
    elif rank == 0 and args.queries > 0:
        rng_idx = torch.arange(args.queries, dtype=torch.int64) * 997 % est.n_packages
        nodes = (rng_idx + est.pkg_base).to(device)
        lat = []
        # serving mode: the query is captured once into a hipGraph and each
        # request replays it (no per-query launch construction)
        if use_gpu:
            engine.enable_query_graph(batch=1, max_hops=4)
        # warm the query path
        engine.blast_radius_query(nodes[:1], max_hops=4)
        if use_gpu:
            torch.cuda.synchronize()
        for i in range(args.queries):
            tq = time.perf_counter()
            engine.blast_radius_query(nodes[i:i + 1], max_hops=4)
            if use_gpu:
                torch.cuda.synchronize()
            lat.append((time.perf_counter() - tq) * 1000.0)
        p50_ms = statistics.median(lat)

    if distributed:
        dist_mod.barrier()

    if rank == 0:
        result = {
            "metric": "findings/sec",
            "value": findings_per_sec,
            "unit": "findings/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "estate-blast-pipeline",
                "global_batch": total_findings_per_step,
                "seq_len": args.packages,
                "parallelism": f"hashpart{n_gpus}" if n_gpus > 1 else "single",
                "estate": (f"synthetic ONE global {args.packages*n_gpus/1e6:.0f}M-pkg/"
                           f"{est.num_nodes/1e6:.1f}M-node estate, "
                           f"hash-partitioned {n_gpus}-way ({args.packages/1e6:.0f}M pkg/GPU)"),
                "nodes_global": est.num_nodes,
                "edges_global": est.num_edges,
                "arena_windows": est.arena.num_windows,
                "findings_per_step": total_findings_per_step,
                "p50_blast_query_ms": p50_ms,
                "estate_gen_s": round(gen_s, 2),
                "engine_build_s": round(build_s, 2),
            },
        }
        print(json.dumps(result))

    if distributed:
        dist_mod.destroy_process_group()


if __name__ == "__main__":
    main()
