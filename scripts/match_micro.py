#!/usr/bin/env python3
"""Microbench: bulk version-range match + blast-radius query latency.

BASELINE config 2 evidence: "Bulk OSV/GHSA version-range match over
1M-package synthetic lockfile, 1x MI355X" — times the match kernel alone
(packages/s) at several scales, plus individual bounded blast-radius query
latency percentiles.  Prints one JSON line per config.
"""

from __future__ import annotations

import argparse
import json
import pathlib
import statistics
import sys
import time

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--sizes", default="1000000,10000000,50000000")
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--queries", type=int, default=501)
    ap.add_argument("--arena-windows", type=int, default=0,
                    help="OSV-dump-shaped arena size (0 = legacy uniform arena)")
    args = ap.parse_args()

    import torch

    assert torch.cuda.is_available(), "requires a GPU"
    from agentbom_amd.graph.gpu_engine import EstateEngine
    from agentbom_amd.ops import native
    from agentbom_amd.scan.synth import generate_estate

    for size in [int(s) for s in args.sizes.split(",")]:
        est = generate_estate(
            n_agents=max(1000, size // 100),
            n_servers=max(5000, size // 20),
            n_packages=size,
            name_catalog=max(10_000, size // 10),
            seed=4242,
            arena_windows=args.arena_windows or None,
        )
        eng = EstateEngine(est, device="cuda")
        torch.cuda.synchronize()

        # match-only timing
        for _ in range(3):
            eng.match()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            pkg_idx, _win = eng.match()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        print(json.dumps({
            "bench": "bulk_match",
            "packages": size,
            "arena_windows": est.arena.num_windows,
            "ms": round(dt * 1000, 3),
            "packages_per_sec": round(size / dt),
            "matches": int(pkg_idx.numel()),
        }), flush=True)

        # blast-radius query latency (individual launches)
        idx = torch.arange(args.queries, dtype=torch.int64) * 1999 % est.n_packages
        nodes = (idx + est.pkg_base).to("cuda")
        eng.blast_radius_query(nodes[:1])
        torch.cuda.synchronize()
        lat = []
        for i in range(args.queries):
            tq = time.perf_counter()
            eng.blast_radius_query(nodes[i:i + 1], max_hops=4)
            torch.cuda.synchronize()
            lat.append((time.perf_counter() - tq) * 1000)
        lat.sort()
        print(json.dumps({
            "bench": "blast_query_latency",
            "nodes": est.num_nodes,
            "queries": args.queries,
            "p50_ms": round(statistics.median(lat), 4),
            "p95_ms": round(lat[int(len(lat) * 0.95)], 4),
            "p99_ms": round(lat[int(len(lat) * 0.99)], 4),
        }), flush=True)

        del eng
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
