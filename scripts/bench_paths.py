#!/usr/bin/env python3
"""Attack/exposure-path DP latency on the estate.

Times EstateEngine.attack_paths end-to-end (relax kernels + top-k +
reconstruction) and the relax-kernel portion alone, on the 10M-pkg estate.
Reference comparator: /v1/graph/paths p50 1,625.6 ms at 10.5k nodes
(BASELINE.md).  Prints one JSON line.
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
    ap.add_argument("--packages", type=int, default=10_000_000)
    ap.add_argument("--arena-windows", type=int, default=2_000_000)
    ap.add_argument("--iters", type=int, default=11)
    ap.add_argument("--k", type=int, default=100)
    ap.add_argument("--max-depth", type=int, default=6)
    args = ap.parse_args()

    import torch

    assert torch.cuda.is_available(), "requires a GPU"
    from agentbom_amd.graph.gpu_engine import EstateEngine
    from agentbom_amd.scan.synth import generate_estate

    est = generate_estate(
        n_agents=max(1000, args.packages // 100),
        n_servers=max(5000, args.packages // 20),
        n_packages=args.packages,
        name_catalog=max(10_000, args.packages // 10),
        seed=1234,
        arena_windows=args.arena_windows or None,
    )
    eng = EstateEngine(est, device="cuda")
    res = eng.step()
    torch.cuda.synchronize()

    # warm (builds the lateral path-edge cache)
    hits = eng.attack_paths(step_res=res, k=args.k, max_depth=args.max_depth)
    torch.cuda.synchronize()

    lat = []
    for _ in range(args.iters):
        t0 = time.perf_counter()
        hits = eng.attack_paths(step_res=res, k=args.k, max_depth=args.max_depth)
        torch.cuda.synchronize()
        lat.append((time.perf_counter() - t0) * 1000.0)

    print(json.dumps({
        "what": "attack_paths_dp_10M_estate",
        "nodes": est.num_nodes,
        "edges_with_lateral": int(est.num_edges + (est.edge_type == 0).sum()),
        "k": args.k, "max_depth": args.max_depth,
        "paths_found": len(hits),
        "top_score": hits[0].score if hits else None,
        "p50_ms": statistics.median(lat),
        "min_ms": min(lat),
        "iters": args.iters,
        "ref_comparator_ms": 1625.6,
    }))


if __name__ == "__main__":
    main()
