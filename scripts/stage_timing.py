#!/usr/bin/env python3
"""Per-stage wall timing of the findings pipeline (cuda-event bracketed)."""

from __future__ import annotations

import argparse
import json
import pathlib
import sys
import time

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--packages", type=int, default=10_000_000)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--arena-windows", type=int, default=0)
    args = ap.parse_args()

    import torch

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
    eng.step()
    torch.cuda.synchronize()

    def timed(fn):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            out = fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / args.iters * 1000, out

    t_match, (pkg_idx, win_idx) = timed(eng.match)
    # dedup-match path (what step() actually runs)
    t_match_dd = None
    if eng.match_dedup is not None:
        from agentbom_amd.ops import native

        dd = eng.match_dedup

        def dd_match():
            sp, sw = native.match(
                dd["u_gk"], dd["u_hi"], dd["u_lo"], dd["u_flags"],
                eng.arena["group_keys"], eng.arena["group_off"],
                eng.arena["windows"], pkg_win_range=dd["u_ranges"])
            from agentbom_amd.graph.gpu_engine import expand_dedup_matches

            return expand_dedup_matches(eng.torch, dd, sp, sw)

        t_match_dd, _ = timed(dd_match)
    t_reach, dist = timed(eng.dependency_reach)
    pkg_nodes = pkg_idx + est.pkg_base
    t_counts, counts = timed(lambda: eng.blast_counts(pkg_nodes))
    t_step, res = timed(eng.step)
    residual = t_step - t_match - t_reach - t_counts
    print(json.dumps({
        "packages": args.packages,
        "findings": int(pkg_idx.numel()),
        "match_ms": round(t_match, 3),
        "match_dedup_ms": round(t_match_dd, 3) if t_match_dd is not None else None,
        "unique_rows": int(eng.match_dedup["u_gk"].numel()) if eng.match_dedup else None,
        "reach_bfs_ms": round(t_reach, 3),
        "blast_counts_ms": round(t_counts, 3),
        "full_step_ms": round(t_step, 3),
        "residual_ms": round(residual, 3),
    }, indent=2))


if __name__ == "__main__":
    main()
