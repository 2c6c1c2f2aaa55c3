"""expand_dedup_matches: randomized oracle equivalence (CPU torch).

The dedup-expand produces the canonical sorted (pkg<<32|win) stream without
an F-sized 64-bit sort; this compares it bit-for-bit against the brute-force
construction over random layouts."""

from __future__ import annotations

import numpy as np
import pytest
import torch

from agentbom_amd.graph.gpu_engine import expand_dedup_matches


def _random_layout(rng, P, U):
    """Random run structure: P packages partitioned into U runs."""
    row_of_pkg = torch.from_numpy(rng.integers(0, U, P)).to(torch.int64)
    perm2 = torch.argsort(row_of_pkg, stable=True)
    sorted_rows = row_of_pkg[perm2]
    counts = torch.bincount(sorted_rows, minlength=U)
    run_off = torch.cat([torch.zeros(1, dtype=torch.int64),
                         torch.cumsum(counts, 0)])
    return {"run_off": run_off, "perm2": perm2}


def _oracle(layout, su, sw):
    run_off, perm2 = layout["run_off"], layout["perm2"]
    pairs = []
    for r, w in zip(su.tolist(), sw.tolist()):
        for k in range(int(run_off[r]), int(run_off[r + 1])):
            pairs.append((int(perm2[k]) << 32) | w)
    pairs.sort()
    pkg = torch.tensor([p >> 32 for p in pairs], dtype=torch.int64)
    win = torch.tensor([p & 0xFFFFFFFF for p in pairs], dtype=torch.int64)
    return pkg, win


@pytest.mark.parametrize("seed", [0, 1, 2, 3])
def test_matches_oracle(seed):
    rng = np.random.default_rng(seed)
    P, U = 500, 60
    layout = _random_layout(rng, P, U)
    # random (row, window) matches incl. duplicate rows with several windows
    m = 80
    su = torch.from_numpy(rng.integers(0, U, m)).to(torch.int64)
    sw = torch.from_numpy(rng.integers(0, 1000, m)).to(torch.int64)
    # distinct (row, win) pairs only — the kernel emits each pair once
    packed = torch.unique((su << 32) | sw)
    su, sw = packed >> 32, packed & 0xFFFFFFFF
    pkg, win = expand_dedup_matches(torch, layout, su, sw)
    opkg, owin = _oracle(layout, su, sw)
    assert torch.equal(pkg, opkg)
    assert torch.equal(win, owin)
    # canonical invariant: packed stream strictly increasing
    if pkg.numel():
        ps = (pkg << 32) | win
        assert bool((ps[1:] > ps[:-1]).all())


def test_empty():
    layout = _random_layout(np.random.default_rng(0), 10, 3)
    e = torch.empty(0, dtype=torch.int64)
    pkg, win = expand_dedup_matches(torch, layout, e, e.clone())
    assert pkg.numel() == 0 and win.numel() == 0


def test_empty_runs_and_multiwindow():
    # rows with zero packages (empty runs) must contribute nothing
    layout = {"run_off": torch.tensor([0, 2, 2, 5]),
              "perm2": torch.tensor([4, 1, 0, 3, 2])}
    su = torch.tensor([0, 1, 2, 2])
    sw = torch.tensor([7, 9, 3, 8])
    pkg, win = expand_dedup_matches(torch, layout, su, sw)
    opkg, owin = _oracle(layout, su, sw)
    assert torch.equal(pkg, opkg) and torch.equal(win, owin)
    # row 1 is empty -> window 9 appears nowhere
    assert 9 not in win.tolist()
    # row 2's three packages each carry both windows 3 and 8, ascending
    assert win.tolist().count(3) == 3 and win.tolist().count(8) == 3
