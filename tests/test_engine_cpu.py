"""CPU pipeline tests: engine correctness at small scale (no GPU needed)."""

from __future__ import annotations

import numpy as np
import pytest
import torch

from agentbom_amd.graph.gpu_engine import EstateEngine
from agentbom_amd.scan.synth import generate_estate


@pytest.fixture(scope="module")
def estate():
    return generate_estate(n_agents=100, n_servers=500, n_packages=10_000,
                           name_catalog=2_000, seed=11)


@pytest.fixture(scope="module")
def engine(estate):
    return EstateEngine(estate, device="cpu")


def test_estate_deterministic():
    a = generate_estate(n_agents=50, n_servers=200, n_packages=2_000, seed=5)
    b = generate_estate(n_agents=50, n_servers=200, n_packages=2_000, seed=5)
    assert np.array_equal(a.edge_src, b.edge_src)
    assert np.array_equal(a.pkg_key_hi, b.pkg_key_hi)
    assert np.array_equal(a.arena.group_keys, b.arena.group_keys)


def test_estate_skew():
    """~1% heavy agents fan out far more servers than the median."""
    est = generate_estate(n_agents=2000, n_servers=20000, n_packages=1000, seed=3)
    from agentbom_amd.scan.synth import ET_USES

    uses = est.edge_type == ET_USES
    deg = np.bincount(est.edge_src[uses], minlength=est.n_agents)
    assert deg.max() > 5 * max(1, int(np.median(deg)))


def test_match_respects_windows(engine, estate):
    """Every reported finding's key really lies in its window (oracle check)."""
    pkg_idx, win_idx = engine.match()
    assert pkg_idx.numel() > 0
    a = estate.arena
    for i in range(0, pkg_idx.numel(), max(1, pkg_idx.numel() // 50)):
        p = int(pkg_idx[i])
        w = int(win_idx[i])
        key = (int(estate.pkg_key_hi[p]), int(estate.pkg_key_lo[p]))
        intro = (int(a.intro_hi[w]), int(a.intro_lo[w]))
        fixed = (int(a.fixed_hi[w]), int(a.fixed_lo[w]))
        assert key >= intro and key < fixed


def test_reach_distances(engine, estate):
    dist = engine.dependency_reach().numpy().view(np.uint32)
    # agents are sources (0); servers 1 hop; most packages 2 hops
    assert (dist[:estate.n_agents] == 0).all()
    srv = dist[estate.server_base:estate.server_base + estate.n_servers]
    assert (srv[srv != 0xFFFFFFFF] == 1).all()
    pkg = dist[estate.pkg_base:]
    assert (pkg[pkg != 0xFFFFFFFF] == 2).all()


def test_blast_counts_match_bruteforce(engine, estate):
    """Distinct agent/cred/tool counts vs a dict-based recount."""
    pkg_idx, _ = engine.match()
    pkg_nodes = pkg_idx + estate.pkg_base
    counts = engine.blast_counts(pkg_nodes)
    uniq = counts["uniq_pkgs"].numpy()

    from agentbom_amd.scan.synth import ET_CONTAINS, ET_HAS_CRED, ET_PROVIDES_TOOL, ET_USES

    # build adjacency dicts
    srv_of_pkg: dict[int, set] = {}
    ag_of_srv: dict[int, set] = {}
    cred_of_srv: dict[int, set] = {}
    tool_of_srv: dict[int, set] = {}
    for s, d, t in zip(estate.edge_src, estate.edge_dst, estate.edge_type):
        if t == ET_CONTAINS:
            srv_of_pkg.setdefault(int(d), set()).add(int(s))
        elif t == ET_USES:
            ag_of_srv.setdefault(int(d), set()).add(int(s))
        elif t == ET_HAS_CRED:
            cred_of_srv.setdefault(int(s), set()).add(int(d))
        elif t == ET_PROVIDES_TOOL:
            tool_of_srv.setdefault(int(s), set()).add(int(d))

    for i in range(0, len(uniq), max(1, len(uniq) // 25)):
        p = int(uniq[i])
        servers = srv_of_pkg.get(p, set())
        agents = set().union(*(ag_of_srv.get(s, set()) for s in servers)) if servers else set()
        creds = set().union(*(cred_of_srv.get(s, set()) for s in servers)) if servers else set()
        tools = set().union(*(tool_of_srv.get(s, set()) for s in servers)) if servers else set()
        assert int(counts["n_servers"][i]) == len(servers)
        assert int(counts["n_agents"][i]) == len(agents)
        assert int(counts["n_creds_all"][i]) == len(creds)
        assert int(counts["n_tools_all"][i]) == len(tools)


def test_step_deterministic(engine):
    r1 = engine.step()
    r2 = engine.step()
    assert r1["n_findings"] == r2["n_findings"]
    assert torch.equal(r1["order"], r2["order"])
    assert torch.equal(r1["scores"], r2["scores"])


def test_impact_filter_zeroes_clientside(estate):
    """Windows with client-side/availability impact expose no creds/tools."""
    eng = EstateEngine(estate, device="cpu")
    res = eng.step()
    impact = estate.arena.impact[res["win_idx"].numpy()]
    none_reach = np.isin(impact, [6, 7])  # availability, client-side
    if none_reach.any():
        assert (res["n_creds"].numpy()[none_reach] == 0).all()
        assert (res["n_tools"].numpy()[none_reach] == 0).all()


class TestOsvShapedArena:
    """The arena_windows mode: exact-window-count OSV-dump
    shape with zipf branch counts and mixed window forms."""

    @pytest.fixture(scope="class")
    def osv_estate(self):
        return generate_estate(n_agents=100, n_servers=500, n_packages=20_000,
                               name_catalog=4_000, seed=21, arena_windows=9_000)

    def test_exact_window_count_and_shape(self, osv_estate):
        a = osv_estate.arena
        assert a.num_windows == 9_000
        gc = np.diff(a.group_off.astype(np.int64))
        assert gc.max() >= 16, "zipf branch heads missing"
        assert (gc >= 1).all()
        from agentbom_amd.ops.cpu_ref import WF_HAS_FIXED, WF_HAS_INTRO, WF_HAS_LAST

        flags = a.flags
        assert ((flags & WF_HAS_INTRO) != 0).all()
        n_last = int(((flags & WF_HAS_LAST) != 0).sum())
        n_open = int((flags == WF_HAS_INTRO).sum())
        assert 0 < n_last < a.num_windows // 5
        assert 0 < n_open < a.num_windows // 20

    def test_match_bruteforce_all_forms(self, osv_estate):
        """CPU match vs per-window brute force incl. last_affected
        (inclusive) and open-ended introduced-only windows."""
        eng = EstateEngine(osv_estate, device="cpu")
        pkg_idx, win_idx = eng.match()
        got = set(zip(pkg_idx.tolist(), win_idx.tolist()))
        assert got, "OSV-shaped arena must produce findings"

        from agentbom_amd.ops.cpu_ref import WF_HAS_FIXED, WF_HAS_LAST

        a = osv_estate.arena
        # brute force: every package x every window of its group
        gk = osv_estate.pkg_name_id
        keys = list(zip(osv_estate.pkg_key_hi.tolist(), osv_estate.pkg_key_lo.tolist()))
        gkeys = a.group_keys
        expect = set()
        import numpy as _np

        for p in range(osv_estate.n_packages):
            g = int(_np.searchsorted(gkeys, gk[p]))
            if g >= len(gkeys) or gkeys[g] != gk[p]:
                continue
            for w in range(int(a.group_off[g]), int(a.group_off[g + 1])):
                key = keys[p]
                if key < (int(a.intro_hi[w]), int(a.intro_lo[w])):
                    continue
                fl = int(a.flags[w])
                if fl & WF_HAS_FIXED and key >= (int(a.fixed_hi[w]), int(a.fixed_lo[w])):
                    continue
                if fl & WF_HAS_LAST and key > (int(a.last_hi[w]), int(a.last_lo[w])):
                    continue
                expect.add((p, w))
        assert got == expect

    def test_step_runs_with_osv_arena(self, osv_estate):
        res = EstateEngine(osv_estate, device="cpu").step()
        rate = res["n_findings"] / osv_estate.n_packages
        assert 0.001 < rate < 0.25, f"finding rate {rate} out of realistic band"


def test_rollup_tree_collapse_bruteforce(estate, engine):
    """Engine rollup (device CONTAINS collapse) vs a dict-based recount."""
    res = engine.step()
    roll = engine.rollup(res)
    from agentbom_amd.scan.synth import ET_CONTAINS, ET_USES

    srv_of_pkg: dict[int, set] = {}
    ag_of_srv: dict[int, set] = {}
    for s, d, t in zip(estate.edge_src, estate.edge_dst, estate.edge_type):
        if t == ET_CONTAINS:
            srv_of_pkg.setdefault(int(d), set()).add(int(s))
        elif t == ET_USES:
            ag_of_srv.setdefault(int(d), set()).add(int(s))

    import numpy as np_

    exp_srv = np_.zeros((estate.n_servers, 6), dtype=np_.int64)
    sev = estate.arena.severity[res["win_idx"].numpy()]
    for pkg_i, sev_code in zip(res["pkg_idx"].numpy(), sev):
        node = int(pkg_i) + estate.pkg_base
        col = max(0, min(5, 5 - int(sev_code)))
        for s in srv_of_pkg.get(node, ()):  # every server containing it
            exp_srv[s - estate.server_base, col] += 1
    assert np_.array_equal(roll["server_hist"].numpy(), exp_srv)

    exp_ag = np_.zeros((estate.n_agents, 6), dtype=np_.int64)
    for s in range(estate.n_servers):
        for a in ag_of_srv.get(s + estate.server_base, ()):  # agents using it
            exp_ag[a] += exp_srv[s]
    assert np_.array_equal(roll["agent_hist"].numpy(), exp_ag)

    # worst codes: 0=critical..5, 6=no findings
    w = roll["server_worst"].numpy()
    for s in range(estate.n_servers):
        nz = np_.nonzero(exp_srv[s])[0]
        assert w[s] == (nz[0] if len(nz) else 6)
