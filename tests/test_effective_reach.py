"""Effective-reach triage scoring: factors, composite, bands."""

from __future__ import annotations

import pytest

from agentbom_amd.graph.effective_reach import (
    ReachScore,
    credential_tier,
    effective_reach_summary,
    reach_score_for_blast,
)


class TestFactors:
    def test_credential_tiers(self):
        assert credential_tier("HOME") == 0.10
        assert credential_tier("AWS_SECRET_ACCESS_KEY") == 1.0
        assert credential_tier("ANTHROPIC_API_KEY") == 1.0
        assert credential_tier("GITHUB_TOKEN") == 0.55
        assert credential_tier("DATABASE_URL") == 0.55
        assert credential_tier("MY_SETTING") == 0.25


class TestComposite:
    def test_low_reach_green(self):
        # CVSS 6.5, no KEV, read tool, HOME env, 1 agent -> green (~25)
        rs = ReachScore(cvss=6.5, epss=0.01, is_kev=False,
                        tool_capability=0.10, cred_visibility=0.10,
                        agent_breadth=1)
        assert rs.band == "green"
        assert 20 <= rs.composite <= 30

    def test_high_reach_pulsing_red(self):
        # CVSS 9.8, KEV, shell tool, AWS_* creds, desktop agent -> ~95+
        rs = ReachScore(cvss=9.8, epss=0.6, is_kev=True,
                        tool_capability=1.0, cred_visibility=1.0,
                        agent_breadth=1)
        assert rs.band == "pulsing-red"
        assert rs.composite >= 90

    def test_kev_alone_flips_band(self):
        base = dict(cvss=7.0, epss=0.1, tool_capability=0.4,
                    cred_visibility=0.55, agent_breadth=2)
        assert ReachScore(is_kev=False, **base).band == "amber"
        assert ReachScore(is_kev=True, **base).band in ("red", "pulsing-red")

    def test_breadth_capped(self):
        a = ReachScore(cvss=5, epss=0, is_kev=False, tool_capability=0,
                       cred_visibility=0, agent_breadth=5)
        b = ReachScore(cvss=5, epss=0, is_kev=False, tool_capability=0,
                       cred_visibility=0, agent_breadth=500)
        assert a.composite == b.composite

    def test_symbol_delta(self):
        base = dict(cvss=8.0, epss=0.2, is_kev=False, tool_capability=0.65,
                    cred_visibility=0.55, agent_breadth=2)
        mid = ReachScore(**base).composite
        up = ReachScore(symbol_reachability="function_reachable",
                        **base).composite
        down = ReachScore(symbol_reachability="unreachable", **base).composite
        assert up == pytest.approx(mid + 15.0)
        assert down == pytest.approx(mid - 30.0)

    def test_breakdown_stable_keys(self):
        bd = ReachScore(cvss=1, epss=0, is_kev=False, tool_capability=0,
                        cred_visibility=0, agent_breadth=0).as_breakdown()
        assert set(bd) >= {"cvss", "epss", "is_kev", "tool_capability",
                           "cred_visibility", "agent_breadth", "composite",
                           "band"}


class TestReportIntegration:
    def test_demo_report_summary(self):
        from agentbom_amd.scan.orchestrator import run_demo_scan

        report = run_demo_scan()
        out = effective_reach_summary(report)
        assert sum(out["bands"].values()) == len(out["findings"]) > 0
        # sorted by composite descending
        comps = [r["composite"] for r in out["findings"]]
        assert comps == sorted(comps, reverse=True)
        # every row carries band + factor evidence
        top = out["findings"][0]
        assert top["band"] in ("green", "amber", "red", "pulsing-red")
        rs = reach_score_for_blast(report.blast_radii[0])
        assert 0.0 <= rs.composite <= 100.0


def test_reach_endpoint():
    from starlette.testclient import TestClient

    from agentbom_amd.api.server import create_app

    client = TestClient(create_app())
    client.post("/v1/scan", json={"demo": True})
    out = client.get("/v1/findings/reach").json()
    assert out["findings"] and "bands" in out
    one_band = out["findings"][0]["band"]
    filtered = client.get(f"/v1/findings/reach?band={one_band}").json()
    assert all(r["band"] == one_band for r in filtered["findings"])


def test_finding_lifecycle_endpoints():
    from starlette.testclient import TestClient

    from agentbom_amd.api.server import create_app

    client = TestClient(create_app())
    client.post("/v1/scan", json={"demo": True})
    assert client.get("/v1/findings/lifecycle").json()["observed_scans"] == 0
    d1 = client.post("/v1/findings/lifecycle/observe").json()
    assert d1["new"] > 0 and d1["active"] == d1["new"]
    d2 = client.post("/v1/findings/lifecycle/observe").json()
    assert d2["new"] == 0 and d2["resolved"] == 0  # same scan, steady state
    s = client.get("/v1/findings/lifecycle").json()
    assert s["tracked_findings"] == d1["active"] and s["resolved"] == 0
