"""Anomaly detection: modified z-score + EWMA spike semantics."""

from __future__ import annotations

from datetime import datetime, timedelta, timezone

from agentbom_amd.api.anomaly import (
    detect_behavior_anomalies,
    detect_cost_anomalies,
    detect_temporal_cost_anomalies,
    modified_z,
)

_NOW = datetime(2026, 9, 14, 12, 0, tzinfo=timezone.utc)


class TestRobustZ:
    def test_runaway_agent_flagged(self):
        spend = {"a": 1.0, "b": 1.2, "c": 0.9, "d": 1.1, "runaway": 50.0}
        hits = detect_cost_anomalies(spend)
        assert [h["subject"] for h in hits] == ["runaway"]
        assert hits[0]["z"] == "inf" or hits[0]["z"] > 3.5
        assert hits[0]["baseline_median"] == 1.1

    def test_mean_based_would_miss_small_sample(self):
        # the documented reason for median+MAD: with mean/std the outlier
        # inflates its own baseline; robust z still flags it
        spend = {"a": 1.0, "b": 1.0, "c": 1.0, "outlier": 10.0}
        assert detect_cost_anomalies(spend)

    def test_too_few_peers_silent(self):
        assert detect_cost_anomalies({"a": 1.0, "b": 99.0}) == []

    def test_uniform_population_clean(self):
        assert detect_cost_anomalies({c: 2.0 for c in "abcdef"}) == []
        assert modified_z(2.0, 2.0, 0.0) == 0.0

    def test_behavior_sessions(self):
        calls = {"s1": 10, "s2": 12, "s3": 9, "s4": 11, "burst": 500}
        hits = detect_behavior_anomalies(calls)
        assert hits and hits[0]["subject"] == "burst"
        assert hits[0]["kind"] == "behavior"


class TestTemporal:
    def _rec(self, hours_ago, cost):
        return {"observed_at": (_NOW - timedelta(hours=hours_ago)).isoformat(),
                "cost_usd": cost}

    def test_spike_over_ewma(self):
        records = [self._rec(h, 1.0) for h in range(8, 1, -1)]
        records.append(self._rec(0, 25.0))  # spike hour
        hits = detect_temporal_cost_anomalies(records)
        assert len(hits) == 1
        assert hits[0]["factor"] >= 3.0
        assert hits[0]["bucket"].endswith("T12")

    def test_spike_does_not_inflate_its_own_baseline(self):
        # baseline uses only PRECEDING buckets: the first hour can never
        # be flagged, steady growth below 3x never trips
        records = [self._rec(h, 1.0 * (1.5 ** (8 - h))) for h in range(8, 0, -1)]
        assert detect_temporal_cost_anomalies(records) == []

    def test_too_few_buckets(self):
        assert detect_temporal_cost_anomalies(
            [self._rec(1, 1.0), self._rec(0, 100.0)]) == []


def test_endpoint():
    from starlette.testclient import TestClient

    from agentbom_amd.api.server import create_app

    client = TestClient(create_app())
    rows = [{"agent": a, "cost_usd": c, "observed_at": _NOW.isoformat()}
            for a, c in (("a", 1.0), ("b", 1.1), ("c", 0.9), ("d", 1.0),
                         ("runaway", 40.0))]
    client.post("/v1/costs/records", json={"records": rows})
    out = client.get("/v1/costs/anomalies").json()
    assert [h["subject"] for h in out["agents"]] == ["runaway"]
