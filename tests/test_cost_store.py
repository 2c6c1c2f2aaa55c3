"""FinOps cost tier: store, rollups, budget status, forecast machine, API."""

from __future__ import annotations

from datetime import datetime, timedelta, timezone

import pytest

from agentbom_amd.api.cost_store import (
    CostBudget,
    LLMCostRecord,
    SQLiteCostStore,
    budget_status,
    check_budget_enforcement,
    forecast_spend,
    rollup,
    rollup_by_tag,
    summarize,
)

_NOW = datetime(2026, 9, 14, 12, 0, tzinfo=timezone.utc)


def _rec(agent="bot", cost=1.0, hours_ago=1.0, model="claude", cc="",
         tags=None):
    return LLMCostRecord(
        tenant_id="t1", agent=agent, cost_usd=cost, model=model,
        tokens_in=100, tokens_out=50, cost_center=cc, tags=tags or {},
        observed_at=(_NOW - timedelta(hours=hours_ago)).isoformat())


class TestStoreAndRollups:
    def test_roundtrip_and_filters(self):
        store = SQLiteCostStore()
        store.add_records([_rec("a", 1.5), _rec("b", 2.0), _rec("a", 0.5)])
        assert len(store.list_records("t1")) == 3
        assert len(store.list_records("t1", agent="a")) == 2
        assert store.list_records("t2") == []
        since = (_NOW - timedelta(minutes=30)).isoformat()
        assert store.list_records("t1", since=since) == []

    def test_rollups(self):
        recs = [_rec("a", 3.0, model="opus"), _rec("b", 1.0, model="haiku"),
                _rec("a", 1.0, model="haiku", tags={"team": "sre"})]
        by_agent = rollup(recs, "agent")
        assert by_agent[0] == {"agent": "a", "cost_usd": 4.0, "records": 2,
                               "tokens": 300}
        assert rollup_by_tag(recs, "team") == [
            {"tag": "(untagged)", "cost_usd": 4.0},
            {"tag": "sre", "cost_usd": 1.0}]
        s = summarize(recs)
        assert s["total_cost_usd"] == 5.0 and s["by_model"][0]["model"] in ("opus", "haiku")


class TestBudget:
    def test_status_thresholds(self):
        b = CostBudget("t1", limit_usd=10.0)
        assert budget_status(5.0, b)["status"] == "ok"
        assert budget_status(8.5, b)["status"] == "warning"
        assert budget_status(10.0, b)["status"] == "exceeded"
        # zero cap = hard cap
        assert budget_status(0.01, CostBudget("t1", 0.0))["status"] == "exceeded"
        assert budget_status(5.0, None)["status"] == "no_budget"

    def test_enforcement_agent_then_tenant(self):
        store = SQLiteCostStore()
        store.add_records([_rec("a", 6.0), _rec("b", 1.0)])
        allowed, budget, spend = check_budget_enforcement(store, "t1", "a")
        assert allowed and budget is None
        store.set_budget(CostBudget("t1", 5.0, agent="a"))
        allowed, budget, spend = check_budget_enforcement(store, "t1", "a")
        assert not allowed and spend == 6.0
        # tenant-wide fallback caps agent b too (total 7.0 > 6.5)
        store.set_budget(CostBudget("t1", 6.5))
        allowed, _, spend = check_budget_enforcement(store, "t1", "b")
        assert not allowed and spend == 7.0


class TestForecast:
    def test_insufficient_history(self):
        out = forecast_spend([_rec(cost=1.0)], now=_NOW)
        assert out["status"] == "insufficient_history"
        assert out["burn_rate_usd_per_day"] is None

    def test_ok_with_runway(self):
        recs = [_rec(cost=1.0, hours_ago=h) for h in (2, 8, 14, 20)]
        out = forecast_spend(recs, CostBudget("t1", 100.0), now=_NOW)
        assert out["status"] == "ok"
        assert out["burn_rate_basis"] == "trailing_24h"
        assert out["burn_rate_usd_per_day"] == pytest.approx(4.0)
        assert out["days_remaining"] == pytest.approx((100 - 4) / 4.0, abs=0.01)
        assert out["projected_exhaustion_at"] > _NOW.isoformat()

    def test_budget_exceeded(self):
        recs = [_rec(cost=60.0, hours_ago=2), _rec(cost=50.0, hours_ago=4)]
        out = forecast_spend(recs, CostBudget("t1", 100.0), now=_NOW)
        assert out["status"] == "budget_exceeded"
        assert out["days_remaining"] == 0.0

    def test_stale_and_no_budget(self):
        old = [_rec(cost=1.0, hours_ago=24 * 30), _rec(cost=1.0, hours_ago=24 * 31)]
        assert forecast_spend(old, CostBudget("t1", 10.0),
                              now=_NOW)["status"] == "stale"
        fresh = [_rec(cost=1.0, hours_ago=1), _rec(cost=1.0, hours_ago=2)]
        assert forecast_spend(fresh, None, now=_NOW)["status"] == "no_budget"

    def test_weekly_fallback_basis(self):
        recs = [_rec(cost=7.0, hours_ago=30), _rec(cost=7.0, hours_ago=100)]
        out = forecast_spend(recs, CostBudget("t1", 1000.0), now=_NOW)
        assert out["burn_rate_basis"] == "trailing_168h"
        assert out["burn_rate_usd_per_day"] == pytest.approx(2.0)


class TestApi:
    @pytest.fixture()
    def client(self):
        from starlette.testclient import TestClient

        from agentbom_amd.api.server import create_app

        return TestClient(create_app())

    def test_ingest_summary_forecast_budget(self, client):
        r = client.post("/v1/costs/records", json={"records": [
            {"agent": "bot", "cost_usd": 2.5, "model": "opus",
             "tokens_in": 10, "tokens_out": 5,
             "observed_at": _NOW.isoformat(), "tags": {"team": "sre"}},
            {"agent": "bot", "cost_usd": 1.5,
             "observed_at": (_NOW - timedelta(hours=3)).isoformat()},
        ]})
        assert r.status_code == 201 and r.json()["ingested"] == 2
        s = client.get("/v1/costs/summary").json()
        assert s["total_cost_usd"] == 4.0
        assert s["by_agent"][0]["agent"] == "bot"
        assert s["budget"]["status"] == "no_budget"
        b = client.put("/v1/costs/budget", json={"limit_usd": 3.0})
        assert b.status_code == 200
        s2 = client.get("/v1/costs/summary").json()
        assert s2["budget"]["status"] == "exceeded"
        f = client.get("/v1/costs/forecast").json()
        assert f["status"] == "budget_exceeded"
        assert client.post("/v1/costs/records", json={"records": []}).status_code == 400
