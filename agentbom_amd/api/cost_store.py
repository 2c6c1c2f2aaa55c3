"""FinOps control plane: LLM cost records, budgets, rollups, forecast.

Reference parity: src/agent_bom/api/{cost_store,cost_forecast,cost_owner}.py
— the observability cost tier: OTel-GenAI-shaped spend records land in a
durable store; budgets cap agents or cost centers; rollups/forecasts drive
the `/v1/costs/*` surface and the gateway's budget enforcement.

Design here: one SQLite-backed store (":memory:" default, same pattern as
identity/lifecycle.py), pure-function rollup/forecast helpers that take a
record list (deterministic, `now` injectable), and a forecast status
machine: insufficient_history / budget_exceeded / stale / no_budget / ok.
"""

from __future__ import annotations

import calendar
import json
import sqlite3
import threading
from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Any, Iterable, Optional

_MIN_RECORDS = 2
_MAX_RUNWAY_DAYS = 3650.0
_WINDOWS_H = (24.0, 7 * 24.0)  # prefer last day; fall back to last week


def _now() -> datetime:
    return datetime.now(timezone.utc)


def _parse_ts(value: str) -> Optional[datetime]:
    try:
        ts = datetime.fromisoformat(str(value).replace("Z", "+00:00"))
    except (TypeError, ValueError):
        return None
    return ts if ts.tzinfo else ts.replace(tzinfo=timezone.utc)


@dataclass
class LLMCostRecord:
    """One LLM spend observation (OTel GenAI span shape)."""

    tenant_id: str
    agent: str
    cost_usd: float
    model: str = ""
    tokens_in: int = 0
    tokens_out: int = 0
    observed_at: str = ""
    cost_center: str = ""
    tags: dict[str, str] = field(default_factory=dict)

    def __post_init__(self) -> None:
        if not self.observed_at:
            self.observed_at = _now().isoformat()

    def to_dict(self) -> dict[str, Any]:
        return {
            "tenant_id": self.tenant_id, "agent": self.agent,
            "cost_usd": self.cost_usd, "model": self.model,
            "tokens_in": self.tokens_in, "tokens_out": self.tokens_out,
            "observed_at": self.observed_at, "cost_center": self.cost_center,
            "tags": dict(self.tags),
        }


@dataclass
class CostBudget:
    """Spend cap for an agent or a cost center (monthly period)."""

    tenant_id: str
    limit_usd: float
    agent: str = ""         # "" = tenant-wide
    cost_center: str = ""

    def to_dict(self) -> dict[str, Any]:
        return {"tenant_id": self.tenant_id, "limit_usd": self.limit_usd,
                "agent": self.agent, "cost_center": self.cost_center}


def budget_status(spend: float, budget: Optional[CostBudget]) -> dict[str, Any]:
    """ok / warning (>=80%) / exceeded; a zero cap is a hard cap."""
    if budget is None:
        return {"status": "no_budget", "spend_usd": round(spend, 6),
                "limit_usd": None, "utilization": None}
    limit = budget.limit_usd
    exceeded = spend >= limit if limit > 0 else spend > 0
    util = (spend / limit) if limit > 0 else None
    status = ("exceeded" if exceeded
              else "warning" if util is not None and util >= 0.8 else "ok")
    return {"status": status, "spend_usd": round(spend, 6),
            "limit_usd": limit,
            "utilization": round(util, 4) if util is not None else None}


def rollup(records: Iterable[LLMCostRecord], dimension: str) -> list[dict[str, Any]]:
    """Spend grouped by a record dimension (agent / model / cost_center)."""
    acc: dict[str, dict[str, Any]] = {}
    for rec in records:
        key = str(getattr(rec, dimension, "") or "(unattributed)")
        row = acc.setdefault(key, {dimension: key, "cost_usd": 0.0,
                                   "records": 0, "tokens": 0})
        row["cost_usd"] += rec.cost_usd
        row["records"] += 1
        row["tokens"] += rec.tokens_in + rec.tokens_out
    out = sorted(acc.values(), key=lambda r: (-r["cost_usd"], r[dimension]))
    for row in out:
        row["cost_usd"] = round(row["cost_usd"], 6)
    return out


def rollup_by_tag(records: Iterable[LLMCostRecord], tag_key: str) -> list[dict[str, Any]]:
    acc: dict[str, float] = {}
    for rec in records:
        key = rec.tags.get(tag_key, "(untagged)")
        acc[key] = acc.get(key, 0.0) + rec.cost_usd
    return [{"tag": k, "cost_usd": round(v, 6)}
            for k, v in sorted(acc.items(), key=lambda kv: (-kv[1], kv[0]))]


def summarize(records: list[LLMCostRecord]) -> dict[str, Any]:
    total = sum(r.cost_usd for r in records)
    return {
        "total_cost_usd": round(total, 6),
        "records": len(records),
        "total_tokens": sum(r.tokens_in + r.tokens_out for r in records),
        "by_agent": rollup(records, "agent"),
        "by_model": rollup(records, "model"),
        "by_cost_center": rollup(records, "cost_center"),
    }


def _period_bounds(now: datetime) -> tuple[datetime, datetime, float]:
    """Calendar-month budget period; returns (start, end, hours_remaining)."""
    start = now.replace(day=1, hour=0, minute=0, second=0, microsecond=0)
    last_day = calendar.monthrange(now.year, now.month)[1]
    end = start.replace(day=last_day, hour=23, minute=59, second=59)
    return start, end, max((end - now).total_seconds() / 3600.0, 0.0)


def _best_daily_rate(timed: list[tuple[datetime, float]],
                     now: datetime) -> tuple[Optional[float], Optional[str]]:
    for hours in _WINDOWS_H:
        window = [c for ts, c in timed
                  if 0.0 <= (now - ts).total_seconds() / 3600.0 <= hours]
        if len(window) >= _MIN_RECORDS:
            return sum(window) * 24.0 / hours, f"trailing_{int(hours)}h"
    return None, None


def forecast_spend(records: Iterable[LLMCostRecord],
                   budget: Optional[CostBudget] = None,
                   now: Optional[datetime] = None) -> dict[str, Any]:
    """Burn rate, projected period spend and budget runway.

    Status machine (reference cost_forecast.forecast_spend):
    insufficient_history (<2 timestamped records) / budget_exceeded
    (runway 0, exhaustion now) / stale (no record in the trailing
    windows) / no_budget (rate but no cap) / ok.
    """
    now = (now or _now()).astimezone(timezone.utc)
    timed: list[tuple[datetime, float]] = []
    total = 0.0
    for rec in records:
        total += rec.cost_usd
        ts = _parse_ts(rec.observed_at)
        if ts is not None:
            timed.append((ts, rec.cost_usd))

    limit = budget.limit_usd if budget else None
    out: dict[str, Any] = {
        "schema_version": "observability.cost_forecast.v1",
        "now": now.isoformat(),
        "current_spend_usd": round(total, 6),
        "budget_limit_usd": limit,
        "burn_rate_usd_per_day": None,
        "burn_rate_basis": None,
        "projected_period_spend_usd": None,
        "days_remaining": None,
        "projected_exhaustion_at": None,
    }
    if len(timed) < _MIN_RECORDS:
        out["status"] = "insufficient_history"
        return out

    _start, _end, hours_left = _period_bounds(now)
    exceeded = limit is not None and (total >= limit if limit > 0 else total > 0)
    rate, basis = _best_daily_rate(timed, now)
    out["burn_rate_usd_per_day"] = round(rate, 6) if rate is not None else None
    out["burn_rate_basis"] = basis
    if exceeded:
        out["status"] = "budget_exceeded"
        out["days_remaining"] = 0.0
        out["projected_exhaustion_at"] = now.isoformat()
        if rate is not None:
            out["projected_period_spend_usd"] = round(total + rate * hours_left / 24.0, 6)
        return out
    if rate is None:
        out["status"] = "stale"
        return out
    out["projected_period_spend_usd"] = round(total + rate * hours_left / 24.0, 6)
    if limit is None:
        out["status"] = "no_budget"
        return out
    days = (limit - total) / rate if rate > 0 else _MAX_RUNWAY_DAYS
    days = max(0.0, min(days, _MAX_RUNWAY_DAYS))
    out["days_remaining"] = round(days, 4)
    out["projected_exhaustion_at"] = datetime.fromtimestamp(
        now.timestamp() + days * 86400.0, tz=timezone.utc).isoformat()
    out["status"] = "ok"
    return out


_SCHEMA = """
CREATE TABLE IF NOT EXISTS cost_records (
    seq INTEGER PRIMARY KEY AUTOINCREMENT,
    tenant_id TEXT NOT NULL,
    agent TEXT NOT NULL,
    observed_at TEXT NOT NULL,
    doc TEXT NOT NULL
);
CREATE INDEX IF NOT EXISTS idx_cost_tenant ON cost_records (tenant_id, agent);
CREATE TABLE IF NOT EXISTS cost_budgets (
    tenant_id TEXT NOT NULL,
    agent TEXT NOT NULL DEFAULT '',
    cost_center TEXT NOT NULL DEFAULT '',
    limit_usd REAL NOT NULL,
    PRIMARY KEY (tenant_id, agent, cost_center)
);
"""


class SQLiteCostStore:
    """Durable cost tier (":memory:" default — same pattern as the
    identity lifecycle store)."""

    def __init__(self, path: str = ":memory:"):
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.executescript(_SCHEMA)
        self._lock = threading.Lock()

    def add_records(self, records: Iterable[LLMCostRecord]) -> int:
        rows = [(r.tenant_id, r.agent, r.observed_at, json.dumps(r.to_dict()))
                for r in records]
        with self._lock:
            self._db.executemany(
                "INSERT INTO cost_records (tenant_id, agent, observed_at, doc)"
                " VALUES (?,?,?,?)", rows)
            self._db.commit()
        return len(rows)

    def list_records(self, tenant_id: str, agent: Optional[str] = None,
                     cost_center: Optional[str] = None,
                     since: Optional[str] = None,
                     limit: int = 100_000) -> list[LLMCostRecord]:
        q = "SELECT doc FROM cost_records WHERE tenant_id=?"
        args: list[Any] = [tenant_id]
        if agent:
            q += " AND agent=?"
            args.append(agent)
        if since:
            q += " AND observed_at>=?"
            args.append(since)
        q += " ORDER BY seq LIMIT ?"
        args.append(limit)
        out = []
        for (doc,) in self._db.execute(q, args):
            d = json.loads(doc)
            rec = LLMCostRecord(**{k: d[k] for k in (
                "tenant_id", "agent", "cost_usd", "model", "tokens_in",
                "tokens_out", "observed_at", "cost_center")} | {"tags": d.get("tags", {})})
            if cost_center and rec.cost_center != cost_center:
                continue
            out.append(rec)
        return out

    def set_budget(self, budget: CostBudget) -> None:
        with self._lock:
            self._db.execute(
                "INSERT OR REPLACE INTO cost_budgets "
                "(tenant_id, agent, cost_center, limit_usd) VALUES (?,?,?,?)",
                (budget.tenant_id, budget.agent, budget.cost_center,
                 budget.limit_usd))
            self._db.commit()

    def get_budget(self, tenant_id: str, agent: str = "",
                   cost_center: str = "") -> Optional[CostBudget]:
        row = self._db.execute(
            "SELECT limit_usd FROM cost_budgets WHERE tenant_id=? AND "
            "agent=? AND cost_center=?",
            (tenant_id, agent, cost_center)).fetchone()
        if row is None:
            return None
        return CostBudget(tenant_id=tenant_id, limit_usd=row[0], agent=agent,
                          cost_center=cost_center)


def check_budget_enforcement(store: SQLiteCostStore, tenant_id: str,
                             agent: str) -> tuple[bool, Optional[CostBudget], float]:
    """(allowed, budget, current_spend) — agent budget first, tenant-wide
    fallback; no budget configured ⇒ allowed.  Feeds the gateway gate."""
    budget = store.get_budget(tenant_id, agent=agent) \
        or store.get_budget(tenant_id)
    if budget is None:
        return True, None, 0.0
    records = store.list_records(
        tenant_id, agent=agent if budget.agent else None)
    spend = sum(r.cost_usd for r in records)
    status = budget_status(spend, budget)
    return status["status"] != "exceeded", budget, spend
