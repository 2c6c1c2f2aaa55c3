"""ClickHouse analytics store (optional tier, reference api/clickhouse_store.py).

Findings-over-time analytics land in ClickHouse via its HTTP interface —
no driver dependency, just SQL over POST with JSONEachRow payloads,
through the offline-guarded retry client (tests inject
httpx.MockTransport).  The control plane works fully without it; this is
the reference's optional analytics tier, mirrored.
"""

from __future__ import annotations

import json
import time
from typing import Any, Optional

from agentbom_amd.utils import config as cfg
from agentbom_amd.utils.http_client import check_offline, create_client, request_with_retry

_SCHEMA = """
CREATE TABLE IF NOT EXISTS {db}.scan_findings (
    ts          DateTime,
    scan_id     String,
    tenant_id   String,
    vuln_id     String,
    package     String,
    ecosystem   String,
    severity    LowCardinality(String),
    risk_score  Float32,
    is_kev      UInt8,
    reachability LowCardinality(String)
) ENGINE = MergeTree()
ORDER BY (tenant_id, ts)
TTL ts + INTERVAL 180 DAY
"""


class ClickHouseAnalyticsStore:
    """Append findings batches; run trend queries.  HTTP interface only."""

    def __init__(self, url: str, database: str = "agentbom",
                 client=None, username: Optional[str] = None,
                 password: Optional[str] = None):
        check_offline(url)
        self.url = url.rstrip("/")
        self.database = database
        self.client = client or create_client(timeout=cfg.CLICKHOUSE_TIMEOUT_S)
        self.headers: dict[str, str] = {}
        if username:
            self.headers["X-ClickHouse-User"] = username
        if password:
            self.headers["X-ClickHouse-Key"] = password

    def _exec(self, sql: str, body: Optional[str] = None):
        params = {"query": sql} if body is not None else {}
        resp = request_with_retry(
            self.client, "POST", self.url, params=params,
            content=(body if body is not None else sql).encode(),
            headers=self.headers)
        if resp is None or resp.status_code != 200:
            raise RuntimeError(
                f"clickhouse query failed: "
                f"{resp.status_code if resp is not None else 'unreachable'} "
                f"{(resp.text[:200] if resp is not None else '')}")
        return resp.text

    def ensure_schema(self) -> None:
        self._exec(f"CREATE DATABASE IF NOT EXISTS {self.database}")
        self._exec(_SCHEMA.format(db=self.database))

    def insert_findings(self, report, tenant_id: str = "default") -> int:
        """One JSONEachRow batch per report (blast-radius rows)."""
        rows = []
        ts = int(time.time())
        for br in report.blast_radii:
            rows.append(json.dumps({
                "ts": ts,
                "scan_id": report.scan_id or "",
                "tenant_id": tenant_id,
                "vuln_id": br.vulnerability.id,
                "package": f"{br.package.name}@{br.package.version}",
                "ecosystem": br.package.ecosystem,
                "severity": br.vulnerability.severity.value,
                "risk_score": float(br.risk_score),
                "is_kev": 1 if br.vulnerability.is_kev else 0,
                "reachability": br.reachability,
            }))
        if not rows:
            return 0
        self._exec(
            f"INSERT INTO {self.database}.scan_findings FORMAT JSONEachRow",
            body="\n".join(rows))
        return len(rows)

    def severity_trend(self, tenant_id: str = "default",
                       days: int = 30) -> list[dict[str, Any]]:
        sql = (
            f"SELECT toDate(ts) AS day, severity, count() AS findings "
            f"FROM {self.database}.scan_findings "
            f"WHERE tenant_id = '{_esc(tenant_id)}' "
            f"AND ts >= now() - INTERVAL {int(days)} DAY "
            f"GROUP BY day, severity ORDER BY day, severity "
            f"FORMAT JSONEachRow")
        return _rows(self._exec(sql))

    def top_risk_packages(self, tenant_id: str = "default",
                          limit: int = 20) -> list[dict[str, Any]]:
        sql = (
            f"SELECT package, max(risk_score) AS max_risk, count() AS findings "
            f"FROM {self.database}.scan_findings "
            f"WHERE tenant_id = '{_esc(tenant_id)}' "
            f"GROUP BY package ORDER BY max_risk DESC, package "
            f"LIMIT {int(limit)} FORMAT JSONEachRow")
        return _rows(self._exec(sql))

    def kev_exposure(self, tenant_id: str = "default") -> list[dict[str, Any]]:
        sql = (
            f"SELECT toDate(ts) AS day, countIf(is_kev = 1) AS kev_findings "
            f"FROM {self.database}.scan_findings "
            f"WHERE tenant_id = '{_esc(tenant_id)}' "
            f"GROUP BY day ORDER BY day FORMAT JSONEachRow")
        return _rows(self._exec(sql))


def _esc(s: str) -> str:
    return s.replace("\\", "\\\\").replace("'", "\\'")


def _rows(text: str) -> list[dict[str, Any]]:
    out = []
    for line in text.splitlines():
        line = line.strip()
        if not line:
            continue
        try:
            out.append(json.loads(line))
        except ValueError:
            continue
    return out
