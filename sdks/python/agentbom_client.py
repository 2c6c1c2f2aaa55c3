"""agent-bom control-plane Python SDK.

Reference: sdks/python (control-plane client, not a scanner).  Thin typed
wrapper over the REST API using httpx; mirrors the /v1 surface.
"""

from __future__ import annotations

from typing import Any, Optional

import httpx

__version__ = "0.1.0"


class AgentBomError(RuntimeError):
    def __init__(self, status_code: int, detail: str):
        super().__init__(f"HTTP {status_code}: {detail}")
        self.status_code = status_code
        self.detail = detail


class AgentBomClient:
    """Synchronous client for the agent-bom REST API."""

    def __init__(self, base_url: str = "http://127.0.0.1:8000",
                 api_key: Optional[str] = None, timeout: float = 60.0,
                 transport: Optional[Any] = None):
        headers = {"x-api-key": api_key} if api_key else {}
        self._http = httpx.Client(base_url=base_url, headers=headers,
                                  timeout=timeout, transport=transport)

    def close(self) -> None:
        self._http.close()

    def __enter__(self) -> "AgentBomClient":
        return self

    def __exit__(self, *exc) -> None:
        self.close()

    def _req(self, method: str, path: str, **kw) -> Any:
        resp = self._http.request(method, path, **kw)
        if resp.status_code >= 400:
            try:
                detail = resp.json().get("detail", resp.text)
            except Exception:  # noqa: BLE001
                detail = resp.text
            raise AgentBomError(resp.status_code, str(detail))
        return resp.json() if resp.content else None

    # ── health / scan ─────────────────────────────────────────────────────

    def health(self) -> dict:
        return self._req("GET", "/healthz")

    def scan(self, inventory: Optional[dict] = None, demo: bool = False,
             blast_radius_depth: int = 1) -> dict:
        return self._req("POST", "/v1/scan", json={
            "inventory": inventory, "demo": demo,
            "blast_radius_depth": blast_radius_depth,
        })

    def scan_job(self, job_id: str) -> dict:
        return self._req("GET", f"/v1/scan/{job_id}")

    def scan_report(self, job_id: str) -> dict:
        return self._req("GET", f"/v1/scan/{job_id}/report")

    def findings(self, severity: Optional[str] = None, limit: int = 100) -> dict:
        params = {"limit": limit}
        if severity:
            params["severity"] = severity
        return self._req("GET", "/v1/findings", params=params)

    # ── graph ─────────────────────────────────────────────────────────────

    def graph(self, limit: int = 100) -> dict:
        return self._req("GET", "/v1/graph", params={"limit": limit})

    def graph_search(self, q: str, entity_type: Optional[str] = None) -> dict:
        params = {"q": q}
        if entity_type:
            params["entity_type"] = entity_type
        return self._req("GET", "/v1/graph/search", params=params)

    def graph_paths(self, limit: int = 25) -> dict:
        return self._req("GET", "/v1/graph/paths", params={"limit": limit})

    def exposure_paths(self, limit: int = 50) -> dict:
        return self._req("GET", "/v1/graph/exposure-paths", params={"limit": limit})

    def graph_query(self, start: str, max_depth: int = 3, max_nodes: int = 500) -> dict:
        return self._req("POST", "/v1/graph/query", json={
            "start": start, "max_depth": max_depth, "max_nodes": max_nodes})

    def impact(self, node_id: str, max_hops: int = 4) -> dict:
        return self._req("GET", f"/v1/graph/impact/{node_id}",
                         params={"max_hops": max_hops})

    def rollup(self) -> dict:
        return self._req("GET", "/v1/graph/rollup")

    def should_i_deploy(self) -> dict:
        return self._req("GET", "/v1/graph/should-i-deploy")

    def evidence_manifest(self) -> dict:
        return self._req("GET", "/v1/graph/evidence-manifest")

    # ── fleet / schedules / compliance ────────────────────────────────────

    def heartbeat(self, member_id: str, **stats) -> dict:
        return self._req("POST", "/v1/fleet/heartbeat",
                         json={"member_id": member_id, "hostname": member_id, **stats})

    def fleet(self) -> dict:
        return self._req("GET", "/v1/fleet")

    def create_schedule(self, schedule_id: str, interval_s: float) -> dict:
        return self._req("POST", "/v1/schedules",
                         json={"schedule_id": schedule_id, "interval_s": interval_s})

    def compliance_report(self, framework: str) -> dict:
        return self._req("GET", f"/v1/compliance/{framework}/report")
