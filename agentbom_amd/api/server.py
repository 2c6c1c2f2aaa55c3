"""Control-plane REST API (FastAPI).

Reference surface: src/agent_bom/api/server.py:695 (app assembly),
api/routes/scan.py (/v1/scan*), api/routes/graph.py (/v1/graph* — 29
endpoints incl. search/paths/attack-paths/exposure-paths/should-i-deploy/
query/rollup/diff/neighbors), api/pipeline.py (job model with
pending/running/done/failed/cancelled), api/metrics.py (/metrics).

Hot graph reads are served from the in-process UnifiedGraph snapshot (and
the GPU engine at estate scale) instead of the reference's Postgres
server-side walks — the read path the BASELINE p50 metric targets.
"""

from __future__ import annotations

import json
import os
import threading
import uuid
from datetime import datetime, timezone
from typing import Any, Optional

from fastapi import Depends, FastAPI, Header, HTTPException, Request
from pydantic import BaseModel, Field

from agentbom_amd import __version__


class ScanRequest(BaseModel):
    inventory: Optional[dict] = None
    demo: bool = False
    offline: bool = True
    blast_radius_depth: int = Field(default=1, ge=1, le=5)


class GraphQueryRequest(BaseModel):
    start: str
    max_depth: int = Field(default=3, ge=1, le=6)
    max_nodes: int = Field(default=500, ge=1, le=10_000)


class _State:
    def __init__(self) -> None:
        self.jobs: dict[str, dict] = {}
        self.reports: dict[str, Any] = {}
        self.graphs: dict[str, Any] = {}
        self.latest_scan: Optional[str] = None
        # tenant scoping: scan ownership + per-tenant latest pointers
        self.scan_tenant: dict[str, str] = {}
        self.latest_by_tenant: dict[str, str] = {}
        self.lock = threading.Lock()
        self.metrics: dict[str, float] = {
            "scans_total": 0, "scan_failures_total": 0, "graph_queries_total": 0,
            "auth_failures_total": 0,
        }
        self.graph_store = None
        store_path = os.environ.get("AGENT_BOM_GRAPH_STORE")
        if store_path:
            from agentbom_amd.graph.store import SQLiteGraphStore

            self.graph_store = SQLiteGraphStore(store_path)
        from agentbom_amd.api.fleet import BackpressureController, FleetRegistry

        self.fleet = FleetRegistry()
        self.backpressure = BackpressureController(
            max_concurrent=int(os.environ.get("AGENT_BOM_MAX_CONCURRENT_SCANS", "8")))
        self.scheduler = None
        self.audit_entries: list[dict] = []
        self.identity_store = None
        from agentbom_amd.api.webhooks import WebhookRegistry

        self.webhooks = WebhookRegistry()
        # Idempotency-Key replay cache: (tenant, key) -> first response
        self.idempotency: dict[tuple, dict] = {}
        self.lifecycle = None  # AssetTracker, created on first observe
        self.exports = None  # ExportManager, created on first use
        self.oauth_as = None  # OAuthAuthorizationServer (AGENT_BOM_OAUTH_AS=1)
        self.model_keys = None  # ModelKeyBroker, created on first use
        self.hub = None  # ComplianceHub, created on first use
        self.datasets = None  # DatasetVersionStore, created on first use
        self.evaluations = None  # EvaluationStore, created on first use


def create_app() -> FastAPI:
    app = FastAPI(title="agent-bom", version=__version__)
    state = _State()
    app.state.abom = state

    api_key = os.environ.get("AGENT_BOM_API_KEY")
    # RBAC: AGENT_BOM_API_KEYS="key1:admin,key2:operator,key3:viewer"
    # (reference: api/middleware.py + rbac.py role model, collapsed to the
    # three-tier contract: viewer=read, operator=read+scan, admin=all)
    key_roles: dict[str, str] = {}
    for part in (os.environ.get("AGENT_BOM_API_KEYS") or "").split(","):
        if ":" in part:
            k, _, role = part.strip().partition(":")
            if k and role in ("admin", "operator", "viewer"):
                key_roles[k] = role
    if api_key and api_key not in key_roles:
        key_roles[api_key] = "admin"  # legacy single-key = admin

    _WRITE_PREFIXES = ("/v1/identities", "/v1/schedules", "/v1/fleet",
                       "/scim", "/v1/delegation-tokens", "/v1/costs/budget",
                       "/v1/exceptions", "/v1/blueprints", "/v1/drift-incidents",
                       "/v1/connections", "/v1/approvals", "/v1/model-keys")

    def _role_allows(role: str, method: str, path: str) -> bool:
        if role == "admin":
            return True
        if method in ("GET", "HEAD", "OPTIONS"):
            return True
        if role == "operator":
            # operators can launch scans and post audit evidence, but not
            # touch identity/fleet/schedule write surfaces
            return not any(path.startswith(p) for p in _WRITE_PREFIXES)
        return False  # viewer: reads only

    from agentbom_amd.api.auth import (
        AuthError,
        DelegationTokens,
        QuotaTracker,
        ScimUserStore,
        role_from_claims,
        verify_oidc_bearer,
    )

    state.delegation = DelegationTokens()
    state.quotas = QuotaTracker()
    state.scim_users = ScimUserStore()
    from agentbom_amd.api.cost_store import SQLiteCostStore
    from agentbom_amd.api.exceptions_store import ExceptionStore

    from agentbom_amd.api.blueprints import BlueprintStore

    from agentbom_amd.api.connections import ConnectionStore

    from agentbom_amd.runtime.hitl import ApprovalQueue

    state.costs = SQLiteCostStore()
    state.exceptions = ExceptionStore()
    state.blueprints = BlueprintStore()
    state.connections = ConnectionStore()
    state.approvals = ApprovalQueue()
    oidc_enabled = bool(os.environ.get("AGENT_BOM_OIDC_SECRET")
                        or os.environ.get("AGENT_BOM_OIDC_JWKS")
                        or os.environ.get("AGENT_BOM_OIDC_JWKS_URL"))

    def _resolve_role(request: Request, x_api_key: Optional[str]) -> Optional[str]:
        """API key (static table or SCIM-bound) > OIDC bearer > delegation
        token.  Returns None when no credential resolves."""
        if x_api_key:
            role = key_roles.get(x_api_key) or state.scim_users.role_for_key(x_api_key)
            if role:
                request.state.principal = f"key:{x_api_key[:8]}"
                return role
        authz = request.headers.get("Authorization") or ""
        if oidc_enabled and authz.startswith("Bearer "):
            try:
                claims = verify_oidc_bearer(authz[7:])
            except AuthError:
                return None
            request.state.principal = f"oidc:{claims.get('sub', '?')}"
            request.state.tenant_id = str(
                claims.get("tenant_id") or claims.get("tid") or "default")
            return role_from_claims(claims)
        dtok = request.headers.get("X-Delegation-Token")
        if dtok and state.delegation.enabled:
            try:
                payload = state.delegation.verify(dtok)
            except AuthError:
                return None
            request.state.principal = f"delegated:{payload.get('jti', '?')[:8]}"
            request.state.tenant_id = payload.get("tenant_id", "default")
            return payload.get("role", "viewer")
        return None

    def auth(request: Request,
             x_api_key: Optional[str] = Header(default=None)) -> None:
        if not key_roles and not oidc_enabled and not state.delegation.enabled:
            request.state.role = "admin"
            request.state.principal = "anonymous"
            return  # auth disabled (nothing configured)
        role = _resolve_role(request, x_api_key)
        if role is None:
            state.metrics["auth_failures_total"] += 1
            raise HTTPException(status_code=401,
                                detail="invalid or missing credential")
        if not _role_allows(role, request.method, request.url.path):
            state.metrics["auth_failures_total"] += 1
            raise HTTPException(
                status_code=403,
                detail=f"role {role!r} may not {request.method} {request.url.path}")
        request.state.role = role

    # ── health + metrics ───────────────────────────────────────────────────

    @app.get("/healthz")
    def healthz() -> dict:
        return {"status": "ok", "version": __version__}

    # ── OAuth 2.1 AS for the MCP auth spec (opt-in: AGENT_BOM_OAUTH_AS=1;
    #    key generation costs seconds, so the broker is lazy + gated) ──────

    def _oauth_as():
        if state.oauth_as is None:
            if os.environ.get("AGENT_BOM_OAUTH_AS", "0") != "1":
                raise HTTPException(status_code=404,
                                    detail="OAuth broker not enabled "
                                           "(AGENT_BOM_OAUTH_AS=1)")
            from agentbom_amd.api.oauth_as import OAuthAuthorizationServer

            issuer = os.environ.get("AGENT_BOM_OAUTH_ISSUER",
                                    "https://agent-bom.local")
            state.oauth_as = OAuthAuthorizationServer(issuer=issuer)
        return state.oauth_as

    @app.get("/.well-known/oauth-authorization-server")
    def oauth_metadata() -> dict:
        return _oauth_as().metadata()

    @app.get("/oauth/jwks.json")
    def oauth_jwks() -> dict:
        return _oauth_as().jwks()

    @app.post("/oauth/register", status_code=201)
    def oauth_register(payload: dict) -> dict:
        from agentbom_amd.api.oauth_as import OAuthError

        try:
            return _oauth_as().register_client(
                [str(u) for u in payload.get("redirect_uris") or []],
                client_name=str(payload.get("client_name", "")),
                confidential=(payload.get("token_endpoint_auth_method")
                              == "client_secret_basic"))
        except OAuthError as exc:
            raise HTTPException(status_code=exc.status, detail=exc.to_dict())

    @app.post("/oauth/authorize")
    def oauth_authorize(payload: dict) -> dict:
        """Programmatic consent (this build has no browser UI); returns
        the code + state for the client's redirect handling."""
        from agentbom_amd.api.oauth_as import OAuthError

        try:
            return _oauth_as().authorize(
                str(payload.get("client_id", "")),
                str(payload.get("redirect_uri", "")),
                str(payload.get("code_challenge", "")),
                str(payload.get("code_challenge_method", "")),
                scope=str(payload.get("scope", "mcp")),
                state=str(payload.get("state", "")))
        except OAuthError as exc:
            raise HTTPException(status_code=exc.status, detail=exc.to_dict())

    @app.post("/oauth/token")
    def oauth_token(payload: dict, request: Request) -> dict:
        import base64 as _b64

        from agentbom_amd.api.oauth_as import OAuthError

        srv = _oauth_as()
        grant = str(payload.get("grant_type", ""))
        try:
            if grant == "authorization_code":
                return srv.token_authorization_code(
                    str(payload.get("code", "")),
                    str(payload.get("client_id", "")),
                    str(payload.get("redirect_uri", "")),
                    str(payload.get("code_verifier", "")))
            if grant == "client_credentials":
                cid = str(payload.get("client_id", ""))
                secret = str(payload.get("client_secret", ""))
                authz = request.headers.get("Authorization", "")
                if authz.startswith("Basic "):
                    try:
                        raw = _b64.b64decode(authz[6:]).decode()
                        cid, _, secret = raw.partition(":")
                    except Exception:
                        pass
                return srv.token_client_credentials(cid, secret,
                                                    scope=str(payload.get(
                                                        "scope", "mcp")))
            raise OAuthError("unsupported_grant_type", grant)
        except OAuthError as exc:
            raise HTTPException(status_code=exc.status, detail=exc.to_dict())

    @app.get("/metrics")
    def metrics() -> Any:
        from fastapi.responses import PlainTextResponse

        lines = []
        for k, v in sorted(state.metrics.items()):
            lines.append(f"# TYPE agent_bom_{k} counter")
            lines.append(f"agent_bom_{k} {v}")
        return PlainTextResponse("\n".join(lines) + "\n")

    # ── scan pipeline ──────────────────────────────────────────────────────

    def _run_scan(job_id: str, req: ScanRequest) -> None:
        from agentbom_amd.graph.builder import build_unified_graph_from_report
        from agentbom_amd.graph.dependency_reach import (
            apply_dependency_reachability_to_blast_radii,
        )
        from agentbom_amd.output.json_fmt import to_json
        from agentbom_amd.scan.orchestrator import (
            ScanOptions,
            inventory_to_agents,
            run_demo_scan,
            scan_agents,
        )

        job = state.jobs[job_id]

        def _cancelled() -> bool:
            """Cooperative cancel checked at phase boundaries."""
            if job.get("cancel_requested"):
                job["status"] = "cancelled"
                job["steps"].append({"step": "cancelled", "at": _now()})
                return True
            return False

        try:
            job["status"] = "running"
            import time as _time

            job["started_monotonic"] = _time.monotonic()
            job["steps"].append({"step": "scan", "at": _now()})
            if _cancelled():
                return
            options = ScanOptions(demo=req.demo, offline=req.offline,
                                  blast_radius_depth=req.blast_radius_depth)
            if req.demo or not req.inventory:
                report = run_demo_scan(options)
            else:
                from agentbom_amd.db.store import load_advisory_windows

                agents = inventory_to_agents(req.inventory)
                report = scan_agents(agents, load_advisory_windows(offline=req.offline), options)
            report.scan_id = job_id
            if _cancelled():
                return
            job["steps"].append({"step": "graph_build", "at": _now()})
            graph = build_unified_graph_from_report(report)
            apply_dependency_reachability_to_blast_radii(report, graph)
            from agentbom_amd.graph.toxic_combos import (
                detect_toxic_combinations,
                toxic_combination_to_finding,
            )

            combos = detect_toxic_combinations(graph)
            report.findings = report.to_findings() + [
                toxic_combination_to_finding(c) for c in combos
            ]
            report.toxic_combination_findings_data = [c.to_dict() for c in combos]
            if _cancelled():
                return
            job["steps"].append({"step": "graph_persist", "at": _now()})
            snapshot_id = None
            tenant = state.scan_tenant.get(job_id, "default")
            if state.graph_store is not None:
                snapshot_id = state.graph_store.save_snapshot(
                    graph, scan_id=job_id, tenant_id=tenant)
            job["snapshot_id"] = snapshot_id
            with state.lock:
                state.reports[job_id] = report
                state.graphs[job_id] = graph
                state.latest_scan = job_id
                state.latest_by_tenant[tenant] = job_id
            job["result"] = {"summary": to_json(report)["summary"]}
            job["status"] = "done"
            job["steps"].append({"step": "done", "at": _now()})
            state.metrics["scans_total"] += 1
            state.webhooks.emit("scan.completed", {
                "job_id": job_id, "summary": job["result"]["summary"],
                "snapshot_id": snapshot_id})
        except Exception as exc:  # noqa: BLE001 — job boundary
            job["status"] = "failed"
            job["error"] = str(exc)
            state.webhooks.emit("scan.failed",
                                {"job_id": job_id, "error": str(exc)})
            state.metrics["scan_failures_total"] += 1

    @app.post("/v1/scan", status_code=201, dependencies=[Depends(auth)])
    def submit_scan(req: ScanRequest, request: Request,
                    idempotency_key: Optional[str] = Header(
                        default=None, alias="Idempotency-Key")) -> dict:
        principal = getattr(request.state, "principal", "anonymous")
        # Idempotency-Key: a retried POST with the same key returns the FIRST
        # submission's response instead of launching a duplicate scan
        # (reference api/idempotency_store.py semantics, per-tenant scoping)
        idem_scope = None
        if idempotency_key:
            idem_scope = (getattr(request.state, "tenant_id", "default"),
                          idempotency_key)
            with state.lock:
                cached = state.idempotency.get(idem_scope)
            if cached is not None:
                return cached
        ok, retry = state.quotas.check_and_record(principal)
        if not ok:
            raise HTTPException(status_code=429,
                                detail="scan quota exceeded for this principal",
                                headers={"Retry-After": str(int(retry))})
        admitted, retry_after = state.backpressure.try_acquire()
        if not admitted:
            from fastapi.responses import JSONResponse

            raise HTTPException(status_code=429, detail="scan concurrency limit reached",
                                headers={"Retry-After": str(retry_after)})
        tenant = getattr(request.state, "tenant_id", "default") or "default"
        try:
            out = _submit_scan_inner(req, tenant)
            if idem_scope is not None:
                with state.lock:
                    if len(state.idempotency) > 10_000:  # bounded replay cache
                        state.idempotency.clear()
                    state.idempotency[idem_scope] = out
            return out
        finally:
            state.backpressure.release()

    def _submit_scan_inner(req: ScanRequest, tenant: str = "default") -> dict:
        job_id = str(uuid.uuid4())
        state.scan_tenant[job_id] = tenant
        state.jobs[job_id] = {
            "id": job_id, "status": "pending", "submitted_at": _now(),
            "steps": [], "result": None, "error": None,
        }
        t = threading.Thread(target=_run_scan, args=(job_id, req), daemon=True)
        t.start()
        t.join(timeout=120)  # scans are fast; keep the API synchronous-ish
        return {"job_id": job_id, "status": state.jobs[job_id]["status"]}

    def reap_stuck_jobs(max_age_s: float = 600.0) -> list[str]:
        """Mark running/pending jobs past the deadline as failed (reaper).

        Reference: api/server.py stuck-job reaper — a crashed worker thread
        must not leave a job 'running' forever.  Called lazily on job reads
        and exposed for schedulers.
        """
        import time as _time

        now = _time.monotonic()
        reaped = []
        with state.lock:
            for job_id, job in state.jobs.items():
                if job["status"] not in ("running", "pending"):
                    continue
                started = job.get("started_monotonic")
                if started is not None and now - started > max_age_s:
                    job["status"] = "failed"
                    job["error"] = f"reaped: exceeded max runtime {max_age_s:.0f}s"
                    job["steps"].append({"step": "reaped", "at": _now()})
                    reaped.append(job_id)
        return reaped

    app.state.reap_stuck_jobs = reap_stuck_jobs

    @app.post("/v1/scan/{job_id}/cancel", dependencies=[Depends(auth)])
    def cancel_scan(job_id: str) -> dict:
        job = state.jobs.get(job_id)
        if not job:
            raise HTTPException(status_code=404, detail="scan job not found")
        if job["status"] in ("done", "failed", "cancelled"):
            return {"job_id": job_id, "status": job["status"],
                    "note": "job already finished"}
        job["cancel_requested"] = True
        return {"job_id": job_id, "status": job["status"],
                "cancel_requested": True}

    @app.get("/v1/scan/{job_id}", dependencies=[Depends(auth)])
    def get_scan(job_id: str) -> dict:
        reap_stuck_jobs(float(os.environ.get("AGENT_BOM_JOB_MAX_RUNTIME_S",
                                             "600")))
        job = state.jobs.get(job_id)
        if not job:
            raise HTTPException(status_code=404, detail="scan job not found")
        return job

    @app.get("/v1/scan/{job_id}/report", dependencies=[Depends(auth)])
    def get_scan_report(job_id: str) -> dict:
        from agentbom_amd.output.json_fmt import to_json

        report = state.reports.get(job_id)
        if report is None:
            raise HTTPException(status_code=404, detail="report not found")
        return json.loads(json.dumps(to_json(report), default=str))

    @app.get("/v1/findings", dependencies=[Depends(auth)])
    def findings(request: Request, severity: Optional[str] = None, limit: int = 100,
                 view: Optional[str] = None) -> dict:
        report = _latest_report(request)
        if view:
            from agentbom_amd.output import finding_views as fv

            views = {"by_severity": fv.by_severity, "by_package": fv.by_package,
                     "by_agent": fv.by_agent, "by_framework": fv.by_framework,
                     "compact": fv.to_compact}
            fn = views.get(view)
            if fn is None:
                raise HTTPException(status_code=400,
                                    detail=f"unknown view {view!r}; "
                                           f"one of {sorted(views)}")
            return fn(report)
        rows = [f.to_dict() for f in report.to_findings()]
        if severity:
            rows = [r for r in rows if r["severity"] == severity]
        return {"total": len(rows), "findings": rows[:limit]}

    # ── graph reads ────────────────────────────────────────────────────────

    def _tenant_of(request: Optional[Request]) -> str:
        if request is None:
            return "default"
        return getattr(request.state, "tenant_id", "default") or "default"

    def _latest_scan_for(request: Optional[Request]) -> str:
        """Tenant-scoped latest scan id (cross-tenant reads 404)."""
        tenant = _tenant_of(request)
        sid = state.latest_by_tenant.get(tenant)
        if sid is None and tenant == "default":
            sid = state.latest_scan  # untenanted/legacy scans
        if sid is None:
            raise HTTPException(status_code=404,
                                detail="no scan yet for this tenant — POST /v1/scan first")
        return sid

    def _latest_report(request: Optional[Request] = None):
        return state.reports[_latest_scan_for(request)]

    def _latest_graph(request: Optional[Request] = None):
        sid = _latest_scan_for(request)
        state.metrics["graph_queries_total"] += 1
        return state.graphs[sid]

    @app.get("/v1/graph", dependencies=[Depends(auth)])
    def graph_summary(request: Request, limit: int = 100) -> dict:
        g = _latest_graph(request)
        node_ids = sorted(g.nodes)[:limit]
        return {
            "node_count": g.node_count,
            "edge_count": g.edge_count,
            "completeness": g.completeness(),
            "nodes": [g.nodes[n].to_dict() for n in node_ids],
        }

    @app.get("/v1/graph/search", dependencies=[Depends(auth)])
    def graph_search(request: Request, q: str = "", entity_type: Optional[str] = None, limit: int = 100) -> dict:
        from agentbom_amd.graph.types import EntityType

        g = _latest_graph(request)
        et = [EntityType(entity_type)] if entity_type else None
        nodes = g.search(query=q, entity_types=et, limit=limit)
        return {"total": len(nodes), "nodes": [n.to_dict() for n in nodes]}

    @app.get("/v1/graph/node/{node_id:path}/neighbors", dependencies=[Depends(auth)])
    def graph_neighbors(request: Request, node_id: str, direction: str = "both") -> dict:
        g = _latest_graph(request)
        if node_id not in g.nodes:
            raise HTTPException(status_code=404, detail="node not found")
        return {
            "id": node_id,
            "neighbors": [g.nodes[n].to_dict() for n in g.neighbors(node_id, direction)],
        }

    @app.get("/v1/graph/node/{node_id:path}/impact", dependencies=[Depends(auth)])
    def graph_node_impact(request: Request, node_id: str, max_hops: int = 4) -> dict:
        """Bounded downstream impact of one node (container.impact_of)."""
        g = _latest_graph(request)
        if node_id not in g.nodes:
            raise HTTPException(status_code=404, detail="node not found")
        return g.impact_of(node_id, max_hops=min(max_hops, 8))

    @app.get("/v1/graph/node/{node_id:path}", dependencies=[Depends(auth)])
    def graph_node(request: Request, node_id: str) -> dict:
        g = _latest_graph(request)
        if node_id not in g.nodes:
            raise HTTPException(status_code=404, detail="node not found")
        edges = [e.to_dict() for e in g.edges
                 if e.source == node_id or e.target == node_id][:200]
        return {"node": g.nodes[node_id].to_dict(), "edges": edges}

    @app.get("/v1/graph/centrality", dependencies=[Depends(auth)])
    def graph_centrality(request: Request, top_n: int = 20) -> dict:
        g = _latest_graph(request)
        return {"centrality": [{"id": nid, "degree": deg}
                               for nid, deg in g.degree_centrality(top_n)]}

    @app.get("/v1/graph/bottlenecks", dependencies=[Depends(auth)])
    def graph_bottlenecks(request: Request, top_n: int = 10,
                          sample: int = 64) -> dict:
        """Approx-betweenness choke points (lateral-movement bottlenecks)."""
        g = _latest_graph(request)
        rows = g.bottlenecks(top_n=top_n, sample=min(sample, 256))
        return {"bottlenecks": [{"id": nid, "score": round(score, 4)}
                                for nid, score in rows]}

    @app.get("/v1/graph/view/{view_name}", dependencies=[Depends(auth)])
    def graph_view(request: Request, view_name: str) -> dict:
        """Typed subgraph views: inventory / attack-path / lateral /
        compliance / runtime."""
        g = _latest_graph(request)
        try:
            return g.view(view_name)
        except (KeyError, ValueError) as exc:
            raise HTTPException(status_code=404, detail=str(exc))

    @app.get("/v1/graph/paths", dependencies=[Depends(auth)])
    def graph_paths(request: Request, source: Optional[str] = None, target: Optional[str] = None,
                    limit: int = 25) -> dict:
        """Attack-path / blast-radius drilldown (the BASELINE p50 metric)."""
        g = _latest_graph(request)
        if source and target:
            path = g.shortest_path(source, target)
            return {"paths": [path] if path else []}
        from agentbom_amd.graph.attack_paths import compute_fused_attack_paths

        paths = compute_fused_attack_paths(g, max_paths=limit)
        return {"path_count": len(paths), "paths": [p.to_dict() for p in paths]}

    @app.get("/v1/graph/attack-paths", dependencies=[Depends(auth)])
    def graph_attack_paths(request: Request, limit: int = 25) -> dict:
        return graph_paths(request, limit=limit)

    @app.get("/v1/graph/exposure-paths", dependencies=[Depends(auth)])
    def graph_exposure_paths(request: Request, limit: int = 50) -> dict:
        from agentbom_amd.models import blast_radius_to_finding
        from agentbom_amd.output.exposure_path import exposure_path_for_finding

        report = _latest_report(request)
        paths = [
            exposure_path_for_finding(blast_radius_to_finding(br), rank=i + 1)
            for i, br in enumerate(report.blast_radii[:limit])
        ]
        return {"schema_version": "1", "source": "blast_radius_output",
                "path_count": len(paths), "paths": paths}

    @app.post("/v1/graph/query", dependencies=[Depends(auth)])
    def graph_query(request: Request, req: GraphQueryRequest) -> dict:
        g = _latest_graph(request)
        if req.start not in g.nodes:
            raise HTTPException(status_code=404, detail="start node not found")
        return g.traverse_subgraph(req.start, max_depth=req.max_depth, max_nodes=req.max_nodes)

    @app.get("/v1/graph/impact/{node_id:path}", dependencies=[Depends(auth)])
    def graph_impact(request: Request, node_id: str, max_hops: int = 4) -> dict:
        g = _latest_graph(request)
        if node_id not in g.nodes:
            raise HTTPException(status_code=404, detail="node not found")
        return g.impact_of(node_id, max_hops=min(max_hops, 4))

    @app.get("/v1/graph/rollup", dependencies=[Depends(auth)])
    def graph_rollup(request: Request) -> dict:
        from agentbom_amd.graph.rollup import rollup_view

        return rollup_view(_latest_graph(request))

    @app.get("/v1/graph/rollup/{container_id:path}", dependencies=[Depends(auth)])
    def graph_drilldown(request: Request, container_id: str) -> dict:
        from agentbom_amd.graph.rollup import drill_down

        return drill_down(_latest_graph(request), container_id)

    @app.get("/v1/graph/should-i-deploy", dependencies=[Depends(auth)])
    def should_i_deploy(request: Request) -> dict:
        from agentbom_amd.utils import config as cfg

        report = _latest_report(request)
        max_risk = max((br.risk_score for br in report.blast_radii), default=0.0) * 10
        has_malicious = any(br.package.is_malicious for br in report.blast_radii)
        has_kev = any(br.vulnerability.is_kev for br in report.blast_radii)
        if has_malicious or max_risk >= cfg.DEPLOY_BLOCK_RISK:
            verdict = "block"
        elif has_kev or max_risk >= cfg.DEPLOY_WARN_RISK:
            verdict = "warn"
        else:
            verdict = "allow"
        return {
            "verdict": verdict,
            "max_risk": max_risk,
            "has_malicious": has_malicious,
            "has_kev": has_kev,
            "warn_threshold": cfg.DEPLOY_WARN_RISK,
            "block_threshold": cfg.DEPLOY_BLOCK_RISK,
        }

    @app.get("/v1/graph/snapshots", dependencies=[Depends(auth)])
    def graph_snapshots() -> dict:
        if state.graph_store is None:
            return {"snapshots": [], "note": "set AGENT_BOM_GRAPH_STORE to persist snapshots"}
        return {"snapshots": state.graph_store.list_snapshots()}

    @app.get("/v1/graph/diff", dependencies=[Depends(auth)])
    def graph_diff(old: Optional[str] = None, new: Optional[str] = None) -> dict:
        if state.graph_store is None:
            raise HTTPException(status_code=404, detail="no graph store configured")
        snaps = state.graph_store.list_snapshots()
        if len(snaps) < 2 and not (old and new):
            raise HTTPException(status_code=404, detail="need two snapshots to diff")
        new = new or snaps[0]["snapshot_id"]
        old = old or snaps[1]["snapshot_id"]
        return state.graph_store.diff_snapshots(old, new)

    @app.post("/v1/findings/lifecycle/observe", dependencies=[Depends(auth)])
    def lifecycle_observe(request: Request) -> dict:
        """Fold the latest scan into the finding-lifecycle tracker
        (first_seen / resolved / reopened / MTTR)."""
        from agentbom_amd.output.json_fmt import to_json

        if state.lifecycle is None:
            from agentbom_amd.scan.history import AssetTracker

            state.lifecycle = AssetTracker(":memory:")
        delta = state.lifecycle.observe_scan(to_json(_latest_report(request)))
        return delta

    @app.get("/v1/findings/lifecycle", dependencies=[Depends(auth)])
    def lifecycle_summary() -> dict:
        if state.lifecycle is None:
            return {"observed_scans": 0, "mttr_seconds": None}
        mttr = state.lifecycle.mttr_seconds()
        rows = state.lifecycle.conn.execute(
            "SELECT COUNT(*), SUM(resolved_at IS NOT NULL) FROM"
            " finding_lifecycle").fetchone()
        return {"tracked_findings": int(rows[0] or 0),
                "resolved": int(rows[1] or 0),
                "mttr_seconds": round(mttr, 1) if mttr is not None else None}

    @app.get("/v1/findings/reach", dependencies=[Depends(auth)])
    def findings_reach(request: Request, band: Optional[str] = None) -> dict:
        """Effective-reach triage view: composite + band per finding."""
        from agentbom_amd.graph.effective_reach import effective_reach_summary

        out = effective_reach_summary(_latest_report(request))
        if band:
            out["findings"] = [r for r in out["findings"] if r["band"] == band]
        return out

    @app.get("/v1/findings/toxic-combinations", dependencies=[Depends(auth)])
    def toxic_combinations(request: Request) -> dict:
        report = _latest_report(request)
        return {"combinations": report.toxic_combination_findings_data or []}

    @app.get("/v1/graph/evidence-manifest", dependencies=[Depends(auth)])
    def graph_evidence_manifest(request: Request) -> dict:
        import hashlib

        if state.graph_store is not None:
            return state.graph_store.evidence_manifest(scan_id=state.latest_scan)
        g = _latest_graph(request)
        digest = hashlib.sha256(
            json.dumps(g.to_dict(), sort_keys=True, default=str).encode()
        ).hexdigest()
        return {
            "schema_version": "agent-bom.graph_evidence_manifest/v1",
            "tenant_id": "default",
            "scan_id": state.latest_scan,
            "generated_at": _now(),
            "graph_digest": digest,
            "counts": {"nodes": g.node_count, "edges": g.edge_count},
            "included_tables": ["graph_nodes", "graph_edges"],
            "excluded_private_fields": ["env_values", "credential_values"],
            "retention_policy": {"snapshots": 10},
        }

    # ── fleet / scheduler / audit ─────────────────────────────────────────

    @app.post("/v1/fleet/heartbeat", dependencies=[Depends(auth)])
    def fleet_heartbeat(payload: dict) -> dict:
        return state.fleet.heartbeat(payload).to_dict()

    @app.get("/v1/fleet", dependencies=[Depends(auth)])
    def fleet_list() -> dict:
        return {"members": state.fleet.list_members(),
                "reconciliation": state.fleet.reconcile()}

    @app.post("/v1/schedules", status_code=201, dependencies=[Depends(auth)])
    def create_schedule(payload: dict) -> dict:
        from agentbom_amd.api.fleet import ScanScheduler

        if state.scheduler is None:
            state.scheduler = ScanScheduler(
                run_scan=lambda p: _run_scan(str(uuid.uuid4()), ScanRequest(demo=True)))
            state.scheduler.start()
        sched = state.scheduler.add(
            payload.get("schedule_id") or str(uuid.uuid4()),
            float(payload.get("interval_s", 3600)),
            demo=bool(payload.get("demo", True)),
        )
        return sched.to_dict()

    @app.get("/v1/schedules", dependencies=[Depends(auth)])
    def list_schedules() -> dict:
        if state.scheduler is None:
            return {"schedules": []}
        return {"schedules": [s.to_dict() for s in state.scheduler.schedules.values()]}

    @app.delete("/v1/schedules/{schedule_id}", status_code=204, dependencies=[Depends(auth)])
    def delete_schedule(schedule_id: str) -> None:
        if state.scheduler is None or not state.scheduler.remove(schedule_id):
            raise HTTPException(status_code=404, detail="schedule not found")

    def _exports(request: Request):
        if state.exports is None:
            from agentbom_amd.api.exports import ExportManager

            # resolve the DEFAULT tenant's latest report at delivery time
            # (binding the creating request would freeze its tenant forever
            # and leak the request object into the scheduler thread)
            state.exports = ExportManager(
                get_report=lambda: _latest_report(None),
                file_root=os.environ.get("AGENT_BOM_EXPORT_ROOT", "."))
        return state.exports

    @app.post("/v1/exports/destinations", status_code=201,
              dependencies=[Depends(auth)])
    def add_export_destination(request: Request, payload: dict) -> dict:
        from agentbom_amd.api.exports import ExportDestination

        if not payload.get("target"):
            raise HTTPException(status_code=400, detail="target required")
        try:
            dest = ExportDestination(
                target=str(payload["target"]),
                format=str(payload.get("format", "json")),
                name=str(payload.get("name", "")),
                tenant_id=_tenant_of(request))
        except ValueError as exc:
            raise HTTPException(status_code=400, detail=str(exc))
        return _exports(request).add_destination(dest).to_dict()

    @app.get("/v1/exports/destinations", dependencies=[Depends(auth)])
    def list_export_destinations(request: Request) -> dict:
        mgr = _exports(request)
        return {"destinations": [d.to_dict()
                                 for d in mgr.destinations.values()]}

    @app.post("/v1/exports/schedules", status_code=201,
              dependencies=[Depends(auth)])
    def add_export_schedule(request: Request, payload: dict) -> dict:
        mgr = _exports(request)
        sched = mgr.add_schedule(str(payload.get("destination_id", "")),
                                 float(payload.get("interval_s", 3600)))
        if sched is None:
            raise HTTPException(status_code=404, detail="unknown destination")
        mgr.start()
        return sched.to_dict()

    @app.get("/v1/exports/schedules", dependencies=[Depends(auth)])
    def list_export_schedules(request: Request) -> dict:
        mgr = _exports(request)
        return {"schedules": [s.to_dict() for s in mgr.schedules.values()],
                "recent_deliveries": mgr.deliveries[-20:]}

    @app.post("/v1/exports/run/{destination_id}", dependencies=[Depends(auth)])
    def run_export_now(request: Request, destination_id: str) -> dict:
        out = _exports(request).run_export(destination_id)
        if not out.get("ok") and out.get("error") == "unknown destination":
            raise HTTPException(status_code=404, detail="unknown destination")
        return out

    def _model_keys(request: Request):
        if state.model_keys is None:
            from agentbom_amd.api.model_keys import ModelKeyBroker

            state.model_keys = ModelKeyBroker()
        return state.model_keys

    @app.post("/v1/model-keys/providers", status_code=201,
              dependencies=[Depends(auth)])
    def register_model_provider_key(request: Request, payload: dict) -> dict:
        from agentbom_amd.api.connections import ConnectionsCryptoUnavailable

        for k in ("provider", "key"):
            if not payload.get(k):
                raise HTTPException(status_code=400, detail=f"{k} required")
        try:
            rec = _model_keys(request).register_provider_key(
                str(payload["provider"]), str(payload["key"]),
                tenant_id=_tenant_of(request),
                label=str(payload.get("label", "")))
        except ConnectionsCryptoUnavailable as exc:
            raise HTTPException(status_code=409, detail=str(exc))
        return rec.to_public_dict()

    @app.post("/v1/model-keys/virtual", status_code=201,
              dependencies=[Depends(auth)])
    def mint_virtual_model_key(request: Request, payload: dict) -> dict:
        from agentbom_amd.api.model_keys import ModelKeyBrokerError

        try:
            rec, raw = _model_keys(request).mint_virtual_key(
                str(payload.get("provider_key_id", "")),
                tenant_id=_tenant_of(request),
                holder=str(payload.get("holder", "")),
                model_allowlist=[str(m) for m in
                                 payload.get("model_allowlist") or []],
                ttl_hours=float(payload.get("ttl_hours", 24.0)))
        except ModelKeyBrokerError as exc:
            raise HTTPException(status_code=404, detail=str(exc))
        return {**rec.to_public_dict(), "virtual_key": raw}  # raw shown ONCE

    @app.get("/v1/model-keys/virtual", dependencies=[Depends(auth)])
    def list_virtual_model_keys(request: Request) -> dict:
        return {"virtual_keys":
                _model_keys(request).list_virtual_keys(_tenant_of(request))}

    @app.post("/v1/model-keys/virtual/{virtual_key_id}/revoke",
              dependencies=[Depends(auth)])
    def revoke_virtual_model_key(request: Request,
                                 virtual_key_id: str) -> dict:
        if not _model_keys(request).revoke_virtual_key(
                virtual_key_id, tenant_id=_tenant_of(request)):
            raise HTTPException(status_code=404, detail="unknown virtual key")
        return {"revoked": virtual_key_id}

    @app.get("/v1/approvals", dependencies=[Depends(auth)])
    def list_approvals(status: Optional[str] = None) -> dict:
        rows = state.approvals.list(status=status)
        return {"total": len(rows), "approvals": [r.to_dict() for r in rows]}

    @app.post("/v1/approvals/{request_id}/approve", dependencies=[Depends(auth)])
    def approve_request(request: Request, request_id: str,
                        payload: Optional[dict] = None) -> dict:
        req = state.approvals.decide(
            request_id, approve=True,
            actor=getattr(request.state, "principal", "?"),
            reason=str((payload or {}).get("reason", "")))
        if req is None:
            raise HTTPException(status_code=409,
                                detail="not pending (unknown/expired/decided)")
        return req.to_dict()

    @app.post("/v1/approvals/{request_id}/deny", dependencies=[Depends(auth)])
    def deny_request(request: Request, request_id: str,
                     payload: Optional[dict] = None) -> dict:
        req = state.approvals.decide(
            request_id, approve=False,
            actor=getattr(request.state, "principal", "?"),
            reason=str((payload or {}).get("reason", "")))
        if req is None:
            raise HTTPException(status_code=409,
                                detail="not pending (unknown/expired/decided)")
        return req.to_dict()

    @app.post("/v1/connections", status_code=201, dependencies=[Depends(auth)])
    def create_connection(request: Request, payload: dict) -> dict:
        from agentbom_amd.api.connections import (
            CloudConnection,
            ConnectionsCryptoUnavailable,
            encrypt_secret,
            validate_credential_ref,
        )

        for key in ("provider", "display_name"):
            if not payload.get(key):
                raise HTTPException(status_code=400, detail=f"{key} required")
        secret_encrypted = ""
        if payload.get("secret"):
            try:
                secret_encrypted = encrypt_secret(str(payload["secret"]))
            except ConnectionsCryptoUnavailable as exc:
                raise HTTPException(status_code=409, detail=str(exc))
        status, detail = validate_credential_ref(
            str(payload["provider"]), str(payload.get("mode", "")),
            str(payload.get("role_ref", "")))
        conn = CloudConnection(
            provider=str(payload["provider"]),
            display_name=str(payload["display_name"]),
            tenant_id=_tenant_of(request),
            role_ref=str(payload.get("role_ref", "")),
            secret_encrypted=secret_encrypted,
            auth_params=dict(payload.get("auth_params") or {}),
            regions=[str(r) for r in payload.get("regions") or []],
            status="pending" if status == "ok" else "degraded",
            status_detail=detail,
            scan_interval_minutes=(int(payload["scan_interval_minutes"])
                                   if payload.get("scan_interval_minutes")
                                   else None))
        return state.connections.put(conn).to_public_dict()

    @app.get("/v1/connections", dependencies=[Depends(auth)])
    def list_connections(request: Request, due: bool = False) -> dict:
        tenant = _tenant_of(request)
        rows = (state.connections.due_connections(tenant) if due
                else state.connections.list(tenant))
        return {"total": len(rows),
                "connections": [c.to_public_dict() for c in rows]}

    @app.delete("/v1/connections/{connection_id}", status_code=204,
                dependencies=[Depends(auth)])
    def delete_connection(request: Request, connection_id: str) -> None:
        if not state.connections.delete(_tenant_of(request), connection_id):
            raise HTTPException(status_code=404, detail="connection not found")

    @app.post("/v1/connections/{connection_id}/scan",
              dependencies=[Depends(auth)])
    def scan_connection(request: Request, connection_id: str) -> dict:
        """Mark a connection scanned (the demo-scan pipeline stands in for
        the provider collector in this build)."""
        out = _submit_scan_inner(ScanRequest(demo=True),
                                 tenant=_tenant_of(request))
        job_id = out["job_id"]
        conn = state.connections.mark_scanned(
            _tenant_of(request), connection_id, job_id)
        if conn is None:
            raise HTTPException(status_code=404, detail="connection not found")
        return {"job_id": job_id, "connection": conn.to_public_dict()}

    @app.post("/v1/blueprints", status_code=201, dependencies=[Depends(auth)])
    def create_blueprint(request: Request, payload: dict) -> dict:
        from agentbom_amd.api.blueprints import BlueprintComposition

        if not payload.get("name"):
            raise HTTPException(status_code=400, detail="name required")
        bp = state.blueprints.create(
            _tenant_of(request), str(payload["name"]),
            BlueprintComposition.from_dict(payload.get("composition") or {}),
            author=getattr(request.state, "principal", "?"),
            owner=str(payload.get("owner", "")),
            seeded_from=str(payload.get("seeded_from", "")))
        return bp.to_dict()

    @app.get("/v1/blueprints", dependencies=[Depends(auth)])
    def list_blueprints(request: Request) -> dict:
        rows = state.blueprints.list(_tenant_of(request))
        return {"total": len(rows), "blueprints": [b.to_dict() for b in rows]}

    @app.get("/v1/blueprints/{blueprint_id}/versions", dependencies=[Depends(auth)])
    def blueprint_versions(request: Request, blueprint_id: str) -> dict:
        if state.blueprints.get(_tenant_of(request), blueprint_id) is None:
            raise HTTPException(status_code=404, detail="blueprint not found")
        return {"versions": [v.to_dict() for v in
                             state.blueprints.list_versions(blueprint_id)]}

    @app.post("/v1/blueprints/{blueprint_id}/versions", status_code=201,
              dependencies=[Depends(auth)])
    def blueprint_new_draft(request: Request, blueprint_id: str,
                            payload: dict) -> dict:
        from agentbom_amd.api.blueprints import BlueprintComposition

        v = state.blueprints.create_draft(
            _tenant_of(request), blueprint_id,
            BlueprintComposition.from_dict(payload.get("composition") or {}),
            author=getattr(request.state, "principal", "?"))
        if v is None:
            raise HTTPException(status_code=404, detail="blueprint not found")
        return v.to_dict()

    @app.post("/v1/blueprints/{blueprint_id}/versions/{version}/submit",
              dependencies=[Depends(auth)])
    def blueprint_submit(blueprint_id: str, version: int) -> dict:
        v = state.blueprints.submit(blueprint_id, version)
        if v is None:
            raise HTTPException(status_code=409, detail="not a draft version")
        return v.to_dict()

    @app.post("/v1/blueprints/{blueprint_id}/versions/{version}/approve",
              dependencies=[Depends(auth)])
    def blueprint_approve(request: Request, blueprint_id: str, version: int,
                          payload: Optional[dict] = None) -> dict:
        from agentbom_amd.api.blueprints import BlueprintApprovalError

        try:
            v = state.blueprints.approve(
                _tenant_of(request), blueprint_id, version,
                approver=getattr(request.state, "principal", ""),
                note=str((payload or {}).get("note", "")))
        except BlueprintApprovalError as exc:
            raise HTTPException(status_code=409, detail=str(exc))
        return v.to_dict()

    @app.get("/v1/blueprints/{blueprint_id}/diff", dependencies=[Depends(auth)])
    def blueprint_diff(blueprint_id: str, from_version: int,
                       to_version: int) -> dict:
        from agentbom_amd.api.blueprints import diff_versions

        d = diff_versions(state.blueprints, blueprint_id, from_version,
                          to_version)
        if d is None:
            raise HTTPException(status_code=404, detail="version not found")
        return d

    @app.post("/v1/blueprints/{blueprint_id}/drift", dependencies=[Depends(auth)])
    def blueprint_drift(request: Request, blueprint_id: str) -> dict:
        """Evaluate the approved composition against the latest scan;
        opens incidents for every deviation."""
        from agentbom_amd.api.blueprints import evaluate_drift

        tenant = _tenant_of(request)
        bp = state.blueprints.get(tenant, blueprint_id)
        if bp is None:
            raise HTTPException(status_code=404, detail="blueprint not found")
        if bp.current_approved_version == 0:
            raise HTTPException(status_code=409, detail="no approved version")
        version = state.blueprints.get_version(blueprint_id,
                                               bp.current_approved_version)
        report = _latest_report(request)
        incidents = [state.blueprints.record_incident(i)
                     for i in evaluate_drift(bp, version, report)]
        return {"incidents_opened": len(incidents), "incidents": incidents}

    @app.get("/v1/drift-incidents", dependencies=[Depends(auth)])
    def list_drift_incidents(request: Request,
                             status: Optional[str] = None) -> dict:
        rows = state.blueprints.list_incidents(_tenant_of(request), status)
        return {"total": len(rows), "incidents": rows}

    @app.post("/v1/drift-incidents/{incident_id}/resolve",
              dependencies=[Depends(auth)])
    def resolve_drift_incident(request: Request, incident_id: str,
                               payload: Optional[dict] = None) -> dict:
        doc = state.blueprints.resolve_incident(
            incident_id, actor=getattr(request.state, "principal", "?"),
            note=str((payload or {}).get("note", "")))
        if doc is None:
            raise HTTPException(status_code=404, detail="incident not found")
        return doc

    @app.post("/v1/exceptions", status_code=201, dependencies=[Depends(auth)])
    def request_exception(request: Request, payload: dict) -> dict:
        from agentbom_amd.api.exceptions_store import VulnException

        for key in ("vuln_id", "package_name", "reason"):
            if not payload.get(key):
                raise HTTPException(status_code=400, detail=f"{key} required")
        exc = VulnException(
            vuln_id=str(payload["vuln_id"]),
            package_name=str(payload["package_name"]),
            server_name=str(payload.get("server_name", "")),
            reason=str(payload["reason"]),
            requested_by=getattr(request.state, "principal", "anonymous"),
            expires_at=str(payload.get("expires_at", "")),
            tenant_id=_tenant_of(request))
        return state.exceptions.request(exc).to_dict()

    @app.get("/v1/exceptions", dependencies=[Depends(auth)])
    def list_exceptions(request: Request, status: Optional[str] = None) -> dict:
        rows = state.exceptions.list(_tenant_of(request), status=status)
        return {"total": len(rows), "exceptions": [e.to_dict() for e in rows]}

    @app.post("/v1/exceptions/{exception_id}/approve", dependencies=[Depends(auth)])
    def approve_exception(request: Request, exception_id: str,
                          payload: Optional[dict] = None) -> dict:
        payload = payload or {}
        exc = state.exceptions.approve(
            exception_id, actor=getattr(request.state, "principal", "?"),
            reason=str(payload.get("reason", "")),
            tenant_id=_tenant_of(request),
            ttl_days=float(payload.get("ttl_days", 90.0)))
        if exc is None:
            raise HTTPException(status_code=409,
                                detail="unknown exception or invalid transition")
        return exc.to_dict()

    @app.post("/v1/exceptions/{exception_id}/reject", dependencies=[Depends(auth)])
    def reject_exception(request: Request, exception_id: str,
                         payload: Optional[dict] = None) -> dict:
        exc = state.exceptions.reject(
            exception_id, actor=getattr(request.state, "principal", "?"),
            reason=str((payload or {}).get("reason", "")),
            tenant_id=_tenant_of(request))
        if exc is None:
            raise HTTPException(status_code=409,
                                detail="unknown exception or invalid transition")
        return exc.to_dict()

    @app.post("/v1/exceptions/{exception_id}/revoke", dependencies=[Depends(auth)])
    def revoke_exception(request: Request, exception_id: str,
                         payload: Optional[dict] = None) -> dict:
        exc = state.exceptions.revoke(
            exception_id, actor=getattr(request.state, "principal", "?"),
            reason=str((payload or {}).get("reason", "")),
            tenant_id=_tenant_of(request))
        if exc is None:
            raise HTTPException(status_code=409,
                                detail="unknown exception or invalid transition")
        return exc.to_dict()

    @app.get("/v1/exceptions/audit", dependencies=[Depends(auth)])
    def exceptions_audit() -> dict:
        return {"chain_valid": state.exceptions.audit_chain_valid()}

    def _datasets(request: Request):
        if state.datasets is None:
            from agentbom_amd.api.dataset_versions import DatasetVersionStore

            state.datasets = DatasetVersionStore()
        return state.datasets

    @app.post("/v1/datasets/{dataset_id}/versions", status_code=201,
              dependencies=[Depends(auth)])
    def register_dataset_version(request: Request, dataset_id: str,
                                 payload: dict) -> dict:
        from agentbom_amd.api.dataset_versions import (
            DatasetVersionConflict,
            DatasetVersionRecord,
        )

        if not payload.get("version_id"):
            raise HTTPException(status_code=400, detail="version_id required")
        rec = DatasetVersionRecord(
            tenant_id=_tenant_of(request), dataset_id=dataset_id,
            version_id=str(payload["version_id"]),
            source=str(payload.get("source", "")),
            artifact_uri=payload.get("artifact_uri"),
            digest=payload.get("digest"),
            digest_algorithm=str(payload.get("digest_algorithm", "sha256")),
            metadata=dict(payload.get("metadata") or {}))
        try:
            return _datasets(request).put(rec).to_dict()
        except DatasetVersionConflict as exc:
            raise HTTPException(status_code=409, detail=str(exc))

    @app.get("/v1/datasets/{dataset_id}/versions", dependencies=[Depends(auth)])
    def list_dataset_versions(request: Request, dataset_id: str) -> dict:
        rows = _datasets(request).list(_tenant_of(request), dataset_id)
        return {"total": len(rows), "versions": [r.to_dict() for r in rows]}

    @app.post("/v1/datasets/{dataset_id}/versions/{version_id}/verify",
              dependencies=[Depends(auth)])
    def verify_dataset_version(request: Request, dataset_id: str,
                               version_id: str, payload: dict) -> dict:
        if not payload.get("digest"):
            raise HTTPException(status_code=400, detail="digest required")
        return _datasets(request).verify(
            _tenant_of(request), dataset_id, version_id,
            str(payload["digest"]))

    def _evals(request: Request):
        if state.evaluations is None:
            from agentbom_amd.api.evaluations import EvaluationStore

            state.evaluations = EvaluationStore(
                dataset_store=_datasets(request))
        return state.evaluations

    @app.post("/v1/evaluations", status_code=201, dependencies=[Depends(auth)])
    def record_evaluation(request: Request, payload: dict) -> dict:
        from agentbom_amd.api.evaluations import EvaluationRun

        if not payload.get("name"):
            raise HTTPException(status_code=400, detail="name required")
        run = EvaluationRun(
            tenant_id=_tenant_of(request), name=str(payload["name"]),
            status=str(payload.get("status", "completed")),
            dataset_id=payload.get("dataset_id"),
            dataset_version_id=payload.get("dataset_version_id"),
            model=payload.get("model"),
            prompt_hash=payload.get("prompt_hash"),
            scores={str(k): float(v) for k, v in
                    (payload.get("scores") or {}).items()},
            cases=list(payload.get("cases") or [])[:1000],
            metadata=dict(payload.get("metadata") or {}))
        try:
            return _evals(request).put(run).to_dict()
        except ValueError as exc:
            raise HTTPException(status_code=409, detail=str(exc))

    @app.get("/v1/evaluations", dependencies=[Depends(auth)])
    def list_evaluations(request: Request, name: Optional[str] = None) -> dict:
        rows = _evals(request).list(_tenant_of(request), name=name)
        return {"total": len(rows), "runs": [r.to_dict() for r in rows]}

    @app.get("/v1/evaluations/compare/{name}", dependencies=[Depends(auth)])
    def compare_evaluations(request: Request, name: str) -> dict:
        return _evals(request).compare(_tenant_of(request), name)

    @app.get("/v1/audit/verify", dependencies=[Depends(auth)])
    def audit_verify() -> dict:
        """Consolidated audit-integrity verdict: every hash chain the
        control plane maintains, verified in one read (a broken chain
        anywhere fails the whole verdict — tamper evidence is only as
        strong as its weakest ledger)."""
        chains = {"exceptions": state.exceptions.audit_chain_valid()}
        if state.identity_store is not None:
            chains["identities"] = state.identity_store.audit_chain_valid()
        return {"chains": chains, "all_valid": all(chains.values()),
                "ingested_proxy_entries": len(state.audit_entries)}

    @app.post("/v1/costs/records", status_code=201, dependencies=[Depends(auth)])
    def ingest_costs(request: Request, payload: dict) -> dict:
        """Batch LLM cost ingest (OTel GenAI span shape)."""
        from agentbom_amd.api.cost_store import LLMCostRecord

        rows = payload.get("records", [])
        if not isinstance(rows, list) or not rows:
            raise HTTPException(status_code=400, detail="records must be a non-empty list")
        tenant = _tenant_of(request)
        recs = []
        for row in rows[:10_000]:
            if not isinstance(row, dict) or "agent" not in row:
                raise HTTPException(status_code=400,
                                    detail="each record needs at least agent+cost_usd")
            recs.append(LLMCostRecord(
                tenant_id=tenant, agent=str(row["agent"]),
                cost_usd=float(row.get("cost_usd", 0.0)),
                model=str(row.get("model", "")),
                tokens_in=int(row.get("tokens_in", 0)),
                tokens_out=int(row.get("tokens_out", 0)),
                observed_at=str(row.get("observed_at", "")),
                cost_center=str(row.get("cost_center", "")),
                tags={str(k): str(v) for k, v in
                      (row.get("tags") or {}).items()}))
        return {"ingested": state.costs.add_records(recs)}

    @app.get("/v1/costs/summary", dependencies=[Depends(auth)])
    def cost_summary(request: Request, agent: Optional[str] = None,
                     since: Optional[str] = None) -> dict:
        from agentbom_amd.api.cost_store import budget_status, summarize

        tenant = _tenant_of(request)
        records = state.costs.list_records(tenant, agent=agent, since=since)
        out = summarize(records)
        budget = state.costs.get_budget(tenant, agent=agent or "")
        out["budget"] = budget_status(
            sum(r.cost_usd for r in records), budget)
        return out

    @app.get("/v1/costs/forecast", dependencies=[Depends(auth)])
    def cost_forecast(request: Request, agent: Optional[str] = None) -> dict:
        from agentbom_amd.api.cost_store import forecast_spend

        tenant = _tenant_of(request)
        records = state.costs.list_records(tenant, agent=agent)
        budget = state.costs.get_budget(tenant, agent=agent or "") \
            or state.costs.get_budget(tenant)
        return forecast_spend(records, budget)

    @app.get("/v1/costs/anomalies", dependencies=[Depends(auth)])
    def cost_anomalies(request: Request,
                       z_threshold: float = 3.5) -> dict:
        """Robust (median+MAD) per-agent spend outliers + hourly EWMA
        spikes over the stored cost records."""
        from agentbom_amd.api.anomaly import (
            detect_cost_anomalies,
            detect_temporal_cost_anomalies,
        )

        records = state.costs.list_records(_tenant_of(request))
        spend: dict[str, float] = {}
        for rec in records:
            spend[rec.agent] = spend.get(rec.agent, 0.0) + rec.cost_usd
        return {"agents": detect_cost_anomalies(spend, z_threshold),
                "temporal": detect_temporal_cost_anomalies(records)}

    @app.put("/v1/costs/budget", dependencies=[Depends(auth)])
    def set_cost_budget(request: Request, payload: dict) -> dict:
        from agentbom_amd.api.cost_store import CostBudget

        if "limit_usd" not in payload:
            raise HTTPException(status_code=400, detail="limit_usd required")
        budget = CostBudget(
            tenant_id=_tenant_of(request),
            limit_usd=float(payload["limit_usd"]),
            agent=str(payload.get("agent", "")),
            cost_center=str(payload.get("cost_center", "")))
        state.costs.set_budget(budget)
        return budget.to_dict()

    @app.post("/v1/proxy/audit", dependencies=[Depends(auth)])
    def ingest_audit(payload: dict) -> dict:
        entries = payload.get("entries", [])
        if not isinstance(entries, list):
            raise HTTPException(status_code=400, detail="entries must be a list")
        state.audit_entries.extend(entries[:10_000])
        return {"ingested": len(entries), "total": len(state.audit_entries)}

    @app.get("/v1/proxy/audit", dependencies=[Depends(auth)])
    def read_audit(limit: int = 100) -> dict:
        return {"total": len(state.audit_entries),
                "entries": state.audit_entries[-limit:]}

    def _hub(request: Request):
        if state.hub is None:
            from agentbom_amd.api.hub import ComplianceHub

            state.hub = ComplianceHub()
        return state.hub

    @app.post("/v1/compliance/ingest", dependencies=[Depends(auth)])
    def hub_ingest(request: Request, payload: dict) -> dict:
        """External-scan / connector finding ingest into the hub."""
        if not payload.get("source"):
            raise HTTPException(status_code=400, detail="source required")
        findings = payload.get("findings")
        if not isinstance(findings, list):
            raise HTTPException(status_code=400,
                                detail="findings must be a list")
        return _hub(request).ingest(
            _tenant_of(request), str(payload["source"]), findings,
            reconcile_absent=bool(payload.get("reconcile_absent", True)))

    @app.get("/v1/compliance/hub/overview", dependencies=[Depends(auth)])
    def hub_overview(request: Request) -> dict:
        return _hub(request).overview(_tenant_of(request))

    @app.get("/v1/compliance/hub/findings", dependencies=[Depends(auth)])
    def hub_findings(request: Request, source: Optional[str] = None,
                     status: str = "open", limit: int = 1000) -> dict:
        rows = _hub(request).findings(_tenant_of(request), source=source,
                                      status=status, limit=min(limit, 10_000))
        return {"total": len(rows), "findings": rows}

    @app.get("/v1/compliance/hub/ledger", dependencies=[Depends(auth)])
    def hub_ledger(request: Request, limit: int = 100) -> dict:
        return {"events": _hub(request).ledger(_tenant_of(request),
                                               limit=min(limit, 1000))}

    @app.get("/v1/compliance/{framework}/report", dependencies=[Depends(auth)])
    def compliance_report(request: Request, framework: str) -> dict:
        from agentbom_amd.models import FRAMEWORK_TAG_FIELDS

        report = _latest_report(request)
        field_name = next((f for f, slug in FRAMEWORK_TAG_FIELDS if slug == framework), None)
        if field_name is None:
            raise HTTPException(status_code=404, detail=f"unknown framework {framework!r}")
        from collections import Counter

        counts: Counter = Counter()
        rows = []
        for br in report.blast_radii:
            tags = getattr(br, field_name)
            counts.update(tags)
            if tags:
                rows.append({"vulnerability_id": br.vulnerability.id,
                             "package": f"{br.package.name}@{br.package.version}",
                             "risk_score": br.risk_score, "controls": tags})
        return {"framework": framework, "tagged_findings": len(rows),
                "controls": dict(sorted(counts.items())), "findings": rows}

    @app.get("/v1/graph/attack-flow", dependencies=[Depends(auth)])
    def attack_flow(request: Request, cve: Optional[str] = None, min_severity: Optional[str] = None,
                    agent: Optional[str] = None) -> dict:
        from agentbom_amd.output.flow_fmt import build_attack_flow

        state.metrics["graph_queries_total"] += 1
        return build_attack_flow(_latest_report(request), cve=cve,
                                 min_severity=min_severity, agent=agent)

    @app.get("/v1/mesh", dependencies=[Depends(auth)])
    def agent_mesh(request: Request) -> dict:
        from agentbom_amd.output.flow_fmt import build_agent_mesh

        return build_agent_mesh(_latest_report(request))

    @app.get("/v1/findings/delta", dependencies=[Depends(auth)])
    def findings_delta(request: Request, format: str = "ndjson") -> dict:
        """Delta events (new/resolved/changed) since the previous call."""
        from agentbom_amd.output.delta_stream import DeltaStreamer

        if getattr(state, "delta_streamer", None) is None:
            state.delta_streamer = DeltaStreamer(fmt="ndjson")
        events = state.delta_streamer.emit(_latest_report(request))
        if format == "ocsf":
            from agentbom_amd.output.delta_stream import _to_ocsf_event

            return {"watermark": state.delta_streamer.watermark,
                    "events": [_to_ocsf_event(e) for e in events]}
        return {"watermark": state.delta_streamer.watermark, "events": events}

    @app.get("/v1/remediation", dependencies=[Depends(auth)])
    def remediation(request: Request, script: bool = False) -> Any:
        from agentbom_amd.scan.remediation import (
            remediation_commands,
            remediation_script,
        )

        report = _latest_report(request)
        if script:
            from fastapi.responses import PlainTextResponse

            return PlainTextResponse(remediation_script(report),
                                     media_type="text/x-shellscript")
        return {"commands": remediation_commands(report)}

    @app.get("/v1/posture", dependencies=[Depends(auth)])
    def posture(request: Request) -> dict:
        from agentbom_amd.scan.auth_posture import assess_a2a, assess_estate
        from agentbom_amd.scan.self_posture import evaluate_self_posture

        report = _latest_report(request)
        return {
            "mcp_auth_posture": assess_estate(report.agents),
            "a2a_auth_posture": assess_a2a(report.agents,
                                           identity_store=state.identity_store),
            "self_posture": evaluate_self_posture(),
        }

    @app.get("/v1/trust/{ecosystem}/{name:path}", dependencies=[Depends(auth)])
    def trust(ecosystem: str, name: str, version: str = "0.0.0") -> dict:
        from agentbom_amd.db.store import load_advisory_windows
        from agentbom_amd.models import Package
        from agentbom_amd.scan.trust import trust_score

        return trust_score(
            Package(name=name, version=version, ecosystem=ecosystem),
            advisory_windows=load_advisory_windows(offline=True))

    # ── webhooks ───────────────────────────────────────────────────────────

    @app.post("/v1/webhooks", status_code=201, dependencies=[Depends(auth)])
    def webhook_subscribe(payload: dict) -> dict:
        try:
            return state.webhooks.subscribe(
                url=str(payload.get("url", "")),
                events=list(payload.get("events") or []),
                secret=payload.get("secret"))
        except ValueError as exc:
            raise HTTPException(status_code=400, detail=str(exc))

    @app.get("/v1/webhooks", dependencies=[Depends(auth)])
    def webhook_list() -> dict:
        return {"webhooks": state.webhooks.list(),
                "dead_letters": len(state.webhooks.dead_letters)}

    @app.delete("/v1/webhooks/{webhook_id}", dependencies=[Depends(auth)])
    def webhook_unsubscribe(webhook_id: str) -> dict:
        if not state.webhooks.unsubscribe(webhook_id):
            raise HTTPException(status_code=404, detail="webhook not found")
        return {"removed": True}

    @app.get("/v1/webhooks/dead-letters", dependencies=[Depends(auth)])
    def webhook_dead_letters(limit: int = 50) -> dict:
        return {"dead_letters": state.webhooks.dead_letters[-limit:]}

    # ── identity lifecycle (reference: api/routes/identities.py) ───────────

    def _identity_store():
        if state.identity_store is None:
            from agentbom_amd.identity import AgentIdentityStore

            state.identity_store = AgentIdentityStore()
        return state.identity_store

    def _gate_identity_write(action: str,
                             x_operator_role: Optional[str],
                             x_operator_scopes: Optional[str],
                             x_audit_reason: Optional[str]) -> str:
        from agentbom_amd.mcp.authz import authorize_write

        ok, blocked = authorize_write(
            action=action, operator_role=x_operator_role or "",
            operator_scopes=x_operator_scopes or "",
            reason=x_audit_reason or "", required_scope="identity:write")
        if not ok:
            raise HTTPException(status_code=403, detail=blocked)
        return (x_audit_reason or "").strip()

    @app.post("/v1/identities", status_code=201, dependencies=[Depends(auth)])
    def identity_issue(payload: dict,
                       x_operator_role: Optional[str] = Header(default=None),
                       x_operator_scopes: Optional[str] = Header(default=None),
                       x_audit_reason: Optional[str] = Header(default=None)) -> dict:
        reason = _gate_identity_write("identity.issue", x_operator_role,
                                      x_operator_scopes, x_audit_reason)
        ident, raw = _identity_store().issue(
            str(payload.get("agent_name", "")),
            scopes=list(payload.get("scopes") or []),
            allowed_tools=list(payload.get("allowed_tools") or []),
            ttl_hours=float(payload.get("ttl_hours", 24)),
            actor="api", reason=reason)
        return {"identity": ident.to_public_dict(), "token": raw,
                "note": "token is shown exactly once"}

    @app.get("/v1/identities", dependencies=[Depends(auth)])
    def identity_list(live_only: bool = False) -> dict:
        idents = _identity_store().list(live_only=live_only)
        return {"total": len(idents),
                "identities": [i.to_public_dict() for i in idents]}

    @app.post("/v1/identities/verify", dependencies=[Depends(auth)])
    def identity_verify(payload: dict) -> dict:
        return _identity_store().verify(
            str(payload.get("token", "")),
            tool=payload.get("tool"),
            source_ip=payload.get("source_ip"))

    @app.get("/v1/identities/reviews/access", dependencies=[Depends(auth)])
    def identity_access_review() -> dict:
        return _identity_store().access_review()

    @app.get("/v1/identities/audit", dependencies=[Depends(auth)])
    def identity_audit(limit: int = 200) -> dict:
        store = _identity_store()
        entries = store.audit_entries()
        return {"total": len(entries), "chain_valid": store.audit_chain_valid(),
                "entries": entries[-limit:]}

    @app.post("/v1/identities/nhi/discover", dependencies=[Depends(auth)])
    def nhi_discover(payload: dict) -> dict:
        """Discover NHIs from inline export payloads (offline connectors)."""
        import json as _json
        import tempfile

        from agentbom_amd.identity.nhi import discover_entra_nhis, discover_okta_nhis

        out: dict[str, Any] = {}
        for provider, fn in (("okta", discover_okta_nhis),
                             ("entra", discover_entra_nhis)):
            export = payload.get(provider)
            if export is None:
                continue
            with tempfile.NamedTemporaryFile("w", suffix=".json",
                                             delete=False) as f:
                _json.dump(export, f)
                path = f.name
            try:
                out[provider] = fn(export_path=path, env={}).to_dict()
            finally:
                os.unlink(path)
        return out or {"note": "provide 'okta' and/or 'entra' export payloads"}

    @app.get("/v1/identities/{identity_id}", dependencies=[Depends(auth)])
    def identity_get(identity_id: str) -> dict:
        ident = _identity_store().get(identity_id)
        if ident is None:
            raise HTTPException(status_code=404, detail="identity not found")
        return {"identity": ident.to_public_dict(),
                "active_scopes": _identity_store().active_scopes(identity_id),
                "jit_grants": [g.to_public_dict() for g in
                               _identity_store().list_jit_grants(identity_id)]}

    @app.post("/v1/identities/{identity_id}/rotate", dependencies=[Depends(auth)])
    def identity_rotate(identity_id: str,
                        x_operator_role: Optional[str] = Header(default=None),
                        x_operator_scopes: Optional[str] = Header(default=None),
                        x_audit_reason: Optional[str] = Header(default=None)) -> dict:
        reason = _gate_identity_write("identity.rotate", x_operator_role,
                                      x_operator_scopes, x_audit_reason)
        new, raw = _identity_store().rotate(identity_id, actor="api",
                                            reason=reason)
        if new is None:
            raise HTTPException(status_code=404,
                                detail="identity not found or not live")
        return {"identity": new.to_public_dict(), "token": raw}

    @app.delete("/v1/identities/{identity_id}", dependencies=[Depends(auth)])
    def identity_revoke(identity_id: str,
                        x_operator_role: Optional[str] = Header(default=None),
                        x_operator_scopes: Optional[str] = Header(default=None),
                        x_audit_reason: Optional[str] = Header(default=None)) -> dict:
        reason = _gate_identity_write("identity.revoke", x_operator_role,
                                      x_operator_scopes, x_audit_reason)
        if not _identity_store().revoke(identity_id, actor="api", reason=reason):
            raise HTTPException(status_code=404, detail="identity not found")
        return {"revoked": True}

    @app.post("/v1/identities/{identity_id}/jit", status_code=201,
              dependencies=[Depends(auth)])
    def identity_grant_jit(identity_id: str, payload: dict,
                           x_operator_role: Optional[str] = Header(default=None),
                           x_operator_scopes: Optional[str] = Header(default=None),
                           x_audit_reason: Optional[str] = Header(default=None)) -> dict:
        reason = _gate_identity_write("identity.grant_jit", x_operator_role,
                                      x_operator_scopes, x_audit_reason)
        g = _identity_store().grant_jit(
            identity_id, list(payload.get("scopes") or []),
            reason=reason, granted_by="api",
            ttl_minutes=float(payload.get("ttl_minutes", 60)))
        if g is None:
            raise HTTPException(status_code=404, detail="identity not found")
        return {"grant": g.to_public_dict()}

    @app.delete("/v1/identities/jit/{grant_id}", dependencies=[Depends(auth)])
    def identity_revoke_jit(grant_id: str,
                            x_operator_role: Optional[str] = Header(default=None),
                            x_operator_scopes: Optional[str] = Header(default=None),
                            x_audit_reason: Optional[str] = Header(default=None)) -> dict:
        reason = _gate_identity_write("identity.revoke_jit", x_operator_role,
                                      x_operator_scopes, x_audit_reason)
        if not _identity_store().revoke_jit(grant_id, actor="api", reason=reason):
            raise HTTPException(status_code=404, detail="grant not found")
        return {"revoked": True}

    # ── delegation tokens + SCIM provisioning (reference: api/{scim,oidc}.py,
    # delegation tokens in middleware; SURVEY §2.6 auth/tenancy row) ───────

    @app.post("/v1/delegation-tokens", status_code=201, dependencies=[Depends(auth)])
    def mint_delegation_token(payload: dict) -> dict:
        if not state.delegation.enabled:
            raise HTTPException(status_code=400,
                                detail="AGENT_BOM_DELEGATION_SECRET not configured")
        try:
            token = state.delegation.mint(
                role=str(payload.get("role") or "viewer"),
                scopes=[str(s) for s in (payload.get("scopes") or [])],
                ttl_s=float(payload.get("ttl_s", 3600)),
                tenant_id=str(payload.get("tenant_id") or "default"))
        except AuthError as exc:
            raise HTTPException(status_code=400, detail=str(exc))
        except (TypeError, ValueError) as exc:
            raise HTTPException(status_code=422, detail=str(exc))
        return {"token": token, "note": "store securely; shown once"}

    @app.delete("/v1/delegation-tokens/{jti}", dependencies=[Depends(auth)])
    def revoke_delegation_token(jti: str) -> dict:
        state.delegation.revoke(jti)
        return {"revoked": jti}

    @app.get("/scim/v2/ServiceProviderConfig", dependencies=[Depends(auth)])
    def scim_config() -> dict:
        return {
            "schemas": ["urn:ietf:params:scim:schemas:core:2.0:ServiceProviderConfig"],
            "patch": {"supported": True}, "bulk": {"supported": False},
            "filter": {"supported": False}, "sort": {"supported": False},
            "authenticationSchemes": [{"type": "httpheader",
                                       "name": "X-API-Key"}],
        }

    @app.get("/scim/v2/Users", dependencies=[Depends(auth)])
    def scim_list_users() -> dict:
        users = state.scim_users.list()
        return {"schemas": ["urn:ietf:params:scim:api:messages:2.0:ListResponse"],
                "totalResults": len(users),
                "Resources": [u.to_scim() for u in users]}

    @app.post("/scim/v2/Users", status_code=201, dependencies=[Depends(auth)])
    def scim_create_user(payload: dict) -> dict:
        try:
            return state.scim_users.create(payload).to_scim()
        except AuthError as exc:
            raise HTTPException(status_code=409, detail=str(exc))

    @app.get("/scim/v2/Users/{uid}", dependencies=[Depends(auth)])
    def scim_get_user(uid: str) -> dict:
        u = state.scim_users.get(uid)
        if u is None:
            raise HTTPException(status_code=404, detail="user not found")
        return u.to_scim()

    @app.patch("/scim/v2/Users/{uid}", dependencies=[Depends(auth)])
    def scim_patch_user(uid: str, payload: dict) -> dict:
        active: Optional[bool] = None
        for op in payload.get("Operations") or []:
            if not isinstance(op, dict):
                continue
            if str(op.get("op", "")).lower() == "replace":
                val = op.get("value")
                if isinstance(val, dict) and "active" in val:
                    active = bool(val["active"])
                elif str(op.get("path", "")).lower() == "active":
                    active = bool(val) if not isinstance(val, str) \
                        else val.lower() == "true"
        if active is None:
            raise HTTPException(status_code=422,
                                detail="only replace-active is supported")
        u = state.scim_users.set_active(uid, active)
        if u is None:
            raise HTTPException(status_code=404, detail="user not found")
        return u.to_scim()

    @app.delete("/scim/v2/Users/{uid}", status_code=204, dependencies=[Depends(auth)])
    def scim_delete_user(uid: str) -> None:
        if not state.scim_users.delete(uid):
            raise HTTPException(status_code=404, detail="user not found")

    @app.post("/scim/v2/Users/{uid}/api-key", status_code=201,
              dependencies=[Depends(auth)])
    def scim_bind_key(uid: str) -> dict:
        """Mint + bind an API key to a SCIM user (hash-only stored)."""
        key = f"abk_{uuid.uuid4().hex}"
        if not state.scim_users.bind_key(uid, key):
            raise HTTPException(status_code=404, detail="user not found")
        return {"api_key": key, "note": "store securely; shown once"}

    return app


def _now() -> str:
    return datetime.now(timezone.utc).isoformat()


app = create_app()
