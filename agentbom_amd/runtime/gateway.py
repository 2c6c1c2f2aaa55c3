"""MCP gateway: central secure-by-default JSON-RPC relay.

Reference: src/agent_bom/gateway_server.py — upstream registry, per-upstream
circuit breaker (:778), DLP/PII redaction (:294-320), quarantine gate
(:354-468); served as FastAPI routes under /mcp/*.
"""

from __future__ import annotations

import re
import time
from dataclasses import dataclass, field
from typing import Any, Callable, Optional

from agentbom_amd.runtime.detectors import DetectorPipeline

_PII_PATTERNS = [
    (re.compile(r"\b\d{3}-\d{2}-\d{4}\b"), "[SSN_REDACTED]"),
    (re.compile(r"\b(?:\d[ -]*?){13,16}\b"), "[CARD_REDACTED]"),
    (re.compile(r"\b[\w.+-]+@[\w-]+\.[\w.]+\b"), "[EMAIL_REDACTED]"),
]


def redact_pii(text: str) -> tuple[str, int]:
    """DLP redaction over outbound content; returns (text, redaction_count)."""
    n = 0
    for pattern, repl in _PII_PATTERNS:
        text, k = pattern.subn(repl, text)
        n += k
    return text, n


@dataclass
class CircuitBreaker:
    """Per-upstream failure breaker: opens after N failures, half-opens
    after a cooldown (reference gateway_server.py:778)."""

    threshold: int = None  # type: ignore[assignment]
    cooldown_s: float = None  # type: ignore[assignment]

    def __post_init__(self):
        from agentbom_amd.utils import config as _cfg

        if self.threshold is None:
            self.threshold = _cfg.GATEWAY_BREAKER_THRESHOLD
        if self.cooldown_s is None:
            self.cooldown_s = _cfg.GATEWAY_BREAKER_COOLDOWN_S
    failures: int = 0
    opened_at: Optional[float] = None

    @property
    def state(self) -> str:
        if self.opened_at is None:
            return "closed"
        if time.monotonic() - self.opened_at >= self.cooldown_s:
            return "half-open"
        return "open"

    def allow(self) -> bool:
        return self.state != "open"

    def record_success(self) -> None:
        self.failures = 0
        self.opened_at = None

    def record_failure(self) -> None:
        self.failures += 1
        if self.failures >= self.threshold:
            self.opened_at = time.monotonic()


@dataclass
class Upstream:
    name: str
    handler: Callable[[dict], dict]  # transport-agnostic for tests; HTTP in prod
    quarantined: bool = False
    breaker: CircuitBreaker = field(default_factory=CircuitBreaker)


class Gateway:
    """Secure-by-default relay: quarantine gate -> detectors -> breaker ->
    upstream -> response inspection -> DLP redaction."""

    def __init__(self, identity_store=None, hitl_queue=None,
                 hitl_escalate: bool = False) -> None:
        self.upstreams: dict[str, Upstream] = {}
        self.pipeline = DetectorPipeline()
        self.identity_gate = IdentityGate()
        self.drift_gate = DriftGate()
        self.cost_gate = CostAnomalyGate()
        # optional human-in-the-loop escalation: warn-level detector hits
        # park the call for approval instead of passing through
        self.hitl = hitl_queue
        self.hitl_escalate = hitl_escalate
        # optional identity.lifecycle.AgentIdentityStore: supplies the ABAC
        # conditional-access policies relay() evaluates per call
        self.identity_store = identity_store
        self.metrics = {"relays_total": 0, "blocked_total": 0, "redactions_total": 0,
                        "breaker_rejections_total": 0}
        # bounded relay-activity history (reference gateway_activity_store):
        # every decision with outcome + latency, newest last
        from collections import deque

        self.activity = deque(maxlen=1000)

    def register(self, upstream: Upstream) -> None:
        self.upstreams[upstream.name] = upstream

    def quarantine(self, name: str, on: bool = True) -> None:
        if name in self.upstreams:
            self.upstreams[name].quarantined = on

    def relay(self, upstream_name: str, frame: dict[str, Any],
              principal: Optional[str] = None,
              cost: float = 1.0, access_ctx=None,
              approval_id: Optional[str] = None) -> dict[str, Any]:
        """Relay with activity recording (outcome + error code + latency)."""
        t0 = time.monotonic()
        out = self._relay_inner(upstream_name, frame, principal=principal,
                                cost=cost, access_ctx=access_ctx,
                                approval_id=approval_id)
        err = out.get("error") if isinstance(out, dict) else None
        self.activity.append({
            "ts": time.time(), "upstream": upstream_name,
            "principal": principal or "anonymous",
            "method": str(frame.get("method", "?")),
            "tool": str((frame.get("params") or {}).get("name", "")),
            "outcome": "error" if err else "ok",
            "code": err.get("code") if err else None,
            "latency_ms": round((time.monotonic() - t0) * 1000, 3)})
        return out

    def activity_summary(self) -> dict[str, Any]:
        """Aggregate view over the bounded history (per-outcome/per-code)."""
        from collections import Counter

        codes = Counter(a["code"] for a in self.activity
                        if a["code"] is not None)
        return {"window": len(self.activity),
                "ok": sum(1 for a in self.activity if a["outcome"] == "ok"),
                "errors_by_code": {str(k): v for k, v in sorted(codes.items())},
                "principals": len({a["principal"] for a in self.activity}),
                "recent": list(self.activity)[-20:]}

    def _relay_inner(self, upstream_name: str, frame: dict[str, Any],
                     principal: Optional[str] = None,
                     cost: float = 1.0, access_ctx=None,
                     approval_id: Optional[str] = None) -> dict[str, Any]:
        """``access_ctx``: an identity.lifecycle.AccessContext — when present
        together with ``self.identity_store`` policies, full ABAC
        conditional-access (deny-wins, fail-closed unknowns) gates the call."""
        up = self.upstreams.get(upstream_name)
        msg_id = frame.get("id")

        def err(code: int, message: str) -> dict:
            return {"jsonrpc": "2.0", "id": msg_id, "error": {"code": code, "message": message}}

        if up is None:
            return err(-32001, f"unknown upstream {upstream_name!r}")
        if up.quarantined:
            self.metrics["blocked_total"] += 1
            return err(-32002, f"upstream {upstream_name!r} is quarantined")
        # conditional-access gates: revoked identity, ABAC policy, catalog
        # drift, cost
        if principal is not None and not self.identity_gate.allow(principal):
            self.metrics["blocked_total"] += 1
            return err(-32005, f"identity {principal!r} is revoked")
        if access_ctx is not None and self.identity_store is not None:
            from agentbom_amd.identity.lifecycle import evaluate_conditional_access

            if not access_ctx.tool_name and frame.get("method") == "tools/call":
                access_ctx.tool_name = str(
                    (frame.get("params") or {}).get("name") or "")
            allowed, why, _pid = evaluate_conditional_access(
                self.identity_store.list_conditional_policies(),
                principal or "*", access_ctx)
            if not allowed:
                self.metrics["blocked_total"] += 1
                return err(-32008, f"conditional access: {why}")
        if not self.drift_gate.allow(upstream_name):
            self.metrics["blocked_total"] += 1
            return err(-32006, f"upstream {upstream_name!r} tool catalog "
                       "drifted; operator re-approval required")
        if principal is not None and not self.cost_gate.record_and_check(
                principal, cost):
            self.metrics["blocked_total"] += 1
            return err(-32007, f"cost budget exhausted for {principal!r}")
        action, alerts = self.pipeline.inspect(frame)
        if action == "block":
            self.metrics["blocked_total"] += 1
            return err(-32000, "blocked by gateway: "
                       + "; ".join(a.message for a in alerts[:3]))
        if action == "warn" and self.hitl is not None and self.hitl_escalate:
            # single-use approval: a granted request_id authorizes exactly
            # one replay of the SAME frame; otherwise the call parks
            if approval_id and self.hitl.consume(approval_id, frame):
                self.metrics["hitl_approved_total"] = \
                    self.metrics.get("hitl_approved_total", 0) + 1
            else:
                req = self.hitl.park(
                    upstream_name, principal or "anonymous", frame,
                    alerts=[{"detector": a.detector, "severity": a.severity,
                             "message": a.message} for a in alerts[:5]])
                self.metrics["hitl_parked_total"] = \
                    self.metrics.get("hitl_parked_total", 0) + 1
                out = err(-32009, "human approval required: "
                          + "; ".join(a.message for a in alerts[:2]))
                out["error"]["data"] = {"approval_request_id": req.request_id}
                return out
        if not up.breaker.allow():
            self.metrics["breaker_rejections_total"] += 1
            return err(-32003, f"upstream {upstream_name!r} circuit open")
        try:
            response = up.handler(frame)
            up.breaker.record_success()
        except Exception as exc:  # noqa: BLE001 — upstream boundary
            up.breaker.record_failure()
            return err(-32004, f"upstream error: {exc}")

        # drift observation: pin/compare tools/list catalogs (rug-pull gate)
        if frame.get("method") == "tools/list" and isinstance(response, dict):
            tools = (response.get("result") or {}).get("tools")
            if tools is not None and not self.drift_gate.observe(upstream_name, tools):
                self.metrics["blocked_total"] += 1
                return err(-32006, f"upstream {upstream_name!r} tool catalog "
                           "drifted; operator re-approval required")

        # response-side inspection + DLP
        r_action, r_alerts = self.pipeline.inspect(response)
        if r_action == "block":
            self.metrics["blocked_total"] += 1
            return err(-32000, "upstream response blocked: "
                       + "; ".join(a.message for a in r_alerts[:3]))
        result = response.get("result")
        if isinstance(result, dict):
            import json as _json

            text = _json.dumps(result)
            redacted, n = redact_pii(text)
            if n:
                self.metrics["redactions_total"] += n
                response = dict(response)
                response["result"] = _json.loads(redacted)
        self.metrics["relays_total"] += 1
        return response


# ── conditional-access gates (reference gateway_server.py:321-703) ─────────


@dataclass
class IdentityGate:
    """Revoked-identity gate: requests from revoked principals never relay."""

    revoked: set = field(default_factory=set)

    def revoke(self, principal: str) -> None:
        self.revoked.add(principal)

    def restore(self, principal: str) -> None:
        self.revoked.discard(principal)

    def allow(self, principal: Optional[str]) -> bool:
        return principal not in self.revoked


@dataclass
class DriftGate:
    """Tool-catalog drift gate: an upstream whose tools/list answer changes
    from its pinned hash is blocked until an operator re-approves
    (rug-pull protection — the reference's drift gate)."""

    pinned: dict = field(default_factory=dict)   # upstream -> catalog hash
    drifted: set = field(default_factory=set)

    @staticmethod
    def catalog_hash(tools: Any) -> str:
        import hashlib
        import json as _json

        canon = _json.dumps(tools, sort_keys=True, default=str)
        return hashlib.sha256(canon.encode()).hexdigest()

    def observe(self, upstream: str, tools: Any) -> bool:
        """Record a tools/list catalog; returns False when drift blocks."""
        h = self.catalog_hash(tools)
        pinned = self.pinned.get(upstream)
        if pinned is None:
            self.pinned[upstream] = h
            return True
        if h != pinned:
            self.drifted.add(upstream)
            return False
        return True

    def approve(self, upstream: str, tools: Any) -> None:
        """Operator re-approval: repin the current catalog."""
        self.pinned[upstream] = self.catalog_hash(tools)
        self.drifted.discard(upstream)

    def allow(self, upstream: str) -> bool:
        return upstream not in self.drifted


@dataclass
class CostAnomalyGate:
    """Per-principal sliding-window cost budget: calls beyond the budget in
    the window are blocked (cost-anomaly gate, gateway_server.py:321)."""

    budget_per_window: float = None  # type: ignore[assignment]
    window_s: float = None  # type: ignore[assignment]

    def __post_init__(self):
        from agentbom_amd.utils import config as _cfg

        if self.budget_per_window is None:
            self.budget_per_window = _cfg.GATEWAY_COST_BUDGET
        if self.window_s is None:
            self.window_s = _cfg.GATEWAY_COST_WINDOW_S
    spend: dict = field(default_factory=dict)  # principal -> [(t, cost)]

    def record_and_check(self, principal: str, cost: float = 1.0,
                         now: Optional[float] = None) -> bool:
        now = time.monotonic() if now is None else now
        hist = [(t, c) for t, c in self.spend.get(principal, [])
                if now - t < self.window_s]
        total = sum(c for _, c in hist) + cost
        hist.append((now, cost))
        self.spend[principal] = hist
        return total <= self.budget_per_window


def load_upstream_registry(path) -> list[dict]:
    """Upstream registry file: [{name, url, quarantined?, budget?}...]
    (yaml or json).  The gateway CLI registers each entry."""
    import json as _json
    from pathlib import Path as _Path

    import yaml as _yaml

    text = _Path(path).read_text()
    data = _yaml.safe_load(text) if str(path).endswith((".yaml", ".yml")) \
        else _json.loads(text)
    ups = data.get("upstreams", data) if isinstance(data, dict) else data
    out = []
    for entry in ups or []:
        if isinstance(entry, dict) and entry.get("name") and entry.get("url"):
            out.append(entry)
    return out
