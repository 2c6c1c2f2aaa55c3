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

    threshold: int = 5
    cooldown_s: float = 30.0
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

    def __init__(self) -> None:
        self.upstreams: dict[str, Upstream] = {}
        self.pipeline = DetectorPipeline()
        self.metrics = {"relays_total": 0, "blocked_total": 0, "redactions_total": 0,
                        "breaker_rejections_total": 0}

    def register(self, upstream: Upstream) -> None:
        self.upstreams[upstream.name] = upstream

    def quarantine(self, name: str, on: bool = True) -> None:
        if name in self.upstreams:
            self.upstreams[name].quarantined = on

    def relay(self, upstream_name: str, frame: dict[str, Any]) -> dict[str, Any]:
        up = self.upstreams.get(upstream_name)
        msg_id = frame.get("id")

        def err(code: int, message: str) -> dict:
            return {"jsonrpc": "2.0", "id": msg_id, "error": {"code": code, "message": message}}

        if up is None:
            return err(-32001, f"unknown upstream {upstream_name!r}")
        if up.quarantined:
            self.metrics["blocked_total"] += 1
            return err(-32002, f"upstream {upstream_name!r} is quarantined")
        action, alerts = self.pipeline.inspect(frame)
        if action == "block":
            self.metrics["blocked_total"] += 1
            return err(-32000, "blocked by gateway: "
                       + "; ".join(a.message for a in alerts[:3]))
        if not up.breaker.allow():
            self.metrics["breaker_rejections_total"] += 1
            return err(-32003, f"upstream {upstream_name!r} circuit open")
        try:
            response = up.handler(frame)
            up.breaker.record_success()
        except Exception as exc:  # noqa: BLE001 — upstream boundary
            up.breaker.record_failure()
            return err(-32004, f"upstream error: {exc}")

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
