"""Firewall / Shield: inline inter-agent decisions.

Reference: src/agent_bom/firewall.py + shield.py — synchronous
allow/warn/block decisions for inter-agent and agent->tool calls; the
three write actions (quarantine upstream, revoke credential, block tool)
are fail-closed: they require admin scope and an audit reason.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Optional

from agentbom_amd.runtime.detectors import DetectorPipeline


@dataclass
class ShieldDecision:
    action: str  # allow | warn | block
    reasons: list[str] = field(default_factory=list)
    detector_alerts: list[dict] = field(default_factory=list)

    def to_dict(self) -> dict[str, Any]:
        return {"action": self.action, "reasons": self.reasons,
                "detector_alerts": self.detector_alerts}


class PermissionError403(RuntimeError):
    pass


class Shield:
    """Inline decision engine with fail-closed write actions."""

    WRITE_ACTIONS = ("quarantine_upstream", "revoke_credential", "block_tool")

    def __init__(self, pipeline: Optional[DetectorPipeline] = None):
        self.pipeline = pipeline or DetectorPipeline()
        self.blocked_tools: set[str] = set()
        self.revoked_credentials: set[str] = set()
        self.quarantined_upstreams: set[str] = set()
        self.audit: list[dict] = []

    # ── inline decision path ──────────────────────────────────────────────

    def decide(self, frame: dict[str, Any], source_agent: str = "",
               target: str = "") -> ShieldDecision:
        reasons: list[str] = []
        method = frame.get("method", "")
        tool = (frame.get("params") or {}).get("name", "") if method == "tools/call" else ""

        if target in self.quarantined_upstreams:
            return ShieldDecision("block", [f"upstream {target!r} quarantined"])
        if tool and tool in self.blocked_tools:
            return ShieldDecision("block", [f"tool {tool!r} blocked by shield"])
        args_text = str((frame.get("params") or {}).get("arguments", ""))
        for cred in self.revoked_credentials:
            if cred and cred in args_text:
                return ShieldDecision("block", [f"revoked credential {cred!r} referenced"])

        action, alerts = self.pipeline.inspect(frame)
        if action != "allow":
            reasons.extend(a.message for a in alerts[:3])
        return ShieldDecision(action, reasons, [a.to_dict() for a in alerts])

    # ── fail-closed write actions ─────────────────────────────────────────

    def apply_write_action(self, action: str, target: str, *, admin: bool,
                           reason: str) -> dict[str, Any]:
        """All three write actions require admin scope AND an audit reason."""
        if action not in self.WRITE_ACTIONS:
            raise ValueError(f"unknown shield action {action!r}")
        if not admin:
            raise PermissionError403(f"shield action {action!r} requires admin scope")
        if not (reason or "").strip():
            raise ValueError(f"shield action {action!r} requires an audit reason")
        if action == "quarantine_upstream":
            self.quarantined_upstreams.add(target)
        elif action == "revoke_credential":
            self.revoked_credentials.add(target)
        else:
            self.blocked_tools.add(target)
        entry = {"ts": time.time(), "action": action, "target": target,
                 "reason": reason.strip()}
        self.audit.append(entry)
        return entry
