"""Operator self-posture: agent-bom audits its OWN deployment hardening.

Reference parity: src/agent_bom/self_posture.py — a product that governs
other estates must report the security posture of its own control plane.
Read-only over the running configuration; honest per-check states:
``hardened`` / ``misconfigured`` / ``acknowledged`` (weakened but
explicitly accepted via env) / ``unknown``.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Any, Callable, Optional

STATE_HARDENED = "hardened"
STATE_MISCONFIGURED = "misconfigured"
STATE_ACKNOWLEDGED = "acknowledged"
STATE_UNKNOWN = "unknown"

_ACK_ENV = "AGENT_BOM_ACCEPTED_WEAKENINGS"  # comma-separated check ids


@dataclass
class SelfCheck:
    check_id: str
    title: str
    state: str
    detail: str

    def to_dict(self) -> dict[str, Any]:
        return {"check_id": self.check_id, "title": self.title,
                "state": self.state, "detail": self.detail}


def _acknowledged(env: dict[str, str]) -> set[str]:
    return {p.strip() for p in env.get(_ACK_ENV, "").split(",") if p.strip()}


def evaluate_self_posture(env: Optional[dict[str, str]] = None) -> dict[str, Any]:
    e = dict(env if env is not None else os.environ)
    acks = _acknowledged(e)
    checks: list[SelfCheck] = []

    def check(check_id: str, title: str, hardened: Optional[bool],
              detail_ok: str, detail_bad: str) -> None:
        if hardened is None:
            state, detail = STATE_UNKNOWN, detail_bad
        elif hardened:
            state, detail = STATE_HARDENED, detail_ok
        elif check_id in acks:
            state, detail = STATE_ACKNOWLEDGED, f"{detail_bad} (accepted by operator)"
        else:
            state, detail = STATE_MISCONFIGURED, detail_bad
        checks.append(SelfCheck(check_id, title, state, detail))

    check("SELF-001", "API authentication configured",
          bool(e.get("AGENT_BOM_API_KEY") or e.get("AGENT_BOM_API_KEYS")),
          "API key auth active",
          "REST API runs without authentication")
    check("SELF-002", "Role-based access (multi-key RBAC) in use",
          bool(e.get("AGENT_BOM_API_KEYS")),
          "per-key roles configured",
          "single shared admin key (no role separation)")
    check("SELF-003", "MCP tenant identity pinned",
          bool(e.get("AGENT_BOM_MCP_TENANT_ID") or e.get("AGENT_BOM_TENANT_ID")),
          "MCP tools run under an explicit tenant",
          "MCP surface falls back to the 'default' tenant")
    check("SELF-004", "Graph snapshots persisted",
          bool(e.get("AGENT_BOM_GRAPH_STORE")),
          "snapshot store configured (diff/evidence available)",
          "no graph store: snapshots and diffs are session-only")
    check("SELF-005", "Offline mode pinned for air-gapped operation",
          None if e.get("AGENT_BOM_OFFLINE") is None
          else e.get("AGENT_BOM_OFFLINE", "").lower() in ("1", "true", "yes"),
          "offline mode on: no egress possible",
          "offline mode not pinned (egress decided per run)")
    check("SELF-006", "Attestation signing key configured",
          bool(e.get("AGENT_BOM_ATTESTATION_KEY")),
          "scan attestations can be signed",
          "no signing key: attestations unavailable")
    check("SELF-007", "Proxy audit trail location writable",
          _audit_path_writable(e),
          "proxy audit JSONL path is writable",
          "proxy audit path not writable — runtime evidence would be lost")

    by_state: dict[str, int] = {}
    for c in checks:
        by_state[c.state] = by_state.get(c.state, 0) + 1
    score = round(100.0 * by_state.get(STATE_HARDENED, 0) / len(checks), 1)
    return {
        "schema_version": "1",
        "score": score,
        "by_state": by_state,
        "checks": [c.to_dict() for c in checks],
    }


def _audit_path_writable(env: dict[str, str]) -> Optional[bool]:
    path = env.get("AGENT_BOM_PROXY_AUDIT_PATH") or os.path.expanduser(
        "~/.agent-bom/proxy_audit.jsonl")
    try:
        parent = os.path.dirname(path) or "."
        os.makedirs(parent, exist_ok=True)
        return os.access(parent, os.W_OK)
    except OSError:
        return False
