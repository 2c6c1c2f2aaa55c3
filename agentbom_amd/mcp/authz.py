"""Shared write-authorization contract for MCP operator tools.

Every write-capable MCP tool (identity lifecycle, shield enforcement,
ticketing, campaigns) is gated identically: the operator must present an
admin role, a scope covering the action, and an audit reason of at least 8
characters.  Reference contract:
src/agent_bom/mcp_tools/identity.py:_authorize_identity_write (admin role +
``identity:write`` scope + reason; blocked results carry structured context).
"""

from __future__ import annotations

from typing import Any


def _csv_set(value: str) -> set[str]:
    return {part.strip() for part in (value or "").split(",") if part.strip()}


def has_scope(operator_scopes: str, required: str) -> bool:
    """``identity:write`` is satisfied by ``*``, ``identity:*`` or itself."""
    scopes = _csv_set(operator_scopes)
    family = required.split(":", 1)[0]
    return bool(scopes & {"*", f"{family}:*", required})


def authorize_write(*, action: str, operator_role: str, operator_scopes: str,
                    reason: str, required_scope: str) -> tuple[bool, dict[str, Any]]:
    """Gate a write tool. Returns (allowed, blocked-context)."""
    role = (operator_role or "").strip().lower()
    clean_reason = (reason or "").strip()
    if role != "admin":
        return False, {
            "error": f"{action} requires admin role",
            "action": action, "required_role": "admin",
            "provided_role": role or "unset", "status": "blocked",
        }
    if not has_scope(operator_scopes, required_scope):
        return False, {
            "error": f"{action} requires {required_scope} scope",
            "action": action, "required_role": "admin",
            "required_scope": required_scope, "status": "blocked",
        }
    if len(clean_reason) < 8:
        return False, {
            "error": f"{action} requires an audit reason of at least 8 characters",
            "action": action, "status": "blocked",
        }
    return True, {}
