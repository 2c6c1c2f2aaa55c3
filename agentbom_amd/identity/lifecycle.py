"""Agent identity lifecycle: issue / rotate / revoke / verify + JIT grants.

The discovery connectors (``identity.nhi``) inventory identities an external
IdP owns; this store is the control plane for agents with *no* IdP — agent-bom
issues time-scoped identities itself, rotates them with an overlap window, and
revokes them.  Parity surface:
reference src/agent_bom/api/agent_identity_store.py (issue/rotate/revoke/JIT/
conditional access; token ``abi_<prefix>_<secret>`` stored hash-only, raw
returned exactly once).

Design here: one SQLite-backed store (``:memory:`` by default) guarded by a
lock; every lifecycle mutation appends to a hash-chained audit log (same chain
scheme as the proxy audit trail) so MCP- and REST-driven writes carry identical
provenance.
"""

from __future__ import annotations

import hashlib
import ipaddress
import json
import secrets
import sqlite3
import threading
from dataclasses import dataclass, field
from datetime import datetime, timedelta, timezone
from typing import Any, Optional

from agentbom_amd.utils import config as cfg

TOKEN_PREFIX = "abi"


def _now() -> datetime:
    return datetime.now(timezone.utc)


def _iso(dt: datetime) -> str:
    return dt.isoformat()


def hash_token(token: str) -> str:
    return hashlib.sha256(token.encode("utf-8")).hexdigest()


def generate_token() -> tuple[str, str, str]:
    """Return ``(raw_token, public_prefix, token_hash)``; raw shown exactly once."""
    public = secrets.token_hex(4)
    secret = secrets.token_urlsafe(32)
    raw = f"{TOKEN_PREFIX}_{public}_{secret}"
    return raw, public, hash_token(raw)


@dataclass
class AgentIdentity:
    """One issued identity. Only the token *hash* is ever stored."""

    identity_id: str
    agent_name: str
    token_prefix: str
    token_hash: str
    scopes: list[str] = field(default_factory=list)
    allowed_tools: list[str] = field(default_factory=list)
    issued_at: str = ""
    expires_at: Optional[str] = None
    revoked_at: Optional[str] = None
    rotated_from: Optional[str] = None
    rotation_overlap_until: Optional[str] = None

    def is_live(self, at: Optional[datetime] = None) -> bool:
        t = at or _now()
        if self.revoked_at and datetime.fromisoformat(self.revoked_at) <= t:
            return False
        if self.expires_at and datetime.fromisoformat(self.expires_at) <= t:
            return False
        return True

    def tool_allowed(self, tool: str) -> bool:
        if not self.allowed_tools:
            return True
        return tool in self.allowed_tools or "*" in self.allowed_tools

    def to_public_dict(self) -> dict[str, Any]:
        """Public view — never includes the token hash."""
        return {
            "identity_id": self.identity_id,
            "agent_name": self.agent_name,
            "token_prefix": self.token_prefix,
            "scopes": list(self.scopes),
            "allowed_tools": list(self.allowed_tools),
            "issued_at": self.issued_at,
            "expires_at": self.expires_at,
            "revoked_at": self.revoked_at,
            "rotated_from": self.rotated_from,
            "rotation_overlap_until": self.rotation_overlap_until,
            "live": self.is_live(),
        }


@dataclass
class AgentJITGrant:
    """A just-in-time scope elevation with a short TTL."""

    grant_id: str
    identity_id: str
    scopes: list[str]
    reason: str
    granted_by: str
    granted_at: str
    expires_at: str
    revoked_at: Optional[str] = None

    def is_live(self, at: Optional[datetime] = None) -> bool:
        t = at or _now()
        if self.revoked_at and datetime.fromisoformat(self.revoked_at) <= t:
            return False
        return datetime.fromisoformat(self.expires_at) > t

    def to_public_dict(self) -> dict[str, Any]:
        return {
            "grant_id": self.grant_id, "identity_id": self.identity_id,
            "scopes": list(self.scopes), "reason": self.reason,
            "granted_by": self.granted_by, "granted_at": self.granted_at,
            "expires_at": self.expires_at, "revoked_at": self.revoked_at,
            "live": self.is_live(),
        }


def _scope_match(pattern: str, scope: str) -> bool:
    """``identity:*`` matches ``identity:write``; ``*`` matches everything."""
    if pattern == "*" or pattern == scope:
        return True
    if pattern.endswith(":*"):
        return scope.startswith(pattern[:-1])
    return False


def ip_in_any_cidr(ip: str, cidrs: list[str]) -> bool:
    try:
        addr = ipaddress.ip_address(ip)
    except ValueError:
        return False
    for c in cidrs:
        try:
            if addr in ipaddress.ip_network(c, strict=False):
                return True
        except ValueError:
            continue
    return False


@dataclass
class AccessContext:
    """Request-time context a conditional-access policy evaluates against.

    Unsupplied attributes ("" / [] / None) FAIL CLOSED against any policy
    condition that constrains them — an unprovable "in prod", "from this
    CIDR" or "device compliant" claim is treated as not met (reference
    agent_identity_store.py AccessContext semantics)."""

    scope: str = ""
    tool_name: str = ""
    environment: str = ""
    source_ip: str = ""
    device_id: str = ""
    groups: list[str] = field(default_factory=list)
    client_id: str = ""
    device_managed: Optional[bool] = None
    device_compliant: Optional[bool] = None
    device_disk_encrypted: Optional[bool] = None
    at: Optional[datetime] = None


@dataclass
class ConditionalAccessPolicy:
    """Context-aware access rule (ABAC) layered over live identities.

    ``effect="require"`` permits only when every configured condition holds;
    ``effect="deny"`` blocks when every condition holds (deny wins, then
    priority/policy_id order).  Empty scope/condition lists mean "any" /
    "unconstrained"."""

    policy_id: str
    name: str
    scopes: list[str] = field(default_factory=list)  # empty = all scopes
    allowed_cidrs: list[str] = field(default_factory=list)
    deny_outside_hours: Optional[tuple[int, int]] = None  # (start_h, end_h) UTC
    effect: str = "require"  # require | deny
    status: str = "active"   # active | disabled
    priority: int = 100
    tools: list[str] = field(default_factory=list)
    allowed_environments: list[str] = field(default_factory=list)
    allowed_hours_utc: list[int] = field(default_factory=list)
    allowed_weekdays: list[int] = field(default_factory=list)  # 0=Mon..6=Sun
    allowed_devices: list[str] = field(default_factory=list)
    allowed_groups: list[str] = field(default_factory=list)
    allowed_clients: list[str] = field(default_factory=list)
    require_device_managed: bool = False
    require_device_compliant: bool = False
    require_device_disk_encrypted: bool = False

    def applies_to(self, scope: str, tool: str = "") -> bool:
        scope_ok = not self.scopes or any(_scope_match(p, scope)
                                          for p in self.scopes)
        tool_ok = not self.tools or "*" in self.tools or tool in self.tools
        return scope_ok and tool_ok

    def conditions_met(self, source_ip: Optional[str] = None,
                       at: Optional[datetime] = None,
                       ctx: Optional[AccessContext] = None) -> tuple[bool, str]:
        """(met, reason_if_not).  Every configured condition must hold."""
        c = ctx or AccessContext(source_ip=source_ip or "", at=at)
        if source_ip and not c.source_ip:
            c.source_ip = source_ip
        now = c.at or at or _now()
        if self.allowed_cidrs:
            if not c.source_ip:
                return False, f"policy {self.name}: source ip required"
            if not ip_in_any_cidr(c.source_ip, self.allowed_cidrs):
                return False, (f"policy {self.name}: ip {c.source_ip} "
                               "outside allowed ranges")
        if self.deny_outside_hours:
            h = now.hour
            lo, hi = self.deny_outside_hours
            inside = lo <= h < hi if lo <= hi else (h >= lo or h < hi)
            if not inside:
                return False, (f"policy {self.name}: outside allowed hours "
                               f"{lo}-{hi} UTC")
        if self.allowed_hours_utc and now.hour not in set(self.allowed_hours_utc):
            return False, f"policy {self.name}: hour {now.hour} UTC not allowed"
        if self.allowed_weekdays and now.weekday() not in set(self.allowed_weekdays):
            return False, f"policy {self.name}: weekday not allowed"
        if self.allowed_environments and c.environment.strip().lower() not in {
                e.strip().lower() for e in self.allowed_environments}:
            return False, (f"policy {self.name}: environment "
                           f"{c.environment or '(unset)'} not allowed")
        if self.allowed_devices and (not c.device_id
                                     or c.device_id not in set(self.allowed_devices)):
            return False, f"policy {self.name}: device not on allow-list"
        if self.allowed_groups and not (set(c.groups) & set(self.allowed_groups)):
            return False, f"policy {self.name}: caller not in an allowed group"
        if self.allowed_clients and (not c.client_id
                                     or c.client_id not in set(self.allowed_clients)):
            return False, f"policy {self.name}: client app not allowed"
        if self.require_device_managed and c.device_managed is not True:
            return False, f"policy {self.name}: unmanaged device"
        if self.require_device_compliant and c.device_compliant is not True:
            return False, f"policy {self.name}: non-compliant device"
        if self.require_device_disk_encrypted and c.device_disk_encrypted is not True:
            return False, f"policy {self.name}: disk encryption not proven"
        return True, ""


def evaluate_conditional_access(policies: list[ConditionalAccessPolicy],
                                scope: str,
                                ctx: AccessContext) -> tuple[bool, str, str]:
    """(allowed, reason, policy_id) with deny precedence.

    Any applying active ``deny`` policy whose conditions hold blocks; then
    any applying ``require`` policy whose conditions do NOT hold blocks.
    Ordering: (priority, policy_id).  Empty/none-applying ⇒ allow."""
    applicable = sorted(
        (p for p in policies
         if p.status == "active" and p.applies_to(scope, ctx.tool_name)),
        key=lambda p: (p.priority, p.policy_id))
    for pol in applicable:
        if pol.effect == "deny":
            met, _ = pol.conditions_met(ctx=ctx)
            if met:
                return False, f"blocked by conditional-access policy '{pol.name}'", \
                    pol.policy_id
    for pol in applicable:
        if pol.effect == "require":
            met, why = pol.conditions_met(ctx=ctx)
            if not met:
                return False, why, pol.policy_id
    return True, "", ""


_SCHEMA = """
CREATE TABLE IF NOT EXISTS identities (
    identity_id TEXT PRIMARY KEY,
    agent_name TEXT NOT NULL,
    token_prefix TEXT NOT NULL,
    token_hash TEXT NOT NULL UNIQUE,
    doc TEXT NOT NULL
);
CREATE TABLE IF NOT EXISTS jit_grants (
    grant_id TEXT PRIMARY KEY,
    identity_id TEXT NOT NULL,
    doc TEXT NOT NULL
);
CREATE TABLE IF NOT EXISTS identity_audit (
    seq INTEGER PRIMARY KEY AUTOINCREMENT,
    ts TEXT NOT NULL,
    action TEXT NOT NULL,
    actor TEXT NOT NULL,
    reason TEXT NOT NULL,
    subject TEXT NOT NULL,
    prev_hash TEXT NOT NULL,
    entry_hash TEXT NOT NULL
);
"""


class AgentIdentityStore:
    """SQLite-backed lifecycle store (``:memory:`` default); thread-safe."""

    def __init__(self, path: str = ":memory:"):
        self._lock = threading.Lock()
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.executescript(_SCHEMA)
        self._policies: dict[str, ConditionalAccessPolicy] = {}

    # ── audit chain ───────────────────────────────────────────────────────

    def _audit(self, action: str, actor: str, reason: str, subject: str) -> None:
        cur = self._db.execute(
            "SELECT entry_hash FROM identity_audit ORDER BY seq DESC LIMIT 1")
        row = cur.fetchone()
        prev = row[0] if row else "genesis"
        ts = _iso(_now())
        entry = hashlib.sha256(
            f"{prev}|{ts}|{action}|{actor}|{reason}|{subject}".encode()).hexdigest()
        self._db.execute(
            "INSERT INTO identity_audit (ts, action, actor, reason, subject,"
            " prev_hash, entry_hash) VALUES (?,?,?,?,?,?,?)",
            (ts, action, actor, reason, subject, prev, entry))
        self._db.commit()

    def audit_entries(self) -> list[dict[str, Any]]:
        cur = self._db.execute(
            "SELECT seq, ts, action, actor, reason, subject, prev_hash, entry_hash"
            " FROM identity_audit ORDER BY seq")
        cols = ["seq", "ts", "action", "actor", "reason", "subject",
                "prev_hash", "entry_hash"]
        return [dict(zip(cols, r)) for r in cur.fetchall()]

    def audit_chain_valid(self) -> bool:
        prev = "genesis"
        for e in self.audit_entries():
            expect = hashlib.sha256(
                f"{prev}|{e['ts']}|{e['action']}|{e['actor']}|{e['reason']}|"
                f"{e['subject']}".encode()).hexdigest()
            if e["prev_hash"] != prev or e["entry_hash"] != expect:
                return False
            prev = e["entry_hash"]
        return True

    # ── persistence helpers ───────────────────────────────────────────────

    def _save(self, ident: AgentIdentity) -> None:
        doc = json.dumps(ident.__dict__)
        self._db.execute(
            "INSERT INTO identities (identity_id, agent_name, token_prefix,"
            " token_hash, doc) VALUES (?,?,?,?,?)"
            " ON CONFLICT(identity_id) DO UPDATE SET doc=excluded.doc,"
            " token_hash=excluded.token_hash, token_prefix=excluded.token_prefix",
            (ident.identity_id, ident.agent_name, ident.token_prefix,
             ident.token_hash, doc))
        self._db.commit()

    @staticmethod
    def _load(doc: str) -> AgentIdentity:
        return AgentIdentity(**json.loads(doc))

    # ── lifecycle ─────────────────────────────────────────────────────────

    def issue(self, agent_name: str, scopes: Optional[list[str]] = None,
              allowed_tools: Optional[list[str]] = None,
              ttl_hours: Optional[float] = None, actor: str = "api",
              reason: str = "initial issue") -> tuple[AgentIdentity, str]:
        """Issue a new identity; returns (identity, raw_token) — raw shown once."""
        if ttl_hours is None:
            ttl_hours = cfg.IDENTITY_TOKEN_TTL_HOURS
        with self._lock:
            raw, prefix, thash = generate_token()
            now = _now()
            ident = AgentIdentity(
                identity_id=f"aid-{secrets.token_hex(6)}",
                agent_name=agent_name,
                token_prefix=prefix,
                token_hash=thash,
                scopes=list(scopes or []),
                allowed_tools=list(allowed_tools or []),
                issued_at=_iso(now),
                expires_at=_iso(now + timedelta(hours=ttl_hours)) if ttl_hours else None,
            )
            self._save(ident)
            self._audit("identity.issue", actor, reason, ident.identity_id)
            return ident, raw

    def rotate(self, identity_id: str,
               overlap_minutes: Optional[float] = None,
               ttl_hours: Optional[float] = None, actor: str = "api",
               reason: str = "rotation") -> tuple[Optional[AgentIdentity], Optional[str]]:
        """Rotate: new token, old identity stays live through the overlap window."""
        if overlap_minutes is None:
            overlap_minutes = cfg.IDENTITY_ROTATE_OVERLAP_MIN
        if ttl_hours is None:
            ttl_hours = cfg.IDENTITY_TOKEN_TTL_HOURS
        with self._lock:
            old = self.get(identity_id)
            if old is None or not old.is_live():
                return None, None
            now = _now()
            old.rotation_overlap_until = _iso(now + timedelta(minutes=overlap_minutes))
            old.expires_at = old.rotation_overlap_until
            self._save(old)
            raw, prefix, thash = generate_token()
            new = AgentIdentity(
                identity_id=f"aid-{secrets.token_hex(6)}",
                agent_name=old.agent_name,
                token_prefix=prefix,
                token_hash=thash,
                scopes=list(old.scopes),
                allowed_tools=list(old.allowed_tools),
                issued_at=_iso(now),
                expires_at=_iso(now + timedelta(hours=ttl_hours)) if ttl_hours else None,
                rotated_from=old.identity_id,
            )
            self._save(new)
            self._audit("identity.rotate", actor, reason, f"{identity_id}->{new.identity_id}")
            return new, raw

    def revoke(self, identity_id: str, actor: str = "api",
               reason: str = "revocation") -> bool:
        with self._lock:
            ident = self.get(identity_id)
            if ident is None:
                return False
            ident.revoked_at = _iso(_now())
            self._save(ident)
            self._audit("identity.revoke", actor, reason, identity_id)
            return True

    def verify(self, raw_token: str, tool: Optional[str] = None,
               source_ip: Optional[str] = None,
               ctx: Optional[AccessContext] = None) -> dict[str, Any]:
        """Verify a presented token: live, tool-scoped, conditional access.

        ``ctx`` carries the full ABAC request context (environment, device,
        groups, client, posture); absent attributes fail closed against any
        policy that constrains them.  Deny-effect policies take precedence
        (evaluate_conditional_access)."""
        ident = self.get_by_token_hash(hash_token(raw_token))
        if ident is None:
            return {"valid": False, "reason": "unknown token"}
        if not ident.is_live():
            return {"valid": False, "reason": "expired or revoked",
                    "identity_id": ident.identity_id}
        if tool and not ident.tool_allowed(tool):
            return {"valid": False, "reason": f"tool {tool!r} not in allowed_tools",
                    "identity_id": ident.identity_id}
        c = ctx or AccessContext()
        if source_ip and not c.source_ip:
            c.source_ip = source_ip
        if tool and not c.tool_name:
            c.tool_name = tool
        policies = list(self._policies.values())
        for scope in ident.scopes or ["*"]:
            allowed, why, pol_id = evaluate_conditional_access(policies, scope, c)
            if not allowed:
                return {"valid": False, "reason": why, "policy_id": pol_id,
                        "identity_id": ident.identity_id}
        return {"valid": True, "identity_id": ident.identity_id,
                "agent_name": ident.agent_name, "scopes": ident.scopes}

    # ── reads ─────────────────────────────────────────────────────────────

    def get(self, identity_id: str) -> Optional[AgentIdentity]:
        cur = self._db.execute("SELECT doc FROM identities WHERE identity_id=?",
                               (identity_id,))
        row = cur.fetchone()
        return self._load(row[0]) if row else None

    def get_by_token_hash(self, token_hash: str) -> Optional[AgentIdentity]:
        cur = self._db.execute("SELECT doc FROM identities WHERE token_hash=?",
                               (token_hash,))
        row = cur.fetchone()
        return self._load(row[0]) if row else None

    def list(self, live_only: bool = False) -> list[AgentIdentity]:
        cur = self._db.execute("SELECT doc FROM identities ORDER BY identity_id")
        out = [self._load(r[0]) for r in cur.fetchall()]
        return [i for i in out if i.is_live()] if live_only else out

    def list_by_agent(self, agent_name: str) -> list[AgentIdentity]:
        return [i for i in self.list() if i.agent_name == agent_name]

    def credential_expiry_report(self, within_hours: float = 72.0) -> list[dict[str, Any]]:
        """Identities whose token expires within the window (or already expired)."""
        horizon = _now() + timedelta(hours=within_hours)
        out = []
        for i in self.list():
            if i.revoked_at or not i.expires_at:
                continue
            exp = datetime.fromisoformat(i.expires_at)
            if exp <= horizon:
                out.append({**i.to_public_dict(),
                            "expired": exp <= _now(),
                            "hours_remaining": round((exp - _now()).total_seconds() / 3600, 2)})
        return sorted(out, key=lambda d: d["hours_remaining"])

    # ── JIT grants ────────────────────────────────────────────────────────

    def grant_jit(self, identity_id: str, scopes: list[str], reason: str,
                  granted_by: str, ttl_minutes: Optional[float] = None) -> Optional[AgentJITGrant]:
        if ttl_minutes is None:
            ttl_minutes = cfg.JIT_GRANT_TTL_MINUTES
        with self._lock:
            if self.get(identity_id) is None:
                return None
            now = _now()
            g = AgentJITGrant(
                grant_id=f"jit-{secrets.token_hex(6)}",
                identity_id=identity_id,
                scopes=list(scopes),
                reason=reason,
                granted_by=granted_by,
                granted_at=_iso(now),
                expires_at=_iso(now + timedelta(minutes=ttl_minutes)),
            )
            self._db.execute(
                "INSERT INTO jit_grants (grant_id, identity_id, doc) VALUES (?,?,?)",
                (g.grant_id, identity_id, json.dumps(g.__dict__)))
            self._db.commit()
            self._audit("jit.grant", granted_by, reason, g.grant_id)
            return g

    def revoke_jit(self, grant_id: str, actor: str = "api",
                   reason: str = "jit revocation") -> bool:
        with self._lock:
            cur = self._db.execute("SELECT doc FROM jit_grants WHERE grant_id=?",
                                   (grant_id,))
            row = cur.fetchone()
            if row is None:
                return False
            g = AgentJITGrant(**json.loads(row[0]))
            g.revoked_at = _iso(_now())
            self._db.execute("UPDATE jit_grants SET doc=? WHERE grant_id=?",
                             (json.dumps(g.__dict__), grant_id))
            self._db.commit()
            self._audit("jit.revoke", actor, reason, grant_id)
            return True

    def list_jit_grants(self, identity_id: Optional[str] = None,
                        live_only: bool = False) -> list[AgentJITGrant]:
        if identity_id:
            cur = self._db.execute(
                "SELECT doc FROM jit_grants WHERE identity_id=? ORDER BY grant_id",
                (identity_id,))
        else:
            cur = self._db.execute("SELECT doc FROM jit_grants ORDER BY grant_id")
        out = [AgentJITGrant(**json.loads(r[0])) for r in cur.fetchall()]
        return [g for g in out if g.is_live()] if live_only else out

    def active_scopes(self, identity_id: str) -> list[str]:
        """Base scopes ∪ live JIT scopes — the effective-permission view."""
        ident = self.get(identity_id)
        if ident is None:
            return []
        scopes = list(ident.scopes)
        for g in self.list_jit_grants(identity_id, live_only=True):
            for s in g.scopes:
                if s not in scopes:
                    scopes.append(s)
        return scopes

    # ── conditional access ────────────────────────────────────────────────

    def put_conditional_policy(self, policy: ConditionalAccessPolicy) -> None:
        self._policies[policy.policy_id] = policy

    def list_conditional_policies(self) -> list[ConditionalAccessPolicy]:
        return list(self._policies.values())

    def access_review(self) -> dict[str, Any]:
        """Estate-wide access posture: live identities, stale, over-scoped, JIT."""
        idents = self.list()
        live = [i for i in idents if i.is_live()]
        wildcard = [i.identity_id for i in live
                    if "*" in i.scopes or "*" in i.allowed_tools or not i.allowed_tools]
        jit_live = self.list_jit_grants(live_only=True)
        return {
            "total_identities": len(idents),
            "live_identities": len(live),
            "revoked": sum(1 for i in idents if i.revoked_at),
            "wildcard_or_unscoped": wildcard,
            "live_jit_grants": [g.to_public_dict() for g in jit_live],
            "expiring_72h": self.credential_expiry_report(72.0),
            "audit_chain_valid": self.audit_chain_valid(),
        }

    def close(self) -> None:
        self._db.close()
