"""Auth/tenancy extensions: OIDC bearer validation, delegation tokens,
request quotas, SCIM user provisioning store.

Reference parity: src/agent_bom/api/{oidc,scim,tenancy}.py + rbac.py
(SURVEY.md §2.6 auth/tenancy row).  Air-gapped constraints shape the
crypto: there is no JOSE/cryptography library in the image, so OIDC
bearer tokens are validated as **HS256 JWTs against an operator-shared
secret** (`AGENT_BOM_OIDC_SECRET`) — the deployment pattern where the
identity provider and agent-bom share an HMAC key or a local token
service re-signs upstream assertions.  RS256 verification requires an
RSA library and is explicitly rejected (fail-closed, never skipped).

Delegation tokens are short-lived HMAC-signed grants an admin mints for
automation (CI jobs, break-glass): scoped, tenant-bound, expiring, and
revocable by jti.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import json
import os
import threading
import time
import uuid
from dataclasses import dataclass, field
from typing import Any, Optional


class AuthError(Exception):
    """Raised with a human-readable reason; callers map to 401/403."""


def _b64url_decode(s: str) -> bytes:
    pad = "=" * (-len(s) % 4)
    return base64.urlsafe_b64decode(s + pad)


def _b64url_encode(b: bytes) -> str:
    return base64.urlsafe_b64encode(b).decode().rstrip("=")


# ── OIDC bearer (HS256) ─────────────────────────────────────────────────────


def verify_oidc_bearer(token: str,
                       secret: Optional[str] = None,
                       issuer: Optional[str] = None,
                       audience: Optional[str] = None,
                       now: Optional[float] = None) -> dict[str, Any]:
    """Validate an OIDC bearer JWT; returns its claims or raises AuthError.

    Dispatch by header alg (strict allow-list, 'none' always rejected):
    - RS256 -> api/oidc.py JWKS verification (pure-stdlib RSA) when
      AGENT_BOM_OIDC_JWKS / AGENT_BOM_OIDC_JWKS_URL is configured;
    - HS256 -> operator-shared-secret HMAC (AGENT_BOM_OIDC_SECRET), the
      air-gapped deployment shape.
    Both paths require a numeric exp (60 s leeway), check nbf, and pin
    iss/aud when configured."""
    issuer = issuer if issuer is not None else os.environ.get("AGENT_BOM_OIDC_ISSUER")
    audience = audience if audience is not None else os.environ.get("AGENT_BOM_OIDC_AUDIENCE")
    parts = token.split(".")
    if len(parts) != 3:
        raise AuthError("malformed JWT")
    try:
        header = json.loads(_b64url_decode(parts[0]))
        claims = json.loads(_b64url_decode(parts[1]))
        sig = _b64url_decode(parts[2])
    except Exception as exc:
        raise AuthError(f"undecodable JWT: {exc}") from None
    alg = header.get("alg")
    if alg == "RS256":
        from agentbom_amd.api.oidc import verify_rs256_bearer

        return verify_rs256_bearer(token, issuer=issuer, audience=audience, now=now)
    secret = secret if secret is not None else os.environ.get("AGENT_BOM_OIDC_SECRET")
    if not secret:
        raise AuthError("OIDC not configured (AGENT_BOM_OIDC_SECRET unset)")
    if alg != "HS256":
        raise AuthError(f"alg {alg!r} not accepted (RS256/HS256 only)")
    expected = hmac.new(secret.encode(), f"{parts[0]}.{parts[1]}".encode(),
                        hashlib.sha256).digest()
    if not hmac.compare_digest(sig, expected):
        raise AuthError("signature mismatch")
    t = now if now is not None else time.time()
    exp = claims.get("exp")
    if not isinstance(exp, (int, float)) or isinstance(exp, bool):
        raise AuthError("token missing numeric exp claim")
    if t > exp + 60:
        raise AuthError("token expired")
    nbf = claims.get("nbf")
    if isinstance(nbf, (int, float)) and t < nbf - 60:
        raise AuthError("token not yet valid")
    if issuer and claims.get("iss") != issuer:
        raise AuthError("issuer mismatch")
    if audience:
        aud = claims.get("aud")
        auds = aud if isinstance(aud, list) else [aud]
        if audience not in auds:
            raise AuthError("audience mismatch")
    return claims


def role_from_claims(claims: dict[str, Any]) -> str:
    """Map OIDC claims to the three-tier role model.  Sources (first hit):
    a ``role`` claim, a ``roles``/``groups`` list, or scope tokens
    (``abom:admin`` > ``abom:operator`` > default viewer)."""
    direct = claims.get("role")
    if direct in ("admin", "operator", "viewer"):
        return direct
    pool: list[str] = []
    for key in ("roles", "groups"):
        v = claims.get(key)
        if isinstance(v, list):
            pool.extend(str(x).lower() for x in v)
    pool.extend(str(claims.get("scope") or "").lower().split())
    for role in ("admin", "operator"):
        if any(p in (role, f"abom:{role}", f"agent-bom-{role}") for p in pool):
            return role
    return "viewer"


def mint_test_jwt(secret: str, claims: dict[str, Any]) -> str:
    """HS256 signer (used by tests and the local token-service pattern)."""
    h = _b64url_encode(json.dumps({"alg": "HS256", "typ": "JWT"}).encode())
    p = _b64url_encode(json.dumps(claims).encode())
    sig = hmac.new(secret.encode(), f"{h}.{p}".encode(), hashlib.sha256).digest()
    return f"{h}.{p}.{_b64url_encode(sig)}"


# ── delegation tokens ───────────────────────────────────────────────────────


class DelegationTokens:
    """Admin-minted, scoped, expiring HMAC grants (reference: delegation
    tokens in api/middleware.py).  Format: ``abd.<payload b64>.<sig b64>``.
    Verification is stateless except revocation (jti deny-list)."""

    def __init__(self, secret: Optional[str] = None):
        self._secret = (secret or os.environ.get("AGENT_BOM_DELEGATION_SECRET")
                        or "").encode()
        self._revoked: set[str] = set()
        self._lock = threading.Lock()

    @property
    def enabled(self) -> bool:
        return bool(self._secret)

    def mint(self, role: str, scopes: list[str], ttl_s: float = 3600,
             tenant_id: str = "default", minted_by: str = "admin") -> str:
        if not self.enabled:
            raise AuthError("delegation tokens not configured")
        if role not in ("operator", "viewer"):
            raise AuthError("delegation grants only operator/viewer roles")
        payload = {
            "jti": str(uuid.uuid4()), "role": role, "scopes": scopes,
            "tenant_id": tenant_id, "minted_by": minted_by,
            "exp": time.time() + max(1.0, ttl_s),
        }
        body = _b64url_encode(json.dumps(payload, sort_keys=True).encode())
        sig = _b64url_encode(hmac.new(self._secret, body.encode(),
                                      hashlib.sha256).digest())
        return f"abd.{body}.{sig}"

    def verify(self, token: str, now: Optional[float] = None) -> dict[str, Any]:
        if not self.enabled:
            raise AuthError("delegation tokens not configured")
        parts = (token or "").split(".")
        if len(parts) != 3 or parts[0] != "abd":
            raise AuthError("malformed delegation token")
        expected = hmac.new(self._secret, parts[1].encode(),
                            hashlib.sha256).digest()
        if not hmac.compare_digest(_b64url_decode(parts[2]), expected):
            raise AuthError("delegation signature mismatch")
        payload = json.loads(_b64url_decode(parts[1]))
        if (now if now is not None else time.time()) > float(payload.get("exp", 0)):
            raise AuthError("delegation token expired")
        with self._lock:
            if payload.get("jti") in self._revoked:
                raise AuthError("delegation token revoked")
        return payload

    def revoke(self, jti: str) -> None:
        with self._lock:
            self._revoked.add(jti)


# ── quotas ──────────────────────────────────────────────────────────────────


class QuotaTracker:
    """Sliding-window per-principal quotas (reference: quotas/entitlements,
    api/tenancy.py).  Configured via ``AGENT_BOM_QUOTA_SCANS_PER_HOUR``;
    0/unset = unlimited."""

    def __init__(self, scans_per_hour: Optional[int] = None,
                 window_s: float = 3600.0):
        if scans_per_hour is None:
            try:
                scans_per_hour = int(os.environ.get(
                    "AGENT_BOM_QUOTA_SCANS_PER_HOUR", "0"))
            except ValueError:
                scans_per_hour = 0
        self.limit = max(0, scans_per_hour)
        self.window_s = window_s
        self._events: dict[str, list[float]] = {}
        self._lock = threading.Lock()

    def check_and_record(self, principal: str,
                         now: Optional[float] = None) -> tuple[bool, float]:
        """(allowed, retry_after_s).  Records the event when allowed."""
        if self.limit <= 0:
            return True, 0.0
        t = now if now is not None else time.time()
        with self._lock:
            evs = [e for e in self._events.get(principal, [])
                   if e > t - self.window_s]
            if len(evs) >= self.limit:
                retry = max(1.0, evs[0] + self.window_s - t)
                self._events[principal] = evs
                return False, retry
            evs.append(t)
            self._events[principal] = evs
            return True, 0.0


# ── SCIM user store ─────────────────────────────────────────────────────────


@dataclass
class ScimUser:
    id: str
    user_name: str
    active: bool = True
    role: str = "viewer"
    api_key_hash: Optional[str] = None
    meta: dict = field(default_factory=dict)

    def to_scim(self) -> dict[str, Any]:
        return {
            "schemas": ["urn:ietf:params:scim:schemas:core:2.0:User"],
            "id": self.id, "userName": self.user_name, "active": self.active,
            "roles": [{"value": self.role}],
            "meta": {"resourceType": "User", **self.meta},
        }


class ScimUserStore:
    """Minimal SCIM 2.0 Users resource (create/list/get/patch-active/delete)
    feeding the API-key role table — deactivated users lose access on the
    next request."""

    def __init__(self) -> None:
        self._users: dict[str, ScimUser] = {}
        self._lock = threading.Lock()

    def create(self, doc: dict[str, Any]) -> ScimUser:
        user_name = str(doc.get("userName") or "").strip()
        if not user_name:
            raise AuthError("userName required")
        with self._lock:
            if any(u.user_name == user_name for u in self._users.values()):
                raise AuthError(f"userName {user_name!r} already exists")
            roles = doc.get("roles") or []
            role = "viewer"
            if roles and isinstance(roles, list):
                cand = str(roles[0].get("value") if isinstance(roles[0], dict)
                           else roles[0]).lower()
                if cand in ("admin", "operator", "viewer"):
                    role = cand
            user = ScimUser(id=str(uuid.uuid4()), user_name=user_name,
                            active=bool(doc.get("active", True)), role=role)
            self._users[user.id] = user
            return user

    def list(self) -> list[ScimUser]:
        with self._lock:
            return sorted(self._users.values(), key=lambda u: u.user_name)

    def get(self, uid: str) -> Optional[ScimUser]:
        return self._users.get(uid)

    def set_active(self, uid: str, active: bool) -> Optional[ScimUser]:
        with self._lock:
            u = self._users.get(uid)
            if u is not None:
                u.active = active
            return u

    def delete(self, uid: str) -> bool:
        with self._lock:
            return self._users.pop(uid, None) is not None

    def bind_key(self, uid: str, api_key: str) -> bool:
        """Associate an API key with a SCIM user (hash only stored)."""
        with self._lock:
            u = self._users.get(uid)
            if u is None:
                return False
            u.api_key_hash = hashlib.sha256(api_key.encode()).hexdigest()
            return True

    def role_for_key(self, api_key: str) -> Optional[str]:
        """Role for a SCIM-bound key; None if unknown or deactivated."""
        h = hashlib.sha256(api_key.encode()).hexdigest()
        with self._lock:
            for u in self._users.values():
                if u.api_key_hash == h:
                    return u.role if u.active else None
        return None
