"""OAuth 2.1 Authorization Server for the MCP auth spec (broker AS).

Reference parity: src/agent_bom/api/oauth_as.py — the subset of OAuth 2.1
+ the MCP authorization spec that lets a standard MCP client discover and
authenticate to gateway-fronted servers:

- RFC 8414 metadata (``/.well-known/oauth-authorization-server``)
- RFC 7591 dynamic client registration (``/oauth/register``)
- PKCE-REQUIRED authorization-code grant (S256 only; ``plain`` and the
  implicit grant are forbidden by OAuth 2.1)
- token endpoint: ``authorization_code`` + ``client_credentials``
  (client_credentials only for confidential clients with a verified
  secret hash)
- JWKS for RS256 access-token validation

Fail-closed + bounded: codes are single-use, short-TTL, bound to
client + redirect_uri + PKCE challenge; every store is LRU-bounded so a
registration/code flood cannot exhaust memory; tokens are RS256 (the
``none`` algorithm does not exist here).  Entirely in-process — no
outbound calls.
"""

from __future__ import annotations

import base64
import hashlib
import json
import secrets
import threading
import time
from collections import OrderedDict
from dataclasses import dataclass, field
from typing import Any, Optional

from agentbom_amd.api.oidc import (
    generate_rsa_keypair,
    jwk_for,
    rs256_sign,
    rsa_verify_pkcs1_sha256,
)

_MAX_CLIENTS = 1000
_MAX_CODES = 5000
CODE_TTL_S = 600
TOKEN_TTL_S = 3600


class OAuthError(Exception):
    def __init__(self, error: str, description: str = "",
                 status: int = 400):
        super().__init__(description or error)
        self.error = error
        self.description = description
        self.status = status

    def to_dict(self) -> dict[str, str]:
        return {"error": self.error, "error_description": self.description}


def _b64url(data: bytes) -> str:
    return base64.urlsafe_b64encode(data).rstrip(b"=").decode()


def _hash_secret(secret: str) -> str:
    return hashlib.sha256(secret.encode()).hexdigest()


@dataclass
class RegisteredClient:
    client_id: str
    redirect_uris: list[str]
    client_name: str = ""
    secret_hash: str = ""  # "" = public client (PKCE code grant only)
    scopes: list[str] = field(default_factory=lambda: ["mcp"])

    @property
    def confidential(self) -> bool:
        return bool(self.secret_hash)


@dataclass
class AuthorizationCode:
    code: str
    client_id: str
    redirect_uri: str
    code_challenge: str
    scope: str
    issued_at: float
    used: bool = False

    def expired(self, now: Optional[float] = None) -> bool:
        return (now or time.time()) - self.issued_at > CODE_TTL_S


class OAuthAuthorizationServer:
    """In-process OAuth 2.1 AS; RS256 tokens verifiable via its JWKS."""

    def __init__(self, issuer: str = "https://agent-bom.local",
                 keypair: Optional[tuple[int, int, int]] = None):
        self.issuer = issuer.rstrip("/")
        self._n, self._e, self._d = keypair or generate_rsa_keypair()
        self.kid = "abgw-1"
        self._clients: OrderedDict[str, RegisteredClient] = OrderedDict()
        self._codes: OrderedDict[str, AuthorizationCode] = OrderedDict()
        self._lock = threading.Lock()

    # ── RFC 8414 metadata ─────────────────────────────────────────────────

    def metadata(self) -> dict[str, Any]:
        return {
            "issuer": self.issuer,
            "authorization_endpoint": f"{self.issuer}/oauth/authorize",
            "token_endpoint": f"{self.issuer}/oauth/token",
            "registration_endpoint": f"{self.issuer}/oauth/register",
            "jwks_uri": f"{self.issuer}/oauth/jwks.json",
            "response_types_supported": ["code"],
            "grant_types_supported": ["authorization_code",
                                      "client_credentials"],
            "code_challenge_methods_supported": ["S256"],  # never "plain"
            "token_endpoint_auth_methods_supported": [
                "client_secret_basic", "none"],
            "scopes_supported": ["mcp"],
        }

    def jwks(self) -> dict[str, Any]:
        return {"keys": [jwk_for(self._n, self._e, kid=self.kid)]}

    # ── RFC 7591 registration ─────────────────────────────────────────────

    def register_client(self, redirect_uris: list[str], client_name: str = "",
                        confidential: bool = False) -> dict[str, Any]:
        uris = [u for u in (redirect_uris or []) if isinstance(u, str)
                and (u.startswith("https://")
                     or u.startswith("http://localhost")
                     or u.startswith("http://127.0.0.1"))]
        if not uris and not confidential:
            raise OAuthError("invalid_redirect_uri",
                             "at least one https:// or localhost redirect_uri"
                             " is required for a public client")
        client = RegisteredClient(
            client_id=f"abc-{secrets.token_hex(8)}",
            redirect_uris=uris, client_name=client_name[:120])
        out: dict[str, Any] = {"client_id": client.client_id,
                               "redirect_uris": uris,
                               "client_name": client.client_name,
                               "token_endpoint_auth_method":
                                   "client_secret_basic" if confidential
                                   else "none"}
        if confidential:
            secret = secrets.token_urlsafe(32)
            client.secret_hash = _hash_secret(secret)
            out["client_secret"] = secret  # returned exactly once
        with self._lock:
            self._clients[client.client_id] = client
            while len(self._clients) > _MAX_CLIENTS:  # bounded registry
                self._clients.popitem(last=False)
        return out

    def get_client(self, client_id: str) -> Optional[RegisteredClient]:
        return self._clients.get(client_id)

    # ── authorization-code grant (PKCE S256 mandatory) ────────────────────

    def authorize(self, client_id: str, redirect_uri: str,
                  code_challenge: str, code_challenge_method: str = "S256",
                  scope: str = "mcp",
                  state: str = "") -> dict[str, str]:
        client = self.get_client(client_id)
        if client is None:
            raise OAuthError("invalid_client", "unknown client_id", 401)
        if redirect_uri not in client.redirect_uris:
            raise OAuthError("invalid_request",
                             "redirect_uri is not registered")
        if code_challenge_method != "S256" or not code_challenge:
            raise OAuthError("invalid_request",
                             "PKCE with S256 is required (OAuth 2.1)")
        code = AuthorizationCode(
            code=f"ac-{secrets.token_urlsafe(24)}", client_id=client_id,
            redirect_uri=redirect_uri, code_challenge=code_challenge,
            scope=scope, issued_at=time.time())
        with self._lock:
            self._codes[code.code] = code
            while len(self._codes) > _MAX_CODES:  # bounded code store
                self._codes.popitem(last=False)
        return {"code": code.code, "state": state}

    # ── token endpoint ────────────────────────────────────────────────────

    def _mint(self, sub: str, scope: str) -> dict[str, Any]:
        now = int(time.time())
        claims = {"iss": self.issuer, "sub": sub, "aud": "mcp",
                  "scope": scope, "iat": now, "exp": now + TOKEN_TTL_S}
        token = rs256_sign(self._n, self._d,
                           {"alg": "RS256", "typ": "JWT", "kid": self.kid},
                           claims)
        return {"access_token": token, "token_type": "Bearer",
                "expires_in": TOKEN_TTL_S, "scope": scope}

    def token_authorization_code(self, code: str, client_id: str,
                                 redirect_uri: str,
                                 code_verifier: str) -> dict[str, Any]:
        with self._lock:
            ac = self._codes.get(code)
            if ac is None or ac.used or ac.expired():
                raise OAuthError("invalid_grant",
                                 "unknown, used, or expired code")
            if ac.client_id != client_id or ac.redirect_uri != redirect_uri:
                raise OAuthError("invalid_grant",
                                 "code is bound to a different client or "
                                 "redirect_uri")
            want = _b64url(hashlib.sha256(code_verifier.encode()).digest())
            if want != ac.code_challenge:
                raise OAuthError("invalid_grant", "PKCE verification failed")
            ac.used = True  # single use
        return self._mint(client_id, ac.scope)

    def token_client_credentials(self, client_id: str,
                                 client_secret: str,
                                 scope: str = "mcp") -> dict[str, Any]:
        client = self.get_client(client_id)
        if client is None or not client.confidential:
            raise OAuthError("invalid_client",
                             "client_credentials requires a registered "
                             "confidential client", 401)
        if _hash_secret(client_secret) != client.secret_hash:
            raise OAuthError("invalid_client", "bad client secret", 401)
        return self._mint(client_id, scope)

    # ── resource-side validation ──────────────────────────────────────────

    def verify_access_token(self, token: str) -> dict[str, Any]:
        try:
            h, c, s = token.split(".")
            pad = lambda x: x + "=" * (-len(x) % 4)  # noqa: E731
            sig = base64.urlsafe_b64decode(pad(s))
            claims = json.loads(base64.urlsafe_b64decode(pad(c)))
        except Exception:
            raise OAuthError("invalid_token", "malformed token", 401)
        if not rsa_verify_pkcs1_sha256(self._n, self._e, sig,
                                       f"{h}.{c}".encode()):
            raise OAuthError("invalid_token", "bad signature", 401)
        if claims.get("iss") != self.issuer:
            raise OAuthError("invalid_token", "wrong issuer", 401)
        exp = claims.get("exp")
        if not isinstance(exp, (int, float)) or exp < time.time():
            raise OAuthError("invalid_token", "expired", 401)
        return claims
