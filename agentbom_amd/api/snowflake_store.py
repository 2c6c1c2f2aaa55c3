"""Snowflake findings store over the SQL API v2 (optional tier).

Reference parity: src/agent_bom/api/snowflake_store.py — selected
persistence/analytics paths on Snowflake.  Speaks the SQL REST API
(POST /api/v2/statements) with a bearer token (OAuth or key-pair JWT the
operator mints); no connector dependency.  Transport-injected for tests,
offline-guarded, parameter bindings used for every value (no SQL
stitching of row data).
"""

from __future__ import annotations

import base64
import hashlib
import secrets
import time
from typing import Any, Optional
from urllib.parse import urlencode

from agentbom_amd.utils import config as cfg
from agentbom_amd.utils.http_client import check_offline, create_client, request_with_retry

_DDL = """
CREATE TABLE IF NOT EXISTS {db}.{schema}.SCAN_FINDINGS (
    TS           TIMESTAMP_NTZ,
    SCAN_ID      VARCHAR,
    TENANT_ID    VARCHAR,
    VULN_ID      VARCHAR,
    PACKAGE      VARCHAR,
    ECOSYSTEM    VARCHAR,
    SEVERITY     VARCHAR,
    RISK_SCORE   FLOAT,
    IS_KEV       BOOLEAN,
    REACHABILITY VARCHAR
)
"""


class SnowflakeStore:
    """Findings batches + posture queries via the SQL API v2."""

    def __init__(self, account_url: str, token: str,
                 database: str = "AGENTBOM", schema: str = "PUBLIC",
                 warehouse: Optional[str] = None, client=None):
        check_offline(account_url)
        self.base = account_url.rstrip("/")
        self.token = token
        self.database = database
        self.schema = schema
        self.warehouse = warehouse
        self.client = client or create_client(timeout=cfg.SNOWFLAKE_TIMEOUT_S)

    def _exec(self, statement: str, bindings: Optional[dict] = None) -> dict:
        payload: dict[str, Any] = {"statement": statement, "timeout": 60}
        if self.warehouse:
            payload["warehouse"] = self.warehouse
        if bindings:
            payload["bindings"] = bindings
        resp = request_with_retry(
            self.client, "POST", f"{self.base}/api/v2/statements",
            json=payload,
            headers={"Authorization": f"Bearer {self.token}",
                     "Content-Type": "application/json",
                     "Accept": "application/json",
                     "X-Snowflake-Authorization-Token-Type": "OAUTH"})
        if resp is None or resp.status_code not in (200, 202):
            raise RuntimeError(
                f"snowflake statement failed: "
                f"{resp.status_code if resp is not None else 'unreachable'} "
                f"{(resp.text[:200] if resp is not None else '')}")
        try:
            return resp.json()
        except ValueError:
            return {}

    def ensure_schema(self) -> None:
        self._exec(f"CREATE DATABASE IF NOT EXISTS {self.database}")
        self._exec(_DDL.format(db=self.database, schema=self.schema))

    def insert_findings(self, report, tenant_id: str = "default") -> int:
        """One bound INSERT per blast-radius row (SQL API bindings, no
        string-stitched values)."""
        n = 0
        sql = (f"INSERT INTO {self.database}.{self.schema}.SCAN_FINDINGS "
               "(TS, SCAN_ID, TENANT_ID, VULN_ID, PACKAGE, ECOSYSTEM, "
               "SEVERITY, RISK_SCORE, IS_KEV, REACHABILITY) "
               "SELECT TO_TIMESTAMP_NTZ(?), ?, ?, ?, ?, ?, ?, ?, ?, ?")
        ts = str(int(time.time()))
        for br in report.blast_radii:
            bindings = {
                "1": {"type": "FIXED", "value": ts},
                "2": {"type": "TEXT", "value": report.scan_id or ""},
                "3": {"type": "TEXT", "value": tenant_id},
                "4": {"type": "TEXT", "value": br.vulnerability.id},
                "5": {"type": "TEXT",
                      "value": f"{br.package.name}@{br.package.version}"},
                "6": {"type": "TEXT", "value": br.package.ecosystem},
                "7": {"type": "TEXT", "value": br.vulnerability.severity.value},
                "8": {"type": "REAL", "value": str(float(br.risk_score))},
                "9": {"type": "BOOLEAN",
                      "value": "true" if br.vulnerability.is_kev else "false"},
                "10": {"type": "TEXT", "value": br.reachability},
            }
            self._exec(sql, bindings)
            n += 1
        return n

    def severity_posture(self, tenant_id: str = "default") -> list[dict[str, Any]]:
        body = self._exec(
            f"SELECT SEVERITY, COUNT(*) AS FINDINGS FROM "
            f"{self.database}.{self.schema}.SCAN_FINDINGS "
            f"WHERE TENANT_ID = ? GROUP BY SEVERITY ORDER BY SEVERITY",
            {"1": {"type": "TEXT", "value": tenant_id}})
        meta = body.get("resultSetMetaData") or {}
        cols = [c.get("name") for c in meta.get("rowType", []) or []]
        return [dict(zip(cols, row)) for row in body.get("data", []) or []]


# ---------------------------------------------------------------------------
# Key-pair JWT (Snowflake's native programmatic auth)
# ---------------------------------------------------------------------------

def _der_len(n: int) -> bytes:
    if n < 0x80:
        return bytes([n])
    raw = n.to_bytes((n.bit_length() + 7) // 8, "big")
    return bytes([0x80 | len(raw)]) + raw


def _der_int(v: int) -> bytes:
    raw = v.to_bytes((v.bit_length() + 7) // 8 or 1, "big")
    if raw[0] & 0x80:
        raw = b"\x00" + raw
    return b"\x02" + _der_len(len(raw)) + raw


def _der_seq(*parts: bytes) -> bytes:
    body = b"".join(parts)
    return b"\x30" + _der_len(len(body)) + body


_RSA_OID = bytes.fromhex("06092a864886f70d010101")  # 1.2.840.113549.1.1.1


def spki_der(n: int, e: int) -> bytes:
    """SubjectPublicKeyInfo DER for an RSA public key (stdlib-only)."""
    pubkey = _der_seq(_der_int(n), _der_int(e))
    bits = b"\x03" + _der_len(len(pubkey) + 1) + b"\x00" + pubkey
    return _der_seq(_der_seq(_RSA_OID, b"\x05\x00"), bits)


def public_key_fingerprint(n: int, e: int) -> str:
    """``SHA256:<b64>`` fingerprint Snowflake expects in the JWT issuer."""
    digest = hashlib.sha256(spki_der(n, e)).digest()
    return "SHA256:" + base64.b64encode(digest).decode()


def keypair_jwt(account: str, user: str, n: int, e: int, d: int,
                lifetime_s: Optional[int] = None, now: Optional[int] = None) -> str:
    """Mint the key-pair bearer JWT for the SQL API
    (iss = ACCOUNT.USER.<fingerprint>, sub = ACCOUNT.USER)."""
    from agentbom_amd.api.oidc import rs256_sign

    if lifetime_s is None:
        lifetime_s = cfg.SNOWFLAKE_JWT_LIFETIME_S
    qualified = f"{account.upper()}.{user.upper()}"
    iat = int(now if now is not None else time.time())
    claims = {"iss": f"{qualified}.{public_key_fingerprint(n, e)}",
              "sub": qualified, "iat": iat, "exp": iat + lifetime_s}
    return rs256_sign(n, d, {"alg": "RS256", "typ": "JWT"}, claims)


# ---------------------------------------------------------------------------
# OAuth authorization-code + PKCE (reference api/snowflake_oauth.py)
# ---------------------------------------------------------------------------

class SnowflakeOAuthError(RuntimeError):
    pass


def pkce_pair() -> tuple[str, str]:
    """(code_verifier, S256 code_challenge)."""
    verifier = base64.urlsafe_b64encode(secrets.token_bytes(32)).rstrip(b"=").decode()
    challenge = base64.urlsafe_b64encode(
        hashlib.sha256(verifier.encode()).digest()).rstrip(b"=").decode()
    return verifier, challenge


def build_authorize_url(account_url: str, client_id: str, redirect_uri: str,
                        state: str, code_challenge: str,
                        scope: str = "") -> str:
    base = account_url.rstrip("/")
    if not base.startswith("https://"):
        raise SnowflakeOAuthError("snowflake account URL must be https")
    params = {"response_type": "code", "client_id": client_id,
              "redirect_uri": redirect_uri, "state": state,
              "code_challenge": code_challenge,
              "code_challenge_method": "S256"}
    if scope:
        params["scope"] = scope
    return f"{base}/oauth/authorize?{urlencode(params)}"


def exchange_code_for_tokens(account_url: str, client_id: str,
                             client_secret: str, redirect_uri: str,
                             code: str, code_verifier: str,
                             client=None) -> dict[str, Any]:
    """POST /oauth/token-request with HTTP Basic client auth + PKCE verifier."""
    if not client_secret:
        raise SnowflakeOAuthError("client secret required for token exchange")
    base = account_url.rstrip("/")
    check_offline(base)
    basic = base64.b64encode(f"{client_id}:{client_secret}".encode()).decode()
    resp = request_with_retry(
        client or create_client(timeout=30.0), "POST",
        f"{base}/oauth/token-request",
        data={"grant_type": "authorization_code", "code": code,
              "redirect_uri": redirect_uri, "code_verifier": code_verifier},
        headers={"Authorization": f"Basic {basic}",
                 "Content-Type": "application/x-www-form-urlencoded"})
    if resp is None or resp.status_code != 200:
        raise SnowflakeOAuthError(
            f"token exchange failed: "
            f"{resp.status_code if resp is not None else 'unreachable'}")
    body = resp.json()
    if "access_token" not in body:
        raise SnowflakeOAuthError("token response missing access_token")
    return body
