"""Stored cloud connections: at-rest-encrypted secret, schedule, validation.

Reference parity: src/agent_bom/api/{connection_store,connection_crypto,
connection_scheduler,credential_ref_validation}.py — a connection is a
durable read-only scan target (cloud account / registry / cluster): one
reversible secret (encrypted at rest, never returned by the API),
non-secret auth params, a polling interval, and lifecycle status.

At-rest crypto (stdlib-only — the runtime has no AES provider): an
encrypt-then-MAC stream construction over HMAC-SHA256 as the PRF:

    enc_key, mac_key  = HMAC(master, "enc"), HMAC(master, "mac")
    keystream block i = HMAC-SHA256(enc_key, nonce || be32(i))
    ct  = pt XOR keystream        (16-byte random nonce per encryption)
    tag = HMAC-SHA256(mac_key, nonce || ct)    (verified constant-time)

This is the textbook PRF-counter-mode + encrypt-then-MAC composition
(the same PRF construction HKDF/CTR-DRBG keystreams use) — chosen over
hand-rolling a block cipher.  With no ``AGENT_BOM_CONNECTIONS_KEY``
configured the module is in DEGRADED mode and refuses to encrypt
(storing plaintext is never a fallback).
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import json
import os
import re
import secrets
import sqlite3
import threading
from dataclasses import dataclass, field
from datetime import datetime, timedelta, timezone
from typing import Any, Optional
from uuid import uuid4

STATUS_PENDING = "pending"
STATUS_ACTIVE = "active"
STATUS_DEGRADED = "degraded"
STATUS_DISABLED = "disabled"


def _now() -> datetime:
    return datetime.now(timezone.utc)


# ── at-rest crypto ──────────────────────────────────────────────────────────


class ConnectionsCryptoUnavailable(RuntimeError):
    """No master key configured — encryption REFUSES (no plaintext fallback)."""


def _master_key(env: Optional[dict] = None) -> Optional[bytes]:
    raw = (env if env is not None else os.environ).get(
        "AGENT_BOM_CONNECTIONS_KEY", "")
    if not raw:
        return None
    try:
        key = base64.urlsafe_b64decode(raw + "=" * (-len(raw) % 4))
    except Exception:
        return None
    return key if len(key) >= 16 else None


def generate_connections_key() -> str:
    """A fresh urlsafe-b64 256-bit master key for AGENT_BOM_CONNECTIONS_KEY."""
    return base64.urlsafe_b64encode(secrets.token_bytes(32)).decode()


def _subkeys(master: bytes) -> tuple[bytes, bytes]:
    return (hmac.new(master, b"enc", hashlib.sha256).digest(),
            hmac.new(master, b"mac", hashlib.sha256).digest())


def encrypt_secret(plaintext: str, env: Optional[dict] = None) -> str:
    master = _master_key(env)
    if master is None:
        raise ConnectionsCryptoUnavailable(
            "AGENT_BOM_CONNECTIONS_KEY is not configured; refusing to store "
            "a connection secret (plaintext at rest is never a fallback)")
    enc_key, mac_key = _subkeys(master)
    nonce = secrets.token_bytes(16)
    pt = plaintext.encode("utf-8")
    stream = b""
    for i in range((len(pt) + 31) // 32):
        stream += hmac.new(enc_key, nonce + i.to_bytes(4, "big"),
                           hashlib.sha256).digest()
    ct = bytes(a ^ b for a, b in zip(pt, stream))
    tag = hmac.new(mac_key, nonce + ct, hashlib.sha256).digest()
    return base64.urlsafe_b64encode(nonce + ct + tag).decode()


def decrypt_secret(token: str, env: Optional[dict] = None) -> str:
    master = _master_key(env)
    if master is None:
        raise ConnectionsCryptoUnavailable("AGENT_BOM_CONNECTIONS_KEY is not configured")
    blob = base64.urlsafe_b64decode(token + "=" * (-len(token) % 4))
    if len(blob) < 48:
        raise ValueError("ciphertext too short")
    nonce, ct, tag = blob[:16], blob[16:-32], blob[-32:]
    enc_key, mac_key = _subkeys(master)
    want = hmac.new(mac_key, nonce + ct, hashlib.sha256).digest()
    if not hmac.compare_digest(want, tag):
        raise ValueError("connection secret authentication failed")
    stream = b""
    for i in range((len(ct) + 31) // 32):
        stream += hmac.new(enc_key, nonce + i.to_bytes(4, "big"),
                           hashlib.sha256).digest()
    return bytes(a ^ b for a, b in zip(ct, stream)).decode("utf-8")


# ── credential-ref validation (metadata-only registry) ─────────────────────

_AWS_ROLE_ARN = re.compile(r"^arn:aws:iam::\d{12}:role/[\w+=,.@-]+$")
_GCP_SA = re.compile(r"^[a-z][-a-z0-9]{4,28}[a-z0-9]@[a-z][-a-z0-9]*\."
                     r"iam\.gserviceaccount\.com$")
_AZURE_SP = re.compile(r"^[0-9a-fA-F-]{36}$")
_REF_SCHEMES = ("env:", "file:", "vault:", "aws-sm:", "azure-kv:", "gcp-sm:")


def validate_credential_ref(provider: str, mode: str,
                            external_ref: str) -> tuple[str, str]:
    """(status, detail) — format checks only; refs never carry secrets."""
    ref = (external_ref or "").strip()
    if not ref:
        return "degraded", "external_ref is required"
    provider = (provider or "").lower()
    mode = (mode or "").lower()
    if provider == "aws" and mode == "role_arn":
        if not _AWS_ROLE_ARN.match(ref):
            return "degraded", "external_ref is not a valid IAM role ARN"
        return "ok", "role ARN format valid"
    if provider == "gcp" and mode == "service_account":
        if not _GCP_SA.match(ref):
            return "degraded", "external_ref is not a service-account email"
        return "ok", "service-account format valid"
    if provider == "azure" and mode == "service_principal":
        if not _AZURE_SP.match(ref):
            return "degraded", "external_ref is not a client (app) id"
        return "ok", "client-id format valid"
    if ref.startswith(_REF_SCHEMES):
        return "ok", f"reference scheme {ref.split(':', 1)[0]}: accepted"
    return "degraded", ("external_ref must use a reference scheme "
                        f"({', '.join(_REF_SCHEMES)}) or a provider-native id")


# ── connection records + store ──────────────────────────────────────────────


@dataclass
class CloudConnection:
    """One read-only scan connection.  ``secret_encrypted`` never leaves
    the process — ``to_public_dict`` is the only API shape."""

    provider: str
    display_name: str
    tenant_id: str = "default"
    connection_id: str = ""
    role_ref: str = ""
    secret_encrypted: str = ""
    auth_params: dict[str, Any] = field(default_factory=dict)
    regions: list[str] = field(default_factory=list)
    status: str = STATUS_PENDING
    status_detail: str = ""
    scan_interval_minutes: Optional[int] = None
    last_scan_at: Optional[str] = None
    last_scan_id: Optional[str] = None
    created_at: str = ""

    def __post_init__(self) -> None:
        if not self.connection_id:
            self.connection_id = f"conn-{uuid4().hex[:12]}"
        if not self.created_at:
            self.created_at = _now().isoformat()

    def to_public_dict(self) -> dict[str, Any]:
        data = {k: getattr(self, k) for k in (
            "connection_id", "tenant_id", "provider", "display_name",
            "role_ref", "auth_params", "regions", "status", "status_detail",
            "scan_interval_minutes", "last_scan_at", "last_scan_id",
            "created_at")}
        data["has_secret"] = bool(self.secret_encrypted)
        return data

    def due(self, at: Optional[datetime] = None) -> bool:
        """True when the polling schedule says this connection needs a scan."""
        if self.status == STATUS_DISABLED or not self.scan_interval_minutes:
            return False
        if not self.last_scan_at:
            return True
        last = datetime.fromisoformat(self.last_scan_at)
        return (at or _now()) >= last + timedelta(
            minutes=self.scan_interval_minutes)


_SCHEMA = """
CREATE TABLE IF NOT EXISTS cloud_connections (
    connection_id TEXT PRIMARY KEY,
    tenant_id TEXT NOT NULL,
    doc TEXT NOT NULL
);
"""


class ConnectionStore:
    def __init__(self, path: str = ":memory:"):
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.executescript(_SCHEMA)
        self._lock = threading.Lock()

    def _save(self, conn: CloudConnection) -> None:
        doc = {**conn.to_public_dict(),
               "secret_encrypted": conn.secret_encrypted}
        doc.pop("has_secret", None)
        self._db.execute(
            "INSERT OR REPLACE INTO cloud_connections (connection_id,"
            " tenant_id, doc) VALUES (?,?,?)",
            (conn.connection_id, conn.tenant_id, json.dumps(doc)))

    def put(self, conn: CloudConnection) -> CloudConnection:
        with self._lock:
            self._save(conn)
            self._db.commit()
        return conn

    def get(self, tenant_id: str,
            connection_id: str) -> Optional[CloudConnection]:
        row = self._db.execute(
            "SELECT doc FROM cloud_connections WHERE connection_id=? AND"
            " tenant_id=?", (connection_id, tenant_id)).fetchone()
        return CloudConnection(**json.loads(row[0])) if row else None

    def list(self, tenant_id: str) -> list[CloudConnection]:
        return [CloudConnection(**json.loads(doc))
                for (doc,) in self._db.execute(
                    "SELECT doc FROM cloud_connections WHERE tenant_id=?"
                    " ORDER BY connection_id", (tenant_id,))]

    def delete(self, tenant_id: str, connection_id: str) -> bool:
        with self._lock:
            cur = self._db.execute(
                "DELETE FROM cloud_connections WHERE connection_id=? AND"
                " tenant_id=?", (connection_id, tenant_id))
            self._db.commit()
            return cur.rowcount > 0

    def mark_scanned(self, tenant_id: str, connection_id: str,
                     scan_id: str) -> Optional[CloudConnection]:
        with self._lock:
            conn = self.get(tenant_id, connection_id)
            if conn is None:
                return None
            conn.last_scan_at = _now().isoformat()
            conn.last_scan_id = scan_id
            conn.status = STATUS_ACTIVE
            self._save(conn)
            self._db.commit()
            return conn

    def due_connections(self, tenant_id: str,
                        at: Optional[datetime] = None) -> list[CloudConnection]:
        return [c for c in self.list(tenant_id) if c.due(at)]
