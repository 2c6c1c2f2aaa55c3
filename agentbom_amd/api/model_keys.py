"""Virtual, scoped, revocable model-provider key broker.

Reference parity: src/agent_bom/api/model_key_broker.py — an operator
registers a REAL model-provider credential once; agent-bom mints
VIRTUAL keys that map to it without ever exposing it:

- real keys are sealed at rest with the SAME encrypt-then-MAC material
  as the connection broker (api/connections.py, AGENT_BOM_CONNECTIONS_KEY)
  and are write-only: never logged, never returned, decrypted only at
  resolve time;
- a virtual key is a bearer ``abvk_<public>_<secret>`` — only its
  SHA-256 hash persists, the raw token is returned exactly once;
- virtual keys are scoped (provider + optional model allowlist +
  optional bound holder), time-boxed, and revocable independently;
- resolution FAILS CLOSED: unknown/revoked/expired virtual key,
  out-of-scope provider/model/holder, or a disabled provider key all
  refuse.
"""

from __future__ import annotations

import hashlib
import json
import secrets
import sqlite3
import threading
from dataclasses import dataclass, field
from datetime import datetime, timedelta, timezone
from typing import Any, Optional
from uuid import uuid4

from agentbom_amd.api.connections import decrypt_secret, encrypt_secret


def _now() -> datetime:
    return datetime.now(timezone.utc)


def hash_token(token: str) -> str:
    return hashlib.sha256(token.encode()).hexdigest()


def generate_virtual_key() -> tuple[str, str, str]:
    public = secrets.token_hex(4)
    raw = f"abvk_{public}_{secrets.token_urlsafe(24)}"
    return raw, public, hash_token(raw)


class ModelKeyBrokerError(RuntimeError):
    pass


@dataclass
class ModelProviderKey:
    provider: str                  # openai | anthropic | azure-openai | ...
    key_sealed: str                # ciphertext — plaintext is write-only
    tenant_id: str = "default"
    provider_key_id: str = ""
    label: str = ""
    enabled: bool = True

    def __post_init__(self) -> None:
        if not self.provider_key_id:
            self.provider_key_id = f"mpk-{uuid4().hex[:10]}"

    def to_public_dict(self) -> dict[str, Any]:
        return {"provider_key_id": self.provider_key_id,
                "provider": self.provider, "label": self.label,
                "tenant_id": self.tenant_id, "enabled": self.enabled}


@dataclass
class VirtualModelKey:
    provider_key_id: str
    token_prefix: str
    token_hash: str
    tenant_id: str = "default"
    virtual_key_id: str = ""
    holder: str = ""                  # bound agent/blueprint ("" = any)
    model_allowlist: list[str] = field(default_factory=list)
    expires_at: str = ""
    revoked_at: str = ""
    uses: int = 0

    def __post_init__(self) -> None:
        if not self.virtual_key_id:
            self.virtual_key_id = f"vmk-{uuid4().hex[:10]}"

    def live(self, at: Optional[datetime] = None) -> bool:
        t = (at or _now()).isoformat()
        if self.revoked_at and self.revoked_at <= t:
            return False
        return not self.expires_at or self.expires_at > t

    def to_public_dict(self) -> dict[str, Any]:
        return {"virtual_key_id": self.virtual_key_id,
                "provider_key_id": self.provider_key_id,
                "token_prefix": self.token_prefix,
                "tenant_id": self.tenant_id, "holder": self.holder,
                "model_allowlist": list(self.model_allowlist),
                "expires_at": self.expires_at,
                "revoked_at": self.revoked_at, "uses": self.uses,
                "live": self.live()}


_SCHEMA = """
CREATE TABLE IF NOT EXISTS provider_keys (
    provider_key_id TEXT PRIMARY KEY,
    tenant_id TEXT NOT NULL,
    doc TEXT NOT NULL
);
CREATE TABLE IF NOT EXISTS virtual_keys (
    virtual_key_id TEXT PRIMARY KEY,
    tenant_id TEXT NOT NULL,
    token_hash TEXT NOT NULL UNIQUE,
    doc TEXT NOT NULL
);
"""


class ModelKeyBroker:
    def __init__(self, path: str = ":memory:",
                 env: Optional[dict] = None):
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.executescript(_SCHEMA)
        self._lock = threading.Lock()
        self._env = env  # crypto key source override (tests)

    # ── provider keys (write-only secrets) ────────────────────────────────

    def register_provider_key(self, provider: str, plaintext_key: str,
                              tenant_id: str = "default",
                              label: str = "") -> ModelProviderKey:
        rec = ModelProviderKey(
            provider=provider.lower(),
            key_sealed=encrypt_secret(plaintext_key, env=self._env),
            tenant_id=tenant_id, label=label)
        with self._lock:
            self._db.execute(
                "INSERT INTO provider_keys (provider_key_id, tenant_id, doc)"
                " VALUES (?,?,?)",
                (rec.provider_key_id, tenant_id,
                 json.dumps({**rec.to_public_dict(),
                             "key_sealed": rec.key_sealed})))
            self._db.commit()
        return rec

    def _provider_key(self, provider_key_id: str,
                      tenant_id: str) -> Optional[ModelProviderKey]:
        row = self._db.execute(
            "SELECT doc FROM provider_keys WHERE provider_key_id=? AND"
            " tenant_id=?", (provider_key_id, tenant_id)).fetchone()
        if row is None:
            return None
        d = json.loads(row[0])
        return ModelProviderKey(provider=d["provider"],
                                key_sealed=d["key_sealed"],
                                tenant_id=d["tenant_id"],
                                provider_key_id=d["provider_key_id"],
                                label=d.get("label", ""),
                                enabled=d.get("enabled", True))

    def disable_provider_key(self, provider_key_id: str,
                             tenant_id: str = "default") -> bool:
        with self._lock:
            rec = self._provider_key(provider_key_id, tenant_id)
            if rec is None:
                return False
            rec.enabled = False
            self._db.execute(
                "UPDATE provider_keys SET doc=? WHERE provider_key_id=?",
                (json.dumps({**rec.to_public_dict(),
                             "key_sealed": rec.key_sealed}),
                 provider_key_id))
            self._db.commit()
            return True

    # ── virtual keys ──────────────────────────────────────────────────────

    def mint_virtual_key(self, provider_key_id: str,
                         tenant_id: str = "default", holder: str = "",
                         model_allowlist: Optional[list[str]] = None,
                         ttl_hours: float = 24.0) -> tuple[VirtualModelKey, str]:
        if self._provider_key(provider_key_id, tenant_id) is None:
            raise ModelKeyBrokerError("unknown provider key")
        raw, public, thash = generate_virtual_key()
        rec = VirtualModelKey(
            provider_key_id=provider_key_id, token_prefix=public,
            token_hash=thash, tenant_id=tenant_id, holder=holder,
            model_allowlist=[m.lower() for m in model_allowlist or []],
            expires_at=(_now() + timedelta(hours=ttl_hours)).isoformat())
        with self._lock:
            self._db.execute(
                "INSERT INTO virtual_keys (virtual_key_id, tenant_id,"
                " token_hash, doc) VALUES (?,?,?,?)",
                (rec.virtual_key_id, tenant_id, thash,
                 json.dumps(rec.to_public_dict() | {"token_hash": thash})))
            self._db.commit()
        return rec, raw  # raw returned exactly once

    def _virtual_by_hash(self, token_hash: str) -> Optional[VirtualModelKey]:
        row = self._db.execute(
            "SELECT doc FROM virtual_keys WHERE token_hash=?",
            (token_hash,)).fetchone()
        if row is None:
            return None
        d = json.loads(row[0])
        d.pop("live", None)
        return VirtualModelKey(**d)

    def revoke_virtual_key(self, virtual_key_id: str,
                           tenant_id: str = "default") -> bool:
        with self._lock:
            row = self._db.execute(
                "SELECT doc FROM virtual_keys WHERE virtual_key_id=? AND"
                " tenant_id=?", (virtual_key_id, tenant_id)).fetchone()
            if row is None:
                return False
            d = json.loads(row[0])
            d["revoked_at"] = _now().isoformat()
            self._db.execute(
                "UPDATE virtual_keys SET doc=? WHERE virtual_key_id=?",
                (json.dumps(d), virtual_key_id))
            self._db.commit()
            return True

    def list_virtual_keys(self, tenant_id: str = "default") -> list[dict]:
        return [json.loads(doc) | {"token_hash": None}
                for (doc,) in self._db.execute(
                    "SELECT doc FROM virtual_keys WHERE tenant_id=?"
                    " ORDER BY virtual_key_id", (tenant_id,))]

    # ── resolution (the gateway-side hot path; FAILS CLOSED) ──────────────

    def resolve(self, raw_token: str, model: str = "",
                holder: str = "") -> dict[str, Any]:
        """virtual bearer -> {provider, real_key} after every scope check.

        The real key is decrypted HERE, immediately before the model
        call — it never persists in the return path beyond the caller.
        """
        rec = self._virtual_by_hash(hash_token(raw_token))
        if rec is None:
            raise ModelKeyBrokerError("unknown virtual key")
        if not rec.live():
            raise ModelKeyBrokerError("virtual key revoked or expired")
        if rec.holder and holder != rec.holder:
            raise ModelKeyBrokerError(
                "virtual key is bound to a different holder")
        if rec.model_allowlist and model.lower() not in rec.model_allowlist:
            raise ModelKeyBrokerError(
                f"model {model!r} is not in the key's allowlist")
        pk = self._provider_key(rec.provider_key_id, rec.tenant_id)
        if pk is None or not pk.enabled:
            raise ModelKeyBrokerError("underlying provider key is disabled")
        with self._lock:
            rec.uses += 1
            self._db.execute(
                "UPDATE virtual_keys SET doc=? WHERE virtual_key_id=?",
                (json.dumps(rec.to_public_dict()
                            | {"token_hash": rec.token_hash}),
                 rec.virtual_key_id))
            self._db.commit()
        return {"provider": pk.provider,
                "real_key": decrypt_secret(pk.key_sealed, env=self._env),
                "virtual_key_id": rec.virtual_key_id,
                "holder": rec.holder, "uses": rec.uses}
