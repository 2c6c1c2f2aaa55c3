"""Non-human identity (NHI) discovery: Okta + Microsoft Entra connectors.

Enumerates *machine* identities an IdP already holds — OAuth2 service apps /
service principals (client-credentials workloads) and API tokens — and
normalizes them into one record shape the graph NHI overlay can project as
``managed_identity`` nodes.

Trust posture (parity with the reference connectors,
reference: src/agent_bom/identity/okta_nhi.py:1-60, entra_nhi.py:1-36):

* Read-only / reference-only — only list reads; no secret material is ever
  captured, only references, timestamps and credential *end dates*.
* Gated, default OFF — ``AGENT_BOM_OKTA_DISCOVERY`` / ``AGENT_BOM_ENTRA_DISCOVERY``
  must be truthy or discovery returns ``DISABLED`` without doing anything.
* Token-authenticated — ``AGENT_BOM_OKTA_TOKEN`` + ``AGENT_BOM_OKTA_ORG_URL``,
  ``AGENT_BOM_ENTRA_TOKEN`` (+ optional ``AGENT_BOM_ENTRA_TENANT_ID``).
* Never raises — every failure degrades to a populated result carrying a
  status and a user-safe warning.

This environment has no egress, so alongside the injectable client protocol
(tests and operators can hand any object with the list methods) each connector
also accepts an **exported inventory file** (``AGENT_BOM_OKTA_EXPORT`` /
``AGENT_BOM_ENTRA_EXPORT``): a JSON dump of the same list API payloads,
produced out-of-band.  File-based discovery follows the identical
normalization path, so posture review works fully offline.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from enum import Enum
from pathlib import Path
from typing import Any, Optional

_TRUTHY = frozenset({"1", "true", "yes", "on"})

OKTA_READ_PERMISSIONS: tuple[str, ...] = ("okta.apps.read", "okta.apiTokens.read")
ENTRA_READ_PERMISSIONS: tuple[str, ...] = ("Application.Read.All", "Directory.Read.All")

_SERVICE_GRANT_TYPES = frozenset({"client_credentials"})
_MAX_RESULTS = 2000


class NHIDiscoveryStatus(str, Enum):
    """Outcome of one NHI discovery run (locked vocabulary)."""

    OK = "ok"
    DISABLED = "disabled"
    MISSING_CREDENTIALS = "missing_credentials"
    MISSING_CLIENT = "missing_client"
    ERROR = "error"


@dataclass(frozen=True)
class DiscoveredNonHumanIdentity:
    """One normalized machine identity. Credential values are never captured."""

    identity_id: str
    name: str
    identity_type: str  # service_account | api_token | service_principal
    provider: str = "okta"
    status: str = "active"
    owner: Optional[str] = None
    created_at: Optional[str] = None
    last_used_at: Optional[str] = None
    credential_expires_at: Optional[str] = None
    scopes: tuple[str, ...] = ()
    raw_identity: dict[str, Any] = field(default_factory=dict)

    def to_dict(self) -> dict[str, Any]:
        return {
            "identity_id": self.identity_id,
            "name": self.name,
            "identity_type": self.identity_type,
            "provider": self.provider,
            "status": self.status,
            "owner": self.owner,
            "created_at": self.created_at,
            "last_used_at": self.last_used_at,
            "credential_expires_at": self.credential_expires_at,
            "scopes": list(self.scopes),
            "raw_identity": dict(self.raw_identity),
        }


@dataclass(frozen=True)
class NHIDiscoveryResult:
    """Result envelope for one discovery run. Never raises into callers."""

    status: NHIDiscoveryStatus
    identities: tuple[DiscoveredNonHumanIdentity, ...] = ()
    warnings: tuple[str, ...] = ()
    org_url: Optional[str] = None

    @property
    def ok(self) -> bool:
        return self.status is NHIDiscoveryStatus.OK

    def to_dict(self) -> dict[str, Any]:
        return {
            "status": self.status.value,
            "identities": [i.to_dict() for i in self.identities],
            "warnings": list(self.warnings),
            "org_url": self.org_url,
        }


def _is_truthy(value: Optional[str]) -> bool:
    return value is not None and value.strip().lower() in _TRUTHY


def _load_export(path: str) -> Optional[dict[str, Any]]:
    try:
        data = json.loads(Path(path).read_text())
        return data if isinstance(data, dict) else None
    except (OSError, json.JSONDecodeError):
        return None


# ── Okta ────────────────────────────────────────────────────────────────────


def _is_okta_service_app(app: dict[str, Any]) -> bool:
    """Non-human OAuth2 app: OIDC sign-on with service type or client_credentials."""
    if str(app.get("signOnMode") or "").upper() != "OPENID_CONNECT":
        return False
    settings = app.get("settings")
    oauth = settings.get("oauthClient", {}) if isinstance(settings, dict) else {}
    app_type = str(oauth.get("application_type") or "").lower()
    grants = {str(g).lower() for g in (oauth.get("grant_types") or [])}
    return app_type == "service" or bool(grants & _SERVICE_GRANT_TYPES)


def _okta_grants(app: dict[str, Any]) -> tuple[str, ...]:
    settings = app.get("settings")
    oauth = settings.get("oauthClient", {}) if isinstance(settings, dict) else {}
    return tuple(str(g) for g in (oauth.get("grant_types") or []) if str(g).strip())


def _normalize_okta_app(app: dict[str, Any]) -> Optional[DiscoveredNonHumanIdentity]:
    app_id = str(app.get("id") or "").strip()
    if not app_id:
        return None
    return DiscoveredNonHumanIdentity(
        identity_id=app_id,
        name=str(app.get("label") or app.get("name") or app_id),
        identity_type="service_account",
        provider="okta",
        status=str(app.get("status") or "ACTIVE").lower(),
        created_at=app.get("created"),
        last_used_at=app.get("lastUpdated"),
        scopes=_okta_grants(app),
        raw_identity={"id": app_id, "signOnMode": app.get("signOnMode"),
                      "created": app.get("created")},
    )


def _normalize_okta_token(tok: dict[str, Any]) -> Optional[DiscoveredNonHumanIdentity]:
    tok_id = str(tok.get("id") or "").strip()
    if not tok_id:
        return None
    return DiscoveredNonHumanIdentity(
        identity_id=tok_id,
        name=str(tok.get("name") or tok_id),
        identity_type="api_token",
        provider="okta",
        status="active",
        owner=str(tok.get("userId") or "") or None,
        created_at=tok.get("created"),
        last_used_at=tok.get("lastUpdated"),
        credential_expires_at=tok.get("expiresAt"),
        raw_identity={"id": tok_id, "tokenWindow": tok.get("tokenWindow")},
    )


def discover_okta_nhis(client: Any = None, export_path: Optional[str] = None,
                       env: Optional[dict[str, str]] = None) -> NHIDiscoveryResult:
    """Discover Okta machine identities (service apps + API tokens).

    ``client`` may be any object exposing ``list_oauth2_service_apps()`` and
    ``list_api_tokens()`` (injected in tests / by live deployments);
    ``export_path`` points at a JSON dump ``{"apps": [...], "api_tokens": [...]}``.
    """
    e = env if env is not None else os.environ
    export_path = export_path or e.get("AGENT_BOM_OKTA_EXPORT")
    if not _is_truthy(e.get("AGENT_BOM_OKTA_DISCOVERY")) and client is None and not export_path:
        return NHIDiscoveryResult(status=NHIDiscoveryStatus.DISABLED)

    warnings: list[str] = []
    apps: list[dict[str, Any]] = []
    tokens: list[dict[str, Any]] = []
    org_url = e.get("AGENT_BOM_OKTA_ORG_URL")

    if client is not None:
        try:
            apps = list(client.list_oauth2_service_apps())
            tokens = list(client.list_api_tokens())
        except Exception as exc:  # noqa: BLE001 — never-raises contract
            return NHIDiscoveryResult(status=NHIDiscoveryStatus.ERROR,
                                      warnings=(f"okta discovery failed: {exc}",),
                                      org_url=org_url)
    elif export_path:
        data = _load_export(export_path)
        if data is None:
            return NHIDiscoveryResult(
                status=NHIDiscoveryStatus.ERROR,
                warnings=(f"okta export unreadable: {export_path}",))
        apps = [a for a in data.get("apps", []) if isinstance(a, dict)]
        tokens = [t for t in data.get("api_tokens", []) if isinstance(t, dict)]
        org_url = org_url or data.get("org_url")
    else:
        if not e.get("AGENT_BOM_OKTA_TOKEN") or not org_url:
            return NHIDiscoveryResult(status=NHIDiscoveryStatus.MISSING_CREDENTIALS,
                                      warnings=("AGENT_BOM_OKTA_TOKEN / _ORG_URL unset",))
        return NHIDiscoveryResult(
            status=NHIDiscoveryStatus.MISSING_CLIENT,
            warnings=("no egress in this environment; pass a client or "
                      "AGENT_BOM_OKTA_EXPORT",),
            org_url=org_url)

    out: list[DiscoveredNonHumanIdentity] = []
    for app in apps[:_MAX_RESULTS]:
        if _is_okta_service_app(app):
            rec = _normalize_okta_app(app)
            if rec:
                out.append(rec)
    for tok in tokens[:_MAX_RESULTS]:
        rec = _normalize_okta_token(tok)
        if rec:
            out.append(rec)
    if len(apps) > _MAX_RESULTS or len(tokens) > _MAX_RESULTS:
        warnings.append(f"truncated to {_MAX_RESULTS} records per list")
    return NHIDiscoveryResult(status=NHIDiscoveryStatus.OK, identities=tuple(out),
                              warnings=tuple(warnings), org_url=org_url)


# ── Microsoft Entra ─────────────────────────────────────────────────────────


def _earliest_credential_end(app: dict[str, Any]) -> Optional[str]:
    """Earliest password/key credential end date — flags stale/long-lived secrets."""
    ends = []
    for kind in ("passwordCredentials", "keyCredentials"):
        for cred in app.get(kind) or []:
            if isinstance(cred, dict) and cred.get("endDateTime"):
                ends.append(str(cred["endDateTime"]))
    return min(ends) if ends else None


def _normalize_entra_sp(sp: dict[str, Any],
                        app_by_appid: dict[str, dict[str, Any]]) -> Optional[DiscoveredNonHumanIdentity]:
    sp_id = str(sp.get("id") or "").strip()
    if not sp_id:
        return None
    app = app_by_appid.get(str(sp.get("appId") or ""), {})
    scopes = tuple(
        str(r.get("value") or r.get("id") or "")
        for res in (sp.get("requiredResourceAccess") or app.get("requiredResourceAccess") or [])
        if isinstance(res, dict)
        for r in (res.get("resourceAccess") or [])
        if isinstance(r, dict)
    )
    return DiscoveredNonHumanIdentity(
        identity_id=sp_id,
        name=str(sp.get("displayName") or sp_id),
        identity_type="service_principal",
        provider="entra",
        status="active" if sp.get("accountEnabled", True) else "disabled",
        owner=str(sp.get("appOwnerOrganizationId") or "") or None,
        created_at=sp.get("createdDateTime") or app.get("createdDateTime"),
        credential_expires_at=_earliest_credential_end(app) or _earliest_credential_end(sp),
        scopes=tuple(s for s in scopes if s),
        raw_identity={"id": sp_id, "appId": sp.get("appId"),
                      "servicePrincipalType": sp.get("servicePrincipalType")},
    )


def discover_entra_nhis(client: Any = None, export_path: Optional[str] = None,
                        env: Optional[dict[str, str]] = None) -> NHIDiscoveryResult:
    """Discover Entra service principals (+ app registrations for credential expiry).

    ``client`` exposes ``list_service_principals()`` and ``list_applications()``;
    ``export_path`` points at ``{"service_principals": [...], "applications": [...]}``.
    """
    e = env if env is not None else os.environ
    export_path = export_path or e.get("AGENT_BOM_ENTRA_EXPORT")
    if not _is_truthy(e.get("AGENT_BOM_ENTRA_DISCOVERY")) and client is None and not export_path:
        return NHIDiscoveryResult(status=NHIDiscoveryStatus.DISABLED)

    sps: list[dict[str, Any]] = []
    apps: list[dict[str, Any]] = []
    tenant = e.get("AGENT_BOM_ENTRA_TENANT_ID")

    if client is not None:
        try:
            sps = list(client.list_service_principals())
            apps = list(client.list_applications())
        except Exception as exc:  # noqa: BLE001 — never-raises contract
            return NHIDiscoveryResult(status=NHIDiscoveryStatus.ERROR,
                                      warnings=(f"entra discovery failed: {exc}",),
                                      org_url=tenant)
    elif export_path:
        data = _load_export(export_path)
        if data is None:
            return NHIDiscoveryResult(
                status=NHIDiscoveryStatus.ERROR,
                warnings=(f"entra export unreadable: {export_path}",))
        sps = [s for s in data.get("service_principals", []) if isinstance(s, dict)]
        apps = [a for a in data.get("applications", []) if isinstance(a, dict)]
        tenant = tenant or data.get("tenant_id")
    else:
        if not e.get("AGENT_BOM_ENTRA_TOKEN"):
            return NHIDiscoveryResult(status=NHIDiscoveryStatus.MISSING_CREDENTIALS,
                                      warnings=("AGENT_BOM_ENTRA_TOKEN unset",))
        return NHIDiscoveryResult(
            status=NHIDiscoveryStatus.MISSING_CLIENT,
            warnings=("no egress in this environment; pass a client or "
                      "AGENT_BOM_ENTRA_EXPORT",),
            org_url=tenant)

    app_by_appid = {str(a.get("appId") or ""): a for a in apps}
    out = []
    for sp in sps[:_MAX_RESULTS]:
        rec = _normalize_entra_sp(sp, app_by_appid)
        if rec:
            out.append(rec)
    warnings = ()
    if len(sps) > _MAX_RESULTS:
        warnings = (f"truncated to {_MAX_RESULTS} service principals",)
    return NHIDiscoveryResult(status=NHIDiscoveryStatus.OK, identities=tuple(out),
                              warnings=warnings, org_url=tenant)


def discover_all_nhis(okta_client: Any = None, entra_client: Any = None,
                      env: Optional[dict[str, str]] = None) -> dict[str, NHIDiscoveryResult]:
    """Run every connector; each result is independent (one failing ≠ all failing)."""
    return {
        "okta": discover_okta_nhis(client=okta_client, env=env),
        "entra": discover_entra_nhis(client=entra_client, env=env),
    }
