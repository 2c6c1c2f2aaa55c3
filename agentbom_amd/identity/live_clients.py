"""Live IdP clients for NHI discovery (Okta + Entra), HTTP-backed.

Implement the injectable client protocols identity/nhi.py consumes
(``list_oauth2_service_apps``/``list_api_tokens`` and
``list_service_principals``/``list_applications``) over the real APIs
(reference: src/agent_bom/identity/{okta_nhi,entra_nhi}.py live modes):

- Okta: org-URL REST with an SSWS API token, cursor pagination via the
  Link: rel="next" header;
- Entra: Microsoft Graph v1.0 with a bearer token, @odata.nextLink
  pagination.

Offline mode refuses at the first request; transports are injected in
tests.  Discovery itself stays behind the nhi.py enable flags.
"""

from __future__ import annotations

import re
from typing import Iterable, Optional

from agentbom_amd.utils.http_client import check_offline, create_client, request_with_retry

_MAX_PAGES = 20


class OktaHttpClient:
    """Okta management REST (read-only endpoints only)."""

    def __init__(self, org_url: str, token: str, client=None):
        check_offline(org_url)
        self.org_url = org_url.rstrip("/")
        self.token = token
        self.client = client or create_client(timeout=30.0)

    def _paged(self, path: str) -> Iterable[dict]:
        url = f"{self.org_url}{path}"
        for _ in range(_MAX_PAGES):
            resp = request_with_retry(
                self.client, "GET", url,
                headers={"Authorization": f"SSWS {self.token}",
                         "Accept": "application/json"})
            if resp is None or resp.status_code != 200:
                raise RuntimeError(
                    f"okta API {path} failed: "
                    f"{resp.status_code if resp is not None else 'unreachable'}")
            body = resp.json()
            if isinstance(body, list):
                yield from (x for x in body if isinstance(x, dict))
            nxt = None
            for link in resp.headers.get_list("Link") if hasattr(
                    resp.headers, "get_list") else [resp.headers.get("Link", "")]:
                m = re.search(r'<([^>]+)>;\s*rel="next"', link or "")
                if m:
                    nxt = m.group(1)
            if not nxt:
                return
            url = nxt

    def list_oauth2_service_apps(self) -> list[dict]:
        return list(self._paged("/api/v1/apps?limit=200"))

    def list_api_tokens(self) -> list[dict]:
        return list(self._paged("/api/v1/api-tokens"))


class EntraHttpClient:
    """Microsoft Graph v1.0 (read-only endpoints only)."""

    GRAPH = "https://graph.microsoft.com/v1.0"

    def __init__(self, token: str, client=None):
        check_offline(self.GRAPH)
        self.token = token
        self.client = client or create_client(timeout=30.0)

    def _paged(self, path: str) -> Iterable[dict]:
        url = f"{self.GRAPH}{path}"
        for _ in range(_MAX_PAGES):
            resp = request_with_retry(
                self.client, "GET", url,
                headers={"Authorization": f"Bearer {self.token}"})
            if resp is None or resp.status_code != 200:
                raise RuntimeError(
                    f"graph API {path} failed: "
                    f"{resp.status_code if resp is not None else 'unreachable'}")
            body = resp.json()
            yield from (x for x in body.get("value", []) or []
                        if isinstance(x, dict))
            url = body.get("@odata.nextLink")
            if not url:
                return

    def list_service_principals(self) -> list[dict]:
        return list(self._paged("/servicePrincipals?$top=999"))

    def list_applications(self) -> list[dict]:
        return list(self._paged("/applications?$top=999"))


def okta_client_from_env(env: Optional[dict[str, str]] = None,
                         client=None) -> Optional[OktaHttpClient]:
    import os

    e = env if env is not None else os.environ
    org = e.get("AGENT_BOM_OKTA_ORG_URL")
    tok = e.get("AGENT_BOM_OKTA_TOKEN")
    if not org or not tok:
        return None
    return OktaHttpClient(org, tok, client=client)


def entra_client_from_env(env: Optional[dict[str, str]] = None,
                          client=None) -> Optional[EntraHttpClient]:
    import os

    e = env if env is not None else os.environ
    tok = e.get("AGENT_BOM_ENTRA_TOKEN")
    if not tok:
        return None
    return EntraHttpClient(tok, client=client)
