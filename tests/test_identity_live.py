"""Live IdP clients (Okta/Entra) over MockTransport, end-to-end into the
NHI discovery that already consumes the injectable-client protocol."""

from __future__ import annotations

import json

import httpx
import pytest

from agentbom_amd.identity.live_clients import (
    EntraHttpClient,
    OktaHttpClient,
    entra_client_from_env,
    okta_client_from_env,
)
from agentbom_amd.identity.nhi import discover_entra_nhis, discover_okta_nhis
from agentbom_amd.utils.http_client import OfflineError, set_offline


@pytest.fixture(autouse=True)
def _online():
    set_offline(False)
    yield
    set_offline(False)


def _okta_handler(request: httpx.Request) -> httpx.Response:
    assert request.headers["Authorization"] == "SSWS tok-okta"
    if request.url.path == "/api/v1/apps":
        # page 1 with a next link
        if "after" not in request.url.params:
            return httpx.Response(200, json=[{
                "id": "app1", "label": "ci-deployer", "status": "ACTIVE",
                "signOnMode": "OPENID_CONNECT",
                "settings": {"oauthClient": {
                    "application_type": "service",
                    "grant_types": ["client_credentials"]}},
            }], headers={"Link": ('<https://org.okta.example/api/v1/apps'
                                  '?limit=200&after=app1>; rel="next"')})
        return httpx.Response(200, json=[{
            "id": "app2", "label": "batch-runner", "status": "ACTIVE",
            "signOnMode": "OPENID_CONNECT",
            "settings": {"oauthClient": {
                "application_type": "service",
                "grant_types": ["client_credentials"]}},
        }])
    if request.url.path == "/api/v1/api-tokens":
        return httpx.Response(200, json=[{
            "id": "tok1", "name": "terraform", "userId": "u1",
            "created": "2023-01-01T00:00:00Z"}])
    return httpx.Response(404)


def test_okta_live_pagination_into_discovery():
    client = httpx.Client(transport=httpx.MockTransport(_okta_handler))
    okta = OktaHttpClient("https://org.okta.example", "tok-okta", client=client)
    apps = okta.list_oauth2_service_apps()
    assert [a["id"] for a in apps] == ["app1", "app2"]  # Link-header paging

    result = discover_okta_nhis(client=okta)
    assert result.ok
    names = {i.name for i in result.identities}
    assert {"ci-deployer", "batch-runner", "terraform"} <= names


def _entra_handler(request: httpx.Request) -> httpx.Response:
    assert request.headers["Authorization"] == "Bearer tok-entra"
    if "/servicePrincipals" in request.url.path:
        if "skiptoken" not in str(request.url):
            return httpx.Response(200, json={
                "value": [{"id": "sp1", "displayName": "pipeline-sp",
                           "appId": "a-1", "accountEnabled": True,
                           "servicePrincipalType": "Application"}],
                "@odata.nextLink": ("https://graph.microsoft.com/v1.0/"
                                    "servicePrincipals?$top=999&$skiptoken=x")})
        return httpx.Response(200, json={"value": [{
            "id": "sp2", "displayName": "etl-sp", "appId": "a-2",
            "accountEnabled": True, "servicePrincipalType": "Application"}]})
    if "/applications" in request.url.path:
        return httpx.Response(200, json={"value": [{
            "appId": "a-1", "displayName": "pipeline-app",
            "passwordCredentials": [{"endDateTime": "2020-01-01T00:00:00Z"}]}]})
    return httpx.Response(404)


def test_entra_live_pagination_into_discovery():
    client = httpx.Client(transport=httpx.MockTransport(_entra_handler))
    entra = EntraHttpClient("tok-entra", client=client)
    sps = entra.list_service_principals()
    assert [s["id"] for s in sps] == ["sp1", "sp2"]  # nextLink paging

    result = discover_entra_nhis(client=entra)
    assert result.ok
    assert {i.name for i in result.identities} >= {"pipeline-sp", "etl-sp"}


def test_env_factories(monkeypatch):
    monkeypatch.delenv("AGENT_BOM_OKTA_ORG_URL", raising=False)
    monkeypatch.delenv("AGENT_BOM_OKTA_TOKEN", raising=False)
    monkeypatch.delenv("AGENT_BOM_ENTRA_TOKEN", raising=False)
    assert okta_client_from_env() is None
    assert entra_client_from_env() is None
    monkeypatch.setenv("AGENT_BOM_OKTA_ORG_URL", "https://org.okta.example")
    monkeypatch.setenv("AGENT_BOM_OKTA_TOKEN", "t")
    assert okta_client_from_env() is not None


def test_api_error_becomes_discovery_error():
    def boom(request):
        return httpx.Response(500)

    client = httpx.Client(transport=httpx.MockTransport(boom))
    okta = OktaHttpClient("https://org.okta.example", "t", client=client)
    result = discover_okta_nhis(client=okta)
    assert not result.ok and result.warnings


def test_offline_refused():
    set_offline(True)
    with pytest.raises(OfflineError):
        OktaHttpClient("https://org.okta.example", "t")
    with pytest.raises(OfflineError):
        EntraHttpClient("t")
