"""OAuth 2.1 AS (MCP auth broker): registration, PKCE grant, tokens."""

from __future__ import annotations

import base64
import hashlib
import time

import pytest

from agentbom_amd.api.oauth_as import (
    CODE_TTL_S,
    OAuthAuthorizationServer,
    OAuthError,
)
from agentbom_amd.api.oidc import generate_rsa_keypair

_KEYS = generate_rsa_keypair(seed=11)  # one 2048-bit gen for the module


@pytest.fixture
def srv():
    return OAuthAuthorizationServer(issuer="https://gw.example",
                                    keypair=_KEYS)


def _pkce():
    verifier = "v" * 48
    challenge = base64.urlsafe_b64encode(
        hashlib.sha256(verifier.encode()).digest()).rstrip(b"=").decode()
    return verifier, challenge


class TestMetadataAndRegistration:
    def test_rfc8414_metadata(self, srv):
        md = srv.metadata()
        assert md["issuer"] == "https://gw.example"
        assert md["code_challenge_methods_supported"] == ["S256"]
        assert "implicit" not in md["grant_types_supported"]
        assert srv.jwks()["keys"][0]["kty"] == "RSA"

    def test_public_client_registration(self, srv):
        out = srv.register_client(["https://app.example/cb"], "mcp-client")
        assert out["client_id"].startswith("abc-")
        assert "client_secret" not in out
        assert out["token_endpoint_auth_method"] == "none"

    def test_confidential_secret_returned_once(self, srv):
        out = srv.register_client([], "svc", confidential=True)
        assert out["client_secret"]
        client = srv.get_client(out["client_id"])
        assert client.secret_hash and out["client_secret"] not in client.secret_hash

    def test_insecure_redirect_refused(self, srv):
        with pytest.raises(OAuthError, match="redirect_uri"):
            srv.register_client(["http://evil.example/cb"])
        # localhost is allowed for native clients
        srv.register_client(["http://localhost:8123/cb"])


class TestCodeGrant:
    def _register(self, srv):
        return srv.register_client(["https://app.example/cb"])["client_id"]

    def test_full_pkce_flow(self, srv):
        cid = self._register(srv)
        verifier, challenge = _pkce()
        out = srv.authorize(cid, "https://app.example/cb", challenge,
                            state="xyz")
        assert out["state"] == "xyz"
        tok = srv.token_authorization_code(out["code"], cid,
                                           "https://app.example/cb", verifier)
        assert tok["token_type"] == "Bearer"
        claims = srv.verify_access_token(tok["access_token"])
        assert claims["sub"] == cid and claims["scope"] == "mcp"

    def test_code_single_use(self, srv):
        cid = self._register(srv)
        verifier, challenge = _pkce()
        code = srv.authorize(cid, "https://app.example/cb", challenge)["code"]
        srv.token_authorization_code(code, cid, "https://app.example/cb",
                                     verifier)
        with pytest.raises(OAuthError, match="used"):
            srv.token_authorization_code(code, cid, "https://app.example/cb",
                                         verifier)

    def test_pkce_mandatory_and_s256_only(self, srv):
        cid = self._register(srv)
        with pytest.raises(OAuthError, match="S256"):
            srv.authorize(cid, "https://app.example/cb", "", "S256")
        with pytest.raises(OAuthError, match="S256"):
            srv.authorize(cid, "https://app.example/cb", "challenge", "plain")

    def test_wrong_verifier_and_binding(self, srv):
        cid = self._register(srv)
        _, challenge = _pkce()
        code = srv.authorize(cid, "https://app.example/cb", challenge)["code"]
        with pytest.raises(OAuthError, match="PKCE"):
            srv.token_authorization_code(code, cid, "https://app.example/cb",
                                         "wrong-verifier")
        # a failed PKCE attempt burns nothing else, but the code is bound:
        other = srv.register_client(["https://app.example/cb"])["client_id"]
        with pytest.raises(OAuthError, match="bound"):
            srv.token_authorization_code(code, other,
                                         "https://app.example/cb", "v" * 48)

    def test_code_expiry(self, srv, monkeypatch):
        cid = self._register(srv)
        verifier, challenge = _pkce()
        code = srv.authorize(cid, "https://app.example/cb", challenge)["code"]
        srv._codes[code].issued_at -= CODE_TTL_S + 1
        with pytest.raises(OAuthError, match="expired"):
            srv.token_authorization_code(code, cid, "https://app.example/cb",
                                         verifier)


class TestClientCredentials:
    def test_confidential_only(self, srv):
        pub = srv.register_client(["https://app.example/cb"])["client_id"]
        with pytest.raises(OAuthError, match="confidential"):
            srv.token_client_credentials(pub, "whatever")
        conf = srv.register_client([], confidential=True)
        tok = srv.token_client_credentials(conf["client_id"],
                                           conf["client_secret"])
        assert srv.verify_access_token(tok["access_token"])["sub"] == \
            conf["client_id"]
        with pytest.raises(OAuthError, match="secret"):
            srv.token_client_credentials(conf["client_id"], "bad")


class TestTokenValidation:
    def test_tampered_and_expired(self, srv):
        conf = srv.register_client([], confidential=True)
        tok = srv.token_client_credentials(conf["client_id"],
                                           conf["client_secret"])["access_token"]
        h, c, s = tok.split(".")
        with pytest.raises(OAuthError, match="signature"):
            srv.verify_access_token(f"{h}.{c}.{'A' * len(s)}")
        other = OAuthAuthorizationServer(issuer="https://other.example",
                                         keypair=_KEYS)
        with pytest.raises(OAuthError, match="issuer"):
            other.verify_access_token(tok)

    def test_bounded_stores(self, srv):
        from agentbom_amd.api import oauth_as as mod

        old = mod._MAX_CODES
        mod._MAX_CODES = 5
        try:
            cid = srv.register_client(["https://app.example/cb"])["client_id"]
            _, ch = _pkce()
            codes = [srv.authorize(cid, "https://app.example/cb", ch)["code"]
                     for _ in range(10)]
            assert len(srv._codes) <= 5
            assert codes[0] not in srv._codes  # oldest evicted
        finally:
            mod._MAX_CODES = old


class TestApi:
    @pytest.fixture()
    def client(self, monkeypatch):
        monkeypatch.setenv("AGENT_BOM_OAUTH_AS", "1")
        from starlette.testclient import TestClient

        from agentbom_amd.api.server import create_app

        c = TestClient(create_app())
        # inject the module's pre-generated keypair (2048-bit gen is slow)
        from agentbom_amd.api.oauth_as import OAuthAuthorizationServer

        c.app.state.abom.oauth_as = OAuthAuthorizationServer(
            issuer="https://gw.example", keypair=_KEYS)
        return c

    def test_discovery_register_flow(self, client):
        md = client.get("/.well-known/oauth-authorization-server").json()
        assert md["issuer"] == "https://gw.example"
        reg = client.post("/oauth/register", json={
            "redirect_uris": ["https://app.example/cb"],
            "client_name": "ide"}).json()
        verifier, challenge = _pkce()
        auth = client.post("/oauth/authorize", json={
            "client_id": reg["client_id"],
            "redirect_uri": "https://app.example/cb",
            "code_challenge": challenge,
            "code_challenge_method": "S256", "state": "s1"}).json()
        tok = client.post("/oauth/token", json={
            "grant_type": "authorization_code", "code": auth["code"],
            "client_id": reg["client_id"],
            "redirect_uri": "https://app.example/cb",
            "code_verifier": verifier}).json()
        assert tok["token_type"] == "Bearer"
        assert client.get("/oauth/jwks.json").json()["keys"]

    def test_disabled_404(self, monkeypatch):
        monkeypatch.delenv("AGENT_BOM_OAUTH_AS", raising=False)
        from starlette.testclient import TestClient

        from agentbom_amd.api.server import create_app

        c = TestClient(create_app())
        assert c.get("/.well-known/oauth-authorization-server").status_code == 404
