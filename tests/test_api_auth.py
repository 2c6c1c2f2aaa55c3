"""Auth/tenancy extensions: OIDC bearer, delegation tokens,
quotas, SCIM provisioning."""

import time

import pytest
from starlette.testclient import TestClient

from agentbom_amd.api.auth import (
    AuthError,
    DelegationTokens,
    QuotaTracker,
    ScimUserStore,
    mint_test_jwt,
    role_from_claims,
    verify_oidc_bearer,
)
from agentbom_amd.api.server import create_app

SECRET = "unit-test-oidc-secret"


class TestOidc:
    def _claims(self, **kw):
        return {"sub": "alice", "iss": "https://idp.local",
                "aud": "agent-bom", "exp": time.time() + 600, **kw}

    def test_valid_token_roundtrip(self):
        tok = mint_test_jwt(SECRET, self._claims(roles=["abom:operator"]))
        claims = verify_oidc_bearer(tok, secret=SECRET,
                                    issuer="https://idp.local",
                                    audience="agent-bom")
        assert claims["sub"] == "alice"
        assert role_from_claims(claims) == "operator"

    def test_signature_tamper_rejected(self):
        tok = mint_test_jwt(SECRET, self._claims())
        bad = tok[:-4] + ("AAAA" if not tok.endswith("AAAA") else "BBBB")
        with pytest.raises(AuthError, match="signature|undecodable"):
            verify_oidc_bearer(bad, secret=SECRET)

    def test_alg_none_and_rs256_rejected(self):
        import base64
        import json as j

        def seg(d):
            return base64.urlsafe_b64encode(j.dumps(d).encode()).decode().rstrip("=")

        tok = f"{seg({'alg': 'none'})}.{seg(self._claims())}."
        with pytest.raises(AuthError, match="not accepted|malformed"):
            verify_oidc_bearer(tok, secret=SECRET)
        # RS256 now dispatches to the JWKS path (api/oidc.py) and fails
        # closed when no key set is configured
        tok = f"{seg({'alg': 'RS256'})}.{seg(self._claims())}."
        with pytest.raises(AuthError, match="RS256 requires|signature|JWKS"):
            verify_oidc_bearer(tok, secret=SECRET)

    def test_expiry_and_issuer_audience(self):
        tok = mint_test_jwt(SECRET, self._claims(exp=time.time() - 3600))
        with pytest.raises(AuthError, match="expired"):
            verify_oidc_bearer(tok, secret=SECRET)
        tok2 = mint_test_jwt(SECRET, self._claims(iss="evil"))
        with pytest.raises(AuthError, match="issuer"):
            verify_oidc_bearer(tok2, secret=SECRET, issuer="https://idp.local")
        tok3 = mint_test_jwt(SECRET, self._claims(aud="other"))
        with pytest.raises(AuthError, match="audience"):
            verify_oidc_bearer(tok3, secret=SECRET, audience="agent-bom")

    def test_unconfigured_fails_closed(self, monkeypatch):
        monkeypatch.delenv("AGENT_BOM_OIDC_SECRET", raising=False)
        with pytest.raises(AuthError, match="not configured|undecodable|malformed"):
            verify_oidc_bearer("a.b.c")

    def test_role_mapping_defaults_viewer(self):
        assert role_from_claims({"sub": "x"}) == "viewer"
        assert role_from_claims({"scope": "openid abom:admin"}) == "admin"
        assert role_from_claims({"role": "operator"}) == "operator"


class TestDelegation:
    def test_mint_verify_revoke(self):
        d = DelegationTokens(secret="s3cret")
        tok = d.mint("operator", ["scan:run"], ttl_s=60, tenant_id="t1")
        payload = d.verify(tok)
        assert payload["role"] == "operator"
        assert payload["tenant_id"] == "t1"
        d.revoke(payload["jti"])
        with pytest.raises(AuthError, match="revoked"):
            d.verify(tok)

    def test_expired_and_tampered(self):
        d = DelegationTokens(secret="s3cret")
        tok = d.mint("viewer", [], ttl_s=5)
        with pytest.raises(AuthError, match="expired"):
            d.verify(tok, now=time.time() + 60)
        tok2 = d.mint("viewer", [], ttl_s=60)
        with pytest.raises(AuthError, match="mismatch|malformed"):
            d.verify(tok2[:-3] + "xyz")

    def test_admin_role_never_delegated(self):
        d = DelegationTokens(secret="s3cret")
        with pytest.raises(AuthError, match="only operator/viewer"):
            d.mint("admin", [])

    def test_cross_secret_rejected(self):
        tok = DelegationTokens(secret="one").mint("viewer", [])
        with pytest.raises(AuthError):
            DelegationTokens(secret="two").verify(tok)


class TestQuota:
    def test_limit_and_retry_after(self):
        q = QuotaTracker(scans_per_hour=2, window_s=100)
        assert q.check_and_record("k", now=0.0) == (True, 0.0)
        assert q.check_and_record("k", now=1.0) == (True, 0.0)
        ok, retry = q.check_and_record("k", now=2.0)
        assert not ok and 97 <= retry <= 99
        # window slides: first event at t=0 expires at t=100
        assert q.check_and_record("k", now=101.0)[0]

    def test_per_principal_isolation_and_unlimited(self):
        q = QuotaTracker(scans_per_hour=1)
        assert q.check_and_record("a")[0]
        assert not q.check_and_record("a")[0]
        assert q.check_and_record("b")[0]
        assert all(QuotaTracker(scans_per_hour=0).check_and_record("x")[0]
                   for _ in range(5))


class TestScimStore:
    def test_lifecycle(self):
        s = ScimUserStore()
        u = s.create({"userName": "bob", "roles": [{"value": "operator"}]})
        assert u.role == "operator" and u.active
        key = "abk_test"
        assert s.bind_key(u.id, key)
        assert s.role_for_key(key) == "operator"
        s.set_active(u.id, False)
        assert s.role_for_key(key) is None  # deactivation cuts access
        assert s.delete(u.id) and not s.delete(u.id)

    def test_duplicate_and_missing_username(self):
        s = ScimUserStore()
        s.create({"userName": "x"})
        with pytest.raises(AuthError, match="exists"):
            s.create({"userName": "x"})
        with pytest.raises(AuthError, match="required"):
            s.create({})


class TestApiIntegration:
    @pytest.fixture()
    def client(self, monkeypatch):
        monkeypatch.setenv("AGENT_BOM_API_KEYS", "adm:admin,op:operator,ro:viewer")
        monkeypatch.setenv("AGENT_BOM_OIDC_SECRET", SECRET)
        monkeypatch.setenv("AGENT_BOM_DELEGATION_SECRET", "deleg-secret")
        return TestClient(create_app())

    def test_oidc_bearer_grants_access(self, client):
        tok = mint_test_jwt(SECRET, {"sub": "alice", "roles": ["abom:operator"],
                                     "exp": time.time() + 600})
        r = client.get("/v1/findings", headers={"Authorization": f"Bearer {tok}"})
        assert r.status_code in (200, 404)  # authenticated (404 = no scan yet)
        r2 = client.get("/v1/findings",
                        headers={"Authorization": "Bearer bad.tok.en"})
        assert r2.status_code == 401

    def test_delegation_token_flow(self, client):
        r = client.post("/v1/delegation-tokens",
                        headers={"X-API-Key": "adm"},
                        json={"role": "viewer", "scopes": ["read"], "ttl_s": 60})
        assert r.status_code == 201, r.text
        tok = r.json()["token"]
        r2 = client.get("/healthz", headers={"X-Delegation-Token": tok})
        assert r2.status_code == 200
        # viewer delegation cannot POST a scan
        r3 = client.post("/v1/scan", json={"demo": True},
                         headers={"X-Delegation-Token": tok})
        assert r3.status_code == 403
        # operators cannot mint delegation tokens
        r4 = client.post("/v1/delegation-tokens", headers={"X-API-Key": "op"},
                         json={"role": "viewer"})
        assert r4.status_code == 403

    def test_scim_provisioning_flow(self, client):
        hdr = {"X-API-Key": "adm"}
        r = client.post("/scim/v2/Users", headers=hdr,
                        json={"userName": "carol",
                              "roles": [{"value": "viewer"}]})
        assert r.status_code == 201, r.text
        uid = r.json()["id"]
        assert client.get("/scim/v2/Users", headers=hdr).json()["totalResults"] == 1
        key = client.post(f"/scim/v2/Users/{uid}/api-key",
                          headers=hdr).json()["api_key"]
        assert client.get("/healthz", headers={"X-API-Key": key}).status_code == 200
        # deactivate via SCIM patch -> key stops working
        r2 = client.patch(f"/scim/v2/Users/{uid}", headers=hdr,
                          json={"Operations": [{"op": "replace",
                                                "value": {"active": False}}]})
        assert r2.status_code == 200 and r2.json()["active"] is False
        assert client.get("/v1/findings",
                          headers={"X-API-Key": key}).status_code == 401
        assert client.delete(f"/scim/v2/Users/{uid}",
                             headers=hdr).status_code == 204
        # operator cannot manage SCIM
        assert client.post("/scim/v2/Users", headers={"X-API-Key": "op"},
                           json={"userName": "eve"}).status_code == 403

    def test_scan_quota_429(self, monkeypatch):
        monkeypatch.setenv("AGENT_BOM_API_KEYS", "adm:admin")
        monkeypatch.setenv("AGENT_BOM_QUOTA_SCANS_PER_HOUR", "1")
        client = TestClient(create_app())
        hdr = {"X-API-Key": "adm"}
        r1 = client.post("/v1/scan", json={"demo": True}, headers=hdr)
        assert r1.status_code == 201
        r2 = client.post("/v1/scan", json={"demo": True}, headers=hdr)
        assert r2.status_code == 429
        assert "Retry-After" in r2.headers
