"""Cloud connections: at-rest crypto, credential-ref validation, schedule."""

from __future__ import annotations

from datetime import datetime, timedelta, timezone

import pytest

from agentbom_amd.api.connections import (
    CloudConnection,
    ConnectionStore,
    ConnectionsCryptoUnavailable,
    decrypt_secret,
    encrypt_secret,
    generate_connections_key,
    validate_credential_ref,
)

_NOW = datetime(2026, 9, 14, 12, 0, tzinfo=timezone.utc)


class TestCrypto:
    def test_roundtrip_and_nonce_freshness(self):
        env = {"AGENT_BOM_CONNECTIONS_KEY": generate_connections_key()}
        pt = "external-id-1234-şecret"  # non-ASCII too
        t1, t2 = encrypt_secret(pt, env), encrypt_secret(pt, env)
        assert t1 != t2  # fresh nonce every call
        assert decrypt_secret(t1, env) == pt
        assert decrypt_secret(t2, env) == pt

    def test_tamper_detected(self):
        env = {"AGENT_BOM_CONNECTIONS_KEY": generate_connections_key()}
        token = encrypt_secret("secret", env)
        flipped = token[:-5] + ("A" if token[-5] != "A" else "B") + token[-4:]
        with pytest.raises(ValueError, match="authentication failed"):
            decrypt_secret(flipped, env)

    def test_wrong_key_rejected(self):
        env1 = {"AGENT_BOM_CONNECTIONS_KEY": generate_connections_key()}
        env2 = {"AGENT_BOM_CONNECTIONS_KEY": generate_connections_key()}
        token = encrypt_secret("secret", env1)
        with pytest.raises(ValueError):
            decrypt_secret(token, env2)

    def test_no_key_refuses_never_plaintext(self):
        with pytest.raises(ConnectionsCryptoUnavailable):
            encrypt_secret("secret", env={})
        with pytest.raises(ConnectionsCryptoUnavailable):
            decrypt_secret("whatever", env={})

    def test_long_secret_multiblock(self):
        env = {"AGENT_BOM_CONNECTIONS_KEY": generate_connections_key()}
        pt = "x" * 1000
        assert decrypt_secret(encrypt_secret(pt, env), env) == pt


class TestCredentialRefs:
    def test_provider_formats(self):
        ok, _ = validate_credential_ref(
            "aws", "role_arn", "arn:aws:iam::123456789012:role/scanner")
        assert ok == "ok"
        bad, why = validate_credential_ref("aws", "role_arn", "arn:aws:nope")
        assert bad == "degraded" and "ARN" in why
        assert validate_credential_ref(
            "gcp", "service_account",
            "scanner-sa@my-project.iam.gserviceaccount.com")[0] == "ok"
        assert validate_credential_ref(
            "azure", "service_principal",
            "12345678-1234-1234-1234-123456789abc")[0] == "ok"

    def test_reference_schemes_only_no_secrets(self):
        assert validate_credential_ref("", "", "env:SCANNER_TOKEN")[0] == "ok"
        assert validate_credential_ref("", "", "vault:kv/scanner")[0] == "ok"
        # a raw secret-looking value is refused
        status, why = validate_credential_ref("", "", "hunter2-plaintext")
        assert status == "degraded" and "scheme" in why

    def test_empty_ref(self):
        assert validate_credential_ref("aws", "role_arn", " ")[0] == "degraded"


class TestStoreAndSchedule:
    def test_public_dict_never_leaks_secret(self):
        env = {"AGENT_BOM_CONNECTIONS_KEY": generate_connections_key()}
        conn = CloudConnection(provider="aws", display_name="prod",
                               secret_encrypted=encrypt_secret("ext-id", env))
        pub = conn.to_public_dict()
        assert pub["has_secret"] is True
        assert "secret_encrypted" not in pub
        assert "ext-id" not in str(pub)

    def test_store_roundtrip_tenancy(self):
        store = ConnectionStore()
        c = store.put(CloudConnection(provider="gcp", display_name="dev",
                                      tenant_id="t1"))
        assert store.get("t1", c.connection_id).provider == "gcp"
        assert store.get("t2", c.connection_id) is None
        assert store.delete("t1", c.connection_id)
        assert store.list("t1") == []

    def test_due_schedule(self):
        store = ConnectionStore()
        c = store.put(CloudConnection(provider="aws", display_name="p",
                                      scan_interval_minutes=60))
        assert c.due(_NOW)  # never scanned -> due
        store.mark_scanned("default", c.connection_id, "scan-1")
        fresh = store.get("default", c.connection_id)
        assert fresh.last_scan_id == "scan-1" and fresh.status == "active"
        assert not fresh.due(datetime.fromisoformat(fresh.last_scan_at)
                             + timedelta(minutes=30))
        assert fresh.due(datetime.fromisoformat(fresh.last_scan_at)
                         + timedelta(minutes=61))
        # no interval -> never polled; disabled -> never due
        c2 = store.put(CloudConnection(provider="aws", display_name="manual"))
        assert not c2.due(_NOW)
        assert [d.connection_id for d in store.due_connections(
            "default", at=_NOW + timedelta(days=1))] == [c.connection_id]


class TestApi:
    @pytest.fixture()
    def client(self, monkeypatch):
        monkeypatch.setenv("AGENT_BOM_CONNECTIONS_KEY",
                           generate_connections_key())
        from starlette.testclient import TestClient

        from agentbom_amd.api.server import create_app

        return TestClient(create_app())

    def test_create_scan_delete(self, client):
        r = client.post("/v1/connections", json={
            "provider": "aws", "display_name": "prod",
            "mode": "role_arn",
            "role_ref": "arn:aws:iam::123456789012:role/scanner",
            "secret": "ext-id-abc", "scan_interval_minutes": 60})
        assert r.status_code == 201
        doc = r.json()
        assert doc["has_secret"] is True and "secret" not in doc
        assert doc["status"] == "pending"
        cid = doc["connection_id"]
        assert client.get("/v1/connections?due=true").json()["total"] == 1
        s = client.post(f"/v1/connections/{cid}/scan").json()
        assert s["connection"]["last_scan_id"] == s["job_id"]
        assert client.get("/v1/connections?due=true").json()["total"] == 0
        assert client.delete(f"/v1/connections/{cid}").status_code == 204
        assert client.post("/v1/connections/conn-x/scan").status_code == 404

    def test_degraded_ref_flagged(self, client):
        r = client.post("/v1/connections", json={
            "provider": "aws", "display_name": "bad", "mode": "role_arn",
            "role_ref": "not-an-arn"})
        assert r.json()["status"] == "degraded"

    def test_secret_without_key_conflicts(self, client, monkeypatch):
        monkeypatch.delenv("AGENT_BOM_CONNECTIONS_KEY")
        r = client.post("/v1/connections", json={
            "provider": "gcp", "display_name": "x", "secret": "oops"})
        assert r.status_code == 409
