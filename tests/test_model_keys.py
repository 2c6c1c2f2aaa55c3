"""Virtual model-key broker: sealing, scoping, revocation, fail-closed."""

from __future__ import annotations

import json

import pytest

from agentbom_amd.api.connections import generate_connections_key
from agentbom_amd.api.model_keys import (
    ModelKeyBroker,
    ModelKeyBrokerError,
)

_ENV = {"AGENT_BOM_CONNECTIONS_KEY": generate_connections_key()}


@pytest.fixture
def broker():
    return ModelKeyBroker(env=_ENV)


@pytest.fixture
def minted(broker):
    pk = broker.register_provider_key("anthropic", "sk-ant-REAL-123",
                                      label="prod")
    vk, raw = broker.mint_virtual_key(pk.provider_key_id, holder="agent-a",
                                      model_allowlist=["claude-opus"])
    return broker, pk, vk, raw


class TestSealing:
    def test_real_key_never_in_public_shapes(self, minted):
        broker, pk, vk, raw = minted
        assert "sk-ant-REAL-123" not in json.dumps(pk.to_public_dict())
        assert "sk-ant-REAL-123" not in json.dumps(vk.to_public_dict())
        assert "sk-ant-REAL-123" not in json.dumps(
            broker.list_virtual_keys())
        # sealed at rest: the db row never carries the plaintext
        row = broker._db.execute("SELECT doc FROM provider_keys").fetchone()
        assert "sk-ant-REAL-123" not in row[0]

    def test_resolve_returns_real_key(self, minted):
        broker, _pk, vk, raw = minted
        out = broker.resolve(raw, model="claude-opus", holder="agent-a")
        assert out["real_key"] == "sk-ant-REAL-123"
        assert out["provider"] == "anthropic" and out["uses"] == 1
        assert broker.resolve(raw, model="claude-opus",
                              holder="agent-a")["uses"] == 2


class TestFailClosed:
    def test_unknown_token(self, broker):
        with pytest.raises(ModelKeyBrokerError, match="unknown"):
            broker.resolve("abvk_dead_beef")

    def test_holder_binding(self, minted):
        broker, _pk, _vk, raw = minted
        with pytest.raises(ModelKeyBrokerError, match="holder"):
            broker.resolve(raw, model="claude-opus", holder="agent-b")

    def test_model_allowlist(self, minted):
        broker, _pk, _vk, raw = minted
        with pytest.raises(ModelKeyBrokerError, match="allowlist"):
            broker.resolve(raw, model="gpt-4", holder="agent-a")

    def test_revocation_and_provider_disable(self, minted):
        broker, pk, vk, raw = minted
        assert broker.revoke_virtual_key(vk.virtual_key_id)
        with pytest.raises(ModelKeyBrokerError, match="revoked"):
            broker.resolve(raw, model="claude-opus", holder="agent-a")
        vk2, raw2 = broker.mint_virtual_key(pk.provider_key_id)
        broker.disable_provider_key(pk.provider_key_id)
        with pytest.raises(ModelKeyBrokerError, match="disabled"):
            broker.resolve(raw2)

    def test_expiry(self, minted):
        broker, pk, _vk, _raw = minted
        vk, raw = broker.mint_virtual_key(pk.provider_key_id, ttl_hours=-1)
        with pytest.raises(ModelKeyBrokerError, match="expired"):
            broker.resolve(raw)

    def test_mint_for_unknown_provider_key(self, broker):
        with pytest.raises(ModelKeyBrokerError, match="unknown provider"):
            broker.mint_virtual_key("mpk-nope")

    def test_no_crypto_key_refuses_registration(self):
        broker = ModelKeyBroker(env={})
        from agentbom_amd.api.connections import ConnectionsCryptoUnavailable

        with pytest.raises(ConnectionsCryptoUnavailable):
            broker.register_provider_key("openai", "sk-x")


class TestApi:
    def test_endpoints(self, monkeypatch):
        monkeypatch.setenv("AGENT_BOM_CONNECTIONS_KEY",
                           _ENV["AGENT_BOM_CONNECTIONS_KEY"])
        from starlette.testclient import TestClient

        from agentbom_amd.api.server import create_app

        client = TestClient(create_app())
        r = client.post("/v1/model-keys/providers", json={
            "provider": "anthropic", "key": "sk-ant-REAL-999",
            "label": "prod"})
        assert r.status_code == 201 and "key" not in r.json()
        pkid = r.json()["provider_key_id"]
        v = client.post("/v1/model-keys/virtual", json={
            "provider_key_id": pkid, "holder": "agent-a"}).json()
        assert v["virtual_key"].startswith("abvk_")
        listed = client.get("/v1/model-keys/virtual").json()["virtual_keys"]
        assert listed and "sk-ant-REAL-999" not in str(listed)
        assert client.post(
            f"/v1/model-keys/virtual/{v['virtual_key_id']}/revoke"
        ).json()["revoked"] == v["virtual_key_id"]
        assert client.post("/v1/model-keys/virtual",
                           json={"provider_key_id": "mpk-x"}).status_code == 404
