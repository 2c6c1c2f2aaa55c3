"""NHI discovery + identity lifecycle tests."""

import json

import pytest

from agentbom_amd.identity import (
    AgentIdentityStore,
    NHIDiscoveryStatus,
    discover_entra_nhis,
    discover_okta_nhis,
    hash_token,
)
from agentbom_amd.identity.lifecycle import ConditionalAccessPolicy

OKTA_EXPORT = {
    "org_url": "https://example.okta.com",
    "apps": [
        {"id": "app1", "label": "ci-deployer", "signOnMode": "OPENID_CONNECT",
         "status": "ACTIVE", "created": "2025-01-01T00:00:00Z",
         "settings": {"oauthClient": {"application_type": "service",
                                      "grant_types": ["client_credentials"]}}},
        {"id": "app2", "label": "human-sso", "signOnMode": "SAML_2_0"},
        {"id": "app3", "label": "web-oidc", "signOnMode": "OPENID_CONNECT",
         "settings": {"oauthClient": {"application_type": "web",
                                      "grant_types": ["authorization_code"]}}},
    ],
    "api_tokens": [
        {"id": "tok1", "name": "terraform-token", "userId": "u1",
         "created": "2025-02-01T00:00:00Z", "expiresAt": "2026-02-01T00:00:00Z"},
    ],
}

ENTRA_EXPORT = {
    "tenant_id": "t-123",
    "service_principals": [
        {"id": "sp1", "appId": "a-1", "displayName": "pipeline-sp",
         "accountEnabled": True, "servicePrincipalType": "Application"},
        {"id": "sp2", "appId": "a-2", "displayName": "disabled-sp",
         "accountEnabled": False},
    ],
    "applications": [
        {"appId": "a-1", "createdDateTime": "2024-06-01T00:00:00Z",
         "passwordCredentials": [{"endDateTime": "2026-06-01T00:00:00Z"},
                                 {"endDateTime": "2025-12-01T00:00:00Z"}]},
    ],
}


class TestNhiDiscovery:
    def test_disabled_by_default(self):
        assert discover_okta_nhis(env={}).status is NHIDiscoveryStatus.DISABLED
        assert discover_entra_nhis(env={}).status is NHIDiscoveryStatus.DISABLED

    def test_missing_credentials(self):
        r = discover_okta_nhis(env={"AGENT_BOM_OKTA_DISCOVERY": "1"})
        assert r.status is NHIDiscoveryStatus.MISSING_CREDENTIALS

    def test_missing_client_with_credentials(self):
        r = discover_okta_nhis(env={"AGENT_BOM_OKTA_DISCOVERY": "1",
                                    "AGENT_BOM_OKTA_TOKEN": "x",
                                    "AGENT_BOM_OKTA_ORG_URL": "https://x.okta.com"})
        assert r.status is NHIDiscoveryStatus.MISSING_CLIENT

    def test_okta_export_file(self, tmp_path):
        p = tmp_path / "okta.json"
        p.write_text(json.dumps(OKTA_EXPORT))
        r = discover_okta_nhis(export_path=str(p), env={})
        assert r.ok
        types = {i.identity_type for i in r.identities}
        assert types == {"service_account", "api_token"}
        # human SSO and auth-code web apps are not machine identities
        ids = {i.identity_id for i in r.identities}
        assert ids == {"app1", "tok1"}
        tok = next(i for i in r.identities if i.identity_id == "tok1")
        assert tok.credential_expires_at == "2026-02-01T00:00:00Z"
        assert tok.owner == "u1"

    def test_okta_injected_client(self):
        class Fake:
            def list_oauth2_service_apps(self):
                return OKTA_EXPORT["apps"]

            def list_api_tokens(self):
                return OKTA_EXPORT["api_tokens"]

        r = discover_okta_nhis(client=Fake(), env={})
        assert r.ok and len(r.identities) == 2

    def test_okta_client_error_degrades(self):
        class Broken:
            def list_oauth2_service_apps(self):
                raise RuntimeError("boom")

            def list_api_tokens(self):
                return []

        r = discover_okta_nhis(client=Broken(), env={})
        assert r.status is NHIDiscoveryStatus.ERROR
        assert "boom" in r.warnings[0]

    def test_entra_export(self, tmp_path):
        p = tmp_path / "entra.json"
        p.write_text(json.dumps(ENTRA_EXPORT))
        r = discover_entra_nhis(export_path=str(p), env={})
        assert r.ok and len(r.identities) == 2
        sp1 = next(i for i in r.identities if i.identity_id == "sp1")
        # earliest credential end date wins
        assert sp1.credential_expires_at == "2025-12-01T00:00:00Z"
        sp2 = next(i for i in r.identities if i.identity_id == "sp2")
        assert sp2.status == "disabled"

    def test_never_captures_secrets(self, tmp_path):
        p = tmp_path / "okta.json"
        export = json.loads(json.dumps(OKTA_EXPORT))
        export["api_tokens"][0]["value"] = "SSWS-supersecret"
        p.write_text(json.dumps(export))
        r = discover_okta_nhis(export_path=str(p), env={})
        dumped = json.dumps(r.to_dict())
        assert "supersecret" not in dumped


class TestLifecycle:
    def test_issue_returns_raw_once_and_stores_hash_only(self):
        store = AgentIdentityStore()
        ident, raw = store.issue("claude-desktop", scopes=["scan:read"])
        assert raw.startswith("abi_")
        assert ident.token_hash == hash_token(raw)
        assert raw not in json.dumps(ident.to_public_dict())
        assert "token_hash" not in ident.to_public_dict()
        assert store.verify(raw)["valid"]

    def test_verify_unknown_and_tool_scoping(self):
        store = AgentIdentityStore()
        ident, raw = store.issue("a", allowed_tools=["read_file"])
        assert store.verify("abi_nope_x")["valid"] is False
        assert store.verify(raw, tool="read_file")["valid"]
        assert store.verify(raw, tool="exec_shell")["valid"] is False

    def test_rotate_overlap_and_revoke(self):
        store = AgentIdentityStore()
        old, raw_old = store.issue("bot")
        new, raw_new = store.rotate(old.identity_id, overlap_minutes=10)
        assert new.rotated_from == old.identity_id
        # both live inside the overlap window
        assert store.verify(raw_old)["valid"]
        assert store.verify(raw_new)["valid"]
        assert store.revoke(new.identity_id)
        assert store.verify(raw_new)["valid"] is False

    def test_rotate_dead_identity_refused(self):
        store = AgentIdentityStore()
        ident, _ = store.issue("bot")
        store.revoke(ident.identity_id)
        assert store.rotate(ident.identity_id) == (None, None)

    def test_jit_grant_lifecycle(self):
        store = AgentIdentityStore()
        ident, _ = store.issue("bot", scopes=["scan:read"])
        g = store.grant_jit(ident.identity_id, ["identity:write"],
                            reason="incident response", granted_by="oncall")
        assert g.is_live()
        assert "identity:write" in store.active_scopes(ident.identity_id)
        assert store.revoke_jit(g.grant_id)
        assert "identity:write" not in store.active_scopes(ident.identity_id)

    def test_jit_unknown_identity(self):
        store = AgentIdentityStore()
        assert store.grant_jit("missing", ["x"], "r", "who") is None

    def test_conditional_access_ip_policy(self):
        store = AgentIdentityStore()
        ident, raw = store.issue("bot", scopes=["scan:read"])
        store.put_conditional_policy(ConditionalAccessPolicy(
            policy_id="p1", name="office-only", scopes=["scan:*"],
            allowed_cidrs=["10.0.0.0/8"]))
        assert store.verify(raw, source_ip="10.1.2.3")["valid"]
        denied = store.verify(raw, source_ip="203.0.113.9")
        assert denied["valid"] is False and "office-only" in denied["reason"]
        assert store.verify(raw)["valid"] is False  # ip required

    def test_credential_expiry_report(self):
        store = AgentIdentityStore()
        store.issue("short", ttl_hours=1)
        store.issue("long", ttl_hours=24 * 30)
        report = store.credential_expiry_report(within_hours=72)
        assert [r["agent_name"] for r in report] == ["short"]

    def test_audit_chain(self):
        store = AgentIdentityStore()
        ident, _ = store.issue("bot")
        store.rotate(ident.identity_id)
        store.revoke(ident.identity_id)
        entries = store.audit_entries()
        assert [e["action"] for e in entries] == [
            "identity.issue", "identity.rotate", "identity.revoke"]
        assert store.audit_chain_valid()
        # tamper detection
        store._db.execute("UPDATE identity_audit SET reason='x' WHERE seq=2")
        store._db.commit()
        assert store.audit_chain_valid() is False

    def test_access_review(self):
        store = AgentIdentityStore()
        store.issue("wild", scopes=["*"])
        scoped, _ = store.issue("narrow", scopes=["scan:read"],
                                allowed_tools=["read_file"])
        review = store.access_review()
        assert review["live_identities"] == 2
        assert any("wild" in store.get(i).agent_name
                   for i in review["wildcard_or_unscoped"])
        assert scoped.identity_id not in review["wildcard_or_unscoped"]
        assert review["audit_chain_valid"]


class TestNhiOverlay:
    def test_overlay_and_posture(self, tmp_path):
        from agentbom_amd.graph.builder import build_unified_graph_from_report
        from agentbom_amd.graph.nhi_overlay import (
            apply_issued_identity_overlay,
            apply_nhi_overlay,
            nhi_posture,
        )
        from agentbom_amd.scan.orchestrator import run_demo_scan

        report = run_demo_scan()
        graph = build_unified_graph_from_report(report)
        p = tmp_path / "okta.json"
        p.write_text(json.dumps(OKTA_EXPORT))
        r = discover_okta_nhis(export_path=str(p), env={})
        stats = apply_nhi_overlay(graph, r.identities)
        assert stats["nodes_added"] == 2
        # idempotent
        stats2 = apply_nhi_overlay(graph, r.identities)
        assert stats2["nodes_added"] == 0

        store = AgentIdentityStore()
        agent_label = report.agents[0].name
        store.issue(agent_label, scopes=["scan:read"])
        stats3 = apply_issued_identity_overlay(graph, store)
        assert stats3["nodes_added"] == 1
        assert stats3["edges_added"] == 1  # linked to its agent by label

        posture = nhi_posture(graph)
        assert posture["total_identities"] == 3
        assert posture["by_provider"]["okta"] == 2
        assert posture["by_provider"]["agent-bom"] == 1


class TestConditionalAccessABAC:
    """Reference agent_identity_store.py ABAC semantics: deny-wins,
    fail-closed unknowns, device/group/client/posture conditions."""

    def _store_with(self, policy):
        from agentbom_amd.identity.lifecycle import AgentIdentityStore

        store = AgentIdentityStore()
        ident, raw = store.issue("bot", scopes=["scan:read"])
        store.put_conditional_policy(policy)
        return store, raw

    def test_deny_effect_wins(self):
        from agentbom_amd.identity.lifecycle import AccessContext

        store, raw = self._store_with(ConditionalAccessPolicy(
            policy_id="d1", name="block-prod-writes", effect="deny",
            allowed_environments=["prod"]))
        # in prod: deny conditions hold -> blocked
        denied = store.verify(raw, ctx=AccessContext(environment="prod"))
        assert denied["valid"] is False and "block-prod-writes" in denied["reason"]
        # in staging: deny conditions do not hold -> allowed
        assert store.verify(raw, ctx=AccessContext(environment="staging"))["valid"]

    def test_group_membership_fails_closed(self):
        from agentbom_amd.identity.lifecycle import AccessContext

        store, raw = self._store_with(ConditionalAccessPolicy(
            policy_id="g1", name="sre-only", allowed_groups=["sre"]))
        # no groups supplied -> fail closed
        assert store.verify(raw)["valid"] is False
        assert store.verify(raw, ctx=AccessContext(groups=["dev"]))["valid"] is False
        assert store.verify(raw, ctx=AccessContext(groups=["dev", "sre"]))["valid"]

    def test_device_posture_fails_closed_on_unknown(self):
        from agentbom_amd.identity.lifecycle import AccessContext

        store, raw = self._store_with(ConditionalAccessPolicy(
            policy_id="p1", name="managed-only", require_device_managed=True,
            require_device_disk_encrypted=True))
        assert store.verify(raw)["valid"] is False  # unknown posture
        partial = AccessContext(device_managed=True)  # encryption unknown
        assert store.verify(raw, ctx=partial)["valid"] is False
        good = AccessContext(device_managed=True, device_disk_encrypted=True)
        assert store.verify(raw, ctx=good)["valid"]

    def test_client_and_device_allowlists(self):
        from agentbom_amd.identity.lifecycle import AccessContext

        store, raw = self._store_with(ConditionalAccessPolicy(
            policy_id="c1", name="claude-only", allowed_clients=["claude-code"],
            allowed_devices=["wks-7"]))
        ok = AccessContext(client_id="claude-code", device_id="wks-7")
        assert store.verify(raw, ctx=ok)["valid"]
        wrong_client = AccessContext(client_id="other", device_id="wks-7")
        assert store.verify(raw, ctx=wrong_client)["valid"] is False

    def test_weekday_and_hour_windows(self):
        import datetime as dt

        from agentbom_amd.identity.lifecycle import AccessContext

        store, raw = self._store_with(ConditionalAccessPolicy(
            policy_id="w1", name="business-hours",
            allowed_weekdays=[0, 1, 2, 3, 4], allowed_hours_utc=list(range(8, 18))))
        monday_noon = dt.datetime(2026, 9, 14, 12, 0,
                                  tzinfo=dt.timezone.utc)  # Monday
        sunday = dt.datetime(2026, 9, 13, 12, 0, tzinfo=dt.timezone.utc)
        assert store.verify(raw, ctx=AccessContext(at=monday_noon))["valid"]
        assert store.verify(raw, ctx=AccessContext(at=sunday))["valid"] is False

    def test_disabled_and_priority(self):
        from agentbom_amd.identity.lifecycle import (
            AccessContext,
            evaluate_conditional_access,
        )

        off = ConditionalAccessPolicy(policy_id="z", name="off", effect="deny",
                                      status="disabled")
        deny = ConditionalAccessPolicy(policy_id="a", name="deny-all",
                                       effect="deny", priority=1)
        req = ConditionalAccessPolicy(policy_id="b", name="needs-prod",
                                      allowed_environments=["prod"], priority=2)
        allowed, reason, pid = evaluate_conditional_access([off], "s",
                                                           AccessContext())
        assert allowed
        allowed, reason, pid = evaluate_conditional_access(
            [req, deny], "s", AccessContext(environment="prod"))
        assert not allowed and pid == "a"  # deny precedence over satisfied require
