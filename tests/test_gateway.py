

class TestGatewayConditionalAccess:
    """ABAC conditional-access gate wired through relay()."""

    def _gw(self):
        from agentbom_amd.identity.lifecycle import (
            AgentIdentityStore,
            ConditionalAccessPolicy,
        )
        from agentbom_amd.runtime.gateway import Gateway, Upstream

        store = AgentIdentityStore()
        store.put_conditional_policy(ConditionalAccessPolicy(
            policy_id="p1", name="prod-freeze", effect="deny",
            allowed_environments=["prod"], tools=["deploy"]))
        gw = Gateway(identity_store=store)
        gw.register(Upstream(name="up", handler=lambda f: {
            "jsonrpc": "2.0", "id": f.get("id"), "result": {"ok": True}}))
        return gw

    def test_deny_policy_blocks_scoped_tool(self):
        from agentbom_amd.identity.lifecycle import AccessContext

        gw = self._gw()
        frame = {"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                 "params": {"name": "deploy"}}
        out = gw.relay("up", frame, principal="bot",
                       access_ctx=AccessContext(environment="prod"))
        assert out["error"]["code"] == -32008
        assert "prod-freeze" in out["error"]["message"]
        assert gw.metrics["blocked_total"] == 1

    def test_out_of_scope_tool_passes(self):
        from agentbom_amd.identity.lifecycle import AccessContext

        gw = self._gw()
        frame = {"jsonrpc": "2.0", "id": 2, "method": "tools/call",
                 "params": {"name": "read_file"}}
        out = gw.relay("up", frame, principal="bot",
                       access_ctx=AccessContext(environment="prod"))
        assert out.get("result") == {"ok": True}

    def test_no_ctx_keeps_legacy_behavior(self):
        gw = self._gw()
        frame = {"jsonrpc": "2.0", "id": 3, "method": "tools/call",
                 "params": {"name": "deploy"}}
        assert gw.relay("up", frame, principal="bot").get("result") == {"ok": True}


class TestActivityHistory:
    def test_relay_activity_recorded_and_summarized(self):
        from agentbom_amd.runtime.gateway import Gateway, Upstream

        gw = Gateway()
        gw.register(Upstream(name="up", handler=lambda f: {
            "jsonrpc": "2.0", "id": f.get("id"), "result": {}}))
        gw.relay("up", {"jsonrpc": "2.0", "id": 1, "method": "tools/list"},
                 principal="bot")
        gw.relay("missing", {"jsonrpc": "2.0", "id": 2,
                             "method": "tools/call",
                             "params": {"name": "x"}}, principal="bot")
        assert len(gw.activity) == 2
        first, second = list(gw.activity)
        assert first["outcome"] == "ok" and first["latency_ms"] >= 0
        assert second["outcome"] == "error" and second["code"] == -32001
        s = gw.activity_summary()
        assert s["ok"] == 1 and s["errors_by_code"] == {"-32001": 1}
        assert s["principals"] == 1 and len(s["recent"]) == 2

    def test_bounded_history(self):
        from agentbom_amd.runtime.gateway import Gateway, Upstream

        gw = Gateway()
        gw.register(Upstream(name="up", handler=lambda f: {
            "jsonrpc": "2.0", "id": f.get("id"), "result": {}}))
        for i in range(1100):
            gw.relay("up", {"jsonrpc": "2.0", "id": i,
                            "method": "tools/list"})
        assert len(gw.activity) == 1000  # ring buffer
