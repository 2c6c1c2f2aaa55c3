

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
