"""Known-server registry + blocklist tests."""

from agentbom_amd.mcp.registry import (
    check_blocklist,
    load_registry,
    lookup_package,
    match_command,
    stamp_server_from_registry,
)
from agentbom_amd.models import MCPServer


class TestRegistryData:
    def test_bundled_registry_loads(self):
        reg = load_registry()
        assert len(reg["servers"]) >= 40
        for entry in reg["servers"].values():
            assert entry["risk_level"] in ("low", "medium", "high",
                                           "critical", "unknown")
            assert isinstance(entry["tools"], list)

    def test_lookup_scoped_and_unscoped(self):
        assert lookup_package("@modelcontextprotocol/server-github")
        # unscoped basename also resolves
        assert lookup_package("server-github")["name"] == "GitHub"
        assert lookup_package("definitely-not-registered") is None

    def test_match_command_longest_pattern(self):
        e = match_command("npx", ["-y", "@modelcontextprotocol/server-filesystem",
                                  "/tmp"])
        assert e and e["name"] == "Filesystem"
        assert match_command("python", ["app.py"]) is None

    def test_blocklist(self):
        assert check_blocklist("mcp-server-backdoor-v2")["severity"] == "critical"
        assert check_blocklist("server-filesystem") is None


class TestStamping:
    def test_stamp_infers_tools_and_credentials(self):
        s = MCPServer(name="github", command="npx",
                      args=["-y", "@modelcontextprotocol/server-github"])
        assert stamp_server_from_registry(s)
        assert s.registry_verified
        assert s.registry_id == "@modelcontextprotocol/server-github"
        assert any(t.name == "create_pull_request" for t in s.tools)
        assert all(t.discovery_source == "registry" for t in s.tools)
        assert "GITHUB_PERSONAL_ACCESS_TOKEN" in s.env
        assert s.env["GITHUB_PERSONAL_ACCESS_TOKEN"] == "***"  # never a value
        assert any("registry risk high" in w for w in s.security_warnings)

    def test_stamp_never_overrides_declared_tools(self):
        from agentbom_amd.models import MCPTool

        s = MCPServer(name="github", command="npx",
                      args=["@modelcontextprotocol/server-github"],
                      tools=[MCPTool(name="only_this", description="")])
        stamp_server_from_registry(s)
        assert [t.name for t in s.tools] == ["only_this"]

    def test_blocklisted_server_fails_closed(self):
        s = MCPServer(name="backdoor", command="npx",
                      args=["mcp-server-backdoor"])
        reg = load_registry()
        reg = {"servers": {"mcp-server-backdoor": {
            "package": "mcp-server-backdoor", "command_patterns":
            ["mcp-server-backdoor"], "tools": [], "verified": False}},
            "blocklist": reg["blocklist"]}
        assert stamp_server_from_registry(s, reg)
        assert s.security_blocked
        assert any("blocklist" in w for w in s.security_warnings)

    def test_discovery_path_stamps(self, tmp_path):
        import json

        from agentbom_amd.scan.discovery import parse_mcp_config

        cfg = tmp_path / "claude_desktop_config.json"
        cfg.write_text(json.dumps({"mcpServers": {
            "fs": {"command": "npx",
                   "args": ["-y", "@modelcontextprotocol/server-filesystem"]}}}))
        servers = parse_mcp_config(cfg)
        assert servers and servers[0].registry_verified


class TestMarketplaceIntegration:
    def test_marketplace_uses_registry(self):
        from agentbom_amd.mcp.server import AgentBomMcpServer

        s = AgentBomMcpServer()
        out = s.tools["marketplace_check"].fn(name="@stripe/mcp")
        assert out["verdict"] == "warn"  # critical risk level -> warn
        assert out["registry"]["risk_level"] == "critical"
        out = s.tools["marketplace_check"].fn(name="mcp-server-backdoor")
        assert out["verdict"] == "block" and out["blocklist_hit"]
