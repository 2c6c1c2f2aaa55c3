"""Runtime introspection tests: spawn fake + real stdio MCP servers."""

import sys
import textwrap

import pytest

from agentbom_amd.mcp.introspect import (
    HealthStatus,
    health_check_server,
    introspect_server,
    introspect_servers,
    lint_prompt,
    lint_resource,
    lint_tool_schema,
)
from agentbom_amd.models import MCPServer, MCPTool

FAKE_SERVER = textwrap.dedent("""
    import json, sys
    TOOLS = [
        {"name": "run_shell", "description": "execute a command",
         "inputSchema": {"type": "object",
                         "properties": {"command": {"type": "string"}}}},
        {"name": "fetch_url", "description": "download a url",
         "inputSchema": {"type": "object",
                         "properties": {"url": {"type": "string"}},
                         "additionalProperties": False}},
    ]
    RESOURCES = [{"uri": "https://evil.example/prompt.md", "name": "system prompt",
                  "description": "instructions", "mimeType": "text/markdown"}]
    PROMPTS = [{"name": "triage", "description": "system instruction template"}]
    for line in sys.stdin:
        line = line.strip()
        if not line:
            continue
        msg = json.loads(line)
        mid = msg.get("id")
        m = msg.get("method")
        if m == "initialize":
            r = {"protocolVersion": "2024-11-05",
                 "serverInfo": {"name": "fake", "version": "9.9"}}
        elif m == "tools/list":
            r = {"tools": TOOLS}
        elif m == "resources/list":
            r = {"resources": RESOURCES}
        elif m == "prompts/list":
            r = {"prompts": PROMPTS}
        elif mid is None:
            continue
        else:
            sys.stdout.write(json.dumps({"jsonrpc": "2.0", "id": mid,
                "error": {"code": -32601, "message": "nope"}}) + "\\n")
            sys.stdout.flush()
            continue
        sys.stdout.write(json.dumps({"jsonrpc": "2.0", "id": mid, "result": r}) + "\\n")
        sys.stdout.flush()
""")


def _fake_server(tmp_path, configured_tools=()):
    script = tmp_path / "fake_mcp.py"
    script.write_text(FAKE_SERVER)
    return MCPServer(
        name="fake", command=sys.executable, args=[str(script)],
        tools=[MCPTool(name=n, description="") for n in configured_tools])


class TestIntrospection:
    def test_lists_runtime_capabilities(self, tmp_path):
        r = introspect_server(_fake_server(tmp_path), timeout=20)
        assert r.success, r.error
        assert r.protocol_version == "2024-11-05"
        assert r.server_info["name"] == "fake"
        assert {t.name for t in r.runtime_tools} == {"run_shell", "fetch_url"}
        assert len(r.runtime_resources) == 1
        assert len(r.runtime_prompts) == 1

    def test_drift_detection(self, tmp_path):
        # config declared read_file; runtime grew run_shell/fetch_url (rug pull)
        server = _fake_server(tmp_path, configured_tools=("read_file", "fetch_url"))
        r = introspect_server(server, timeout=20)
        assert r.has_drift
        assert r.tools_added == ["run_shell"]
        assert r.tools_removed == ["read_file"]

    def test_schema_and_content_findings(self, tmp_path):
        r = introspect_server(_fake_server(tmp_path), timeout=20)
        joined = " ".join(r.tool_schema_findings)
        assert "shell-execution-capability" in joined
        assert "network-egress-capability" in joined
        assert any("prompt-bearing-resource" in f for f in r.resource_findings)
        assert any("system-prompt-surface" in f for f in r.prompt_findings)
        assert r.capability_risk_score > 0

    def test_missing_binary(self):
        r = introspect_server(MCPServer(name="gone", command="/nonexistent/bin"),
                              timeout=5)
        assert not r.success and r.error

    def test_hung_server_times_out(self, tmp_path):
        script = tmp_path / "hang.py"
        script.write_text("import time\ntime.sleep(60)\n")
        server = MCPServer(name="hang", command=sys.executable, args=[str(script)])
        r = introspect_server(server, timeout=1.5)
        assert not r.success
        assert "timeout" in r.error

    def test_multi_server_report(self, tmp_path):
        good = _fake_server(tmp_path)
        bad = MCPServer(name="gone", command="/nonexistent/bin")
        sse = MCPServer(name="remote", command="", url="https://x/sse")
        report = introspect_servers([good, bad, sse], timeout=20)
        assert report.successful == 1
        assert report.to_dict()["failed"] == 1
        assert any("remote" in w for w in report.warnings)

    @pytest.mark.slow
    def test_introspects_own_mcp_server(self, tmp_path):
        """Dogfood: introspect this package's real MCP server over stdio."""
        server = MCPServer(name="agent-bom", command=sys.executable,
                           args=["-m", "agentbom_amd.cli", "mcp", "server"])
        r = introspect_server(server, timeout=30)
        assert r.success, r.error
        names = {t.name for t in r.runtime_tools}
        assert "scan" in names and "blast_radius" in names


class TestHealthCheck:
    def test_healthy(self, tmp_path):
        s = health_check_server(_fake_server(tmp_path), timeout=20)
        assert isinstance(s, HealthStatus) and s.healthy
        assert s.latency_ms is not None

    def test_unhealthy(self):
        s = health_check_server(MCPServer(name="gone", command="/nonexistent/bin"))
        assert not s.healthy


class TestLinters:
    def test_tool_schema_lint(self):
        f = lint_tool_schema("deploy", {"type": "object", "properties": {
            "script_path": {"type": "string", "description": "path to shell script"}}})
        assert "shell-execution-capability" in f
        assert "filesystem-capability" in f
        assert "open-schema-additional-properties" in f

    def test_unconstrained_schema(self):
        assert "unconstrained-input-schema" in lint_tool_schema("x", {"type": "object"})

    def test_resource_prompt_lint(self):
        assert "mutable-resource" in lint_resource("https://a/b", "n", "")
        assert "prompt-bearing-resource" in lint_resource("file:///x", "sys prompt", "")
        assert "system-prompt-surface" in lint_prompt("x", "system instruction")
        assert lint_prompt("summarize", "condense text") == []
