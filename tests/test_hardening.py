"""Strict-args, tenant, attestation, auth-posture, self-posture tests."""

import json
import time

import pytest

from agentbom_amd.mcp.server import AgentBomMcpServer, resolve_mcp_tenant_id
from agentbom_amd.models import Agent, AgentType, MCPServer
from agentbom_amd.scan.auth_posture import assess_estate, assess_server
from agentbom_amd.scan.self_posture import evaluate_self_posture
from agentbom_amd.utils.attestation import (
    attest_scanned_server,
    build_attestation_statement,
    sign_attestation,
    verify_attestation,
)

KEY = b"0" * 32
KEY_HEX = KEY.hex()


def _call_raw(server, name, args):
    return server.handle({"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                          "params": {"name": name, "arguments": args}})


class TestStrictArgs:
    def test_unknown_arg_rejected_not_dropped(self):
        s = AgentBomMcpServer()
        # the reference P1: check with a typo'd extra arg must NOT return
        # a clean verdict — it must fail loudly with the accepted keys
        resp = _call_raw(s, "check_package",
                         {"name": "flask", "version": "2.0.0",
                          "ecosystem": "PyPI", "Version": "1.0"})
        assert resp["result"]["isError"]
        payload = json.loads(resp["result"]["content"][0]["text"])
        assert payload["unknown"] == ["Version"]
        assert "version" in payload["accepted"]

    def test_missing_required_rejected(self):
        s = AgentBomMcpServer()
        resp = _call_raw(s, "check_package", {"name": "flask"})
        assert resp["result"]["isError"]
        payload = json.loads(resp["result"]["content"][0]["text"])
        assert set(payload["missing"]) == {"ecosystem", "version"}

    def test_valid_call_still_works(self):
        s = AgentBomMcpServer()
        resp = _call_raw(s, "runtime_blueprints", {})
        assert not resp["result"].get("isError")


class TestTenant:
    def test_resolution_order(self, monkeypatch):
        monkeypatch.delenv("AGENT_BOM_MCP_TENANT_ID", raising=False)
        monkeypatch.delenv("AGENT_BOM_TENANT_ID", raising=False)
        assert resolve_mcp_tenant_id() == "default"
        monkeypatch.setenv("AGENT_BOM_TENANT_ID", "org-a")
        assert resolve_mcp_tenant_id() == "org-a"
        monkeypatch.setenv("AGENT_BOM_MCP_TENANT_ID", "org-b")
        assert resolve_mcp_tenant_id() == "org-b"

    def test_initialize_reports_tenant(self, monkeypatch):
        monkeypatch.setenv("AGENT_BOM_MCP_TENANT_ID", "org-x")
        s = AgentBomMcpServer()
        resp = s.handle({"jsonrpc": "2.0", "id": 1, "method": "initialize"})
        assert resp["result"]["serverInfo"]["tenant"] == "org-x"


class TestAttestation:
    def _envelope(self, **kw):
        stmt = build_attestation_statement(
            server_name="fs", instance_digest="d" * 64, verdict="pass",
            tool_names=["read_file", "write_file"], tenant_id="org-a", **kw)
        return sign_attestation(stmt, KEY, "k1")

    def test_roundtrip(self):
        env = self._envelope()
        out = verify_attestation(env, {"keys": {"k1": KEY_HEX}})
        assert out["valid"], out
        pred = out["statement"]["predicate"]
        assert pred["verdict"] == "pass"
        assert pred["tool_count"] == 2

    def test_unpinned_signer_rejected(self):
        env = self._envelope()
        out = verify_attestation(env, {"keys": {"other": KEY_HEX}})
        assert not out["valid"] and "pinned" in out["reason"]

    def test_tamper_detected(self):
        import base64

        env = self._envelope()
        payload = json.loads(base64.b64decode(env["payload"]))
        payload["predicate"]["verdict"] = "block"
        env["payload"] = base64.b64encode(
            json.dumps(payload, sort_keys=True).encode()).decode()
        out = verify_attestation(env, {"keys": {"k1": KEY_HEX}})
        assert not out["valid"]

    def test_tenant_and_freshness_policy(self):
        env = self._envelope()
        out = verify_attestation(env, {"keys": {"k1": KEY_HEX},
                                       "expected_tenant": "org-b"})
        assert not out["valid"] and "tenant" in out["reason"]
        stale = self._envelope(observed_at=time.time() - 10 * 24 * 3600)
        out = verify_attestation(stale, {"keys": {"k1": KEY_HEX}})
        assert not out["valid"] and "stale" in out["reason"]

    def test_mcp_tools_end_to_end(self, monkeypatch):
        monkeypatch.setenv("AGENT_BOM_ATTESTATION_KEY", KEY_HEX)
        monkeypatch.setenv("AGENT_BOM_MCP_TENANT_ID", "org-a")
        s = AgentBomMcpServer()
        report, _g = s._ensure_scan()
        name = report.agents[0].mcp_servers[0].name
        env = s.tools["attest_server"].fn(server_name=name)
        assert "signatures" in env
        out = s.tools["verify_attestation"].fn(envelope=env,
                                               expected_tenant="org-a")
        assert out["valid"]


class TestAuthPosture:
    def _agent(self, servers):
        return Agent(name="a", agent_type=AgentType.CLAUDE_DESKTOP,
                     config_path="/x", mcp_servers=servers)

    def test_remote_no_auth_is_critical(self):
        srv = MCPServer(name="rmt", command="", url="https://mcp.example.com/sse")
        f = assess_server(srv, ["a"])
        assert f.posture == "no_auth_remote" and f.severity == "critical"

    def test_plaintext_transport(self):
        srv = MCPServer(name="rmt", command="", url="http://mcp.example.com/sse")
        assert assess_server(srv, []).posture == "plaintext_transport"

    def test_token_oauth_local_stdio(self):
        tok = MCPServer(name="t", command="", url="https://x/sse",
                        env={"API_TOKEN": "***"})
        assert assess_server(tok, []).posture == "static_token"
        oauth = MCPServer(name="o", command="", url="https://x/sse",
                          env={"OAUTH_CLIENT_ID": "***"})
        assert assess_server(oauth, []).posture == "oauth"
        local = MCPServer(name="l", command="", url="http://localhost:3000/sse")
        assert assess_server(local, []).posture == "no_auth_local"
        stdio = MCPServer(name="s", command="npx", args=["x"])
        assert assess_server(stdio, []).posture == "stdio_local"

    def test_estate_rollup(self):
        agents = [self._agent([
            MCPServer(name="rmt", command="", url="https://evil.example/sse"),
            MCPServer(name="s", command="npx", args=["x"])])]
        out = assess_estate(agents)
        assert out["servers_assessed"] == 2
        assert out["critical_exposures"][0]["server_name"] == "rmt"
        assert out["posture_counts"]["stdio_local"] == 1


class TestSelfPosture:
    def test_unhardened_default(self):
        out = evaluate_self_posture(env={})
        states = {c["check_id"]: c["state"] for c in out["checks"]}
        assert states["SELF-001"] == "misconfigured"
        assert states["SELF-005"] == "unknown"  # offline unpinned

    def test_hardened_and_acknowledged(self):
        out = evaluate_self_posture(env={
            "AGENT_BOM_API_KEYS": "k:admin",
            "AGENT_BOM_API_KEY": "k",
            "AGENT_BOM_MCP_TENANT_ID": "t",
            "AGENT_BOM_GRAPH_STORE": "/tmp/g.db",
            "AGENT_BOM_OFFLINE": "1",
            "AGENT_BOM_ATTESTATION_KEY": KEY_HEX,
            "AGENT_BOM_ACCEPTED_WEAKENINGS": "SELF-007",
        })
        states = {c["check_id"]: c["state"] for c in out["checks"]}
        assert states["SELF-001"] == "hardened"
        assert states["SELF-005"] == "hardened"
        assert out["score"] > 50


class TestA2APosture:
    def test_shared_credential_and_mutual_auth(self):
        from agentbom_amd.identity import AgentIdentityStore
        from agentbom_amd.scan.auth_posture import assess_a2a
        from agentbom_amd.scan.orchestrator import run_demo_scan

        report = run_demo_scan()
        store = AgentIdentityStore()
        out = assess_a2a(report.agents, identity_store=store)
        # the demo estate shares credentials across agents by design
        assert out["by_weakness"].get("shared_static_credential", 0) >= 1
        shared = next(f for f in out["findings"]
                      if f["weakness"] == "shared_static_credential")
        assert len(shared["agents"]) > 1

    def test_issued_identity_clears_mutual_auth(self):
        from agentbom_amd.identity import AgentIdentityStore
        from agentbom_amd.scan.auth_posture import assess_a2a
        from agentbom_amd.scan.orchestrator import run_demo_scan

        report = run_demo_scan()
        store = AgentIdentityStore()
        for a in report.agents:
            store.issue(a.name, scopes=["scan:read"],
                        allowed_tools=["read_file"])
        out = assess_a2a(report.agents, identity_store=store)
        assert out["by_weakness"].get("missing_mutual_auth", 0) == 0

    def test_wildcard_identity_flagged(self):
        from agentbom_amd.identity import AgentIdentityStore
        from agentbom_amd.scan.auth_posture import assess_a2a
        from agentbom_amd.scan.orchestrator import run_demo_scan

        report = run_demo_scan()
        store = AgentIdentityStore()
        store.issue("x", scopes=["*"])
        out = assess_a2a(report.agents, identity_store=store)
        assert out["by_weakness"].get("over_broad_delegation", 0) >= 1


class TestPostureCli:
    def test_posture_command(self):
        from click.testing import CliRunner

        from agentbom_amd.cli import main

        out = CliRunner().invoke(main, ["posture", "--demo", "--a2a"])
        assert out.exit_code in (0, 1), out.output
        doc = json.loads(out.output[out.output.index("{"):])
        assert "mcp_auth_posture" in doc and "self_posture" in doc
        assert "a2a_auth_posture" in doc

    def test_compliance_bundle_command(self, tmp_path, monkeypatch):
        from click.testing import CliRunner

        from agentbom_amd.cli import main

        monkeypatch.setenv("AGENT_BOM_AUDIT_HMAC_KEY", "k")
        out = CliRunner().invoke(main, [
            "compliance-bundle", "owasp_llm", "--demo",
            "-o", str(tmp_path / "b.json")])
        assert out.exit_code == 0, out.output
        doc = json.loads((tmp_path / "b.json").read_text())
        assert doc["signature"]["status"] == "signed"

    def test_compliance_bundle_unknown_framework(self):
        from click.testing import CliRunner

        from agentbom_amd.cli import main

        out = CliRunner().invoke(main, ["compliance-bundle", "nope", "--demo"])
        assert out.exit_code == 2


class TestExecScoreFreshness:
    def test_demo_estate_scores_poorly(self):
        from agentbom_amd.scan.orchestrator import run_demo_scan
        from agentbom_amd.scan.risk import estate_exec_score

        out = estate_exec_score(run_demo_scan())
        assert out["grade"] in ("D", "F")  # KEV + malicious by design
        drivers = {d["driver"] for d in out["drivers"]}
        assert {"known_exploited", "malicious_packages"} <= drivers

    def test_clean_estate_scores_high(self):
        from agentbom_amd.models import AIBOMReport
        from agentbom_amd.scan.risk import estate_exec_score

        out = estate_exec_score(AIBOMReport(agents=[], blast_radii=[]))
        assert out["score"] == 100.0 and out["grade"] == "A"

    def test_partial_scan_caps_score(self):
        from agentbom_amd.models import AIBOMReport
        from agentbom_amd.models.report import ScanIssue, ScanRun

        from agentbom_amd.scan.risk import estate_exec_score

        report = AIBOMReport(agents=[], blast_radii=[])
        report.scan_run = ScanRun(issues=[ScanIssue(
            code="x", stage="s", source="t", message="m",
            severity="error", affects_coverage=True)])
        out = estate_exec_score(report)
        assert out["score"] <= 60.0

    def test_freshness_grading(self):
        from datetime import datetime, timedelta, timezone

        from agentbom_amd.db.store import grade_freshness

        now = datetime.now(timezone.utc)
        sync = {
            "osv": {"last_synced": (now - timedelta(days=1)).isoformat()},
            "epss": {"last_synced": (now - timedelta(days=10)).isoformat()},
            "kev": {"last_synced": (now - timedelta(days=90)).isoformat()},
            "ghsa": {"last_synced": None},
        }
        out = grade_freshness(sync, now=now)
        assert out["sources"]["osv"]["grade"] == "fresh"
        assert out["sources"]["epss"]["grade"] == "stale"
        assert out["sources"]["kev"]["grade"] == "expired"
        assert out["sources"]["ghsa"]["grade"] == "unknown"
        assert out["overall"] == "unknown"
        assert grade_freshness({}, now=now)["overall"] == "unknown"
