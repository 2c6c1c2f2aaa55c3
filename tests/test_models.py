"""Core data-model behavior tests: identities, risk scoring, reachability."""

import pytest

from agentbom_amd.models import (
    Agent,
    AgentType,
    BlastRadius,
    MCPServer,
    MCPTool,
    Package,
    Severity,
    Vulnerability,
    blast_radius_to_finding,
    classify_cwe_impact,
    compute_confidence,
    expand_blast_radius_hops,
    filter_credentials_by_impact,
    filter_tools_by_impact,
    is_credential_key,
)
from agentbom_amd.utils.canonical_ids import canonical_package_id


def make_vuln(**kw):
    base = dict(id="CVE-2024-0001", summary="test vuln", severity=Severity.HIGH)
    base.update(kw)
    return Vulnerability(**base)


class TestCanonicalIds:
    def test_package_id_reference_parity(self):
        # Byte-level expectations pinned from the reference implementation.
        assert canonical_package_id("PyYAML", "5.3.1", "pypi") == "edce0d7f-6f2c-5b06-a5da-8cae090b42fd"

    def test_pypi_name_normalization(self):
        assert canonical_package_id("Py_YAML.x", "1", "pypi") == canonical_package_id("py-yaml-x", "1", "pypi")

    def test_stable_across_instances(self):
        p1 = Package(name="torch", version="2.3.0", ecosystem="pypi")
        p2 = Package(name="torch", version="2.3.0", ecosystem="pypi")
        assert p1.stable_id == p2.stable_id

    def test_tool_scoped_to_server(self):
        s1 = MCPServer(name="a", command="npx", tools=[MCPTool(name="read", description="")])
        s2 = MCPServer(name="b", command="uvx", tools=[MCPTool(name="read", description="")])
        assert s1.tools[0].stable_id != s2.tools[0].stable_id


class TestVulnerability:
    def test_fixed_version_sha_sanitized(self):
        v = make_vuln(fixed_version="deadbeefdeadbeefdeadbeefdeadbeefdeadbeef")
        assert v.fixed_version is None
        v = make_vuln(fixed_version="deadbee")
        assert v.fixed_version is None
        v = make_vuln(fixed_version="nodigits")
        assert v.fixed_version is None
        v = make_vuln(fixed_version="1.2.3")
        assert v.fixed_version == "1.2.3"

    def test_exploit_likelihood_gradation(self):
        assert make_vuln(is_kev=True).exploit_likelihood == "actively_exploited"
        assert make_vuln().exploit_likelihood == "unassessed"
        assert make_vuln(epss_score=0.6).exploit_likelihood == "likely_exploited"
        assert make_vuln(epss_percentile=96.0).exploit_likelihood == "likely_exploited"
        assert make_vuln(epss_percentile=85.0).exploit_likelihood == "public_exploit"
        assert make_vuln(epss_score=0.01).exploit_likelihood == "theoretical"

    def test_cvss_vector_signals(self):
        v = make_vuln(cvss_vector="CVSS:3.1/AV:N/AC:L/PR:N/UI:N/S:U/C:H/I:H/A:H")
        assert v.attack_vector == "network"
        assert v.network_exploitable

    def test_advisory_sources_derived(self):
        v = make_vuln(id="GHSA-xxxx-yyyy-zzzz", epss_score=0.1, is_kev=True)
        srcs = v.all_advisory_sources
        assert "ghsa" in srcs and "epss" in srcs and "cisa_kev" in srcs
        assert v.advisory_coverage_state == "enriched"

    def test_confidence(self):
        v = make_vuln(cvss_score=9.8, severity_source="cvss", epss_score=0.5,
                      cwe_ids=["CWE-94"], fixed_version="1.2.3", cvss_vector="CVSS:3.1/AV:N")
        assert compute_confidence(v) == pytest.approx(1.0)


class TestRiskScore:
    def test_base_by_severity(self):
        for sev, expected in [(Severity.CRITICAL, 8.0), (Severity.HIGH, 6.0),
                              (Severity.MEDIUM, 4.0), (Severity.LOW, 2.0)]:
            br = BlastRadius(make_vuln(severity=sev), Package("a", "1", "npm"), [], [], [], [])
            assert br.calculate_risk_score() == expected

    def test_reach_amplifiers_capped(self):
        agents = [Agent(name=f"a{i}", agent_type=AgentType.CURSOR, config_path=f"/c{i}") for i in range(10)]
        br = BlastRadius(make_vuln(severity=Severity.LOW), Package("a", "1", "npm"),
                         [], agents, [f"K{i}_TOKEN" for i in range(10)], [])
        # base 2.0 + agent cap 2.0 + cred cap 1.5 + ai_boost 0 (1 signal: creds)
        assert br.calculate_risk_score() == pytest.approx(2.0 + 2.0 + 1.5)

    def test_kev_epss_boosts(self):
        br = BlastRadius(make_vuln(severity=Severity.CRITICAL, is_kev=True, epss_score=0.9),
                         Package("a", "1", "npm"), [], [], [], [])
        # 8.0 + 1.0 + 0.5 = 9.5
        assert br.calculate_risk_score() == pytest.approx(9.5)

    def test_clamped_to_10(self):
        agents = [Agent(name=f"a{i}", agent_type=AgentType.CURSOR, config_path=f"/c{i}") for i in range(10)]
        tools = [MCPTool(name=f"t{i}", description="") for i in range(20)]
        br = BlastRadius(make_vuln(severity=Severity.CRITICAL, is_kev=True, epss_score=0.9),
                         Package("a", "1", "npm", scorecard_score=1.0), [], agents,
                         ["A_TOKEN", "B_KEY", "C_SECRET", "D_PASSWORD", "E_TOKEN"], tools,
                         ai_risk_context="ai framework")
        assert br.calculate_risk_score() == 10.0

    def test_vex_suppression_zeroes(self):
        br = BlastRadius(make_vuln(vex_status="not_affected"), Package("a", "1", "npm"), [], [], [], [])
        assert br.calculate_risk_score() == 0.0
        assert br.is_actionable is False

    def test_graph_reachability_nudge(self):
        br = BlastRadius(make_vuln(severity=Severity.MEDIUM), Package("a", "1", "npm"), [], [], [], [])
        base = br.calculate_risk_score()
        br.graph_reachable = True
        assert br.calculate_risk_score() == pytest.approx(base + 0.5)
        br.graph_reachable = False
        assert br.calculate_risk_score() == pytest.approx(base - 0.5)


class TestReachability:
    def test_confirmed(self):
        br = BlastRadius(make_vuln(), Package("a", "1", "npm", is_direct=True), [], [], ["API_KEY"], [])
        assert br.reachability == "confirmed"

    def test_unlikely(self):
        br = BlastRadius(make_vuln(severity=Severity.LOW), Package("a", "1", "npm", is_direct=False), [], [], [], [])
        assert br.reachability == "unlikely"

    def test_declaration_only_unknown(self):
        br = BlastRadius(make_vuln(severity=Severity.LOW),
                         Package("a", "1", "npm", is_direct=True, reachability_evidence="declaration_only"),
                         [], [], [], [])
        assert br.reachability == "unknown"


class TestCweImpact:
    def test_classification(self):
        assert classify_cwe_impact(["CWE-94"]) == "code-execution"
        assert classify_cwe_impact(["CWE-79"]) == "client-side"
        assert classify_cwe_impact(["CWE-79", "CWE-94"]) == "code-execution"  # worst wins
        assert classify_cwe_impact([]) == "unknown"
        assert classify_cwe_impact(["CWE-99999"]) == "unknown"

    def test_credential_filter(self):
        creds = ["DATABASE_URL", "OPENAI_API_KEY"]
        assert filter_credentials_by_impact("code-execution", creds) == creds
        assert filter_credentials_by_impact("injection", creds) == ["DATABASE_URL"]
        assert filter_credentials_by_impact("client-side", creds) == []
        assert filter_credentials_by_impact("unknown", creds) == []

    def test_tool_filter(self):
        tools = [MCPTool(name="query_db", description=""), MCPTool(name="send_email", description="")]
        assert len(filter_tools_by_impact("code-execution", tools)) == 2
        assert [t.name for t in filter_tools_by_impact("injection", tools)] == ["query_db"]
        assert filter_tools_by_impact("availability", tools) == []


class TestCredentialHeuristic:
    @pytest.mark.parametrize("name,expected", [
        ("OPENAI_API_KEY", True),
        ("GITHUB_TOKEN", True),
        ("DB_PASSWORD", True),
        ("PGPASSWORD", True),
        ("ID_RSA", True),
        ("DATABASE_URL", True),
        ("CONNECTION_STRING", True),
        ("CERTIFICATE_PATH", False),
        ("OAUTH_CLIENT_ID", False),
        ("LOG_LEVEL", False),
        ("TIMEOUT_SECONDS", False),
    ])
    def test_cases(self, name, expected):
        assert is_credential_key(name) is expected


class TestHopExpansion:
    def _estate(self):
        shared = MCPServer(name="shared-srv", command="npx", env={"SHARED_TOKEN": "x"})
        s1 = MCPServer(name="s1", command="npx", env={"A_KEY": "x"})
        s2 = MCPServer(name="s2", command="uvx", env={"B_SECRET": "x"})
        a1 = Agent(name="agent1", agent_type=AgentType.CURSOR, config_path="/a1",
                   mcp_servers=[s1, shared])
        a2 = Agent(name="agent2", agent_type=AgentType.CLAUDE_DESKTOP, config_path="/a2",
                   mcp_servers=[shared, s2])
        return a1, a2, s1

    def test_two_hop_delegation(self):
        a1, a2, s1 = self._estate()
        br = BlastRadius(make_vuln(), Package("p", "1", "npm"), [s1], [a1], [], [])
        br.calculate_risk_score()
        expand_blast_radius_hops([br], [a1, a2], max_depth=3)
        assert br.hop_depth == 2
        names = [t["name"] for t in br.transitive_agents]
        assert "agent2" in names
        assert "B_SECRET" in br.transitive_credentials
        assert br.transitive_risk_score == pytest.approx(round(br.risk_score * 0.7, 2))

    def test_depth_one_noop(self):
        a1, a2, s1 = self._estate()
        br = BlastRadius(make_vuln(), Package("p", "1", "npm"), [s1], [a1], [], [])
        expand_blast_radius_hops([br], [a1, a2], max_depth=1)
        assert br.transitive_agents == []


class TestFindingConversion:
    def test_blast_radius_to_finding_roundtrip(self):
        pkg = Package("lodash", "4.17.20", "npm")
        v = make_vuln(id="CVE-2021-23337", cwe_ids=["CWE-94"], cvss_score=7.2)
        srv = MCPServer(name="srv", command="node", env={"NPM_TOKEN": "x"})
        ag = Agent(name="ag", agent_type=AgentType.CURSOR, config_path="/c")
        br = BlastRadius(v, pkg, [srv], [ag], ["NPM_TOKEN"], [])
        br.calculate_risk_score()
        f = blast_radius_to_finding(br)
        assert f.cve_id == "CVE-2021-23337"
        assert f.severity == "high"
        assert f.affected_agents == ["ag"]
        assert f.exposed_credentials == ["NPM_TOKEN"]
        assert f.evidence["package_name"] == "lodash"
        assert f.node_id.startswith("pkg:npm:lodash@")
        assert f.finding_node_id == "vuln:CVE-2021-23337"
        # deterministic id
        f2 = blast_radius_to_finding(br)
        assert f.id == f2.id

    def test_triage_priority_band(self):
        pkg = Package("p", "1", "npm")
        v = make_vuln(severity=Severity.CRITICAL, is_kev=True, epss_score=0.9, cwe_ids=["CWE-94"])
        br = BlastRadius(v, pkg, [], [], ["K_TOKEN"], [])
        br.impact_category = "code-execution"
        f = blast_radius_to_finding(br)
        tp = f.evidence["triage_priority"]
        assert tp["band"] in ("urgent", "high")
        assert "cisa_kev" in tp["reasons"]
