"""Python AST security analysis + symbol reachability tests."""

import textwrap

import pytest

from agentbom_amd.scan.ast_analysis import (
    analyze_python_source,
    apply_symbol_reachability,
    ast_finding_to_finding,
    build_symbol_index,
)


class TestDangerousSinks:
    def test_eval_exec(self):
        findings, calls = analyze_python_source(
            "def handler(req):\n    return eval(req.body)\n", "app.py")
        assert len(findings) == 1
        f = findings[0]
        assert f.call == "eval" and f.category == "code-injection"
        assert f.severity == "high" and f.tainted
        assert f.entrypoint == "handler"

    def test_literal_arg_downgraded(self):
        findings, _ = analyze_python_source("x = eval('1+1')\n")
        assert findings[0].severity == "medium" and not findings[0].tainted

    def test_yaml_load_with_alias(self):
        src = "import yaml as y\ndata = y.unsafe_load(blob)\n"
        findings, _ = analyze_python_source(src)
        assert findings[0].call == "yaml.unsafe_load"
        assert findings[0].severity == "critical"

    def test_subprocess_shell_true_only(self):
        ok, _ = analyze_python_source("import subprocess\nsubprocess.run(['ls'])\n")
        assert not ok
        bad, _ = analyze_python_source(
            "import subprocess\nsubprocess.run(cmd, shell=True)\n")
        assert bad and bad[0].severity == "high"

    def test_requests_verify_false(self):
        findings, _ = analyze_python_source(
            "import requests\nrequests.get(url, verify=False)\n")
        assert findings[0].category == "tls-verification-disabled"

    def test_syntax_error_tolerated(self):
        findings, calls = analyze_python_source("def broken(:\n")
        assert findings == [] and calls == set()

    def test_to_finding(self):
        findings, _ = analyze_python_source("exec(payload)\n", "x.py")
        f = ast_finding_to_finding(findings[0])
        assert f.finding_type.value == "SAST"
        assert f.cwe_ids == ["CWE-95"]


class TestSymbolIndex:
    @pytest.fixture
    def project(self, tmp_path):
        (tmp_path / "app.py").write_text(textwrap.dedent("""\
            import yaml
            import requests

            def load_config(path):
                return yaml.full_load(open(path).read())
        """))
        (tmp_path / "util.py").write_text("import lodash_py\n")
        return tmp_path

    def test_index(self, project):
        idx = build_symbol_index(project)
        assert idx.files_scanned == 2
        assert idx.calls_symbol("yaml.full_load")
        assert idx.calls_symbol("full_load")
        assert idx.imports_module("requests")
        assert not idx.imports_module("flask")
        assert idx.findings  # yaml.full_load is a sink

    def test_reachability_stamping(self, project):
        from agentbom_amd.scan.demo import demo_advisory_windows
        from agentbom_amd.scan.orchestrator import ScanOptions, inventory_to_agents, scan_agents

        inv = {"agents": [{"name": "a", "agent_type": "custom", "mcp_servers": [
            {"name": "s", "command": "x", "packages": [
                {"name": "pyyaml", "version": "5.3", "ecosystem": "pypi"},
                {"name": "flask", "version": "2.2.0", "ecosystem": "pypi"},
            ]}]}]}
        report = scan_agents(inventory_to_agents(inv), demo_advisory_windows(), ScanOptions())
        # give the pyyaml advisory symbol data so the function-level join runs
        for br in report.blast_radii:
            if br.package.name == "pyyaml":
                br.vulnerability.affected_symbols = ["yaml.full_load"]
        idx = build_symbol_index(project)
        n = apply_symbol_reachability(report, idx)
        assert n >= 2
        yaml_br = next(b for b in report.blast_radii if b.package.name == "pyyaml")
        assert yaml_br.symbol_reachability == "function_reachable"
        assert yaml_br.reachable_affected_symbols == ["yaml.full_load"]
        flask_br = next(b for b in report.blast_radii if b.package.name == "flask")
        assert flask_br.symbol_reachability == "unreachable"
        # reachability nudge: function_reachable boosted above baseline
        assert yaml_br.risk_score >= flask_br.risk_score
