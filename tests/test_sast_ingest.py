"""Semgrep ingestion → SAST findings + symbol-index joins."""

import json

from agentbom_amd.models.finding import FindingSource, FindingType
from agentbom_amd.scan.ast_analysis import SymbolIndex
from agentbom_amd.scan.sast_ingest import (
    extend_symbol_index_from_semgrep,
    load_semgrep_file,
    looks_like_semgrep,
    parse_semgrep_json,
    semgrep_to_findings,
)

SEMGREP_DOC = {
    "version": "1.50.0",
    "results": [
        {
            "check_id": "python.lang.security.audit.dangerous-subprocess-use",
            "path": "app/run.py",
            "start": {"line": 42, "col": 5},
            "end": {"line": 42, "col": 60},
            "extra": {
                "severity": "ERROR",
                "message": "subprocess call with shell=True",
                "lines": "    subprocess.call(cmd, shell=True)",
                "metadata": {"cwe": ["CWE-78: OS Command Injection"],
                             "owasp": ["A03:2021"],
                             "confidence": "HIGH"},
                "metavars": {"$CMD": {"abstract_content": "render_template(x)"}},
            },
        },
        {
            "check_id": "js.express.xss",
            "path": "web/server.js",
            "start": {"line": 7},
            "end": {"line": 7},
            "extra": {"severity": "WARNING", "message": "reflected xss",
                      "metadata": {"cwe": "CWE-79"}},
        },
        {"check_id": 12345, "path": "bad-row"},
        "not-a-dict",
    ],
    "errors": [],
}


class TestParse:
    def test_parses_valid_rows_skips_malformed(self):
        rows = parse_semgrep_json(SEMGREP_DOC)
        assert len(rows) == 2
        r = rows[0]
        assert r.line == 42 and r.severity == "high"
        assert r.cwe_ids == ["CWE-78"]
        assert r.owasp == ["A03:2021"]
        assert r.confidence == "high"
        assert rows[1].severity == "medium" and rows[1].cwe_ids == ["CWE-79"]

    def test_accepts_json_text_and_garbage(self):
        assert parse_semgrep_json(json.dumps(SEMGREP_DOC))
        assert parse_semgrep_json("{broken") == []
        assert parse_semgrep_json(None) == []
        assert parse_semgrep_json([1, 2]) == []

    def test_load_missing_file(self, tmp_path):
        assert load_semgrep_file(tmp_path / "nope.json") == []

    def test_format_detection_vs_sarif(self):
        assert looks_like_semgrep(SEMGREP_DOC)
        sarif = {"runs": [{"results": []}], "results": []}
        assert not looks_like_semgrep(sarif)
        assert not looks_like_semgrep("junk")


class TestFindings:
    def test_unified_finding_shape(self):
        fs = semgrep_to_findings(parse_semgrep_json(SEMGREP_DOC))
        assert len(fs) == 2
        f = fs[0]
        assert f.finding_type == FindingType.SAST
        assert f.source == FindingSource.SAST
        assert f.severity == "high"
        assert f.asset.location == "app/run.py:42"
        assert f.evidence["tool"] == "semgrep"
        assert "CWE-78" in f.cwe_ids

    def test_dedup_by_rule_path_line(self):
        rows = parse_semgrep_json(SEMGREP_DOC) * 3
        assert len(semgrep_to_findings(rows)) == 2


class TestSymbolJoin:
    def test_matched_lines_and_metavars_feed_calls(self):
        idx = SymbolIndex()
        added = extend_symbol_index_from_semgrep(
            idx, parse_semgrep_json(SEMGREP_DOC))
        assert added >= 2
        assert "subprocess.call" in idx.calls
        assert "render_template" in idx.calls

    def test_keywords_not_added(self):
        idx = SymbolIndex()
        doc = {"results": [{"check_id": "x", "path": "a.py",
                            "extra": {"lines": "if (x) return foo(y)"}}]}
        extend_symbol_index_from_semgrep(idx, parse_semgrep_json(doc))
        assert "if" not in idx.calls and "return" not in idx.calls
        assert "foo" in idx.calls

    def test_reachability_join_end_to_end(self):
        """A semgrep-flagged call site makes an advisory symbol
        function_reachable."""
        from agentbom_amd.scan.ast_analysis import apply_symbol_reachability
        from agentbom_amd.scan.orchestrator import run_demo_scan

        report = run_demo_scan()
        br = report.blast_radii[0]
        br.vulnerability.affected_symbols = ["pickle.loads"]
        idx = SymbolIndex()
        doc = {"results": [{"check_id": "x", "path": "a.py",
                            "extra": {"lines": "data = pickle.loads(blob)"}}]}
        extend_symbol_index_from_semgrep(idx, parse_semgrep_json(doc))
        apply_symbol_reachability(report, idx)
        assert br.symbol_reachability == "function_reachable"
        assert br.reachable_affected_symbols == ["pickle.loads"]


class TestCliFlag:
    def test_scan_with_semgrep_flag(self, tmp_path):
        from click.testing import CliRunner

        from agentbom_amd.cli import main

        f = tmp_path / "semgrep.json"
        f.write_text(json.dumps(SEMGREP_DOC))
        out = tmp_path / "report.json"
        res = CliRunner().invoke(main, [
            "agents", "--demo", "--offline", "--exit-zero",
            "--semgrep", str(f), "-f", "json", "-o", str(out)])
        # demo estate trips the malicious-package gate (fails closed, not
        # overridable by --exit-zero) — exit 1 with the report still written
        assert res.exit_code in (0, 1), res.output
        doc = json.loads(out.read_text())
        sast = [x for x in doc["findings"] if x["finding_type"] == "SAST"]
        assert len(sast) >= 2

    def test_mcp_external_ingest_autodetects_semgrep(self, tmp_path):
        from agentbom_amd.mcp.server import AgentBomMcpServer

        f = tmp_path / "semgrep.json"
        f.write_text(json.dumps(SEMGREP_DOC))
        s = AgentBomMcpServer()
        resp = s.handle({"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                         "params": {"name": "ingest_external_scan",
                                    "arguments": {"sarif_path": str(f)}}})
        payload = json.loads(resp["result"]["content"][0]["text"])
        assert payload["ingested"] == 2
        assert payload["by_level"].get("error") == 1
