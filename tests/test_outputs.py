"""Output contract tests: JSON / SARIF / CycloneDX / SPDX / CSV / etc."""

import json

import pytest

from agentbom_amd.output import json_fmt, misc_fmt
from agentbom_amd.output.cyclonedx_fmt import to_cyclonedx
from agentbom_amd.output.sarif import to_sarif
from agentbom_amd.output.spdx_fmt import to_spdx
from agentbom_amd.scan.orchestrator import run_demo_scan

REQUIRED_TOP_KEYS = {
    "schema_version", "canonical_id_schema_version", "document_type", "spec_version",
    "scan_id", "ai_bom_version", "generated_at", "scan_run", "warnings",
    "scan_sources", "has_mcp_context", "has_agent_context", "ai_bom_entities",
    "packages", "summary", "finding_summary", "assets", "inventory_snapshot",
    "agents", "blast_radius", "exposure_paths", "findings",
    "threat_framework_summary", "scorecard_summary", "remediation_plan",
}

REQUIRED_BLAST_KEYS = {
    "schema_version", "canonical_id", "asset", "exposure_path", "package_name",
    "package_version", "package_stable_id", "package_canonical_id", "risk_score",
    "reachability", "actionable", "vulnerability_id", "severity", "severity_label",
    "severity_state", "advisory_sources", "primary_advisory_source",
    "advisory_coverage_state", "match_confidence_tier", "cvss_score", "epss_score",
    "is_kev", "exploit_likelihood", "vex_status", "vex_suppressed", "suppressed",
    "compliance_tags", "package", "ecosystem", "layer_attribution",
    "is_malicious", "malicious_reason", "affected_agents", "affected_servers",
    "exposed_credentials", "exposed_tools", "phantom_tools", "framework_tags",
    "impact_category", "triage_priority", "all_server_credentials",
    "attack_vector_summary", "fixed_version", "hop_depth", "delegation_chain",
    "transitive_agents", "transitive_credentials", "transitive_risk_score",
    "dependency_reachable", "dependency_min_hop_distance",
    "dependency_reachable_from_agents", "graph_reachable", "symbol_reachability",
    "owasp_tags", "atlas_tags", "attack_tags", "nist_ai_rmf_tags",
}


@pytest.fixture(scope="module")
def report():
    return run_demo_scan()


@pytest.fixture(scope="module")
def doc(report):
    return json_fmt.to_json(report)


class TestJsonContract:
    def test_schema_version(self, doc):
        assert doc["schema_version"] == "1.0"
        assert doc["document_type"] == "AI-BOM"

    def test_top_level_keys(self, doc):
        missing = REQUIRED_TOP_KEYS - set(doc)
        assert not missing, f"missing top-level keys: {missing}"

    def test_blast_radius_entry_keys(self, doc):
        entry = doc["blast_radius"][0]
        missing = REQUIRED_BLAST_KEYS - set(entry)
        assert not missing, f"missing blast keys: {missing}"

    def test_serializable(self, doc):
        assert json.loads(json.dumps(doc, default=str))

    def test_exposure_paths_parallel_block(self, doc):
        ep = doc["exposure_paths"]
        assert ep["schema_version"] == "1"
        assert ep["source"] == "blast_radius_output"
        assert ep["path_count"] == len(doc["blast_radius"])
        path = ep["paths"][0]
        for key in ("id", "rank", "label", "riskScore", "hops", "relationships",
                    "nodeIds", "edgeIds", "fix", "provenance"):
            assert key in path

    def test_summary_counts(self, doc, report):
        s = doc["summary"]
        assert s["total_agents"] == report.total_agents
        assert s["total_mcp_servers"] == report.total_servers
        assert s["total_findings"] == len(doc["findings"])

    def test_blast_rank_order_is_risk_desc(self, doc):
        scores = [b["risk_score"] for b in doc["blast_radius"]]
        assert scores == sorted(scores, reverse=True)

    def test_severity_label_advisory_for_unknown(self):
        from agentbom_amd.models import Severity
        from agentbom_amd.output.json_fmt import _severity_label, _severity_state

        assert _severity_label(Severity.UNKNOWN) == "advisory"
        assert _severity_state(Severity.UNKNOWN) == "pending"
        assert _severity_label(Severity.HIGH) == "high"


class TestSarif:
    def test_valid_shape(self, report):
        doc = to_sarif(report)
        assert doc["version"] == "2.1.0"
        run = doc["runs"][0]
        assert run["tool"]["driver"]["name"] == "agent-bom"
        assert len(run["results"]) == len(report.blast_radii)
        result = run["results"][0]
        assert result["level"] in ("error", "warning", "note")
        assert "risk_score" in result["properties"]

    def test_kev_tagged(self, report):
        doc = to_sarif(report)
        rules = {r["id"]: r for r in doc["runs"][0]["tool"]["driver"]["rules"]}
        assert "cisa-kev" in rules["CVE-2023-4863"]["properties"]["tags"]


class TestSboms:
    def test_cyclonedx(self, report):
        doc = to_cyclonedx(report)
        assert doc["bomFormat"] == "CycloneDX"
        assert doc["specVersion"] == "1.6"
        assert len(doc["components"]) == 17  # unique packages
        assert doc["vulnerabilities"]
        v = next(v for v in doc["vulnerabilities"] if v["id"] == "CVE-2020-14343")
        assert v["affects"]

    def test_spdx(self, report):
        doc = to_spdx(report)
        assert doc["spdxVersion"] == "SPDX-2.3"
        assert doc["packages"]
        assert all(p["SPDXID"].startswith("SPDXRef-") for p in doc["packages"])


class TestMiscFormats:
    def test_csv(self, report):
        text = misc_fmt.to_csv(report)
        lines = text.strip().splitlines()
        assert len(lines) == 1 + len(report.blast_radii)
        assert "vulnerability_id" in lines[0]

    def test_markdown(self, report):
        text = misc_fmt.to_markdown(report)
        assert "CVE-2020-14343" in text
        assert "**KEV**" in text

    def test_prometheus(self, report):
        text = misc_fmt.to_prometheus(report)
        assert 'agent_bom_vulnerabilities_total{severity="critical"} 3' in text
        assert "agent_bom_kev_findings_total 1" in text

    def test_junit(self, report):
        text = misc_fmt.to_junit(report)
        assert text.startswith("<?xml")
        assert "<testsuite" in text and "<failure" in text


class TestGraphExports:
    def test_dot_mermaid_graphml_cypher(self, report):
        from agentbom_amd.output import graph_export

        assert "digraph" in graph_export.to_dot(report)
        assert graph_export.to_mermaid(report).startswith("graph LR")
        assert "<graphml" in graph_export.to_graphml(report)
        assert "MERGE" in graph_export.to_cypher(report)

    def test_console_renders(self, report):
        import io

        from rich.console import Console

        from agentbom_amd.output.console_render import render_report

        buf = io.StringIO()
        render_report(report, console=Console(file=buf, width=120))
        out = buf.getvalue()
        assert "CVE-2020-14343" in out
        assert "AI-BOM Scan Report" in out


class TestParquetBadge:
    def test_parquet(self, report, tmp_path):
        import pyarrow.parquet as pq

        data = misc_fmt.to_parquet_bytes(report)
        p = tmp_path / "f.parquet"
        p.write_bytes(data)
        table = pq.read_table(p)
        assert table.num_rows == len(report.to_findings())
        assert "risk_score" in table.column_names

    def test_badge(self, report):
        svg = misc_fmt.to_badge_svg(report)
        assert svg.startswith("<svg") and "critical" in svg


class TestPdf:
    def test_structurally_valid(self, report):
        import re
        import zlib

        from agentbom_amd.output.pdf_fmt import to_pdf_bytes

        data = to_pdf_bytes(report)
        assert data.startswith(b"%PDF-1.4")
        assert data.rstrip().endswith(b"%%EOF")
        text = data.decode("latin-1")
        # every xref offset points at the matching "N 0 obj" header
        xref_at = int(text.rsplit("startxref\n", 1)[1].split("\n")[0])
        assert text[xref_at:xref_at + 4] == "xref"
        entries = re.findall(r"^(\d{10}) 00000 n",
                             text[xref_at:], flags=re.M)
        for i, off in enumerate(entries, start=1):
            assert text[int(off):].startswith(f"{i} 0 obj")
        # content streams decompress and carry the headline text
        streams = re.findall(rb"stream\n(.*?)\nendstream", data, re.S)
        assert streams
        first = zlib.decompress(streams[0]).decode("latin-1")
        assert "AI-BOM Scan Report" in first
        assert "CVE-2020-14343" in first

    def test_many_findings_paginate(self, report):
        import copy
        import re

        from agentbom_amd.output.pdf_fmt import to_pdf_bytes

        big = copy.copy(report)
        big.blast_radii = report.blast_radii * 20  # ~320 rows
        data = to_pdf_bytes(big)
        n_pages = len(re.findall(rb"/Type /Page\b(?! s)", data))
        assert n_pages >= 4

    def test_cli_pdf_output(self, report, tmp_path):
        from click.testing import CliRunner

        from agentbom_amd.cli import main

        out = CliRunner().invoke(main, ["scan", "--demo", "--offline",
                                        "-f", "pdf",
                                        "-o", str(tmp_path / "r.pdf")])
        assert out.exit_code == 1  # demo estate gates by design
        assert (tmp_path / "r.pdf").read_bytes().startswith(b"%PDF")


class TestFindingViews:
    def test_views_and_compact(self, report):
        from agentbom_amd.output import finding_views as fv

        sev = fv.by_severity(report)
        assert list(sev["groups"]) == [s for s in
                                       ("critical", "high", "medium", "low",
                                        "unknown", "none") if s in sev["groups"]]
        total = sum(len(v) for v in sev["groups"].values())
        assert total == len(report.blast_radii)

        pkg = fv.by_package(report)
        firsts = [max(r["risk_score"] for r in v)
                  for v in pkg["groups"].values()]
        assert firsts == sorted(firsts, reverse=True)

        ag = fv.by_agent(report)
        assert ag["groups"]

        fw = fv.by_framework(report)
        assert all(rows for rows in fw["groups"].values())

        compact = fv.to_compact(report, top=5)
        assert len(compact["top_findings"]) == 5
        assert compact["summary"]["total_findings"] == len(report.blast_radii)
        assert compact["summary"]["kev_count"] >= 1


class TestNarratives:
    def test_advisory_text(self, report):
        from agentbom_amd.output.narratives import advisory_text

        text = advisory_text(report)
        assert "CVE-2020-14343" in text
        assert "KNOWN MALICIOUS" in text  # reqeusts typosquat
        assert "Known Exploited Vulnerabilities" in text  # KEV Pillow
        assert "Fix: upgrade to" in text

    def test_compliance_narrative(self, report):
        from agentbom_amd.output.narratives import compliance_narrative

        text = compliance_narrative(report)
        assert "OWASP LLM Top 10" in text
        assert "worst residual risk" in text
        empty = compliance_narrative(report, framework_field="no_such_field",
                                     framework_label="X")
        assert "No findings mapped" in empty

    def test_cis_posture_text(self):
        from agentbom_amd.output.narratives import cis_posture_text

        rows = [
            {"check_id": "CIS-1.1", "title": "root MFA", "severity": "critical",
             "status": "fail", "resource": "iam:root", "detail": "no MFA"},
            {"check_id": "CIS-2.1", "title": "bucket public", "severity": "high",
             "status": "pass", "resource": "s3:a"},
            {"check_id": "CIS-3.1", "title": "trail", "severity": "high",
             "status": "error", "resource": "ct"},
        ]
        text = cis_posture_text(rows)
        assert "1/3 checks pass" in text
        assert "CIS-1.1" in text and "no MFA" in text
        assert "UNVERIFIED" in text


class TestConsoleRenderSections:
    """Console render parity sections: remediation
    plan, compliance posture, exposure-path tree, coverage-gap panel."""

    def _render(self, report, verbose=True, width=120):
        import io

        from rich.console import Console

        from agentbom_amd.output.console_render import render_report

        buf = io.StringIO()
        render_report(report, console=Console(file=buf, width=width),
                      verbose=verbose)
        return buf.getvalue()

    def test_sections_present(self):
        from agentbom_amd.scan.orchestrator import run_demo_scan

        out = self._render(run_demo_scan())
        assert "Remediation plan" in out
        assert "Compliance posture" in out
        assert "Top exposure paths" in out
        assert "Findings (by risk)" in out
        # remediation carries a concrete action for the demo PyYAML RCE
        assert "pyyaml" in out

    def test_other_findings_table(self):
        from agentbom_amd.models.finding import (
            Asset,
            Finding,
            FindingSource,
            FindingType,
        )
        from agentbom_amd.scan.orchestrator import run_demo_scan

        report = run_demo_scan()
        report.findings.append(Finding(
            finding_type=FindingType.CREDENTIAL_EXPOSURE,
            source=FindingSource.SECRET_SCAN,
            asset=Asset(name="env", asset_type="file", location="/app/.env"),
            severity="high", title="AWS key in .env", description="x",
            id="sec-1"))
        out = self._render(report)
        assert "Other findings" in out
        assert "CREDENTIAL_EXPOSURE" in out


def test_spdx3_jsonld_shape():
    from agentbom_amd.output.spdx_fmt import to_spdx3
    from agentbom_amd.scan.orchestrator import run_demo_scan

    doc = to_spdx3(run_demo_scan())
    assert doc["@context"].endswith("spdx-context.jsonld")
    graph = doc["@graph"]
    types = [e.get("type") for e in graph]
    assert "CreationInfo" in types and "SpdxDocument" in types
    pkgs = [e for e in graph if e.get("type") == "software_Package"]
    assert pkgs, "no packages exported"
    for p in pkgs:
        assert p["spdxId"].startswith("urn:agent-bom:pkg:")
        purl = p["externalIdentifier"][0]["identifier"]
        assert purl.startswith("pkg:")
    sdoc = next(e for e in graph if e.get("type") == "SpdxDocument")
    rels = [e for e in graph if e.get("type") == "Relationship"]
    assert len(rels) == len(sdoc["rootElement"]) == len(pkgs)
    assert {r["to"][0] for r in rels} == {p["spdxId"] for p in pkgs}


def test_html_report_sections():
    from agentbom_amd.output.html_fmt import to_html
    from agentbom_amd.scan.orchestrator import run_demo_scan

    h = to_html(run_demo_scan())
    for section in ("Remediation plan", "Compliance posture",
                    "Top exposure paths", "Blast radius"):
        assert section in h, section
    assert "<script" not in h  # self-contained, no external/active content
    assert "pyyaml" in h
