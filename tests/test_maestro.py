"""MAESTRO KC-layer classification + report summary."""

from __future__ import annotations

from agentbom_amd.models.maestro import (
    MaestroLayer,
    classify_finding,
    layer_label,
    maestro_summary,
)


def _f(ftype, asset_type="package", name="pkg", severity="high"):
    return {"finding_type": ftype, "severity": severity,
            "asset": {"asset_type": asset_type, "name": name}}


class TestClassification:
    def test_type_table(self):
        assert classify_finding(_f("TOOL_DRIFT")) == MaestroLayer.KC3_AGENTIC_PATTERNS
        assert classify_finding(_f("MODEL_INTEGRITY")) == MaestroLayer.KC1_AI_MODELS
        assert classify_finding(_f("CIS_FAIL")) == MaestroLayer.KC6_INFRASTRUCTURE
        assert classify_finding(_f("CREDENTIAL_EXPOSURE")) == \
            MaestroLayer.KC5_TOOLS_CAPABILITIES
        assert classify_finding(_f("SENSITIVE_DATA")) == \
            MaestroLayer.KC4_MEMORY_CONTEXT

    def test_asset_fallbacks(self):
        assert classify_finding(_f("CVE", "agent", "cursor")) == \
            MaestroLayer.KC2_AGENT_ARCHITECTURE
        assert classify_finding(_f("CVE", "mcp_server", "fs")) == \
            MaestroLayer.KC5_TOOLS_CAPABILITIES
        assert classify_finding(_f("CVE", "package", "qdrant-client")) == \
            MaestroLayer.KC4_MEMORY_CONTEXT
        assert classify_finding(_f("CVE", "package", "safetensors")) == \
            MaestroLayer.KC1_AI_MODELS
        assert classify_finding(_f("CVE", "package", "requests")) == \
            MaestroLayer.KC5_TOOLS_CAPABILITIES
        assert classify_finding(_f("CVE", "cloud_resource", "s3")) == \
            MaestroLayer.KC6_INFRASTRUCTURE

    def test_label(self):
        assert layer_label(MaestroLayer.KC1_AI_MODELS).startswith(
            "KC1: AI Models (LLM")


class TestSummaryAndReport:
    def test_summary_histogram(self):
        rows = maestro_summary([
            _f("CVE"), _f("CVE", severity="critical"), _f("CIS_FAIL",
                                                          "cloud_resource"),
        ])["layers"]
        by = {r["layer"]: r for r in rows}
        assert by["KC5: Tools & Capabilities"]["findings"] == 2
        assert by["KC5: Tools & Capabilities"]["worst_severity"] == "critical"
        assert by["KC6: Infrastructure"]["findings"] == 1

    def test_report_json_carries_layers(self):
        from agentbom_amd.output.json_fmt import to_json
        from agentbom_amd.scan.orchestrator import run_demo_scan

        doc = to_json(run_demo_scan())
        assert doc["maestro_summary"]["layers"]
        assert all("maestro_layer" in f for f in doc["findings"])
        layers = {f["maestro_layer"] for f in doc["findings"]}
        assert layers <= {l.value for l in MaestroLayer}
