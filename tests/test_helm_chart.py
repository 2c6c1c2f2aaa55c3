"""Helm chart consistency checks (no helm binary in this environment)."""

import re
from pathlib import Path

import yaml

CHART = Path(__file__).resolve().parents[1] / "deploy/helm/agent-bom"


def test_chart_and_values_parse():
    chart = yaml.safe_load((CHART / "Chart.yaml").read_text())
    assert chart["name"] == "agent-bom"
    values = yaml.safe_load((CHART / "values.yaml").read_text())
    assert values["service"]["port"] == 8080
    assert values["amdGpu"]["enabled"] is False  # GPU off by default


def test_every_values_reference_exists():
    values = yaml.safe_load((CHART / "values.yaml").read_text())

    def has_path(doc, parts):
        for p in parts:
            if not isinstance(doc, dict) or p not in doc:
                return False
            doc = doc[p]
        return True

    missing = []
    for tpl in (CHART / "templates").glob("*.yaml"):
        for ref in re.findall(r"\.Values\.([A-Za-z0-9_.]+)", tpl.read_text()):
            if not has_path(values, ref.split(".")):
                missing.append(f"{tpl.name}: .Values.{ref}")
    assert not missing, f"templates reference undefined values: {missing}"


def test_gpu_resource_wiring():
    text = (CHART / "templates/deployment.yaml").read_text()
    assert "amd.com/gpu" in text  # AMD device-plugin resource, not nvidia.com
    assert "nvidia.com" not in text
