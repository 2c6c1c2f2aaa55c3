"""Standalone console printers (agent tree, severity chart, CIS, diff,
policy, posture) over the demo estate, recorded and text-asserted."""

from __future__ import annotations

import pytest
from rich.console import Console

from agentbom_amd.output.console_render import (
    print_agent_tree,
    print_cis_findings,
    print_diff,
    print_policy_results,
    print_posture_summary,
    print_severity_chart,
    render_report,
)


@pytest.fixture(scope="module")
def report():
    from agentbom_amd.scan.orchestrator import run_demo_scan

    return run_demo_scan()


def _text(fn, *args, **kw) -> str:
    console = Console(record=True, width=140, force_terminal=False)
    fn(*args, console=console, **kw)
    return console.export_text()


def test_agent_tree(report):
    text = _text(print_agent_tree, report)
    assert "Estate inventory" in text
    for agent in report.agents[:2]:
        assert agent.name in text
    assert "packages (" in text


def test_severity_chart(report):
    text = _text(print_severity_chart, report)
    assert "critical" in text and "█" in text


def test_posture_summary(report):
    text = _text(print_posture_summary, report)
    assert "estate grade" in text and "security posture" in text


def test_cis_findings(report):
    report.extra_data["cis_benchmark_data"] = [
        {"check_id": "CIS-1.10", "status": "fail", "title": "MFA missing",
         "evidence": "user bob"},
        {"check_id": "CIS-2.1", "status": "pass", "title": "ok"},
    ]
    text = _text(print_cis_findings, report)
    assert "CIS-1.10" in text and "MFA missing" in text
    assert "CIS-2.1" not in text  # passed checks hidden by default
    shown = _text(print_cis_findings, report, show_passed=True)
    assert "CIS-2.1" in shown
    report.extra_data.pop("cis_benchmark_data")


def test_diff_printer():
    diff = {"new_findings": [{"vulnerability_id": "CVE-2024-1", "severity":
                              "high", "package_name": "a@1", "risk_score": 7}],
            "resolved_findings": [{"vulnerability_id": "CVE-2020-9",
                                   "package": "b@2"}],
            "unchanged_count": 3,
            "packages_added": ["pypi:c@3"], "packages_removed": []}
    text = _text(print_diff, diff)
    assert "New findings (1)" in text and "CVE-2024-1" in text
    assert "Resolved (1)" in text and "CVE-2020-9" in text
    assert "unchanged: 3" in text and "pypi:c@3" in text


def test_policy_printer():
    fail = {"passed": False, "rules_evaluated": 2,
            "violations": [{"rule_id": "no-kev", "action": "fail",
                            "vulnerability_id": "CVE-1", "package": "p@1",
                            "risk_score": 9.1}],
            "warnings": []}
    text = _text(print_policy_results, fail)
    assert "no-kev" in text and "FAIL" in text
    ok = _text(print_policy_results, {"passed": True, "rules_evaluated": 2})
    assert "PASS" in ok


def test_render_report_verbose_includes_tree(report):
    console = Console(record=True, width=140, force_terminal=False)
    render_report(report, console=console, verbose=True)
    text = console.export_text()
    assert "Estate inventory" in text       # agent tree
    assert "Severity distribution" in text  # chart
    assert "security posture" in text       # posture panel


def test_html_report_sections(report):
    from agentbom_amd.output.html_fmt import to_html

    h = to_html(report)
    assert "<svg" in h and 'aria-label="severity distribution"' in h
    assert "estate score" in h and 'class="grade"' in h
    assert "Estate inventory" in h
    for agent in report.agents[:2]:
        assert agent.name in h
