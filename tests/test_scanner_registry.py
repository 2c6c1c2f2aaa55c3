"""Scanner registry/executor: declared failure modes + coverage contract."""

import pytest

from agentbom_amd.models.report import ScanOutcome
from agentbom_amd.scan.orchestrator import run_demo_scan
from agentbom_amd.scan.registry import (
    FAIL_CLOSED,
    SKIP,
    WARN_CONTINUE,
    ScannerRegistration,
    list_scanners,
    register_scanner,
    run_scanner_driver,
)


class TestRegistry:
    def test_builtins_registered(self):
        names = {r.name for r in list_scanners()}
        assert {"secrets", "model_files", "iac", "cloud_cis",
                "endpoint"} <= names

    def test_invalid_mode_rejected(self):
        with pytest.raises(ValueError):
            ScannerRegistration("x", "c", lambda r, t: 0, failure_mode="explode")


class TestExecutor:
    def test_complete_scope(self, tmp_path):
        (tmp_path / "main.tf").write_text(
            'resource "aws_s3_bucket" "b" { acl = "public-read" }')
        report = run_demo_scan()
        scope = run_scanner_driver("iac", report, str(tmp_path))
        assert scope.status.value == "complete"
        assert any(s.name == "iac" for s in report.scan_run.scopes)
        assert report.scan_run.outcome is ScanOutcome.COMPLETE

    def test_fail_closed_marks_partial(self):
        register_scanner(ScannerRegistration(
            "boom_closed", "test", lambda r, t: 1 / 0,
            failure_mode=FAIL_CLOSED))
        report = run_demo_scan()
        scope = run_scanner_driver("boom_closed", report, "x")
        assert scope.status.value == "partial"
        assert report.scan_run.outcome is ScanOutcome.PARTIAL
        issue = report.scan_run.issues[-1]
        assert issue.severity == "error" and issue.affects_coverage

    def test_warn_continue_keeps_coverage(self):
        register_scanner(ScannerRegistration(
            "boom_warn", "test", lambda r, t: 1 / 0,
            failure_mode=WARN_CONTINUE))
        report = run_demo_scan()
        run_scanner_driver("boom_warn", report, "x")
        assert report.scan_run.outcome is ScanOutcome.COMPLETE
        assert report.scan_run.issues[-1].severity == "warning"

    def test_skip_records_unavailable_only(self):
        register_scanner(ScannerRegistration(
            "boom_skip", "test", lambda r, t: 1 / 0, failure_mode=SKIP))
        report = run_demo_scan()
        scope = run_scanner_driver("boom_skip", report, "x")
        assert scope.status.value == "unavailable"
        assert not report.scan_run.issues

    def test_missing_target_skipped(self):
        report = run_demo_scan()
        scope = run_scanner_driver("iac", report, None)
        assert scope.status.value == "skipped"

    def test_unknown_scanner_unsupported(self):
        report = run_demo_scan()
        assert run_scanner_driver("nope", report, "x").status.value == "unsupported"

    def test_partial_evidence_fails_exit_code(self):
        """Incomplete evidence must fail closed at the gate (exit 1)."""
        from agentbom_amd.scan.orchestrator import ScanOptions, compute_exit_code

        register_scanner(ScannerRegistration(
            "boom_gate", "test", lambda r, t: 1 / 0, failure_mode=FAIL_CLOSED))
        report = run_demo_scan()
        # strip findings so only the coverage gate can fire
        report.blast_radii = []
        report.findings = []
        assert compute_exit_code(report, ScanOptions(exit_zero=True)) == 0
        run_scanner_driver("boom_gate", report, "x")
        assert compute_exit_code(report, ScanOptions(exit_zero=True)) == 1
