"""Scanner registry + executor with declared failure modes.

Reference parity: src/agent_bom/scanners/{registry,executor}.py — every
side scanner registers a capability descriptor with a DECLARED failure
mode, and the executor applies that mode uniformly instead of each call
site inventing its own try/except policy:

- ``fail_closed``   — a crash marks the scan PARTIAL with an error issue
  (``affects_coverage=True``): missing evidence must never read as clean;
- ``warn_continue`` — a crash records a warning issue, coverage intact;
- ``skip``          — an unavailable scanner records a skipped scope only.

The executor also stamps a :class:`ScanScope` per driver run (complete /
partial / unavailable / skipped) so the report's coverage contract shows
exactly what ran.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Callable, Optional

from agentbom_amd.models.report import (
    AIBOMReport,
    ScanIssue,
    ScanScope,
    ScanScopeStatus,
)

FAIL_CLOSED = "fail_closed"
WARN_CONTINUE = "warn_continue"
SKIP = "skip"

_MODES = (FAIL_CLOSED, WARN_CONTINUE, SKIP)


@dataclass
class ScannerRegistration:
    """One registered side scanner."""

    name: str
    capability: str  # secrets | models | iac | cloud_cis | code | endpoint | image
    run: Callable[..., Any]  # (report, target) -> item_count
    failure_mode: str = WARN_CONTINUE
    requires_target: bool = True

    def __post_init__(self) -> None:
        if self.failure_mode not in _MODES:
            raise ValueError(f"unknown failure mode {self.failure_mode!r}")


_REGISTRY: dict[str, ScannerRegistration] = {}


def register_scanner(reg: ScannerRegistration) -> None:
    _REGISTRY[reg.name] = reg


def get_scanner(name: str) -> Optional[ScannerRegistration]:
    _ensure_builtins()
    return _REGISTRY.get(name)


def list_scanners() -> list[ScannerRegistration]:
    _ensure_builtins()
    return sorted(_REGISTRY.values(), key=lambda r: r.name)


def run_scanner_driver(name: str, report: AIBOMReport,
                       target: Optional[str] = None) -> ScanScope:
    """Run one registered scanner, applying its declared failure mode.

    The returned scope is also appended to ``report.scan_run``; fail-closed
    crashes additionally append an error issue with ``affects_coverage``
    so the outcome derives to PARTIAL (never silently clean).
    """
    reg = get_scanner(name)
    if reg is None:
        scope = ScanScope(name=name, status=ScanScopeStatus.UNSUPPORTED,
                          message="no such scanner registered")
        _attach(report, scope)
        return scope
    if reg.requires_target and not target:
        scope = ScanScope(name=name, status=ScanScopeStatus.SKIPPED,
                          requested=False, message="no target supplied")
        _attach(report, scope)
        return scope

    t0 = time.perf_counter()
    try:
        count = reg.run(report, target)
        scope = ScanScope(
            name=name, status=ScanScopeStatus.COMPLETE,
            item_count=int(count or 0),
            message=f"completed in {(time.perf_counter() - t0) * 1000:.0f} ms")
        _attach(report, scope)
        return scope
    except Exception as exc:  # noqa: BLE001 — the executor IS the boundary
        detail = f"{type(exc).__name__}: {exc}"
        if reg.failure_mode == SKIP:
            scope = ScanScope(name=name, status=ScanScopeStatus.UNAVAILABLE,
                              message=detail)
            _attach(report, scope)
            return scope
        fail_closed = reg.failure_mode == FAIL_CLOSED
        issue = ScanIssue(
            code=f"{reg.capability}_scanner_failed", stage=name,
            source=f"scanner:{name}", message=detail,
            severity="error" if fail_closed else "warning",
            affects_coverage=fail_closed)
        # fail-closed -> PARTIAL scope degrades the outcome (gates exit 1);
        # warn-continue -> SKIPPED scope + warning issue: recorded honestly
        # but the scan outcome (and the exit gate) stay intact.
        scope = ScanScope(
            name=name,
            status=ScanScopeStatus.PARTIAL if fail_closed
            else ScanScopeStatus.SKIPPED,
            message=detail)
        _attach(report, scope, issue)
        return scope


def _attach(report: AIBOMReport, scope: ScanScope,
            issue: Optional[ScanIssue] = None) -> None:
    from agentbom_amd.models.report import ScanRun

    run = report.scan_run or ScanRun()
    issues = list(run.issues) + ([issue] if issue else [])
    report.scan_run = ScanRun(outcome=run.outcome, issues=issues,
                              scopes=list(run.scopes) + [scope])


# ── builtin registrations ───────────────────────────────────────────────────


def _run_secrets(report: AIBOMReport, target: str) -> int:
    from agentbom_amd.scan.secrets import scan_paths, secret_hit_to_finding

    hits = scan_paths(target)
    report.findings.extend(secret_hit_to_finding(h) for h in hits)
    report.ai_inventory_data = {
        "secrets": {"findings": [h.to_dict() for h in hits]}}
    return len(hits)


def _run_models(report: AIBOMReport, target: str) -> int:
    from agentbom_amd.scan.model_scan import model_result_to_finding, scan_model_tree

    results = scan_model_tree(target)
    report.extra_data["model_files"] = [r.to_dict() for r in results]
    report.findings.extend(
        f for f in (model_result_to_finding(r) for r in results) if f)
    return len(results)


def _run_iac(report: AIBOMReport, target: str) -> int:
    from agentbom_amd.scan.iac import iac_finding_to_finding, scan_iac_tree

    hits = scan_iac_tree(target)
    report.iac_findings_data = {"findings": [h.to_dict() for h in hits]}
    report.findings.extend(iac_finding_to_finding(h) for h in hits)
    return len(hits)


def _run_cloud_cis(report: AIBOMReport, target: str) -> int:
    from agentbom_amd.scan.cloud import cis_result_to_finding, scan_cloud_inventory

    cis = scan_cloud_inventory(target)
    report.extra_data["cis_benchmark_data"] = [r.to_dict() for r in cis]
    report.findings.extend(
        f for f in (cis_result_to_finding(r) for r in cis) if f)
    return len(cis)


def _run_cloud_estate(report: AIBOMReport, target: str) -> int:
    """target = 'provider:path' (azure/gcp/snowflake/databricks/aws) — the
    multi-cloud inventory evaluator incl. IAM / audit-trail / DSPM sections."""
    from agentbom_amd.scan.cloud import cis_result_to_finding
    from agentbom_amd.scan.cloud_estate import scan_cloud_estate

    provider, _, path = target.partition(":")
    if not path:
        provider, path = "aws", provider
    cis = scan_cloud_estate(path, provider)
    rows = report.extra_data.setdefault("cis_benchmark_data", [])
    rows.extend(r.to_dict() for r in cis)
    report.findings.extend(
        f for f in (cis_result_to_finding(r, provider) for r in cis) if f)
    return len(cis)


def _run_endpoint(report: AIBOMReport, target: Optional[str]) -> int:
    from agentbom_amd.scan.endpoint import collect_endpoint_inventory

    inv = collect_endpoint_inventory()
    report.extra_data["endpoint_inventory_data"] = inv.to_dict()
    return len(inv.processes)


def _run_notebooks(report: AIBOMReport, target: str) -> int:
    from agentbom_amd.scan.notebooks import scan_notebook_tree

    results = scan_notebook_tree(target)
    report.extra_data["notebooks"] = [r.to_dict() for r in results]
    return sum(r.cells_scanned for r in results)


def _run_skills(report: AIBOMReport, target: str) -> int:
    from agentbom_amd.scan.skills import scan_skills_tree

    bundles = scan_skills_tree(target)
    report.extra_data["skills"] = [b.to_dict() for b in bundles]
    return len(bundles)


def _run_floating_refs(report: AIBOMReport, target: str) -> int:
    from agentbom_amd.scan.floating_refs import scan_floating_refs

    refs = scan_floating_refs(target)
    report.extra_data["floating_refs"] = [r.to_dict() for r in refs]
    return len(refs)


def _run_kspm(report: AIBOMReport, target: str) -> int:
    from agentbom_amd.scan.kspm import kspm_finding_to_finding, scan_cluster_posture

    res = scan_cluster_posture(target)
    report.extra_data["kspm_posture"] = res.to_evidence_dict()
    report.findings.extend(kspm_finding_to_finding(f) for f in res.findings)
    if res.status == "partial":
        report.warnings.append(
            "KSPM posture is PARTIAL: one or more collectors were denied or "
            "failed — absent evidence is never a clean pass")
    return len(res.findings)


def _run_ci_workflows(report: AIBOMReport, target: str) -> int:
    from agentbom_amd.scan.ci_workflows import scan_github_actions

    agents, warnings = scan_github_actions(target)
    report.agents.extend(agents)
    report.warnings.extend(warnings)
    if "github-actions" not in report.scan_sources:
        report.scan_sources.append("github-actions")
    return len(agents)


def _run_repo_inventory(report: AIBOMReport, target: str) -> int:
    from agentbom_amd.scan.repo_inventory import collect_project_inventory

    inv = collect_project_inventory(target)
    if inv is None:
        return 0
    report.project_inventory_data = inv
    return len(inv["files"])


def _ensure_builtins() -> None:
    if "secrets" in _REGISTRY:
        return
    register_scanner(ScannerRegistration(
        "secrets", "secrets", _run_secrets, failure_mode=WARN_CONTINUE))
    register_scanner(ScannerRegistration(
        "model_files", "models", _run_models, failure_mode=FAIL_CLOSED))
    register_scanner(ScannerRegistration(
        "iac", "iac", _run_iac, failure_mode=WARN_CONTINUE))
    register_scanner(ScannerRegistration(
        "cloud_cis", "cloud_cis", _run_cloud_cis, failure_mode=FAIL_CLOSED))
    register_scanner(ScannerRegistration(
        "cloud_estate", "cloud_cis", _run_cloud_estate, failure_mode=FAIL_CLOSED))
    register_scanner(ScannerRegistration(
        "endpoint", "endpoint", _run_endpoint, failure_mode=SKIP,
        requires_target=False))
    register_scanner(ScannerRegistration(
        "notebooks", "notebooks", _run_notebooks, failure_mode=WARN_CONTINUE))
    register_scanner(ScannerRegistration(
        "skills", "skills", _run_skills, failure_mode=WARN_CONTINUE))
    register_scanner(ScannerRegistration(
        "floating_refs", "pinning", _run_floating_refs,
        failure_mode=WARN_CONTINUE))
    register_scanner(ScannerRegistration(
        "ci_workflows", "code", _run_ci_workflows,
        failure_mode=WARN_CONTINUE))
    register_scanner(ScannerRegistration(
        "kspm", "cloud_cis", _run_kspm, failure_mode=FAIL_CLOSED))
    register_scanner(ScannerRegistration(
        "repo_inventory", "code", _run_repo_inventory,
        failure_mode=WARN_CONTINUE))
