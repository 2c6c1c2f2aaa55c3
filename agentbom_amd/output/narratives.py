"""Human-language narratives: advisory text, compliance prose, CIS posture.

Reference parity: src/agent_bom/output/{advisory_text,
compliance_narrative,cis_posture}.py — deterministic prose built from the
same structures the JSON exports use (no model calls): per-finding
advisory paragraphs, an auditor-facing framework narrative, and a CIS
posture summary.
"""

from __future__ import annotations

from collections import Counter
from typing import Any, Optional

from agentbom_amd.models import AIBOMReport


def advisory_text(report: AIBOMReport, limit: int = 25) -> str:
    """One advisory paragraph per finding, risk-ordered."""
    lines = [
        f"agent-bom advisory digest — {len(report.blast_radii)} findings "
        f"across {report.total_packages} packages / {report.total_agents} agents.",
        "",
    ]
    for br in report.blast_radii[:limit]:
        v, p = br.vulnerability, br.package
        sentences = [
            f"{v.id} affects {p.name} {p.version} ({p.ecosystem}), "
            f"severity {v.severity.value}, risk {br.risk_score:.1f}/10."
        ]
        if p.is_malicious:
            sentences.append(
                f"The package itself is KNOWN MALICIOUS ({p.malicious_reason}); "
                "remove it immediately — do not merely upgrade.")
        if v.is_kev:
            sentences.append("It is in the CISA Known Exploited Vulnerabilities "
                             "catalog: exploitation is confirmed in the wild.")
        if v.epss_score is not None:
            sentences.append(f"EPSS puts 30-day exploitation probability at "
                             f"{v.epss_score:.0%}.")
        reach = br.reachability
        if reach == "unreachable":
            sentences.append("Dependency analysis shows no path from any agent "
                             "to this package, which lowers its priority.")
        elif br.affected_agents:
            agents = ", ".join(a.name for a in br.affected_agents[:4])
            sentences.append(f"It is reachable from {len(br.affected_agents)} "
                             f"agent(s) ({agents}"
                             f"{', …' if len(br.affected_agents) > 4 else ''})")
        if br.exposed_credentials:
            sentences.append(
                f"A compromise would expose {len(br.exposed_credentials)} "
                f"credential(s) including {br.exposed_credentials[0]}.")
        if v.fixed_version:
            sentences.append(f"Fix: upgrade to {v.fixed_version}.")
        elif not p.is_malicious:
            sentences.append("No fixed version is published yet; apply "
                             "compensating controls and watch the advisory.")
        lines.append(" ".join(sentences))
        lines.append("")
    if len(report.blast_radii) > limit:
        lines.append(f"… and {len(report.blast_radii) - limit} further "
                     "findings (see the full report).")
    return "\n".join(lines).strip() + "\n"


def compliance_narrative(report: AIBOMReport, framework_field: str = "owasp_tags",
                         framework_label: str = "OWASP LLM Top 10") -> str:
    """Auditor-facing prose for one framework's control coverage."""
    control_rows: Counter = Counter()
    worst: dict[str, float] = {}
    for br in report.blast_radii:
        for tag in getattr(br, framework_field, []) or []:
            control_rows[str(tag)] += 1
            worst[str(tag)] = max(worst.get(str(tag), 0.0), br.risk_score)

    lines = [
        f"{framework_label} evidence narrative",
        "=" * 40,
        "",
        f"Scan {report.scan_id or '(unsaved)'} mapped "
        f"{sum(control_rows.values())} findings onto "
        f"{len(control_rows)} {framework_label} controls.",
        "",
    ]
    if not control_rows:
        lines.append("No findings mapped to this framework: either the estate "
                     "is clean for these control classes or the relevant "
                     "scanners did not run (check scan_run.scopes).")
        return "\n".join(lines) + "\n"
    for control, n in control_rows.most_common():
        lines.append(
            f"- {control}: {n} finding(s), worst residual risk "
            f"{worst[control]:.1f}/10. Evidence rows are carried in the "
            f"signed compliance bundle for this framework.")
    lines += [
        "",
        "Method: controls are stamped during scanning from vulnerability, "
        "capability and reachability signals; every row above is backed by "
        "a concrete finding with a canonical id — nothing is attested "
        "without evidence.",
    ]
    return "\n".join(lines) + "\n"


def cis_posture_text(cis_results: list[dict[str, Any]],
                     provider: str = "aws") -> str:
    """Posture summary for a CIS benchmark run (dict rows from to_dict())."""
    total = len(cis_results)
    failed = [r for r in cis_results if r.get("status") == "fail"]
    errored = [r for r in cis_results if r.get("status") == "error"]
    passed = total - len(failed) - len(errored)
    pct = 100.0 * passed / total if total else 0.0
    lines = [
        f"CIS posture ({provider}): {passed}/{total} checks pass "
        f"({pct:.0f}%).",
        "",
    ]
    by_sev = Counter(r.get("severity", "medium") for r in failed)
    if failed:
        lines.append("Failing checks by severity: " + ", ".join(
            f"{sev} {n}" for sev, n in by_sev.most_common()) + ".")
        lines.append("")
        for r in sorted(failed, key=lambda r: ({"critical": 0, "high": 1,
                                                "medium": 2, "low": 3}
                                               .get(r.get("severity"), 4)))[:15]:
            lines.append(f"- [{r.get('severity')}] {r.get('check_id')}: "
                         f"{r.get('title')} ({r.get('resource')})"
                         + (f" — {r['detail']}" if r.get("detail") else ""))
    else:
        lines.append("No failing checks in the evaluated inventory.")
    if errored:
        lines.append("")
        lines.append(f"{len(errored)} check(s) could not be evaluated — "
                     "treat their controls as UNVERIFIED, not passing.")
    return "\n".join(lines) + "\n"
