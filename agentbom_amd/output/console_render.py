"""Rich console report incl. the blast-radius tree.

Reference: src/agent_bom/output/console_render.py (2,203 LoC) — summary
panel, per-severity findings table, blast-radius tree (vuln -> package ->
server -> agent -> credential/tool), remediation plan.
"""

from __future__ import annotations

from typing import Optional

from rich.console import Console
from rich.panel import Panel
from rich.table import Table
from rich.tree import Tree

from agentbom_amd.models import AIBOMReport, Severity, active_blast_radii

_SEV_STYLE = {
    "critical": "bold red",
    "high": "red",
    "medium": "yellow",
    "low": "cyan",
    "none": "green",
    "unknown": "dim",
}


def render_report(report: AIBOMReport, console: Optional[Console] = None,
                  verbose: bool = False, max_blast: int = 10) -> None:
    console = console or Console()
    counts = report.severity_counts()

    console.print(
        Panel.fit(
            f"[bold]AI-BOM Scan Report[/bold]\n"
            f"agents: {report.total_agents} · MCP servers: {report.total_servers} · "
            f"packages: {report.total_packages} · vulnerabilities: {report.total_vulnerabilities}\n"
            f"[bold red]critical: {counts['critical']}[/bold red] · [red]high: {counts['high']}[/red] · "
            f"[yellow]medium: {counts['medium']}[/yellow] · low: {counts['low']}",
            title="agent-bom",
        )
    )

    active = active_blast_radii(report.blast_radii)
    if not active:
        console.print("[green]No vulnerabilities found.[/green]")
        return

    table = Table(title="Findings (by risk)", show_lines=False)
    table.add_column("risk", justify="right")
    table.add_column("severity")
    table.add_column("vulnerability")
    table.add_column("package")
    table.add_column("reach")
    table.add_column("agents", justify="right")
    table.add_column("creds", justify="right")
    table.add_column("tools", justify="right")
    table.add_column("fix")
    for br in active:
        sev = br.vulnerability.severity.value
        style = _SEV_STYLE.get(sev, "")
        kev = " [bold magenta]KEV[/bold magenta]" if br.vulnerability.is_kev else ""
        mal = " [bold red]MALICIOUS[/bold red]" if br.package.is_malicious else ""
        table.add_row(
            f"{br.risk_score:.1f}",
            f"[{style}]{sev}[/{style}]{kev}{mal}",
            br.vulnerability.id,
            f"{br.package.name}@{br.package.version}",
            br.reachability,
            str(len(br.affected_agents)),
            str(len(br.exposed_credentials)),
            str(len(br.exposed_tools)),
            br.vulnerability.fixed_version or "-",
        )
    console.print(table)

    # blast-radius tree for the top findings
    for br in active[:max_blast if not verbose else len(active)]:
        sev = br.vulnerability.severity.value
        style = _SEV_STYLE.get(sev, "")
        root = Tree(
            f"[{style}]{br.vulnerability.id}[/{style}] "
            f"(risk {br.risk_score:.1f}, {br.impact_category}) — {br.vulnerability.summary[:100]}"
        )
        pkg_node = root.add(f"package [bold]{br.package.name}@{br.package.version}[/bold] ({br.package.ecosystem})")
        for srv in br.affected_servers:
            s_node = pkg_node.add(f"server [cyan]{srv.name}[/cyan]")
            for agent in br.affected_agents:
                if srv in agent.mcp_servers:
                    s_node.add(f"agent [magenta]{agent.name}[/magenta] ({agent.agent_type.value})")
        if br.exposed_credentials:
            c_node = root.add(f"[red]exposed credentials ({len(br.exposed_credentials)})[/red]")
            for c in br.exposed_credentials[:8]:
                c_node.add(c)
        if br.exposed_tools:
            t_node = root.add(f"[yellow]reachable tools ({len(br.exposed_tools)})[/yellow]")
            for t in br.exposed_tools[:8]:
                t_node.add(t.name)
        if br.delegation_chain:
            d_node = root.add(f"delegation chains (hop depth {br.hop_depth})")
            for chain in br.delegation_chain[:5]:
                d_node.add(chain)
        console.print(root)

    _render_remediation(report, console)
    _render_framework_summary(report, console)
    _render_exposure_paths(report, console, max_paths=5 if not verbose else 25)
    _render_other_findings(report, console, verbose)
    _render_scan_performance(report, console, verbose)
    if verbose:
        print_severity_chart(report, console)
        print_agent_tree(report, console)
        print_cis_findings(report, console)
        print_posture_summary(report, console)

    if report.warnings:
        console.print(Panel("\n".join(report.warnings[:10]), title="warnings", style="yellow"))
    gaps = [i for i in report.scan_run.issues
            if getattr(i, "kind", "") == "coverage_gap"] if report.scan_run else []
    if gaps:
        console.print(Panel(
            "\n".join(f"{g.scanner}: {g.message}" for g in gaps[:10]),
            title="coverage gaps — PARTIAL evidence never renders as clean",
            style="red"))


def _render_remediation(report: AIBOMReport, console: Console) -> None:
    """Prioritized remediation plan: one row per (package, fix), findings
    grouped — the operator's 'what do I upgrade first' view."""
    from agentbom_amd.output.json_fmt import _build_remediation_json

    plan = _build_remediation_json(report)
    if not plan:
        return
    table = Table(title="Remediation plan (highest risk first)")
    table.add_column("#", justify="right")
    table.add_column("package")
    table.add_column("upgrade to")
    table.add_column("vulns", justify="right")
    table.add_column("action")
    table.add_column("risk", justify="right")
    for i, item in enumerate(plan[:15], 1):
        risk = item.get("max_risk_score", 0)
        style = ("bold red" if risk >= 8 else "red" if risk >= 6
                 else "yellow" if risk >= 4 else "")
        table.add_row(
            str(i), item.get("package", "?"),
            item.get("fix_version") or "[dim]no fix yet[/dim]",
            str(len(item.get("vulns", []))),
            f"[{style}]{item.get('action', '')[:40]}[/{style}]" if style
            else item.get("action", "")[:40],
            f"{risk:.1f}",
        )
    console.print(table)


def _render_framework_summary(report: AIBOMReport, console: Console) -> None:
    from agentbom_amd.output.json_fmt import _build_framework_summary

    summary = _build_framework_summary(report.blast_radii)
    rows = [(fw, data) for fw, data in sorted(summary.items())
            if isinstance(data, dict) and data.get("tagged_findings")]
    if not rows:
        return
    table = Table(title="Compliance posture (16 frameworks)")
    table.add_column("framework")
    table.add_column("tagged findings", justify="right")
    table.add_column("top controls")
    for fw, data in rows[:16]:
        controls = data.get("controls") or {}
        top = sorted(controls.items(), key=lambda kv: -kv[1])[:4] \
            if isinstance(controls, dict) else [(c, "") for c in list(controls)[:4]]
        table.add_row(fw, str(data.get("tagged_findings", 0)),
                      ", ".join(f"{c} ({n})" if n != "" else str(c)
                                for c, n in top))
    console.print(table)


def _render_exposure_paths(report: AIBOMReport, console: Console,
                           max_paths: int = 5) -> None:
    """Ranked exposure paths: entry -> hops -> crown jewel, with fix."""
    from agentbom_amd.models import blast_radius_to_finding
    from agentbom_amd.output.exposure_path import exposure_path_for_finding

    active = active_blast_radii(report.blast_radii)
    if not active:
        return
    tree = Tree("[bold]Top exposure paths[/bold]")
    for i, br in enumerate(active[:max_paths], 1):
        path = exposure_path_for_finding(blast_radius_to_finding(br), rank=i)
        hops = path.get("hops") or []
        label = " [dim]->[/dim] ".join(str(h.get("label", h)) if isinstance(h, dict)
                                       else str(h) for h in hops) or path.get("label", "")
        node = tree.add(f"#{i} [bold]{path.get('riskScore', br.risk_score):.1f}[/bold] {label}")
        fix = path.get("fix")
        if fix:
            node.add(f"[green]fix:[/green] {fix}")
    console.print(tree)


def _render_other_findings(report: AIBOMReport, console: Console,
                           verbose: bool) -> None:
    """Non-CVE findings stream: secrets, IaC, prompts, toxic combos, ..."""
    others = [f for f in (report.findings or [])
              if f.finding_type.value not in ("CVE",)]
    if not others:
        return
    by_type: dict[str, list] = {}
    for f in others:
        by_type.setdefault(f.finding_type.value, []).append(f)
    table = Table(title="Other findings")
    table.add_column("type")
    table.add_column("count", justify="right")
    table.add_column("worst", justify="center")
    table.add_column("example")
    order = {"critical": 0, "high": 1, "medium": 2, "low": 3}
    for ftype, items in sorted(by_type.items()):
        worst = min((f.severity for f in items), key=lambda s: order.get(s, 9))
        style = _SEV_STYLE.get(worst, "")
        example = items[0].title if items else ""
        table.add_row(ftype, str(len(items)), f"[{style}]{worst}[/{style}]",
                      example[:60])
    console.print(table)


def _render_scan_performance(report: AIBOMReport, console: Console,
                             verbose: bool) -> None:
    perf = report.scan_performance_data or {}
    if not perf or not verbose:
        return
    parts = [f"{k.removesuffix('_ms')} {v:.1f}ms" for k, v in perf.items()
             if isinstance(v, (int, float))]
    if parts:
        console.print(Panel(" · ".join(parts), title="scan performance",
                            style="dim"))


# ── standalone printers (reference console_render.py public surface) ───────


def print_agent_tree(report: AIBOMReport,
                     console: Optional[Console] = None) -> None:
    """Full inventory tree: agent → server → (credentials, tools, packages),
    vulnerable packages highlighted (reference print_agent_tree, :540)."""
    console = console or Console()
    root = Tree(f"[bold]Estate inventory[/bold] — {report.total_agents} "
                f"agents, {report.total_servers} servers")
    for agent in report.agents:
        a_node = root.add(
            f"[magenta]{agent.name}[/magenta] ({agent.agent_type.value})"
            + (f" [dim]{agent.config_path}[/dim]" if agent.config_path else ""))
        for srv in agent.mcp_servers:
            vuln_ct = srv.total_vulnerabilities
            badge = f" [red]{vuln_ct} vulns[/red]" if vuln_ct else ""
            s_node = a_node.add(f"[cyan]{srv.name}[/cyan] "
                                f"({srv.transport.value}){badge}")
            if srv.credential_names:
                c = s_node.add(f"[red]credentials "
                               f"({len(srv.credential_names)})[/red]")
                for name in srv.credential_names[:6]:
                    c.add(name)
            if srv.tools:
                t = s_node.add(f"tools ({len(srv.tools)})")
                for tool in srv.tools[:6]:
                    t.add(tool.name)
            vuln_pkgs = srv.vulnerable_packages
            shown = vuln_pkgs[:8] if vuln_pkgs else srv.packages[:5]
            if srv.packages:
                p = s_node.add(f"packages ({len(srv.packages)})")
                for pkg in shown:
                    sevs = {v.severity.value for v in pkg.vulnerabilities}
                    worst = next((s for s in ("critical", "high", "medium",
                                              "low") if s in sevs), None)
                    style = _SEV_STYLE.get(worst or "", "")
                    mark = f" [{style}]{worst}[/{style}]" if worst else ""
                    p.add(f"{pkg.name}@{pkg.version}{mark}")
    console.print(root)


def print_severity_chart(report: AIBOMReport,
                         console: Optional[Console] = None,
                         width: int = 40) -> None:
    """Horizontal severity bar chart (reference print_severity_chart, :2168)."""
    console = console or Console()
    counts = report.severity_counts()
    total = max(sum(counts.get(s, 0) for s in
                    ("critical", "high", "medium", "low")), 1)
    table = Table(title="Severity distribution", show_header=False,
                  show_lines=False, box=None, padding=(0, 1))
    table.add_column("sev", width=9)
    table.add_column("bar")
    table.add_column("n", justify="right")
    for sev in ("critical", "high", "medium", "low"):
        n = counts.get(sev, 0)
        bar = "█" * max(round(width * n / total), 1 if n else 0)
        style = _SEV_STYLE.get(sev, "")
        table.add_row(f"[{style}]{sev}[/{style}]",
                      f"[{style}]{bar}[/{style}]", str(n))
    console.print(table)


def print_cis_findings(report: AIBOMReport,
                       console: Optional[Console] = None,
                       show_passed: bool = False) -> None:
    """Cloud CIS benchmark results table (reference print_cis_findings,
    :1691): failures first, per-check evidence + remediation line."""
    console = console or Console()
    rows = report.extra_data.get("cis_benchmark_data") or []
    if not rows:
        return
    order = {"fail": 0, "warn": 1, "manual": 2, "pass": 3}
    rows = sorted((r for r in rows if isinstance(r, dict)),
                  key=lambda r: (order.get(str(r.get("status")), 9),
                                 str(r.get("check_id"))))
    failed = [r for r in rows if r.get("status") == "fail"]
    table = Table(title=f"Cloud CIS benchmark — {len(failed)} failing "
                        f"of {len(rows)} checks")
    table.add_column("check")
    table.add_column("status")
    table.add_column("title")
    table.add_column("evidence")
    for r in rows:
        status = str(r.get("status", "?"))
        if status == "pass" and not show_passed:
            continue
        style = {"fail": "bold red", "warn": "yellow",
                 "manual": "dim", "pass": "green"}.get(status, "")
        table.add_row(str(r.get("check_id", "?")),
                      f"[{style}]{status}[/{style}]",
                      str(r.get("title", ""))[:60],
                      str(r.get("evidence", r.get("detail", "")))[:50])
    console.print(table)


def print_diff(diff: dict, console: Optional[Console] = None,
               quiet: bool = False) -> None:
    """Render a scan-over-scan diff (scan/history.diff_reports output):
    new findings red, resolved green, package inventory delta."""
    console = console or Console()
    new = diff.get("new_findings") or []
    resolved = diff.get("resolved_findings") or []
    if new:
        table = Table(title=f"[red]New findings ({len(new)})[/red]")
        table.add_column("vulnerability")
        table.add_column("package")
        table.add_column("severity")
        table.add_column("risk", justify="right")
        for row in new:
            sev = str(row.get("severity", "?"))
            style = _SEV_STYLE.get(sev, "")
            table.add_row(str(row.get("vulnerability_id", "?")),
                          str(row.get("package_name",
                                      row.get("package", "?"))),
                          f"[{style}]{sev}[/{style}]",
                          f"{float(row.get('risk_score', 0) or 0):.1f}")
        console.print(table)
    if resolved and not quiet:
        table = Table(title=f"[green]Resolved ({len(resolved)})[/green]")
        table.add_column("vulnerability")
        table.add_column("package")
        for row in resolved:
            table.add_row(str(row.get("vulnerability_id", "?")),
                          str(row.get("package", "?")))
        console.print(table)
    added = diff.get("packages_added") or []
    removed = diff.get("packages_removed") or []
    console.print(
        f"unchanged: {diff.get('unchanged_count', 0)} · "
        f"packages [green]+{len(added)}[/green]/[red]-{len(removed)}[/red]"
        + (f" · drift: {', '.join(added[:5])}" if added else ""))


def print_policy_results(policy_result: dict,
                         console: Optional[Console] = None) -> None:
    """Policy gate outcome (reference print_policy_results, :2131)."""
    console = console or Console()
    violations = policy_result.get("violations") or []
    warnings = policy_result.get("warnings") or []
    if not violations and not warnings:
        console.print(f"[green]policy: PASS[/green] "
                      f"({policy_result.get('rules_evaluated', 0)} rules)")
        return
    table = Table(title="Policy gate")
    table.add_column("rule")
    table.add_column("action")
    table.add_column("vulnerability")
    table.add_column("package")
    table.add_column("risk", justify="right")
    for hit in (*violations, *warnings):
        action = str(hit.get("action", "fail"))
        style = "bold red" if action == "fail" else "yellow"
        table.add_row(str(hit.get("rule_id", "?")),
                      f"[{style}]{action}[/{style}]",
                      str(hit.get("vulnerability_id", "")),
                      str(hit.get("package", "")),
                      f"{float(hit.get('risk_score', 0) or 0):.1f}")
    console.print(table)
    verdict = "[bold red]FAIL[/bold red]" if violations else "[yellow]WARN[/yellow]"
    console.print(f"policy: {verdict} — {len(violations)} violations, "
                  f"{len(warnings)} warnings")


def print_posture_summary(report: AIBOMReport,
                          console: Optional[Console] = None) -> None:
    """One-panel estate posture: grade + the dominant risk drivers
    (reference print_posture_summary, :323)."""
    from agentbom_amd.output.json_fmt import _estate_score

    console = console or Console()
    score = _estate_score(report)
    grade = str(score.get("grade", "?"))
    style = {"A": "green", "B": "green", "C": "yellow",
             "D": "red", "F": "bold red"}.get(grade[:1], "")
    drivers = score.get("drivers") or score.get("top_drivers") or []
    lines = [f"estate grade: [{style}]{grade}[/{style}] "
             f"(score {score.get('score', '?')})"]
    lines += [f"• {d}" for d in list(drivers)[:5]]
    counts = report.severity_counts()
    lines.append(f"critical {counts['critical']} · high {counts['high']} · "
                 f"KEV {sum(1 for br in report.blast_radii if br.vulnerability.is_kev)}")
    console.print(Panel("\n".join(lines), title="security posture"))
