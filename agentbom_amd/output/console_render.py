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

    if report.warnings:
        console.print(Panel("\n".join(report.warnings[:10]), title="warnings", style="yellow"))
