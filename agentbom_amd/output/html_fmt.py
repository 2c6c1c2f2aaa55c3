"""Self-contained HTML report (no external assets).

Reference: src/agent_bom/output/html*.py — a standalone single-file report
with summary cards, findings table, and per-finding blast-radius details.
"""

from __future__ import annotations

import html
from typing import Any

from agentbom_amd.models import AIBOMReport

_CSS = """
body { font-family: -apple-system, 'Segoe UI', Roboto, sans-serif; margin: 2rem;
       background: #0d1117; color: #e6edf3; }
h1, h2 { color: #f0f6fc; }
.cards { display: flex; gap: 1rem; flex-wrap: wrap; margin: 1rem 0; }
.card { background: #161b22; border: 1px solid #30363d; border-radius: 8px;
        padding: 1rem 1.5rem; min-width: 8rem; }
.card .num { font-size: 1.8rem; font-weight: 700; }
.card.crit .num { color: #f85149; } .card.high .num { color: #ff8d67; }
.card.med .num { color: #d29922; } .card.ok .num { color: #3fb950; }
table { border-collapse: collapse; width: 100%; margin: 1rem 0; }
th, td { border: 1px solid #30363d; padding: 0.4rem 0.7rem; text-align: left;
         font-size: 0.9rem; }
th { background: #161b22; }
.sev-critical { color: #f85149; font-weight: 700; }
.sev-high { color: #ff8d67; } .sev-medium { color: #d29922; } .sev-low { color: #79c0ff; }
.badge { display: inline-block; padding: 0 0.4rem; border-radius: 6px;
         font-size: 0.75rem; font-weight: 700; }
.badge.kev { background: #8957e5; color: white; }
.badge.mal { background: #f85149; color: white; }
details { margin: 0.5rem 0; background: #161b22; border: 1px solid #30363d;
          border-radius: 8px; padding: 0.5rem 1rem; }
summary { cursor: pointer; }
code { background: #21262d; padding: 0.1rem 0.3rem; border-radius: 4px; }
.posture { background: #161b22; border-radius: 8px; padding: 0.8rem 1.2rem;
           margin: 1rem 0; display: flex; align-items: center; gap: 1rem; }
.posture .grade { font-size: 2.2rem; font-weight: 800; }
.posture ul { margin: 0; font-size: 0.85rem; color: #8b949e; }
svg text { fill: #e6edf3; }
"""


def _e(x: Any) -> str:
    return html.escape(str(x if x is not None else ""))


def to_html(report: AIBOMReport) -> str:
    counts = report.severity_counts()
    rows = []
    details = []
    for br in report.blast_radii:
        v = br.vulnerability
        badges = ""
        if v.is_kev:
            badges += ' <span class="badge kev">KEV</span>'
        if br.package.is_malicious:
            badges += ' <span class="badge mal">MALICIOUS</span>'
        rows.append(
            f"<tr><td>{br.risk_score:.1f}</td>"
            f'<td class="sev-{_e(v.severity.value)}">{_e(v.severity.value)}{badges}</td>'
            f"<td><code>{_e(v.id)}</code></td>"
            f"<td>{_e(br.package.name)}@{_e(br.package.version)}</td>"
            f"<td>{_e(br.reachability)}</td>"
            f"<td>{len(br.affected_agents)}</td><td>{len(br.exposed_credentials)}</td>"
            f"<td>{len(br.exposed_tools)}</td>"
            f"<td>{_e(v.fixed_version or '-')}</td></tr>"
        )
        details.append(
            f"<details><summary><code>{_e(v.id)}</code> — {_e(br.package.name)} "
            f"(risk {br.risk_score:.1f})</summary>"
            f"<p>{_e(v.summary)}</p>"
            f"<p><b>Attack vector:</b> {_e(br.attack_vector_summary or '')}</p>"
            f"<p><b>Affected agents:</b> {_e(', '.join(a.name for a in br.affected_agents))}</p>"
            f"<p><b>Affected servers:</b> {_e(', '.join(s.name for s in br.affected_servers))}</p>"
            f"<p><b>Exposed credentials:</b> {_e(', '.join(br.exposed_credentials) or 'none')}</p>"
            f"<p><b>Reachable tools:</b> {_e(', '.join(t.name for t in br.exposed_tools) or 'none')}</p>"
            "</details>"
        )

    return f"""<!DOCTYPE html>
<html><head><meta charset="utf-8"><title>agent-bom scan report</title>
<style>{_CSS}</style></head><body>
<h1>AI-BOM Scan Report</h1>
<p>generated {_e(report.generated_at.isoformat())} · scan {_e(report.scan_id or 'local')}</p>
<div class="cards">
<div class="card"><div class="num">{report.total_agents}</div>agents</div>
<div class="card"><div class="num">{report.total_servers}</div>MCP servers</div>
<div class="card"><div class="num">{report.total_packages}</div>packages</div>
<div class="card crit"><div class="num">{counts['critical']}</div>critical</div>
<div class="card high"><div class="num">{counts['high']}</div>high</div>
<div class="card med"><div class="num">{counts['medium']}</div>medium</div>
<div class="card ok"><div class="num">{counts['low']}</div>low</div>
</div>
{_posture_html(report)}
{_severity_chart_svg(counts)}
<h2>Findings</h2>
<table><tr><th>risk</th><th>severity</th><th>vulnerability</th><th>package</th>
<th>reach</th><th>agents</th><th>creds</th><th>tools</th><th>fix</th></tr>
{''.join(rows)}
</table>
<h2>Blast radius</h2>
{''.join(details)}
{_remediation_html(report)}
{_frameworks_html(report)}
{_exposure_paths_html(report)}
{_inventory_tree_html(report)}
{_other_findings_html(report)}
{_warnings_html(report)}
</body></html>"""


def _posture_html(report: AIBOMReport) -> str:
    """Estate grade banner (scan/risk.estate_exec_score)."""
    from agentbom_amd.scan.risk import estate_exec_score

    score = estate_exec_score(report)
    grade = str(score.get("grade", "?"))
    color = {"A": "#2e7d32", "B": "#558b2f", "C": "#f9a825",
             "D": "#e65100", "F": "#c62828"}.get(grade[:1], "#666")
    drivers = "".join(f"<li>{_e(d)}</li>" for d in (score.get("drivers") or [])[:5])
    return (f'<div class="posture" style="border-left:6px solid {color}">'
            f'<span class="grade" style="color:{color}">{_e(grade)}</span>'
            f'<span>estate score {score.get("score", "?")}/100</span>'
            f"<ul>{drivers}</ul></div>")


def _severity_chart_svg(counts: dict) -> str:
    """Inline SVG horizontal severity bars — no external assets."""
    sev_colors = [("critical", "#c62828"), ("high", "#e65100"),
                  ("medium", "#f9a825"), ("low", "#2e7d32")]
    total = max(sum(counts.get(s, 0) for s, _ in sev_colors), 1)
    bars = []
    for i, (sev, color) in enumerate(sev_colors):
        n = counts.get(sev, 0)
        w = max(round(360 * n / total), 2 if n else 0)
        y = 8 + i * 26
        bars.append(
            f'<text x="0" y="{y + 13}" font-size="12">{sev}</text>'
            f'<rect x="70" y="{y}" width="{w}" height="18" fill="{color}"/>'
            f'<text x="{76 + w}" y="{y + 13}" font-size="12">{n}</text>')
    return (f'<svg width="480" height="118" role="img" '
            f'aria-label="severity distribution">{"".join(bars)}</svg>')


def _inventory_tree_html(report: AIBOMReport) -> str:
    """Nested inventory: agent → server → creds/tools/vulnerable pkgs."""
    if not report.agents:
        return ""
    parts = ["<h2>Estate inventory</h2>"]
    for agent in report.agents[:200]:
        servers = []
        for srv in agent.mcp_servers:
            vulns = srv.total_vulnerabilities
            badge = (f' <span class="badge kev">{vulns} vulns</span>'
                     if vulns else "")
            inner = []
            if srv.credential_names:
                inner.append("<p><b>credentials:</b> "
                             + _e(", ".join(srv.credential_names[:8])) + "</p>")
            if srv.tools:
                inner.append("<p><b>tools:</b> "
                             + _e(", ".join(t.name for t in srv.tools[:8]))
                             + "</p>")
            vp = srv.vulnerable_packages
            if vp:
                inner.append("<p><b>vulnerable:</b> " + _e(", ".join(
                    f"{p.name}@{p.version}" for p in vp[:10])) + "</p>")
            servers.append(
                f"<details><summary><code>{_e(srv.name)}</code> "
                f"({_e(srv.transport.value)}, {len(srv.packages)} pkgs)"
                f"{badge}</summary>{''.join(inner)}</details>")
        parts.append(
            f"<details><summary><b>{_e(agent.name)}</b> "
            f"({_e(agent.agent_type.value)}, {len(agent.mcp_servers)} "
            f"servers)</summary>{''.join(servers)}</details>")
    return "".join(parts)


def _remediation_html(report: AIBOMReport) -> str:
    from agentbom_amd.output.json_fmt import _build_remediation_json

    plan = _build_remediation_json(report)
    if not plan:
        return ""
    rows = "".join(
        f"<tr><td>{i}</td><td><code>{_e(p.get('package'))}</code></td>"
        f"<td>{_e(p.get('fix_version') or 'no fix yet')}</td>"
        f"<td>{len(p.get('vulns', []))}</td>"
        f"<td>{_e(p.get('action', ''))}</td>"
        f"<td>{p.get('max_risk_score', 0):.1f}</td></tr>"
        for i, p in enumerate(plan[:20], 1))
    return ("<h2>Remediation plan</h2><table><tr><th>#</th><th>package</th>"
            "<th>upgrade to</th><th>vulns</th><th>action</th><th>risk</th></tr>"
            f"{rows}</table>")


def _frameworks_html(report: AIBOMReport) -> str:
    from agentbom_amd.output.json_fmt import _build_framework_summary

    summary = _build_framework_summary(report.blast_radii)
    rows = []
    for fw, data in sorted(summary.items()):
        if not isinstance(data, dict) or not data.get("tagged_findings"):
            continue
        controls = data.get("controls") or {}
        top = sorted(controls.items(), key=lambda kv: -kv[1])[:4] \
            if isinstance(controls, dict) else []
        rows.append(f"<tr><td>{_e(fw)}</td><td>{data['tagged_findings']}</td>"
                    f"<td>{_e(', '.join(f'{c} ({n})' for c, n in top))}</td></tr>")
    if not rows:
        return ""
    return ("<h2>Compliance posture</h2><table><tr><th>framework</th>"
            "<th>tagged findings</th><th>top controls</th></tr>"
            + "".join(rows) + "</table>")


def _exposure_paths_html(report: AIBOMReport) -> str:
    from agentbom_amd.models import active_blast_radii, blast_radius_to_finding
    from agentbom_amd.output.exposure_path import exposure_path_for_finding

    active = active_blast_radii(report.blast_radii)[:10]
    if not active:
        return ""
    items = []
    for i, br in enumerate(active, 1):
        path = exposure_path_for_finding(blast_radius_to_finding(br), rank=i)
        hops = path.get("hops") or []
        label = " → ".join(_e(h.get("label", h)) if isinstance(h, dict) else _e(h)
                           for h in hops) or _e(path.get("label", ""))
        fix = path.get("fix")
        items.append(f"<li><b>{path.get('riskScore', br.risk_score):.1f}</b> {label}"
                     + (f" — <i>fix: {_e(fix)}</i>" if fix else "") + "</li>")
    return "<h2>Top exposure paths</h2><ol>" + "".join(items) + "</ol>"


def _other_findings_html(report: AIBOMReport) -> str:
    others = [f for f in (report.findings or [])
              if f.finding_type.value != "CVE"]
    if not others:
        return ""
    by_type: dict[str, list] = {}
    for f in others:
        by_type.setdefault(f.finding_type.value, []).append(f)
    rows = "".join(
        f"<tr><td>{_e(t)}</td><td>{len(items)}</td>"
        f"<td>{_e(items[0].title[:80])}</td></tr>"
        for t, items in sorted(by_type.items()))
    return ("<h2>Other findings</h2><table><tr><th>type</th><th>count</th>"
            f"<th>example</th></tr>{rows}</table>")


def _warnings_html(report: AIBOMReport) -> str:
    if not report.warnings:
        return ""
    items = "".join(f"<li>{_e(w)}</li>" for w in report.warnings[:15])
    return f"<h2>Warnings</h2><ul>{items}</ul>"
