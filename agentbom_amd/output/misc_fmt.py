"""CSV / Markdown / Prometheus / JUnit / plain exports.

Reference: src/agent_bom/output/{csv,markdown,prometheus,junit}.py shapes.
"""

from __future__ import annotations

import csv
import io
from typing import Any

from agentbom_amd.models import AIBOMReport

_CSV_FIELDS = [
    "finding_id", "finding_type", "severity", "risk_score", "vulnerability_id",
    "package", "version", "ecosystem", "fixed_version", "is_kev", "epss_score",
    "cvss_score", "reachability", "impact_category", "is_malicious",
    "affected_agents", "affected_servers", "exposed_credentials", "exposed_tools",
    "asset_name", "asset_type",
]


def to_csv(report: AIBOMReport) -> str:
    buf = io.StringIO()
    w = csv.DictWriter(buf, fieldnames=_CSV_FIELDS)
    w.writeheader()
    for f in report.to_findings():
        ev = f.evidence if isinstance(f.evidence, dict) else {}
        w.writerow(
            {
                "finding_id": f.id,
                "finding_type": f.finding_type.value,
                "severity": f.severity,
                "risk_score": f.risk_score,
                "vulnerability_id": f.vulnerability_id or "",
                "package": ev.get("package_name", ""),
                "version": ev.get("package_version", ""),
                "ecosystem": ev.get("ecosystem", ""),
                "fixed_version": f.fixed_version or "",
                "is_kev": f.is_kev,
                "epss_score": f.epss_score if f.epss_score is not None else "",
                "cvss_score": f.cvss_score if f.cvss_score is not None else "",
                "reachability": f.reachability or "",
                "impact_category": f.impact_category or "",
                "is_malicious": f.is_malicious,
                "affected_agents": ";".join(f.affected_agents),
                "affected_servers": ";".join(f.affected_servers),
                "exposed_credentials": ";".join(f.exposed_credentials),
                "exposed_tools": ";".join(f.exposed_tools),
                "asset_name": f.asset.name,
                "asset_type": f.asset.asset_type,
            }
        )
    return buf.getvalue()


def to_markdown(report: AIBOMReport) -> str:
    counts = report.severity_counts()
    lines = [
        "# AI-BOM Scan Report",
        "",
        f"- agents: **{report.total_agents}** · MCP servers: **{report.total_servers}**"
        f" · packages: **{report.total_packages}**",
        f"- vulnerabilities: **{report.total_vulnerabilities}**"
        f" (critical {counts['critical']} / high {counts['high']} /"
        f" medium {counts['medium']} / low {counts['low']})",
        "",
        "| risk | severity | vulnerability | package | reachability | fix |",
        "|---:|---|---|---|---|---|",
    ]
    for br in report.blast_radii:
        kev = " **KEV**" if br.vulnerability.is_kev else ""
        mal = " **MALICIOUS**" if br.package.is_malicious else ""
        lines.append(
            f"| {br.risk_score:.1f} | {br.vulnerability.severity.value}{kev}{mal} "
            f"| {br.vulnerability.id} | {br.package.name}@{br.package.version} "
            f"| {br.reachability} | {br.vulnerability.fixed_version or '-'} |"
        )
    return "\n".join(lines) + "\n"


def to_prometheus(report: AIBOMReport) -> str:
    """Prometheus text exposition (reference: output/prometheus.py)."""
    counts = report.severity_counts()
    lines = [
        "# HELP agent_bom_vulnerabilities_total Vulnerabilities by severity",
        "# TYPE agent_bom_vulnerabilities_total gauge",
    ]
    for sev, n in counts.items():
        lines.append(f'agent_bom_vulnerabilities_total{{severity="{sev}"}} {n}')
    lines += [
        "# HELP agent_bom_agents_total Discovered agents",
        "# TYPE agent_bom_agents_total gauge",
        f"agent_bom_agents_total {report.total_agents}",
        "# HELP agent_bom_mcp_servers_total Discovered MCP servers",
        "# TYPE agent_bom_mcp_servers_total gauge",
        f"agent_bom_mcp_servers_total {report.total_servers}",
        "# HELP agent_bom_packages_total Inventoried packages",
        "# TYPE agent_bom_packages_total gauge",
        f"agent_bom_packages_total {report.total_packages}",
        "# HELP agent_bom_malicious_packages_total Known-malicious package hits",
        "# TYPE agent_bom_malicious_packages_total gauge",
        f"agent_bom_malicious_packages_total "
        f"{sum(1 for br in report.blast_radii if br.package.is_malicious)}",
        "# HELP agent_bom_kev_findings_total CISA KEV findings",
        "# TYPE agent_bom_kev_findings_total gauge",
        f"agent_bom_kev_findings_total "
        f"{sum(1 for br in report.blast_radii if br.vulnerability.is_kev)}",
        "# HELP agent_bom_max_risk_score Highest finding risk score",
        "# TYPE agent_bom_max_risk_score gauge",
        f"agent_bom_max_risk_score "
        f"{max((br.risk_score for br in report.blast_radii), default=0.0)}",
    ]
    return "\n".join(lines) + "\n"


def to_junit(report: AIBOMReport) -> str:
    """JUnit XML: one testcase per finding; failures = actionable findings."""
    from xml.sax.saxutils import escape

    findings = report.to_findings()
    failures = sum(1 for f in findings if f.is_actionable)
    out = [
        '<?xml version="1.0" encoding="UTF-8"?>',
        f'<testsuite name="agent-bom" tests="{len(findings)}" failures="{failures}">',
    ]
    for f in findings:
        name = escape(f.title or f.id)
        out.append(f'  <testcase name="{name}" classname="{escape(f.asset.name)}">')
        if f.is_actionable:
            msg = escape(f"{f.severity}: {f.description[:300]}")
            out.append(f'    <failure message="{msg}" type="{f.finding_type.value}"/>')
        out.append("  </testcase>")
    out.append("</testsuite>")
    return "\n".join(out) + "\n"


def to_parquet_bytes(report: AIBOMReport) -> bytes:
    """Findings table as Parquet (pyarrow) — the analytics export."""
    import io as _io

    import pyarrow as pa
    import pyarrow.parquet as pq

    rows = []
    for f in report.to_findings():
        ev = f.evidence if isinstance(f.evidence, dict) else {}
        rows.append({
            "finding_id": f.id, "finding_type": f.finding_type.value,
            "severity": f.severity, "risk_score": float(f.risk_score),
            "vulnerability_id": f.vulnerability_id or "",
            "package": str(ev.get("package_name", "")),
            "version": str(ev.get("package_version", "")),
            "ecosystem": str(ev.get("ecosystem", "")),
            "is_kev": bool(f.is_kev), "is_malicious": bool(f.is_malicious),
            "epss_score": float(f.epss_score) if f.epss_score is not None else None,
            "cvss_score": float(f.cvss_score) if f.cvss_score is not None else None,
            "reachability": f.reachability or "",
            "impact_category": f.impact_category or "",
            "affected_agents": list(f.affected_agents),
            "exposed_credentials": list(f.exposed_credentials),
            "asset_name": f.asset.name, "asset_type": f.asset.asset_type,
        })
    table = pa.Table.from_pylist(rows)
    buf = _io.BytesIO()
    pq.write_table(table, buf)
    return buf.getvalue()


_BADGE_COLORS = {"critical": "#e05d44", "high": "#fe7d37", "medium": "#dfb317",
                 "clean": "#4c1"}


def to_badge_svg(report: AIBOMReport) -> str:
    """Shields-style SVG badge summarizing the scan verdict."""
    counts = report.severity_counts()
    if counts["critical"]:
        label, color = f"{counts['critical']} critical", _BADGE_COLORS["critical"]
    elif counts["high"]:
        label, color = f"{counts['high']} high", _BADGE_COLORS["high"]
    elif counts["medium"]:
        label, color = f"{counts['medium']} medium", _BADGE_COLORS["medium"]
    else:
        label, color = "clean", _BADGE_COLORS["clean"]
    left, right = "agent-bom", label
    lw, rw = 6 * len(left) + 10, 6 * len(right) + 10
    return f"""<svg xmlns="http://www.w3.org/2000/svg" width="{lw + rw}" height="20">
<rect width="{lw}" height="20" fill="#555"/>
<rect x="{lw}" width="{rw}" height="20" fill="{color}"/>
<g fill="#fff" text-anchor="middle" font-family="Verdana,sans-serif" font-size="11">
<text x="{lw / 2}" y="14">{left}</text>
<text x="{lw + rw / 2}" y="14">{right}</text>
</g></svg>
"""
