"""PDF report export — hand-rolled PDF 1.4 writer, zero dependencies.

Reference parity: src/agent_bom/output/pdf.py.  This environment ships no
PDF library, so the document is emitted directly: a valid PDF 1.4 with
Helvetica text objects, one content stream per page, a correct xref
table, and flate-compressed streams (zlib is stdlib).  Layout: a summary
header page plus paginated finding rows.
"""

from __future__ import annotations

import zlib
from typing import Any

from agentbom_amd.models import AIBOMReport

_PAGE_W, _PAGE_H = 612, 792  # US Letter, points
_MARGIN = 54
_LINE_H = 14
_ROWS_PER_PAGE = int((_PAGE_H - 2 * _MARGIN - 60) / _LINE_H)


def _esc(text: str) -> str:
    return (text.replace("\\", r"\\").replace("(", r"\(")
            .replace(")", r"\)"))[:110]


def _page_stream(lines: list[tuple[int, int, str, int]]) -> bytes:
    """lines: (x, y, text, size) -> one content stream."""
    parts = ["BT"]
    for x, y, text, size in lines:
        parts.append(f"/F1 {size} Tf 1 0 0 1 {x} {y} Tm ({_esc(text)}) Tj")
    parts.append("ET")
    return zlib.compress("\n".join(parts).encode("latin-1", "replace"))


def to_pdf_bytes(report: AIBOMReport) -> bytes:
    """Render the scan report as a multi-page PDF."""
    counts = report.severity_counts()
    pages: list[list[tuple[int, int, str, int]]] = []

    from agentbom_amd.scan.risk import estate_exec_score

    score = estate_exec_score(report)
    header = [
        (_MARGIN, _PAGE_H - _MARGIN, "agent-bom AI-BOM Scan Report", 18),
        (_MARGIN, _PAGE_H - _MARGIN - 24,
         f"agents: {report.total_agents}   MCP servers: {report.total_servers}"
         f"   packages: {report.total_packages}", 11),
        (_MARGIN, _PAGE_H - _MARGIN - 40,
         f"findings: {report.total_vulnerabilities}   critical: "
         f"{counts['critical']}   high: {counts['high']}   medium: "
         f"{counts['medium']}   low: {counts['low']}", 11),
        (_MARGIN, _PAGE_H - _MARGIN - 56,
         f"estate posture: grade {score.get('grade', '?')} "
         f"(score {score.get('score', '?')}/100)", 11),
        (_MARGIN, _PAGE_H - _MARGIN - 80, "risk  severity  vulnerability"
         "          package                    reachability", 10),
    ]
    y = _PAGE_H - _MARGIN - 80 - _LINE_H
    current = list(header)
    for br in report.blast_radii:
        if y < _MARGIN:
            pages.append(current)
            current = [(_MARGIN, _PAGE_H - _MARGIN,
                        "agent-bom scan report (continued)", 12)]
            y = _PAGE_H - _MARGIN - 2 * _LINE_H
        kev = " [KEV]" if br.vulnerability.is_kev else ""
        mal = " [MALICIOUS]" if br.package.is_malicious else ""
        current.append((
            _MARGIN, y,
            f"{br.risk_score:>4.1f}  {br.vulnerability.severity.value:<9s}"
            f"{br.vulnerability.id:<22s}"
            f"{br.package.name}@{br.package.version:<12s}"
            f"  {br.reachability}{kev}{mal}", 9))
        y -= _LINE_H

    # remediation section (same prioritized plan as console/HTML)
    from agentbom_amd.output.json_fmt import _build_remediation_json

    plan = _build_remediation_json(report)
    if plan:
        needed = (len(plan[:15]) + 3) * _LINE_H
        if y - needed < _MARGIN:
            pages.append(current)
            current = [(_MARGIN, _PAGE_H - _MARGIN,
                        "agent-bom scan report (continued)", 12)]
            y = _PAGE_H - _MARGIN - 2 * _LINE_H
        y -= _LINE_H
        current.append((_MARGIN, y, "Remediation plan (highest risk first)",
                        12))
        y -= _LINE_H
        for i, item in enumerate(plan[:15], 1):
            current.append((
                _MARGIN, y,
                f"{i:>2d}. {item.get('package', '?'):<34s} -> "
                f"{item.get('fix_version') or 'no fix yet':<14s} "
                f"({len(item.get('vulns', []))} vulns, max risk "
                f"{item.get('max_risk_score', 0):.1f})", 9))
            y -= _LINE_H
    pages.append(current)

    # assemble objects: 1 catalog, 2 pages-tree, 3 font, then per page
    objects: list[bytes] = []

    def add(obj: bytes) -> int:
        objects.append(obj)
        return len(objects)

    font_id = 3
    page_ids = []
    content_ids = []
    first_page_obj = 4
    n_pages = len(pages)
    # object ids: pages occupy first_page_obj..+2*n-1 alternating page/content
    for i in range(n_pages):
        page_ids.append(first_page_obj + 2 * i)
        content_ids.append(first_page_obj + 2 * i + 1)

    add(b"<< /Type /Catalog /Pages 2 0 R >>")  # 1
    kids = " ".join(f"{pid} 0 R" for pid in page_ids)
    add(f"<< /Type /Pages /Kids [{kids}] /Count {n_pages} >>".encode())  # 2
    add(b"<< /Type /Font /Subtype /Type1 /BaseFont /Helvetica >>")  # 3
    for i, lines in enumerate(pages):
        add(f"<< /Type /Page /Parent 2 0 R /MediaBox [0 0 {_PAGE_W} {_PAGE_H}]"
            f" /Resources << /Font << /F1 {font_id} 0 R >> >>"
            f" /Contents {content_ids[i]} 0 R >>".encode())
        stream = _page_stream(lines)
        add(f"<< /Length {len(stream)} /Filter /FlateDecode >>\nstream\n"
            .encode() + stream + b"\nendstream")

    out = bytearray(b"%PDF-1.4\n%\xe2\xe3\xcf\xd3\n")
    offsets = [0]
    for i, obj in enumerate(objects, start=1):
        offsets.append(len(out))
        out += f"{i} 0 obj\n".encode() + obj + b"\nendobj\n"
    xref_at = len(out)
    out += f"xref\n0 {len(objects) + 1}\n".encode()
    out += b"0000000000 65535 f \n"
    for off in offsets[1:]:
        out += f"{off:010d} 00000 n \n".encode()
    out += (f"trailer\n<< /Size {len(objects) + 1} /Root 1 0 R >>\n"
            f"startxref\n{xref_at}\n%%EOF\n".encode())
    return bytes(out)
