"""Semgrep result ingestion → unified findings + symbol-index joins.

Reference parity: the SAST row of SURVEY.md §2.2 — the reference ingests
Semgrep output alongside its own AST analyzers and feeds the results into
the ``affected_symbols`` reachability join.  This module consumes
``semgrep --json`` documents (native format; Semgrep's SARIF goes through
the existing SARIF ingester) without running semgrep itself — air-gapped
deployments ship result files produced in CI.

Tolerant by contract: any malformed result row is skipped, never fatal.
"""

from __future__ import annotations

import json
import re
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Optional

from agentbom_amd.models.finding import Asset, Finding, FindingSource, FindingType

_SEVERITY = {"ERROR": "high", "WARNING": "medium", "INFO": "low"}

# identifier immediately followed by a call — feeds the call-symbol join
_CALL_RE = re.compile(r"\b([A-Za-z_][A-Za-z0-9_]*(?:\.[A-Za-z_][A-Za-z0-9_]*)*)\s*\(")
_KEYWORDS = {"if", "for", "while", "return", "switch", "catch", "function",
             "def", "print", "with", "assert", "raise", "new", "in", "not"}


@dataclass
class SemgrepResult:
    check_id: str
    path: str
    line: int = 0
    end_line: int = 0
    severity: str = "medium"
    message: str = ""
    cwe_ids: list[str] = field(default_factory=list)
    owasp: list[str] = field(default_factory=list)
    confidence: str = ""
    matched_lines: str = ""
    metavars: dict[str, str] = field(default_factory=dict)


def parse_semgrep_json(doc: Any) -> list[SemgrepResult]:
    """Parse a ``semgrep --json`` document.  Fail-soft per result row."""
    if isinstance(doc, (str, bytes)):
        try:
            doc = json.loads(doc)
        except (ValueError, TypeError):
            return []
    if not isinstance(doc, dict):
        return []
    out: list[SemgrepResult] = []
    for res in doc.get("results") or []:
        if not isinstance(res, dict):
            continue
        check_id = res.get("check_id")
        path = res.get("path")
        if not isinstance(check_id, str) or not isinstance(path, str):
            continue
        extra = res.get("extra") if isinstance(res.get("extra"), dict) else {}
        meta = extra.get("metadata") if isinstance(extra.get("metadata"), dict) else {}

        def _line(key: str) -> int:
            v = res.get(key)
            if isinstance(v, dict) and isinstance(v.get("line"), int):
                return v["line"]
            return 0

        cwe = meta.get("cwe")
        cwe_ids = []
        for c in (cwe if isinstance(cwe, list) else [cwe]):
            if isinstance(c, str):
                m = re.search(r"CWE-\d+", c)
                cwe_ids.append(m.group(0) if m else c)
        owasp = meta.get("owasp")
        owasp_ids = [o for o in (owasp if isinstance(owasp, list) else [owasp])
                     if isinstance(o, str)]
        metavars = {}
        mv = extra.get("metavars")
        if isinstance(mv, dict):
            for k, v in mv.items():
                if isinstance(v, dict) and isinstance(v.get("abstract_content"), str):
                    metavars[k] = v["abstract_content"]
        out.append(SemgrepResult(
            check_id=check_id,
            path=path,
            line=_line("start"),
            end_line=_line("end"),
            severity=_SEVERITY.get(str(extra.get("severity") or "").upper(), "medium"),
            message=str(extra.get("message") or "")[:500],
            cwe_ids=cwe_ids,
            owasp=owasp_ids,
            confidence=str(meta.get("confidence") or "").lower(),
            matched_lines=str(extra.get("lines") or "")[:400],
            metavars=metavars,
        ))
    return out


def load_semgrep_file(path: str | Path) -> list[SemgrepResult]:
    try:
        return parse_semgrep_json(Path(path).read_text())
    except OSError:
        return []


def looks_like_semgrep(doc: Any) -> bool:
    """Distinguish semgrep-native JSON from SARIF (both arrive at the same
    external-ingest seam)."""
    if isinstance(doc, (str, bytes)):
        try:
            doc = json.loads(doc)
        except (ValueError, TypeError):
            return False
    return (isinstance(doc, dict) and isinstance(doc.get("results"), list)
            and "runs" not in doc)


def semgrep_to_findings(results: list[SemgrepResult]) -> list[Finding]:
    """Semgrep rows → unified SAST findings (dedup by rule+path+line)."""
    out: list[Finding] = []
    seen: set[tuple] = set()
    for r in results:
        key = (r.check_id, r.path, r.line)
        if key in seen:
            continue
        seen.add(key)
        out.append(Finding(
            finding_type=FindingType.SAST,
            source=FindingSource.SAST,
            asset=Asset(name=r.path, asset_type="package",
                        location=f"{r.path}:{r.line}" if r.line else r.path),
            severity=r.severity,
            title=f"semgrep: {r.check_id.rsplit('.', 1)[-1]}",
            description=r.message,
            cwe_ids=list(r.cwe_ids),
            compliance_tags=list(r.owasp),
            evidence={"rule_id": r.check_id, "line": r.line,
                      "end_line": r.end_line, "confidence": r.confidence,
                      "matched": r.matched_lines, "tool": "semgrep"},
        ))
    return out


def extend_symbol_index_from_semgrep(index, results: list[SemgrepResult]) -> int:
    """Feed semgrep-matched call sites into the SymbolIndex so advisory
    ``affected_symbols`` joins see code semgrep flagged even in languages
    the built-in AST pass missed.  Returns symbols added."""
    added = 0
    for r in results:
        texts = [r.matched_lines] + list(r.metavars.values())
        for text in texts:
            for m in _CALL_RE.finditer(text or ""):
                sym = m.group(1)
                head = sym.split(".")[0]
                if head in _KEYWORDS or len(sym) <= 1:
                    continue
                if sym not in index.calls:
                    index.calls.add(sym)
                    added += 1
    return added
