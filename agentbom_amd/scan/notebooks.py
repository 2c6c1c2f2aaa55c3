"""Jupyter notebook scanning: packages, secrets, risky calls per cell.

Reference parity: src/agent_bom/jupyter.py — notebooks are a first-class
AI-estate surface (they install packages imperatively and leak keys in
outputs).  Extracted signals:

- ``!pip install`` / ``%pip install`` lines → Package rows (pypi);
- code cells run through the Python AST sink analysis;
- SOURCE AND OUTPUTS run through the secret scanner (a leaked key in a
  stored cell output is the classic notebook incident);
- risky magics (``%%bash``, ``!curl … | sh``).
"""

from __future__ import annotations

import json
import re
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any

from agentbom_amd.models import Package

_PIP_RE = re.compile(
    r"^\s*[!%]\s*pip3?\s+install\s+(?:-U\s+|--upgrade\s+|-q\s+)*(.+)$")
_SPEC_RE = re.compile(r"^([A-Za-z0-9_.@/-]+)\s*(?:==\s*([A-Za-z0-9_.!+-]+))?$")
_SHELL_PIPE_RE = re.compile(r"(?i)(?:curl|wget)[^|\n]*\|\s*(?:bash|sh|python)")


@dataclass
class NotebookScanResult:
    path: str
    packages: list[Package] = field(default_factory=list)
    secret_hits: list[dict[str, Any]] = field(default_factory=list)
    ast_findings: list[dict[str, Any]] = field(default_factory=list)
    risky_cells: list[dict[str, Any]] = field(default_factory=list)
    cells_scanned: int = 0

    def to_dict(self) -> dict[str, Any]:
        return {
            "path": self.path,
            "cells_scanned": self.cells_scanned,
            "packages": [f"{p.name}@{p.version or '?'}" for p in self.packages],
            "secret_hits": self.secret_hits,
            "ast_findings": self.ast_findings,
            "risky_cells": self.risky_cells,
        }


def _cell_text(cell: dict[str, Any]) -> str:
    src = cell.get("source", "")
    return "".join(src) if isinstance(src, list) else str(src)


def _outputs_text(cell: dict[str, Any]) -> str:
    parts = []
    for out in cell.get("outputs", []) or []:
        if not isinstance(out, dict):
            continue
        text = out.get("text")
        if isinstance(text, list):
            parts.append("".join(text))
        elif isinstance(text, str):
            parts.append(text)
        data = out.get("data")
        if isinstance(data, dict):
            plain = data.get("text/plain")
            if isinstance(plain, list):
                parts.append("".join(plain))
            elif isinstance(plain, str):
                parts.append(plain)
    return "\n".join(parts)


def scan_notebook(path: str | Path) -> NotebookScanResult:
    from agentbom_amd.scan.ast_analysis import analyze_python_source
    from agentbom_amd.scan.secrets import scan_text

    path = Path(path)
    result = NotebookScanResult(path=str(path))
    try:
        doc = json.loads(path.read_text())
    except (OSError, json.JSONDecodeError):
        return result

    for i, cell in enumerate(doc.get("cells", []) or []):
        if not isinstance(cell, dict):
            continue
        result.cells_scanned += 1
        text = _cell_text(cell)
        out_text = _outputs_text(cell)

        for line in text.splitlines():
            m = _PIP_RE.match(line)
            if not m:
                continue
            for spec in m.group(1).split():
                if spec.startswith("-"):
                    continue
                sm = _SPEC_RE.match(spec)
                if sm:
                    result.packages.append(Package(
                        name=sm.group(1), version=sm.group(2) or "",
                        ecosystem="pypi"))

        if _SHELL_PIPE_RE.search(text):
            result.risky_cells.append({
                "cell": i, "rule": "remote-script-piped-to-shell",
                "severity": "critical"})

        if cell.get("cell_type") == "code":
            code = "\n".join(l for l in text.splitlines()
                             if not l.lstrip().startswith(("!", "%")))
            try:
                hits, _syms = analyze_python_source(code, f"{path}:cell{i}")
                result.ast_findings.extend(h.to_dict() for h in hits)
            except SyntaxError:
                pass

        for blob, where in ((text, "source"), (out_text, "output")):
            for hit in scan_text(blob, f"{path}:cell{i}"):
                entry = hit.to_dict()
                entry["where"] = where
                entry["cell"] = i
                result.secret_hits.append(entry)
    return result


def scan_notebook_tree(root: str | Path, cap: int = 200) -> list[NotebookScanResult]:
    root = Path(root)
    paths = [root] if root.is_file() else sorted(root.rglob("*.ipynb"))[:cap]
    return [scan_notebook(p) for p in paths]
