"""Floating version-reference detection (supply-chain pinning hygiene).

Reference parity: src/agent_bom/floating_refs.py — a dependency declared
as ``latest``, ``*``, a bare range (``^``/``~``/``>=``) or a mutable git
branch resolves DIFFERENTLY tomorrow: the classic npm/PyPI supply-chain
takeover precondition.  Manifests (not lockfiles) are scanned for
floating specs; each hit carries the spec class and a pin suggestion.
"""

from __future__ import annotations

import json
import re
from dataclasses import dataclass
from pathlib import Path
from typing import Any, Iterator, Optional

# spec-class detection, ordered most to least severe
_CLASSES = [
    ("wildcard", re.compile(r"^\s*(\*|x|latest)\s*$", re.I), "critical"),
    ("git-branch", re.compile(r"(git\+|github:|git://).*[#@](main|master|HEAD)$|"
                              r"(git\+|github:|git://)(?!.*[#@]).*$"), "high"),
    ("url", re.compile(r"^https?://"), "high"),
    ("open-lower-bound", re.compile(r"^\s*>=?[^,]*$"), "medium"),
    ("caret-range", re.compile(r"^\s*\^"), "low"),
    ("tilde-range", re.compile(r"^\s*~(?!=)"), "low"),
]


@dataclass
class FloatingRef:
    file: str
    ecosystem: str
    name: str
    spec: str
    spec_class: str
    severity: str

    def to_dict(self) -> dict[str, Any]:
        return {"file": self.file, "ecosystem": self.ecosystem,
                "name": self.name, "spec": self.spec,
                "spec_class": self.spec_class, "severity": self.severity,
                "suggestion": f"pin {self.name} to an exact version "
                              "(and commit a lockfile)"}


def classify_spec(spec: str) -> Optional[tuple[str, str]]:
    spec = (spec or "").strip()
    if not spec:
        return ("wildcard", "critical")
    for cls, pattern, severity in _CLASSES:
        if pattern.search(spec):
            return (cls, severity)
    return None


def _npm_specs(path: Path) -> Iterator[tuple[str, str]]:
    try:
        doc = json.loads(path.read_text())
    except (OSError, json.JSONDecodeError):
        return
    if not isinstance(doc, dict):
        return
    for section in ("dependencies", "devDependencies", "optionalDependencies"):
        deps = doc.get(section)
        if isinstance(deps, dict):
            for name, spec in deps.items():
                yield str(name), str(spec)


_REQ_LINE_RE = re.compile(r"^\s*([A-Za-z0-9_.\[\]-]+)\s*(.*?)\s*(?:#.*)?$")


def _requirements_specs(path: Path) -> Iterator[tuple[str, str]]:
    try:
        text = path.read_text(errors="replace")
    except OSError:
        return
    for line in text.splitlines():
        line = line.strip()
        if not line or line.startswith(("#", "-")):
            continue
        m = _REQ_LINE_RE.match(line)
        if m and "==" not in (m.group(2) or ""):
            yield m.group(1), m.group(2) or ""


def _pyproject_specs(path: Path) -> Iterator[tuple[str, str]]:
    from agentbom_amd.utils.compat import tomllib

    try:
        doc = tomllib.loads(path.read_text())
    except Exception:  # noqa: BLE001 — fail-soft parse boundary
        return
    deps = (doc.get("project") or {}).get("dependencies") or []
    for dep in deps:
        m = _REQ_LINE_RE.match(str(dep))
        if m and "==" not in (m.group(2) or ""):
            yield m.group(1), m.group(2) or ""
    poetry = (((doc.get("tool") or {}).get("poetry") or {})
              .get("dependencies") or {})
    for name, spec in poetry.items():
        if name.lower() == "python":
            continue
        if isinstance(spec, dict):
            spec = spec.get("version", "")
        yield str(name), str(spec)


_MANIFESTS = {
    "package.json": ("npm", _npm_specs),
    "requirements.txt": ("pypi", _requirements_specs),
    "pyproject.toml": ("pypi", _pyproject_specs),
}


def _lockfile_present(manifest: Path) -> bool:
    locks = {"package.json": ("package-lock.json", "yarn.lock",
                              "pnpm-lock.yaml", "npm-shrinkwrap.json"),
             "requirements.txt": (),
             "pyproject.toml": ("poetry.lock", "uv.lock", "pdm.lock")}
    return any((manifest.parent / l).exists()
               for l in locks.get(manifest.name, ()))


def scan_floating_refs(root: str | Path, cap: int = 500) -> list[FloatingRef]:
    """Scan manifests under ``root``; lock-covered manifests are skipped
    (the lockfile pins resolution, so ranges there are fine)."""
    root = Path(root)
    if root.is_file():
        candidates = [root]
    else:
        candidates = [p for name in _MANIFESTS
                      for p in sorted(root.rglob(name))[:cap]
                      if not {"node_modules", ".git", ".venv",
                              "venv"} & set(p.parts)]
    out: list[FloatingRef] = []
    for manifest in candidates:
        entry = _MANIFESTS.get(manifest.name)
        if entry is None or _lockfile_present(manifest):
            continue
        eco, extract = entry
        for name, spec in extract(manifest):
            cls = classify_spec(spec)
            if cls:
                out.append(FloatingRef(
                    file=str(manifest), ecosystem=eco, name=name,
                    spec=spec, spec_class=cls[0], severity=cls[1]))
    return out
