"""Project repository inventory: directory tree + source files + imports.

Produces the report's ``project_inventory`` side block consumed by the
repo-structure / code-graph overlays (reference counterpart:
src/agent_bom/graph/repo_structure_overlay.py reads the same block; the
per-file Python import lists here go further than the reference — they are
extracted with the real ``ast`` module so the code-graph overlay can emit
IMPORTS edges to EXTERNAL_IMPORT nodes, not just module containment).

Bounded walk: depth, per-directory entries and total files are capped so a
giant monorepo cannot blow up the report; skip lists cover VCS/venv/build
dirs.  Deterministic output (sorted) for snapshot-stable graphs.
"""

from __future__ import annotations

import ast
from pathlib import Path
from typing import Any, Optional

_SKIP_DIRS = {".git", ".hg", ".svn", "node_modules", "__pycache__", ".venv",
              "venv", ".tox", ".mypy_cache", ".pytest_cache", "dist", "build",
              ".eggs", ".ruff_cache", "target", ".next", ".cache"}
_SOURCE_EXT = {".py", ".js", ".ts", ".tsx", ".jsx", ".go", ".rs", ".java",
               ".rb", ".c", ".cc", ".cpp", ".h", ".hpp", ".hip", ".cs",
               ".php", ".swift", ".kt", ".scala", ".sh"}
_CONFIG_EXT = {".yml", ".yaml", ".json", ".toml", ".ini", ".cfg", ".env"}
_MANIFESTS = {"package.json", "requirements.txt", "pyproject.toml", "setup.py",
              "Pipfile", "poetry.lock", "package-lock.json", "yarn.lock",
              "go.mod", "Cargo.toml", "pom.xml", "build.gradle", "Gemfile",
              "composer.json", "environment.yml"}


def _py_imports(path: Path, limit: int = 64) -> list[str]:
    """Top-level module names imported by one Python file (real AST)."""
    try:
        tree = ast.parse(path.read_text(encoding="utf-8", errors="replace"))
    except (OSError, SyntaxError, ValueError):
        return []
    mods: set[str] = set()
    for node in ast.walk(tree):
        if isinstance(node, ast.Import):
            mods.update(alias.name.split(".")[0] for alias in node.names)
        elif isinstance(node, ast.ImportFrom) and node.module and node.level == 0:
            mods.add(node.module.split(".")[0])
    return sorted(mods)[:limit]


def collect_project_inventory(root: str, *, max_depth: Optional[int] = None,
                              max_files: Optional[int] = None,
                              max_entries_per_dir: Optional[int] = None,
                              parse_imports: bool = True) -> Optional[dict[str, Any]]:
    """Walk ``root`` → {"directories": [...], "files": [...], "truncated"}.

    Directory records: path (repo-relative, "" = root), counts per class,
    manifest names.  File records: path, kind (source|config|manifest),
    language for sources, and ``imports`` for Python files.
    """
    from agentbom_amd.utils import config as cfg

    max_depth = cfg.REPO_INVENTORY_MAX_DEPTH if max_depth is None else max_depth
    max_files = cfg.REPO_INVENTORY_MAX_FILES if max_files is None else max_files
    max_entries_per_dir = (cfg.REPO_INVENTORY_MAX_DIR_ENTRIES
                           if max_entries_per_dir is None else max_entries_per_dir)
    base = Path(root).expanduser()
    if not base.is_dir():
        return None

    directories: list[dict[str, Any]] = []
    files: list[dict[str, Any]] = []
    truncated = False

    def walk(d: Path, rel: str, depth: int) -> None:
        nonlocal truncated
        if depth > max_depth or len(files) >= max_files:
            truncated = True
            return
        try:
            entries = sorted(d.iterdir(), key=lambda p: p.name)
        except OSError:
            return
        if len(entries) > max_entries_per_dir:
            entries = entries[:max_entries_per_dir]
            truncated = True
        rec = {"path": rel, "source_files": 0, "config_files": 0,
               "manifests": []}
        subdirs = []
        for entry in entries:
            name = entry.name
            if entry.is_dir():
                # hidden dirs are skipped except .github (CI workflows feed
                # the CI-graph overlay's CONFIGURES stitching)
                if name not in _SKIP_DIRS and (not name.startswith(".")
                                               or name == ".github"):
                    subdirs.append(entry)
                continue
            if len(files) >= max_files:
                truncated = True
                break
            fr = rel + "/" + name if rel else name
            if name in _MANIFESTS:
                rec["manifests"].append(name)
                files.append({"path": fr, "kind": "manifest"})
            elif entry.suffix in _SOURCE_EXT:
                rec["source_files"] += 1
                frec: dict[str, Any] = {"path": fr, "kind": "source",
                                        "language": entry.suffix.lstrip(".")}
                if parse_imports and entry.suffix == ".py":
                    imports = _py_imports(entry)
                    if imports:
                        frec["imports"] = imports
                files.append(frec)
            elif entry.suffix in _CONFIG_EXT:
                rec["config_files"] += 1
                files.append({"path": fr, "kind": "config"})
        if rec["source_files"] or rec["config_files"] or rec["manifests"] \
                or rel == "":
            directories.append(rec)
        for sub in subdirs:
            walk(sub, rel + "/" + sub.name if rel else sub.name, depth + 1)

    walk(base, "", 0)
    return {"directories": directories, "files": files,
            "truncated": truncated, "root": str(base)}
