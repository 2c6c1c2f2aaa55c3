"""Python AST security analysis + symbol-level CVE reachability.

Reference: src/agent_bom/ast_python_analysis.py (2,036 LoC — dangerous-sink
flow findings), sast.py, reachability_cve.py (classify findings as
function_reachable / package_reachable / unreachable by joining advisory
``affected_symbols`` against the project's imported symbols/call sites).

Two products:
- ``analyze_python_source``: dangerous-sink findings (eval/exec/pickle/
  yaml.load/subprocess-shell/os.system/requests-verify-off/...) with
  taint-lite qualification (literal args downgrade severity);
- ``SymbolIndex`` + ``apply_symbol_reachability``: which advisory symbols
  a codebase actually calls -> upgrades/downgrades finding reachability
  (models/blast.py consumes symbol_reachability in scoring).
"""

from __future__ import annotations

import ast
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Optional

from agentbom_amd.models.finding import Asset, Finding, FindingSource, FindingType, stable_id

# (qualified call, category, severity, cwe)
DANGEROUS_CALLS: dict[str, tuple[str, str, str]] = {
    "eval": ("code-injection", "high", "CWE-95"),
    "exec": ("code-injection", "high", "CWE-95"),
    "compile": ("code-injection", "medium", "CWE-94"),
    "os.system": ("command-injection", "high", "CWE-78"),
    "os.popen": ("command-injection", "high", "CWE-78"),
    "subprocess.call": ("command-injection", "medium", "CWE-78"),
    "subprocess.run": ("command-injection", "medium", "CWE-78"),
    "subprocess.Popen": ("command-injection", "medium", "CWE-78"),
    "pickle.load": ("unsafe-deserialization", "high", "CWE-502"),
    "pickle.loads": ("unsafe-deserialization", "high", "CWE-502"),
    "yaml.load": ("unsafe-deserialization", "high", "CWE-502"),
    "yaml.full_load": ("unsafe-deserialization", "high", "CWE-502"),
    "yaml.unsafe_load": ("unsafe-deserialization", "critical", "CWE-502"),
    "marshal.load": ("unsafe-deserialization", "high", "CWE-502"),
    "marshal.loads": ("unsafe-deserialization", "high", "CWE-502"),
    "shelve.open": ("unsafe-deserialization", "medium", "CWE-502"),
    "tempfile.mktemp": ("insecure-temp-file", "medium", "CWE-377"),
    "random.random": ("weak-randomness", "low", "CWE-338"),
    "hashlib.md5": ("weak-crypto", "low", "CWE-327"),
    "hashlib.sha1": ("weak-crypto", "low", "CWE-327"),
    "ssl._create_unverified_context": ("tls-verification-disabled", "high", "CWE-295"),
    "torch.load": ("unsafe-deserialization", "medium", "CWE-502"),
}


@dataclass
class AstFinding:
    file: str
    line: int
    call: str
    category: str
    severity: str
    cwe: str
    entrypoint: str  # enclosing function, or <module>
    snippet: str
    tainted: bool  # non-literal argument observed

    def to_dict(self) -> dict[str, Any]:
        return {
            "file": self.file, "line": self.line, "call": self.call,
            "category": self.category, "severity": self.severity, "cwe": self.cwe,
            "entrypoint": self.entrypoint, "snippet": self.snippet,
            "tainted": self.tainted,
        }


def _qualified_name(node: ast.AST) -> Optional[str]:
    if isinstance(node, ast.Name):
        return node.id
    if isinstance(node, ast.Attribute):
        base = _qualified_name(node.value)
        return f"{base}.{node.attr}" if base else node.attr
    return None


class _Analyzer(ast.NodeVisitor):
    def __init__(self, path: str, source_lines: list[str], aliases: dict[str, str]):
        self.path = path
        self.lines = source_lines
        self.aliases = aliases
        self.findings: list[AstFinding] = []
        self.calls: set[str] = set()
        self.stack: list[str] = []

    def _resolve(self, name: str) -> str:
        head, _, rest = name.partition(".")
        head = self.aliases.get(head, head)
        return f"{head}.{rest}" if rest else head

    def visit_FunctionDef(self, node: ast.FunctionDef) -> None:
        self.stack.append(node.name)
        self.generic_visit(node)
        self.stack.pop()

    visit_AsyncFunctionDef = visit_FunctionDef  # type: ignore[assignment]

    def visit_Call(self, node: ast.Call) -> None:
        qn = _qualified_name(node.func)
        if qn:
            resolved = self._resolve(qn)
            self.calls.add(resolved)
            hit = DANGEROUS_CALLS.get(resolved)
            if hit:
                category, severity, cwe = hit
                tainted = any(
                    not isinstance(a, ast.Constant) for a in node.args
                ) or bool(node.keywords)
                # subprocess only matters with shell=True or tainted strings
                if resolved.startswith("subprocess."):
                    shell = any(
                        kw.arg == "shell" and isinstance(kw.value, ast.Constant)
                        and kw.value.value is True
                        for kw in node.keywords
                    )
                    if not shell:
                        self.generic_visit(node)
                        return
                    severity = "high"
                if not tainted and severity in ("high", "critical"):
                    severity = "medium"  # literal-only args: lower risk
                line = node.lineno
                self.findings.append(AstFinding(
                    file=self.path, line=line, call=resolved, category=category,
                    severity=severity, cwe=cwe,
                    entrypoint=self.stack[-1] if self.stack else "<module>",
                    snippet=self.lines[line - 1].strip()[:160] if line <= len(self.lines) else "",
                    tainted=tainted,
                ))
            # requests verify=False
            if resolved.startswith("requests.") and any(
                kw.arg == "verify" and isinstance(kw.value, ast.Constant)
                and kw.value.value is False for kw in node.keywords
            ):
                self.findings.append(AstFinding(
                    file=self.path, line=node.lineno, call=resolved,
                    category="tls-verification-disabled", severity="high",
                    cwe="CWE-295",
                    entrypoint=self.stack[-1] if self.stack else "<module>",
                    snippet=self.lines[node.lineno - 1].strip()[:160],
                    tainted=True,
                ))
        self.generic_visit(node)


def _collect_aliases(tree: ast.AST) -> dict[str, str]:
    aliases: dict[str, str] = {}
    for node in ast.walk(tree):
        if isinstance(node, ast.Import):
            for a in node.names:
                aliases[a.asname or a.name.split(".")[0]] = a.name.split(".")[0]
        elif isinstance(node, ast.ImportFrom) and node.module:
            for a in node.names:
                aliases[a.asname or a.name] = f"{node.module}.{a.name}"
    return aliases


def analyze_python_source(source: str, path: str = "<memory>") -> tuple[list[AstFinding], set[str]]:
    """(dangerous-sink findings, set of called qualified symbols)."""
    try:
        tree = ast.parse(source)
    except SyntaxError:
        return [], set()
    aliases = _collect_aliases(tree)
    analyzer = _Analyzer(path, source.splitlines(), aliases)
    analyzer.visit(tree)
    return analyzer.findings, analyzer.calls


@dataclass
class SymbolIndex:
    """Called symbols + imported modules across a project tree."""

    calls: set[str] = field(default_factory=set)
    imports: set[str] = field(default_factory=set)
    findings: list[AstFinding] = field(default_factory=list)
    files_scanned: int = 0

    def calls_symbol(self, symbol: str) -> bool:
        """Advisory symbols come as 'mod.Class.method' or bare 'func'."""
        if symbol in self.calls:
            return True
        tail = symbol.split(".")[-1]
        return any(c == tail or c.endswith("." + tail) for c in self.calls)

    def imports_module(self, module: str) -> bool:
        head = module.split(".")[0].split("/")[-1]
        return any(i == head or i.startswith(head + ".") for i in self.imports)


def build_symbol_index(root: str | Path, max_files: int = 2000) -> SymbolIndex:
    root = Path(root)
    index = SymbolIndex()
    files = [root] if root.is_file() else sorted(root.rglob("*.py"))[:max_files]
    for f in files:
        try:
            source = f.read_text(errors="replace")
        except OSError:
            continue
        findings, calls = analyze_python_source(source, str(f))
        index.findings.extend(findings)
        index.calls |= calls
        try:
            tree = ast.parse(source)
            for node in ast.walk(tree):
                if isinstance(node, ast.Import):
                    index.imports |= {a.name.split(".")[0] for a in node.names}
                elif isinstance(node, ast.ImportFrom) and node.module:
                    index.imports.add(node.module.split(".")[0])
        except SyntaxError:
            continue
        index.files_scanned += 1
    # polyglot sources (JS/TS, Go, Java, Ruby, Rust, PHP, C#) feed the same
    # index so symbol reachability spans the whole repository
    from agentbom_amd.scan.ast_polyglot import extend_symbol_index

    extend_symbol_index(index, root, max_files=max_files)
    return index


def apply_symbol_reachability(report, index: SymbolIndex) -> int:
    """Stamp symbol_reachability on blast radii from the advisory's
    affected_symbols vs the project's call index; rescore after.

    function_reachable: an advisory symbol is actually called;
    package_reachable: the package is imported but no named symbol called;
    unreachable: package never imported.  Advisories without symbol data
    stay None (never fabricate reachability)."""
    updated = 0
    for br in report.blast_radii:
        pkg_module = br.package.name.replace("-", "_")
        symbols = br.vulnerability.affected_symbols
        if symbols:
            matched = [s for s in symbols if index.calls_symbol(s)]
            if matched:
                br.symbol_reachability = "function_reachable"
                br.reachable_affected_symbols = matched
            elif index.imports_module(pkg_module):
                br.symbol_reachability = "package_reachable"
            else:
                br.symbol_reachability = "unreachable"
        elif not index.imports_module(pkg_module) and index.files_scanned:
            # no symbol data: only the import-level downgrade is safe
            br.symbol_reachability = "unreachable"
        else:
            continue
        br.calculate_risk_score()
        updated += 1
    report.blast_radii.sort(key=lambda b: -b.risk_score)
    return updated


def ast_finding_to_finding(f: AstFinding) -> Finding:
    return Finding(
        finding_type=FindingType.SAST,
        source=FindingSource.SAST,
        asset=Asset(name=Path(f.file).name, asset_type="source_file", location=f.file),
        severity=f.severity,
        title=f"{f.category}: {f.call} at {Path(f.file).name}:{f.line}",
        description=f"{f.call} ({f.cwe}) in {f.entrypoint}: {f.snippet}",
        cwe_ids=[f.cwe],
        evidence=f.to_dict(),
        is_actionable=f.severity in ("critical", "high"),
        id=stable_id("ast-flow", f.file, str(f.line), f.entrypoint, f.call, f.category),
    )
