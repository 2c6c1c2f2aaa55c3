"""Polyglot source analysis: JS/TS, Go, Java, Ruby, Rust, PHP, C#.

Reference parity: src/agent_bom/ast_{js_ts,go,java,ruby,rust,php,csharp}.py
— per-language dangerous-sink detection plus call/import extraction that
feeds the same :class:`~agentbom_amd.scan.ast_analysis.SymbolIndex` the
Python analyzer fills, so symbol-level CVE reachability works across a
polyglot repository.  The non-Python analyzers are pattern-structural
(no parser dependency): sink rules anchor on call syntax, and
function/import extraction uses each language's declaration grammar.
"""

from __future__ import annotations

import re
from pathlib import Path
from typing import Any

from agentbom_amd.scan.ast_analysis import AstFinding, SymbolIndex

# (regex, call, category, severity, cwe) — per language
_SINKS: dict[str, list[tuple] ] = {
    "js": [
        (re.compile(r"\beval\s*\("), "eval", "code-injection", "critical", "CWE-95"),
        (re.compile(r"new\s+Function\s*\("), "Function", "code-injection", "critical", "CWE-95"),
        (re.compile(r"child_process[.)]\s*(?:exec|execSync|spawn)\s*\(|"
                    r"\b(?:exec|execSync)\s*\("), "child_process.exec",
         "command-injection", "critical", "CWE-78"),
        (re.compile(r"\.innerHTML\s*="), "innerHTML", "xss", "high", "CWE-79"),
        (re.compile(r"document\.write\s*\("), "document.write", "xss", "high", "CWE-79"),
        (re.compile(r"\bvm\.(?:runInContext|runInNewContext|runInThisContext)\s*\("),
         "vm.runIn*Context", "code-injection", "critical", "CWE-95"),
        (re.compile(r"JSON\.parse\s*\(\s*(?:await\s+)?(?:req|request|res)\b"),
         "JSON.parse(request)", "deserialization", "medium", "CWE-502"),
    ],
    "go": [
        (re.compile(r"exec\.Command(?:Context)?\s*\("), "exec.Command",
         "command-injection", "high", "CWE-78"),
        (re.compile(r"syscall\.Exec\s*\("), "syscall.Exec",
         "command-injection", "critical", "CWE-78"),
        (re.compile(r"template\.HTML\s*\("), "template.HTML", "xss", "high", "CWE-79"),
        (re.compile(r"\bunsafe\.Pointer\b"), "unsafe.Pointer",
         "memory-safety", "medium", "CWE-119"),
        (re.compile(r"sql\.(?:Open|Query|Exec)[^;\n]*\+\s*"), "sql string concat",
         "sql-injection", "high", "CWE-89"),
        (re.compile(r"gob\.NewDecoder\s*\("), "gob.Decode",
         "deserialization", "medium", "CWE-502"),
    ],
    "java": [
        (re.compile(r"Runtime\.getRuntime\(\)\s*\.\s*exec\s*\("), "Runtime.exec",
         "command-injection", "critical", "CWE-78"),
        (re.compile(r"new\s+ProcessBuilder\s*\("), "ProcessBuilder",
         "command-injection", "high", "CWE-78"),
        (re.compile(r"new\s+ObjectInputStream\s*\(|\.readObject\s*\("),
         "ObjectInputStream.readObject", "deserialization", "critical", "CWE-502"),
        (re.compile(r"ScriptEngine\w*\s*\.\s*eval\s*\(|getEngineByName"),
         "ScriptEngine.eval", "code-injection", "high", "CWE-95"),
        (re.compile(r"createStatement\(\)[^;\n]*executeQuery\s*\([^)]*\+"),
         "Statement concat", "sql-injection", "high", "CWE-89"),
        (re.compile(r"XMLDecoder\s*\("), "XMLDecoder", "deserialization",
         "critical", "CWE-502"),
    ],
    "ruby": [
        (re.compile(r"(?<![\w.])eval\s*\("), "eval", "code-injection", "critical", "CWE-95"),
        (re.compile(r"(?<![\w.])system\s*\(|`[^`\n]+`|%x\{"), "system/backtick",
         "command-injection", "high", "CWE-78"),
        (re.compile(r"Marshal\.load\s*\("), "Marshal.load",
         "deserialization", "critical", "CWE-502"),
        (re.compile(r"YAML\.load\s*\((?![^)]*safe)"), "YAML.load",
         "deserialization", "high", "CWE-502"),
        (re.compile(r"send\s*\(\s*params"), "send(params)",
         "code-injection", "high", "CWE-95"),
    ],
    "rust": [
        (re.compile(r"\bunsafe\s*\{"), "unsafe block", "memory-safety",
         "medium", "CWE-119"),
        (re.compile(r"Command::new\s*\("), "Command::new",
         "command-injection", "medium", "CWE-78"),
        (re.compile(r"std::mem::transmute"), "mem::transmute",
         "memory-safety", "high", "CWE-843"),
    ],
    "php": [
        (re.compile(r"(?<![\w$])eval\s*\("), "eval", "code-injection", "critical", "CWE-95"),
        (re.compile(r"(?<![\w$])(?:system|exec|shell_exec|passthru|popen)\s*\("),
         "system/exec", "command-injection", "critical", "CWE-78"),
        (re.compile(r"unserialize\s*\("), "unserialize",
         "deserialization", "critical", "CWE-502"),
        (re.compile(r"include(?:_once)?\s*\(\s*\$_"), "include($_...)",
         "file-inclusion", "critical", "CWE-98"),
        (re.compile(r"\$_(?:GET|POST|REQUEST)\[[^\]]+\][^;\n]*(?:mysql_query|->query)\b|"
                    r"(?:mysql_query|->query)\s*\([^)]*\$_"), "query($_...)",
         "sql-injection", "high", "CWE-89"),
    ],
    "csharp": [
        (re.compile(r"Process\.Start\s*\("), "Process.Start",
         "command-injection", "high", "CWE-78"),
        (re.compile(r"BinaryFormatter\b"), "BinaryFormatter",
         "deserialization", "critical", "CWE-502"),
        (re.compile(r"new\s+SqlCommand\s*\([^)]*\+"), "SqlCommand concat",
         "sql-injection", "high", "CWE-89"),
    ],
}

# function/method declarations — feed the symbol index for reachability
_DEFS: dict[str, re.Pattern] = {
    "js": re.compile(r"(?:function\s+([A-Za-z_$][\w$]*)|"
                     r"(?:const|let|var)\s+([A-Za-z_$][\w$]*)\s*=\s*"
                     r"(?:async\s*)?(?:function|\())"),
    "go": re.compile(r"^func\s+(?:\([^)]*\)\s*)?([A-Za-z_]\w*)\s*\(", re.M),
    "java": re.compile(r"(?:public|private|protected|static|\s)+[\w<>\[\]]+\s+"
                       r"([a-z]\w*)\s*\([^;{]*\)\s*\{"),
    "ruby": re.compile(r"^\s*def\s+(?:self\.)?([a-z_]\w*[?!]?)", re.M),
    "rust": re.compile(r"\bfn\s+([a-z_]\w*)\s*[(<]"),
    "php": re.compile(r"\bfunction\s+([A-Za-z_]\w*)\s*\("),
    "csharp": re.compile(r"(?:public|private|protected|internal|static|\s)+"
                         r"[\w<>\[\]]+\s+([A-Z]\w*)\s*\([^;{]*\)\s*\{"),
}

# call sites (best-effort): identifier followed by '('
_CALL_RE = re.compile(r"\b([A-Za-z_][\w.]{1,60})\s*\(")
_CALL_STOPWORDS = frozenset({
    "if", "for", "while", "switch", "return", "catch", "function", "fn",
    "def", "new", "class", "match", "loop", "print", "println",
})

_IMPORTS: dict[str, re.Pattern] = {
    "js": re.compile(r"(?:import\s+(?:[\w{},*\s]+\s+from\s+)?|require\s*\(\s*)"
                     r"['\"]([^'\"]+)['\"]"),
    "go": re.compile(r"(?:^\s*(?:import\s+)?(?:\w+\s+)?\"([^\"]+)\")", re.M),
    "java": re.compile(r"^import\s+(?:static\s+)?([\w.]+)", re.M),
    "ruby": re.compile(r"^\s*require(?:_relative)?\s+['\"]([^'\"]+)['\"]", re.M),
    "rust": re.compile(r"^\s*use\s+([\w:]+)", re.M),
    "php": re.compile(r"^use\s+([\w\\]+)", re.M),
    "csharp": re.compile(r"^using\s+([\w.]+)\s*;", re.M),
}

_LANG_BY_SUFFIX = {
    ".js": "js", ".jsx": "js", ".mjs": "js", ".cjs": "js",
    ".ts": "js", ".tsx": "js",
    ".go": "go", ".java": "java", ".kt": "java",
    ".rb": "ruby", ".rs": "rust", ".php": "php", ".cs": "csharp",
}

_COMMENT_LINE = {
    "js": "//", "go": "//", "java": "//", "rust": "//", "csharp": "//",
    "ruby": "#", "php": "//",
}




# languages whose line comments start with '#' (others use '//')
_HASH_COMMENT_LANGS = {"ruby", "php"}  # php supports both # and //


def strip_comments_and_strings(text: str, lang: str) -> str:
    """Blank out string literals and comments, preserving line structure.

    A small shared lexer (state machine) rather than per-line regex: sinks
    inside strings ("use eval() carefully"), line comments and /* block
    comments */ no longer false-positive, and block comments spanning
    lines are handled.  Stripped characters become spaces so line/column
    numbers survive."""
    out = list(text)
    i = 0
    n = len(text)
    hash_comments = lang in _HASH_COMMENT_LANGS

    def blank(a: int, b: int) -> None:
        for j in range(a, min(b, n)):
            if out[j] != "\n":
                out[j] = " "

    while i < n:
        c = text[i]
        nxt = text[i + 1] if i + 1 < n else ""
        if c == "/" and nxt == "/" and lang != "ruby":
            j = text.find("\n", i)
            j = n if j == -1 else j
            blank(i, j)
            i = j
        elif c == "#" and hash_comments:
            j = text.find("\n", i)
            j = n if j == -1 else j
            blank(i, j)
            i = j
        elif c == "/" and nxt == "*":
            j = text.find("*/", i + 2)
            j = n if j == -1 else j + 2
            blank(i, j)
            i = j
        elif c in ("\"", "'", "`"):
            q = c
            j = i + 1
            while j < n:
                if text[j] == "\\":
                    j += 2
                    continue
                if text[j] == q or (q != "`" and text[j] == "\n"):
                    break
                j += 1
            blank(i + 1, j)  # keep the quotes, blank the contents
            i = min(j + 1, n)
        else:
            i += 1
    return "".join(out)


# user-input source identifiers per language family (taint heuristic)
_TAINT_SOURCES = re.compile(
    r"\b(req|request|params|query|body|argv|args|stdin|input|r\.URL|"
    r"r\.Form|getenv|environ|os\.Args|\$_GET|\$_POST|\$_REQUEST|"
    r"event|payload|untrusted)\b", re.IGNORECASE)


def language_for(path: str | Path) -> str | None:
    return _LANG_BY_SUFFIX.get(Path(path).suffix.lower())


def _enclosing_function(lines_before: list[str], lang: str) -> str:
    pat = _DEFS.get(lang)
    if pat is None:
        return "<module>"
    for line in reversed(lines_before):
        m = pat.search(line)
        if m:
            return next((g for g in m.groups() if g), "<module>")
    return "<module>"


def analyze_source(text: str, path: str, lang: str) -> tuple[list[AstFinding], set[str]]:
    """(findings, called symbols) for one non-Python source file.

    Analysis runs over a comment/string-stripped view (lexer pass) so
    sinks quoted in strings or commented out never fire; snippets come
    from the ORIGINAL text for readable evidence."""
    findings: list[AstFinding] = []
    calls: set[str] = set()
    code = strip_comments_and_strings(text, lang)
    lines = code.splitlines()
    orig_lines = text.splitlines()
    for ln, line in enumerate(lines, start=1):
        for m in _CALL_RE.finditer(line):
            name = m.group(1)
            if name not in _CALL_STOPWORDS:
                calls.add(name)
        for pattern, call, category, severity, cwe in _SINKS.get(lang, []):
            sm = pattern.search(line)
            if sm:
                tail = line[sm.start():]
                # taint: a user-input source identifier in the call, or a
                # non-literal argument (anything but blanked quotes/consts)
                arg_region = tail[: tail.find(")") + 1 or len(tail)]
                tainted = bool(_TAINT_SOURCES.search(tail)) or bool(
                    re.search(r"\(\s*[A-Za-z_][\w.\[\]]*", arg_region))
                findings.append(AstFinding(
                    file=path, line=ln, call=call, category=category,
                    severity=severity, cwe=cwe,
                    entrypoint=_enclosing_function(lines[:ln], lang),
                    snippet=(orig_lines[ln - 1].strip()[:160]
                             if ln <= len(orig_lines) else ""),
                    tainted=tainted,
                ))
    for dm in _DEFS.get(lang, re.compile(r"$^")).finditer(code):
        name = next((g for g in dm.groups() if g), None)
        if name:
            calls.add(name)

    # taint-lite structural pass (Go/JS): argument-level untrusted-identifier
    # escalation + guard de-escalation over the SAME stripped view.  Upgrades
    # severity of regex hits on the same line; adds sink calls the line
    # patterns missed (method-style spellings).
    if lang in ("go", "js"):
        from agentbom_amd.scan.ast_taint import analyze_taint

        seen = {(f.line, f.category) for f in findings}
        by_line_cat = {(f.line, f.category): f for f in findings}
        for tf in analyze_taint(code, path, lang):
            key = (tf.line, tf.category)
            if key in seen:
                hit = by_line_cat[key]
                if tf.untrusted_args and not tf.guarded:
                    hit.severity = "critical"
                    hit.tainted = True
                continue
            findings.append(AstFinding(
                file=path, line=tf.line, call=tf.call, category=tf.category,
                severity=tf.severity, cwe=tf.cwe, entrypoint=tf.caller,
                snippet=(orig_lines[tf.line - 1].strip()[:160]
                         if tf.line <= len(orig_lines) else ""),
                tainted=bool(tf.untrusted_args)))
    return findings, calls


def extract_imports(text: str, lang: str) -> set[str]:
    pat = _IMPORTS.get(lang)
    if pat is None:
        return set()
    out = set()
    for m in pat.finditer(text):
        mod = m.group(1)
        out.add(mod.split("/")[-1].split(".")[0] if lang == "go"
                else mod.split(".")[0].split("\\")[0].split("::")[0]
                .split("/")[0])
    return out


def extend_symbol_index(index: SymbolIndex, root: str | Path,
                        max_files: int = 2000) -> int:
    """Add non-Python sources under ``root`` to an existing SymbolIndex."""
    root = Path(root)
    if root.is_file():
        files = [root]
    else:
        files = sorted(p for p in root.rglob("*")
                       if p.suffix.lower() in _LANG_BY_SUFFIX
                       and not {"node_modules", ".git", "vendor", "target",
                                "dist", "build"} & set(p.parts))[:max_files]
    added = 0
    for f in files:
        lang = language_for(f)
        if lang is None:
            continue
        try:
            text = f.read_text(errors="replace")
        except OSError:
            continue
        findings, calls = analyze_source(text, str(f), lang)
        index.findings.extend(findings)
        index.calls |= calls
        index.imports |= extract_imports(text, lang)
        index.files_scanned += 1
        added += 1
    return added
