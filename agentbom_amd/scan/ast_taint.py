"""Structural taint-lite analysis for Go and JS/TS sources.

Reference parity: src/agent_bom/ast_go.py (function/param extraction,
top-level argument splitting, untrusted-identifier heuristics, guarded-call
detection) and ast/js_ts (the tree-sitter engine — unavailable offline, so
the JS path here is the same lexer-structural approach as Go).

What this adds over the pattern sinks in ast_polyglot.py:

- **function extraction** with parameter names and body spans (brace
  matching with language-correct string states: Go backtick raw strings,
  JS template literals);
- **call-site extraction** with balanced-paren argument capture and
  top-level argument splitting;
- **taint-lite verdicts**: a sink call whose arguments mention an
  untrusted-looking identifier (request/input/user/env/...) or a parameter
  of the enclosing HTTP-handler-shaped function escalates; a call guarded
  by a validation branch (if + validate/sanitize/allowlist) de-escalates;
- **intra-file call edges** (caller -> callee) so symbol-level CVE
  reachability can walk through polyglot code, not just observe leaf calls.

No parser dependency; all lexer-structural, deterministic.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Optional

_UNTRUSTED = re.compile(
    r"\b(req|request|input|query|body|payload|param|params|arg|args|user|"
    r"username|untrusted|external|form|header|headers|cookie|cookies|env|"
    r"stdin|argv|message|msg|prompt|data)\w*\b", re.IGNORECASE)

_GUARD = re.compile(
    r"\b(validat\w+|sanitiz\w+|escape\w*|allowlist|whitelist|is_safe|"
    r"issafe|clean\w*|shlex\.quote|filepath\.Clean)\b", re.IGNORECASE)

# handler-shaped functions: their params are untrusted by construction
_HANDLER_PARAM = re.compile(
    r"http\.ResponseWriter|\*http\.Request|\bctx\b|\breq\b|\bres\b|"
    r"\brequest\b|\bresponse\b|gin\.Context|fiber\.Ctx|echo\.Context")


def balanced_segment(source: str, open_index: int, lang: str,
                     open_char: str = "(", close_char: str = ")"
                     ) -> Optional[tuple[str, int]]:
    """Return (segment incl. delimiters, end_index) or None.

    String states: '"', "'" with backslash escapes; Go adds backtick raw
    strings (backslash literal); JS adds backtick template literals
    (escapes active).
    """
    if open_index < 0 or open_index >= len(source) \
            or source[open_index] != open_char:
        return None
    depth = 0
    quote = ""
    escaped = False
    raw_quote = lang == "go"  # backticks: no escapes in Go, escapes in JS
    i = open_index
    n = len(source)
    while i < n:
        ch = source[i]
        if quote:
            if not (raw_quote and quote == "`"):
                if escaped:
                    escaped = False
                    i += 1
                    continue
                if ch == "\\":
                    escaped = True
                    i += 1
                    continue
            if ch == quote:
                quote = ""
        elif ch in ('"', "'", "`"):
            quote = ch
        elif ch == open_char:
            depth += 1
        elif ch == close_char:
            depth -= 1
            if depth == 0:
                return source[open_index:i + 1], i + 1
        i += 1
    return None


def split_top_level_args(arg_body: str) -> list[str]:
    """Split 'a, f(b, c), {d: e}' on top-level commas only."""
    out: list[str] = []
    depth = 0
    quote = ""
    escaped = False
    cur: list[str] = []
    for ch in arg_body:
        if quote:
            cur.append(ch)
            if escaped:
                escaped = False
            elif ch == "\\" and quote != "`":
                escaped = True
            elif ch == quote:
                quote = ""
            continue
        if ch in ('"', "'", "`"):
            quote = ch
        elif ch in "([{":
            depth += 1
        elif ch in ")]}":
            depth -= 1
        elif ch == "," and depth == 0:
            out.append("".join(cur).strip())
            cur = []
            continue
        cur.append(ch)
    tail = "".join(cur).strip()
    if tail:
        out.append(tail)
    return out


@dataclass
class FunctionDecl:
    name: str
    params: list[str]
    start: int  # char offset of body "{"
    end: int    # char offset past closing "}"
    line: int
    handler_shaped: bool = False


@dataclass
class CallSite:
    callee: str
    args: list[str]
    line: int
    caller: str  # enclosing function name or "<module>"


_GO_FUNC = re.compile(
    r"^func\s+(?:\([^)]*\)\s+)?(?P<name>[A-Za-z_]\w*)\s*\(", re.MULTILINE)
_JS_FUNCS = [
    re.compile(r"\bfunction\s+(?P<name>[A-Za-z_$][\w$]*)\s*\("),
    re.compile(r"\b(?:const|let|var)\s+(?P<name>[A-Za-z_$][\w$]*)\s*=\s*"
               r"(?:async\s*)?(?:function\s*)?\("),
    re.compile(r"(?P<name>[A-Za-z_$][\w$]*)\s*:\s*(?:async\s*)?function\s*\("),
]
_CALL = re.compile(r"(?<![\w$.])(?P<name>[A-Za-z_$][\w$]*(?:\.[A-Za-z_$][\w$]*)*)\s*\(")

_NOT_CALLS = {"if", "for", "while", "switch", "return", "func", "function",
              "catch", "defer", "go", "new", "typeof", "await", "delete"}


def _param_names(params_segment: str, lang: str) -> list[str]:
    names: list[str] = []
    for part in split_top_level_args(params_segment):
        part = part.strip()
        if not part or part in ("...",):
            continue
        if lang == "go":
            # "name Type", "a, b Type", "name ...Type"
            first = part.split()[0].rstrip(",")
            if first and first not in ("_",):
                names.append(first)
        else:
            # JS: "name", "name = default", "{destructured}", "...rest"
            tok = re.match(r"(?:\.\.\.)?\s*([A-Za-z_$][\w$]*)", part)
            if tok:
                names.append(tok.group(1))
    return names


def extract_functions(text: str, lang: str) -> list[FunctionDecl]:
    decls: list[FunctionDecl] = []
    patterns = [_GO_FUNC] if lang == "go" else _JS_FUNCS if lang == "js" else []
    for pat in patterns:
        for m in pat.finditer(text):
            seg = balanced_segment(text, text.find("(", m.end() - 1), lang)
            if seg is None:
                continue
            params_raw, after = seg[0][1:-1], seg[1]
            # find the body "{" (skip JS arrow "=>", Go return types)
            brace = text.find("{", after)
            if brace < 0 or brace - after > 200:
                continue
            body = balanced_segment(text, brace, lang, "{", "}")
            if body is None:
                continue
            params = _param_names(params_raw, lang)
            decls.append(FunctionDecl(
                name=m.group("name"), params=params, start=brace,
                end=body[1], line=text.count("\n", 0, m.start()) + 1,
                handler_shaped=bool(_HANDLER_PARAM.search(params_raw))))
    decls.sort(key=lambda d: d.start)
    return decls


def extract_calls(text: str, lang: str,
                  functions: Optional[list[FunctionDecl]] = None
                  ) -> list[CallSite]:
    functions = functions if functions is not None \
        else extract_functions(text, lang)

    def enclosing(offset: int) -> str:
        best = "<module>"
        for d in functions:
            if d.start <= offset < d.end:
                best = d.name  # innermost wins (sorted by start)
        return best

    calls: list[CallSite] = []
    for m in _CALL.finditer(text):
        name = m.group("name")
        if name.split(".")[0] in _NOT_CALLS:
            continue
        seg = balanced_segment(text, m.end() - 1, lang)
        if seg is None:
            continue
        calls.append(CallSite(
            callee=name, args=split_top_level_args(seg[0][1:-1]),
            line=text.count("\n", 0, m.start()) + 1,
            caller=enclosing(m.start())))
    return calls


# sink call-name tables (canonical names after alias resolution is out of
# scope for the lexer pass; the common import spellings are listed directly)
_SINK_CALLS = {
    "go": {
        "exec.Command": ("command-injection", "CWE-78"),
        "exec.CommandContext": ("command-injection", "CWE-78"),
        "syscall.Exec": ("command-injection", "CWE-78"),
        "os.ReadFile": ("path-traversal", "CWE-22"),
        "os.Open": ("path-traversal", "CWE-22"),
        "ioutil.ReadFile": ("path-traversal", "CWE-22"),
        "template.HTML": ("xss", "CWE-79"),
        "template.JS": ("xss", "CWE-79"),
        "db.Query": ("sql-injection", "CWE-89"),
        "db.Exec": ("sql-injection", "CWE-89"),
    },
    "js": {
        "eval": ("code-injection", "CWE-95"),
        "Function": ("code-injection", "CWE-95"),
        "child_process.exec": ("command-injection", "CWE-78"),
        "child_process.execSync": ("command-injection", "CWE-78"),
        "exec": ("command-injection", "CWE-78"),
        "execSync": ("command-injection", "CWE-78"),
        "fs.readFile": ("path-traversal", "CWE-22"),
        "fs.readFileSync": ("path-traversal", "CWE-22"),
        "db.query": ("sql-injection", "CWE-89"),
    },
}


@dataclass
class TaintFinding:
    file: str
    line: int
    call: str
    category: str
    cwe: str
    severity: str
    caller: str
    untrusted_args: list[str] = field(default_factory=list)
    guarded: bool = False

    def to_dict(self) -> dict:
        return self.__dict__.copy()


def _line_of(text: str, line_no: int) -> str:
    lines = text.splitlines()
    return lines[line_no - 1] if 0 < line_no <= len(lines) else ""


def _is_guarded(text: str, call_line: int) -> bool:
    """Validation branch within the 4 lines above the call."""
    lines = text.splitlines()
    lo = max(0, call_line - 5)
    window = "\n".join(lines[lo:call_line])
    return bool(_GUARD.search(window))


_ASSIGN = re.compile(
    r"^\s*(?:var\s+|let\s+|const\s+)?([A-Za-z_$][\w$]*)\s*(?::=|=)\s*(.+)$")


def _tainted_locals(body: str, seeds: set[str]) -> set[str]:
    """Local names assigned from untrusted expressions (2 propagation
    rounds — enough for the assign-then-use chains a lexer pass can see)."""
    from agentbom_amd.utils import config as cfg

    tainted = set(seeds)
    for _ in range(max(1, cfg.TAINT_PROPAGATION_ROUNDS)):
        grew = False
        for line in body.splitlines():
            m = _ASSIGN.match(line)
            if not m or m.group(1) in tainted:
                continue
            rhs = m.group(2)
            rhs_ids = set(re.findall(r"[A-Za-z_$][\w$]*", rhs))
            if _UNTRUSTED.search(rhs) or (rhs_ids & tainted):
                tainted.add(m.group(1))
                grew = True
        if not grew:
            break
    return tainted


def analyze_taint(text: str, path: str, lang: str) -> list[TaintFinding]:
    """Sink calls with untrusted-argument escalation + guard de-escalation."""
    if lang not in _SINK_CALLS:
        return []
    functions = extract_functions(text, lang)
    by_name = {d.name: d for d in functions}
    taint_by_fn: dict[str, set[str]] = {}
    for d in functions:
        seeds = set(d.params) if d.handler_shaped else set()
        taint_by_fn[d.name] = _tainted_locals(text[d.start:d.end], seeds)
    findings: list[TaintFinding] = []
    for call in extract_calls(text, lang, functions):
        sink = _SINK_CALLS[lang].get(call.callee)
        if sink is None:
            # suffix match for method-style spellings (cmd.Run, pool.query)
            tail = call.callee.split(".")[-1]
            sink = next((v for k, v in _SINK_CALLS[lang].items()
                         if k.split(".")[-1] == tail and "." in call.callee
                         and "." in k), None)
        if sink is None:
            continue
        category, cwe = sink
        untrusted = []
        tainted_names = taint_by_fn.get(call.caller, set())
        for arg in call.args:
            ids = set(re.findall(r"[A-Za-z_$][\w$]*", arg))
            if _UNTRUSTED.search(arg) or (ids & tainted_names):
                untrusted.append(arg[:60])
        guarded = _is_guarded(text, call.line)
        if untrusted and not guarded:
            severity = "critical"
        elif untrusted:
            severity = "high"   # tainted but a validation branch precedes
        elif guarded:
            severity = "low"
        else:
            severity = "medium"
        findings.append(TaintFinding(
            file=path, line=call.line, call=call.callee, category=category,
            cwe=cwe, severity=severity, caller=call.caller,
            untrusted_args=untrusted, guarded=guarded))
    return findings


def call_edges(text: str, lang: str) -> set[tuple[str, str]]:
    """Intra-file (caller, callee) pairs for symbol reachability."""
    return {(c.caller, c.callee) for c in extract_calls(text, lang)}
