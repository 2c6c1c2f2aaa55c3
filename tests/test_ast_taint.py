"""Structural taint-lite pass (Go/JS): functions, args, escalation, guards."""

from __future__ import annotations

import textwrap

from agentbom_amd.scan.ast_polyglot import analyze_source
from agentbom_amd.scan.ast_taint import (
    analyze_taint,
    balanced_segment,
    call_edges,
    extract_calls,
    extract_functions,
    split_top_level_args,
)

GO = textwrap.dedent('''\
    package main

    import (
        "net/http"
        "os/exec"
    )

    func handler(w http.ResponseWriter, r *http.Request) {
        cmd := r.URL.Query().Get("cmd")
        exec.Command("sh", "-c", cmd).Run()
    }

    func safe() {
        exec.Command("ls", "-l").Run()
    }

    func guarded(input string) {
        if !validateInput(input) {
            return
        }
        exec.Command("convert", input).Run()
    }
''')

JS = textwrap.dedent('''\
    const { exec } = require("child_process");

    function run(req, res) {
        exec("convert " + req.query.file, () => {});
    }

    const ok = () => {
        exec("ls -l");
    };
''')


class TestLexing:
    def test_split_top_level_args(self):
        assert split_top_level_args('a, f(b, c), {d: 1, e: 2}, "x,y"') == \
            ["a", "f(b, c)", "{d: 1, e: 2}", '"x,y"']

    def test_balanced_go_backtick(self):
        src = 'f(`raw ) \\ string`, x)'
        seg, end = balanced_segment(src, 1, "go")
        assert seg == '(`raw ) \\ string`, x)'

    def test_balanced_js_template(self):
        src = 'f(`tpl ${g(1, ")")} end`, y)'
        seg, _ = balanced_segment(src, 1, "js")
        assert seg.endswith(", y)")


class TestFunctions:
    def test_go_functions_and_handler_shape(self):
        fns = {d.name: d for d in extract_functions(GO, "go")}
        assert set(fns) == {"handler", "safe", "guarded"}
        assert fns["handler"].handler_shaped
        assert not fns["safe"].handler_shaped
        assert fns["guarded"].params == ["input"]

    def test_js_functions(self):
        fns = {d.name: d for d in extract_functions(JS, "js")}
        assert "run" in fns and "ok" in fns
        assert fns["run"].params == ["req", "res"]

    def test_call_enclosure(self):
        calls = extract_calls(GO, "go")
        by_callee = {}
        for c in calls:
            by_callee.setdefault(c.callee, []).append(c)
        assert {c.caller for c in by_callee["exec.Command"]} == \
            {"handler", "safe", "guarded"}


class TestTaint:
    def test_untrusted_escalates_guard_deescalates(self):
        fs = {(f.caller, f.severity, bool(f.untrusted_args), f.guarded)
              for f in analyze_taint(GO, "m.go", "go")
              if f.call == "exec.Command"}
        assert ("handler", "critical", True, False) in fs  # query input
        assert ("safe", "medium", False, False) in fs      # literals only
        # guarded: untrusted param but validation branch above
        assert ("guarded", "high", True, True) in fs

    def test_js_request_arg(self):
        fs = [f for f in analyze_taint(JS, "a.js", "js") if f.call == "exec"]
        sev = {f.caller: f.severity for f in fs}
        assert sev["run"] == "critical" and sev["ok"] == "medium"

    def test_call_edges(self):
        edges = call_edges(GO, "go")
        assert ("handler", "exec.Command") in edges
        assert ("guarded", "validateInput") in edges


class TestIntegration:
    def test_polyglot_merge_escalates_regex_hit(self):
        findings, calls = analyze_source(GO, "m.go", "go")
        cmd = [f for f in findings if f.category == "command-injection"
               and f.entrypoint == "handler"]
        assert cmd and cmd[0].severity == "critical" and cmd[0].tainted

    def test_polyglot_adds_path_sink(self):
        src = ('package main\nimport "os"\n'
               'func f(name string) { os.ReadFile(name) }\n')
        findings, _ = analyze_source(src, "p.go", "go")
        assert any(f.category == "path-traversal" for f in findings)
