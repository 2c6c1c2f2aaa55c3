#!/usr/bin/env python3
"""Fuzz harness: the policy expression language.

Reference parity: .clusterfuzzlite/ + fuzz/fuzz_policy.py — a standalone
coverage-guided entry point for the policy evaluator.  Runs under atheris
when available, and falls back to a deterministic random driver in the
air-gapped image (`python fuzz/fuzz_policy.py [iterations]`).

Contract under test: ``evaluate_expression`` returns bool or raises
ValueError — anything else is a crash.
"""

from __future__ import annotations

import random
import string
import sys

from agentbom_amd.scan.policy import evaluate_expression


class _Br:
    class vulnerability:
        severity = type("S", (), {"value": "high"})()
        is_kev = True
        cvss_score = 8.1
        epss_score = 0.5
        id = "CVE-X"

    class package:
        is_malicious = False
        name = "p"
        ecosystem = "PyPI"

    risk_score = 7.5
    reachability = "reachable"
    impact_category = "code-execution"


ALPHABET = string.ascii_letters + string.digits + " .<>=!()'\"&|_-"


def one_input(data: bytes) -> None:
    try:
        expr = data.decode("utf-8", errors="replace")[:200]
    except Exception:
        return
    try:
        out = evaluate_expression(expr, _Br())
    except ValueError:
        return
    assert isinstance(out, bool), (expr, out)


def main() -> int:
    try:
        import atheris  # type: ignore

        atheris.Setup(sys.argv, one_input)
        atheris.Fuzz()
        return 0
    except ImportError:
        pass
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 20000
    rng = random.Random(1234)
    fragments = ["severity", ">=", "high", "and", "or", "not", "is_kev",
                 "risk_score", ">", "9", "(", ")", "cvss_score", "epss_score",
                 "reachability", "==", "'reachable'", "impact_category"]
    for i in range(n):
        if i % 3 == 0:
            expr = " ".join(rng.choices(fragments, k=rng.randint(1, 12)))
        else:
            expr = "".join(rng.choices(ALPHABET, k=rng.randint(0, 80)))
        one_input(expr.encode())
    print(f"fuzz_policy: {n} iterations, no crashes")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
