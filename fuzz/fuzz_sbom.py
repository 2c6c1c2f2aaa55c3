#!/usr/bin/env python3
"""Fuzz harness: SBOM + lockfile parsing (reference fuzz/fuzz_sbom.py).

Feeds mutated JSON/text at ``ingest_sbom`` and the manifest parsers; the
contract is fail-soft (a list back, or a controlled skip) — any other
exception is a crash.  atheris when present, deterministic driver otherwise.
"""

from __future__ import annotations

import json
import random
import sys


def one_input(data: bytes) -> None:
    from agentbom_amd.scan import parsers as P
    import json as _json

    from agentbom_amd.scan.sbom_ingest import parse_cyclonedx, parse_spdx

    text = data.decode("utf-8", errors="replace")[:4000]
    try:
        doc = _json.loads(text)
    except ValueError:
        doc = None
    if isinstance(doc, dict):
        for parse in (parse_cyclonedx, parse_spdx):
            try:
                out = parse(doc)
                assert isinstance(out, list)
            except (ValueError, KeyError, TypeError, AttributeError):
                continue
    for fn in (P.parse_package_lock, P.parse_requirements_txt,
               P.parse_go_mod, P.parse_gemfile_lock):
        try:
            out = fn(text, "<fuzz>")
            assert isinstance(out, list)
        except (ValueError, KeyError, TypeError, AttributeError):
            continue


SEEDS = [
    {"bomFormat": "CycloneDX", "specVersion": "1.6",
     "components": [{"type": "library", "name": "a", "version": "1"}]},
    {"spdxVersion": "SPDX-2.3", "packages": [{"name": "b", "versionInfo": "2"}]},
    {"lockfileVersion": 3, "packages": {"node_modules/x": {"version": "1.0"}}},
]


def _mutate(doc_text: str, rng: random.Random) -> str:
    ops = [
        lambda s: s[:rng.randint(0, len(s))],
        lambda s: s.replace('"', "'", rng.randint(1, 3)),
        lambda s: s + rng.choice(["}", "]", '"', "\x00", "é中"]),
        lambda s: s.replace(":", "=", 1),
        lambda s: s[::-1][:200],
    ]
    return rng.choice(ops)(doc_text)


def main() -> int:
    try:
        import atheris  # type: ignore

        atheris.Setup(sys.argv, one_input)
        atheris.Fuzz()
        return 0
    except ImportError:
        pass
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 5000
    rng = random.Random(99)
    for i in range(n):
        base = json.dumps(rng.choice(SEEDS))
        for _ in range(rng.randint(0, 4)):
            base = _mutate(base, rng)
        one_input(base.encode())
    print(f"fuzz_sbom: {n} iterations, no crashes")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
