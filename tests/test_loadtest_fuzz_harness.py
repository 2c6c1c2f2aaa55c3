"""Distribution & ops: load-test driver + fuzz harnesses."""

import json
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parents[1]


class TestLoadtest:
    def test_in_process_read_heavy(self):
        sys.path.insert(0, str(ROOT / "deploy" / "loadtest"))
        try:
            from loadtest import make_inprocess_requester, run_load
        finally:
            sys.path.pop(0)
        req = make_inprocess_requester()
        summary = run_load(req, "read-heavy", vus=3, duration_s=1.5)
        assert summary["requests"] > 10
        assert summary["errors"] == 0
        assert summary["latency_ms"]["p95"] >= summary["latency_ms"]["p50"] >= 0
        assert set(summary["status_codes"]) <= {"200"}

    def test_scan_burst_hits_backpressure_or_succeeds(self):
        sys.path.insert(0, str(ROOT / "deploy" / "loadtest"))
        try:
            from loadtest import make_inprocess_requester, run_load
        finally:
            sys.path.pop(0)
        req = make_inprocess_requester()
        summary = run_load(req, "scan-burst", vus=4, duration_s=1.0)
        codes = {int(k) for k in summary["status_codes"]}
        # every submission either lands (201) or is throttled with a
        # Retry-After (429) — never 5xx
        assert codes <= {201, 429}, summary
        assert summary["errors"] == 0

    def test_cli_entrypoint(self):
        out = subprocess.run(
            [sys.executable, str(ROOT / "deploy" / "loadtest" / "loadtest.py"),
             "--in-process", "--vus", "2", "--duration", "1"],
            capture_output=True, text=True, timeout=120)
        assert out.returncode == 0, out.stderr[-500:]
        summary = json.loads(out.stdout.strip().splitlines()[-1])
        assert summary["scenario"] == "read-heavy"


class TestFuzzHarnesses:
    """The standalone harnesses (reference .clusterfuzzlite/ + fuzz/) must
    run their deterministic fallback clean."""

    def _run(self, name: str, iters: str):
        out = subprocess.run(
            [sys.executable, str(ROOT / "fuzz" / name), iters],
            capture_output=True, text=True, timeout=300)
        assert out.returncode == 0, out.stderr[-800:]
        assert "no crashes" in out.stdout

    def test_fuzz_policy(self):
        self._run("fuzz_policy.py", "2000")

    def test_fuzz_sbom(self):
        self._run("fuzz_sbom.py", "400")

    def test_fuzz_skill_parser(self):
        self._run("fuzz_skill_parser.py", "30")
