"""bench.py driver contract: one JSON line, required keys, sane values.

The round driver depends on this exact interface (single + distributed
launch); these tests run the real script at toy sizes on CPU.
"""

from __future__ import annotations

import json
import os
import socket
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parents[1]
ARGS = ["--device", "cpu", "--packages", "8000", "--agents", "200",
        "--servers", "800", "--name-catalog", "1500", "--arena-windows",
        "3000", "--steps", "2", "--warmup", "1", "--queries", "2"]

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _parse(stdout: str) -> dict:
    lines = [ln for ln in stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, f"expected exactly one JSON line, got {len(lines)}"
    return json.loads(lines[0])


@pytest.mark.timeout(300)
def test_single_process_contract():
    out = subprocess.run(
        [sys.executable, str(ROOT / "bench.py"), *ARGS],
        capture_output=True, text=True, timeout=240, cwd=str(ROOT))
    assert out.returncode == 0, out.stderr[-800:]
    doc = _parse(out.stdout)
    assert REQUIRED <= set(doc)
    assert doc["metric"] == "findings/sec"
    assert doc["n_gpus"] == 1 and doc["steps"] == 2 and doc["warmup"] == 1
    assert doc["higher_is_better"] is True and doc["scaling"] == "weak"
    assert doc["data"] == "synthetic"
    assert doc["value"] > 0 and doc["ms_per_step"] > 0
    cfg = doc["config"]
    assert cfg["parallelism"] == "single"
    assert cfg["findings_per_step"] > 0
    assert cfg["p50_blast_query_ms"] is not None


@pytest.mark.timeout(420)
def test_distributed_launch_contract():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), str(ROOT / "bench.py"), "--gpus", "2",
         *ARGS],
        capture_output=True, text=True, timeout=360, cwd=str(ROOT),
        env={**os.environ, "MASTER_ADDR": "127.0.0.1"})
    assert out.returncode == 0, out.stderr[-1200:]
    doc = _parse(out.stdout)
    assert doc["n_gpus"] == 2
    assert doc["config"]["parallelism"] == "hashpart2"
    # whole-job aggregate: both ranks' findings summed over ONE estate
    assert doc["config"]["findings_per_step"] > 0
