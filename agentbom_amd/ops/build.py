"""Build the gfx950 HIP engine in-tree.

``python -m agentbom_amd.ops.build`` (or ``__graft_entry__.build()``) compiles
``csrc/*.hip`` with hipcc into ``agentbom_amd/ops/_abom_gpu.so``.  hipcc
cross-compiles for gfx950 without a GPU present; the built .so travels with
the source tree to the GPU box.
"""

from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
CSRC = OPS_DIR / "csrc"
SO_PATH = OPS_DIR / "_abom_gpu.so"
ARCH = os.environ.get("AGENT_BOM_GPU_ARCH", "gfx950")


def _hipcc() -> str:
    for cand in (os.environ.get("HIPCC"), "/opt/rocm/bin/hipcc", "hipcc"):
        if not cand:
            continue
        try:
            subprocess.run([cand, "--version"], capture_output=True, check=True)
            return cand
        except (OSError, subprocess.CalledProcessError):
            continue
    raise RuntimeError("hipcc not found — install ROCm or set HIPCC")


def needs_rebuild() -> bool:
    if not SO_PATH.exists():
        return True
    so_mtime = SO_PATH.stat().st_mtime
    return any(p.stat().st_mtime > so_mtime for p in CSRC.glob("*") if p.is_file())


def build(force: bool = False, verbose: bool = True) -> Path:
    if not force and not needs_rebuild():
        return SO_PATH
    sources = sorted(str(p) for p in CSRC.glob("*.hip"))
    cmd = [
        _hipcc(),
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        f"-I{CSRC}",
        *sources,
        "-o",
        str(SO_PATH),
    ]
    if verbose:
        print("[abom build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(SO_PATH)
