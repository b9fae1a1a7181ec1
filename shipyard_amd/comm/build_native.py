"""Build the native RCCL benchmark binary (gfx950)."""
from __future__ import annotations

import os
import shutil
import subprocess
import sys
from pathlib import Path

HERE = Path(__file__).resolve().parent
CSRC = HERE / "csrc"
BIN = HERE / "rccl_allreduce_bench"
ARCH = os.environ.get("SHIPYARD_GPU_ARCH", "gfx950")


def build(force: bool = False, verbose: bool = True) -> Path:
    src = CSRC / "rccl_bench.cpp"
    if BIN.exists() and not force and \
            BIN.stat().st_mtime >= src.stat().st_mtime:
        return BIN
    hipcc = shutil.which("hipcc") or "/opt/rocm/bin/hipcc"
    cmd = [
        hipcc, f"--offload-arch={ARCH}", "-O2", "-std=c++17",
        str(src), "-I/opt/rocm/include",
        "-L/opt/rocm/lib", "-lrccl", "-o", str(BIN),
    ]
    if verbose:
        print("[shipyard_amd.comm.build_native]", " ".join(cmd),
              file=sys.stderr)
    subprocess.run(cmd, check=True)
    return BIN


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(BIN)
