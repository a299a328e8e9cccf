"""Build driver for the in-tree CDNA4 GPU library.

Builds ``covalent_ssh_plugin_amd/ops/libcsp_gpu.so`` from
``ops/hip/csp_gpu.hip`` with hipcc for **gfx950 only** (MI355X; no
multi-arch fatbin, no CUDA path).  hipcc cross-compiles without a GPU, so
this runs in CPU-only CI; the built .so travels to the GPU box in the
repo snapshot and to the remote execution host via the executor's
content-addressed lib provisioning (ssh.py _provision_gpu_lib).
"""

from __future__ import annotations

import os
import shutil
import subprocess
import sys
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
SRC = OPS_DIR / "hip" / "csp_gpu.hip"
OUT = OPS_DIR / "libcsp_gpu.so"

HIPCC = shutil.which("hipcc") or "/opt/rocm/bin/hipcc"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def needs_build() -> bool:
    return not OUT.exists() or OUT.stat().st_mtime < SRC.stat().st_mtime


def build(force: bool = False, verbose: bool = True) -> Path:
    if not force and not needs_build():
        return OUT
    cmd = [
        HIPCC,
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        str(SRC),
        "-o",
        str(OUT),
    ]
    if verbose:
        print("+", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(OUT)
