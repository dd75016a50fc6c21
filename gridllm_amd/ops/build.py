"""In-tree build of the HIP kernel library for gfx950.

Builds csrc/gridllm_kernels.hip -> gridllm_amd/ops/libgridllm_hip.so with a
direct hipcc invocation (no hipify, no CUDA-compat layer). The .so is
git-ignored but ships with the repo snapshot to GPU boxes.
"""

from __future__ import annotations

import os
import subprocess
import sys

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
REPO_ROOT = os.path.dirname(os.path.dirname(PKG_DIR))
SRC = os.path.join(REPO_ROOT, "csrc", "gridllm_kernels.hip")
OUT = os.path.join(PKG_DIR, "libgridllm_hip.so")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def needs_build() -> bool:
    if not os.path.exists(OUT):
        return True
    return os.path.getmtime(SRC) > os.path.getmtime(OUT)


def build(force: bool = False) -> str:
    if not force and not needs_build():
        return OUT
    cmd = [
        HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-shared", "-fPIC", SRC, "-o", OUT,
    ]
    proc = subprocess.run(cmd, capture_output=True, text=True)
    if proc.returncode != 0:
        raise RuntimeError(
            f"hipcc build failed (rc={proc.returncode}):\n{proc.stderr[-4000:]}")
    return OUT


if __name__ == "__main__":
    path = build(force="--force" in sys.argv)
    print(f"built {path}")
