"""Build the dpo_amd HIP extension in-tree.

Compiles dpo_amd/ops/hip/*.hip with hipcc for gfx950 (MI355X) into
dpo_amd/ops/hip/libdpo_hip_ops.so. Cross-compiles fine on a box without
a GPU. Run:  python -m dpo_amd.ops.build
"""
from __future__ import annotations

import os
import subprocess
import sys

HIP_DIR = os.path.dirname(os.path.abspath(__file__)) + "/hip"
SOURCES = ["dpo_ops.hip", "dpo_partition.cpp"]
OUT = os.path.join(HIP_DIR, "libdpo_hip_ops.so")
ARCH = os.environ.get("DPO_GFX_ARCH", "gfx950")


def _mtime(p):
    try:
        return os.path.getmtime(p)
    except OSError:
        return 0.0


def needs_build() -> bool:
    out_t = _mtime(OUT)
    return any(_mtime(os.path.join(HIP_DIR, s)) >= out_t for s in SOURCES)


def build(force: bool = False, verbose: bool = True) -> str:
    if not force and not needs_build():
        return OUT
    hipcc = os.environ.get("HIPCC", "hipcc")
    cmd = [hipcc, f"--offload-arch={ARCH}", "-O3", "-std=c++17",
           "-shared", "-fPIC", "-o", OUT]
    cmd += [os.path.join(HIP_DIR, s) for s in SOURCES]
    if verbose:
        print("+", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(OUT)
