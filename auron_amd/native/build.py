"""Build the native HIP engine in-tree: auron_amd/native/libauron_hip.so.

gfx950-only by design (MI355X/CDNA4). hipcc cross-compiles on CPU-only
hosts, so this runs in no-GPU CI as the does-it-build check.
"""
from __future__ import annotations

import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")
OUT = os.path.join(HERE, "libauron_hip.so")
SOURCES = [os.path.join(CSRC, "kernels.hip"), os.path.join(CSRC, "parquet.hip"),
           os.path.join(CSRC, "agg.hip"),
           os.path.join(CSRC, "fused.hip")]
HIPCC = os.environ.get("HIPCC", "hipcc")
ARCH = os.environ.get("AURON_OFFLOAD_ARCH", "gfx950")


def needs_build() -> bool:
    if not os.path.exists(OUT):
        return True
    out_m = os.path.getmtime(OUT)
    return any(os.path.getmtime(s) > out_m for s in SOURCES)


def build(force: bool = False, verbose: bool = True) -> str:
    if not force and not needs_build():
        return OUT
    cmd = [
        HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-fPIC", "-shared", "-fvisibility=hidden",
        *SOURCES, "-o", OUT,
    ]
    if verbose:
        print("[auron build]", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(OUT)
