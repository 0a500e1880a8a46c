"""Build the in-tree HIP library for gfx950 (and its host entry points).

Called by __graft_entry__.build() and on-demand by pathway_amd.ops.  Uses
explicit hipcc (no hipify, no multi-backend dispatch): the kernels are
native HIP/CDNA4 source.  The built .so lives in-tree so the gpurun
snapshot ships it.
"""

from __future__ import annotations

import os
import subprocess
import sys

_THIS = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(_THIS, "csrc", "hip", "pw_kernels.hip")
OUT = os.path.join(_THIS, "libpwhip.so")
IO_SRC = os.path.join(_THIS, "csrc", "cpp", "pw_io.cpp")
IO_OUT = os.path.join(_THIS, "libpwio.so")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def needs_build() -> bool:
    if not os.path.exists(OUT):
        return True
    srcs = [SRC, os.path.join(_THIS, "csrc", "xxhash_common.h")]
    out_m = os.path.getmtime(OUT)
    return any(os.path.getmtime(s) > out_m for s in srcs)


def build(verbose: bool = True) -> str:
    if not needs_build():
        return OUT
    hipcc = os.environ.get("HIPCC", "hipcc")
    cmd = [
        hipcc,
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        SRC,
        "-o",
        OUT,
    ]
    if verbose:
        print("[pathway_amd.ops]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return OUT


def build_io(verbose: bool = True) -> str:
    """Host-only native IO scanner (g++; reference data_storage analog)."""
    if os.path.exists(IO_OUT) and os.path.getmtime(IO_OUT) >= os.path.getmtime(
        IO_SRC
    ):
        return IO_OUT
    cmd = ["g++", "-O3", "-std=c++17", "-fPIC", "-shared", IO_SRC, "-o", IO_OUT]
    if verbose:
        print("[pathway_amd.ops]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return IO_OUT


if __name__ == "__main__":
    build()
    build_io()
