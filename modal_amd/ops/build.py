"""Build the HIP ops library in-tree with hipcc (gfx950 only, no multi-arch).

The built .so is git-ignored but ships to GPU boxes with the repo snapshot.
"""

from __future__ import annotations

import os
import shutil
import subprocess
import sys

_THIS_DIR = os.path.dirname(os.path.abspath(__file__))
_CSRC = os.path.join(_THIS_DIR, "csrc")
SOURCES = ["sha256.hip", "pack.hip", "lz4.hip", "rawmem.hip"]


def _hipcc() -> str:
    hipcc = shutil.which("hipcc") or "/opt/rocm/bin/hipcc"
    if not os.path.exists(hipcc):
        raise RuntimeError("hipcc not found; ROCm toolchain required to build modal_amd ops")
    return hipcc


def needs_build() -> bool:
    from . import lib_path

    out = lib_path()
    if not os.path.exists(out):
        return True
    out_mtime = os.path.getmtime(out)
    for src in SOURCES:
        if os.path.getmtime(os.path.join(_CSRC, src)) > out_mtime:
            return True
    return False


def build(verbose: bool = True, force: bool = False) -> str:
    from . import lib_path

    out = lib_path()
    if not force and not needs_build():
        return out
    cmd = [
        _hipcc(),
        "--offload-arch=gfx950",
        "-O3",
        "-std=c++17",
        "-shared",
        "-fPIC",
        *[os.path.join(_CSRC, src) for src in SOURCES],
        "-o",
        out,
    ]
    if verbose:
        print("[modal_amd.ops]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True, capture_output=not verbose)
    return out


if __name__ == "__main__":
    build()
