"""Build modal_amd/_core.so (pybind11, C++20) in-tree."""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

_THIS = os.path.dirname(os.path.abspath(__file__))
_REPO = os.path.dirname(_THIS)
_OUT = os.path.join(_REPO, "modal_amd", f"_core{sysconfig.get_config_var('EXT_SUFFIX')}")


def needs_build() -> bool:
    src = os.path.join(_THIS, "core.cpp")
    return not os.path.exists(_OUT) or os.path.getmtime(src) > os.path.getmtime(_OUT)


def build(verbose: bool = True, force: bool = False) -> str:
    if not force and not needs_build():
        return _OUT
    import pybind11

    cmd = [
        "g++",
        "-O3",
        "-std=c++20",
        "-shared",
        "-fPIC",
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        os.path.join(_THIS, "core.cpp"),
        "-o",
        _OUT,
    ]
    if verbose:
        print("[modal_amd._core]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True, capture_output=not verbose)
    return _OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
