"""rocTX tracing ranges for rocprofv3 timelines.

SURVEY §5.1's MI355X obligation: emit rocprof/rocTX ranges around kernel
launches and data-plane operations so `rocprofv3 --sys-trace` timelines show
framework phases next to GPU kernels. Enabled by config
``runtime_perf_record`` / ``MODAL_AMD_RUNTIME_PERF_RECORD=1`` (parity:
the reference's runtime_perf_record passthrough, config.py:321); no-op when
libroctx64 is unavailable or tracing is off.
"""

from __future__ import annotations

import contextlib
import ctypes
import os
from typing import Iterator, Optional

_lib: Optional[ctypes.CDLL] = None
_checked = False


def _load() -> Optional[ctypes.CDLL]:
    global _lib, _checked
    if _checked:
        return _lib
    _checked = True
    if os.environ.get("MODAL_AMD_RUNTIME_PERF_RECORD", "").lower() not in ("1", "true"):
        from ..config import config

        if not config.get("runtime_perf_record"):
            return None
    for path in ("libroctx64.so", "/opt/rocm/lib/libroctx64.so"):
        try:
            lib = ctypes.CDLL(path)
            lib.roctxRangePushA.argtypes = [ctypes.c_char_p]
            lib.roctxRangePushA.restype = ctypes.c_int
            lib.roctxRangePop.restype = ctypes.c_int
            lib.roctxMarkA.argtypes = [ctypes.c_char_p]
            _lib = lib
            return lib
        except OSError:
            continue
    return None


def enabled() -> bool:
    return _load() is not None


@contextlib.contextmanager
def trace_range(name: str) -> Iterator[None]:
    lib = _load()
    if lib is None:
        yield
        return
    lib.roctxRangePushA(name.encode())
    try:
        yield
    finally:
        lib.roctxRangePop()


def mark(name: str) -> None:
    lib = _load()
    if lib is not None:
        lib.roctxMarkA(name.encode())
