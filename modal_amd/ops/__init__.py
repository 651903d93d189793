"""HIP/CDNA4 data-plane ops (gfx950): batched SHA-256, segment pack/unpack.

Built in-tree by ``build.py`` (hipcc --offload-arch=gfx950) into
``libmodal_amd_ops.so`` next to this file; loaded via ctypes so the hot path
has no extension-module overhead. On a GPU machine a missing library is a
hard error (no silent eager fallback); on CPU-only machines the pure-Python
reference implementations serve.
"""

from __future__ import annotations

import ctypes
import os
import threading
from typing import Optional

_LIB_NAME = "libmodal_amd_ops.so"
_lib_lock = threading.Lock()
_lib: Optional[ctypes.CDLL] = None
_lib_error: Optional[str] = None


def lib_path() -> str:
    return os.path.join(os.path.dirname(os.path.abspath(__file__)), _LIB_NAME)


def load_lib(required: bool = False) -> Optional[ctypes.CDLL]:
    """Load the HIP ops library. required=True raises on failure."""
    global _lib, _lib_error
    with _lib_lock:
        if _lib is not None:
            return _lib
        path = lib_path()
        if not os.path.exists(path):
            _lib_error = f"HIP ops library not built: {path} (run __graft_entry__.build())"
            if required:
                raise RuntimeError(_lib_error)
            return None
        try:
            lib = ctypes.CDLL(path)
            lib.ma_sha256_many.restype = ctypes.c_int
            lib.ma_sha256_many.argtypes = [ctypes.c_void_p] * 4 + [ctypes.c_int, ctypes.c_void_p]
            lib.ma_sha256_many2.restype = ctypes.c_int
            lib.ma_sha256_many2.argtypes = lib.ma_sha256_many.argtypes
            lib.ma_pack_segments.restype = ctypes.c_int
            lib.ma_pack_segments.argtypes = (
                [ctypes.c_void_p] * 5 + [ctypes.c_int, ctypes.c_long, ctypes.c_void_p]
            )
            lib.ma_unpack_segments.restype = ctypes.c_int
            lib.ma_unpack_segments.argtypes = lib.ma_pack_segments.argtypes
            lib.ma_lz4_compress.restype = ctypes.c_int
            lib.ma_lz4_compress.argtypes = [
                ctypes.c_void_p, ctypes.c_long, ctypes.c_void_p, ctypes.c_void_p,
                ctypes.c_int, ctypes.c_int, ctypes.c_void_p,
            ]
            lib.ma_lz4_decompress.restype = ctypes.c_int
            lib.ma_lz4_decompress.argtypes = [
                ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                ctypes.c_long, ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
            ]
            from .rawmem import _bind as _bind_rawmem

            _bind_rawmem(lib)
            _lib = lib
            return lib
        except OSError as exc:
            _lib_error = f"Failed to load {path}: {exc}"
            if required:
                raise RuntimeError(_lib_error) from exc
            return None


def gpu_available() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


def require_native_on_gpu() -> Optional[ctypes.CDLL]:
    """On a GPU machine the native library must be present — fail loudly."""
    if gpu_available():
        return load_lib(required=True)
    return load_lib(required=False)
