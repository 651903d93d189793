"""Tracked raw device memory: the snapshot-visible hipMalloc surface.

The round-1 review (Missing #4) flagged that the torch-gc-walk snapshot
misses raw hipMalloc allocations. ROCm has no cuda-checkpoint, so raw device
memory that must survive a snapshot is allocated THROUGH this registry
(csrc/rawmem.hip): each buffer has a stable string key; snapshot enumerates
and pages D2H; restore re-allocates by key (same key, new pointer — keys are
the durable identity, pointers are process-local).

``fidelity_report()`` compares the device's used bytes against
torch-reserved + tracked to surface UNTRACKED allocations — the honest
failure mode the reference gets for free from cuda-checkpoint's whole-PID
scope (reference gpu_memory_snapshot.py:158-300).
"""

from __future__ import annotations

import ctypes
from typing import Optional

from . import load_lib


class RawDeviceBuffer:
    """A tracked hipMalloc allocation addressed by a stable key."""

    def __init__(self, key: str, size: int):
        lib = load_lib(required=True)
        self.key = key.encode()
        self.size = size
        ptr = ctypes.c_void_p()
        rc = lib.ma_tracked_alloc(self.key, ctypes.c_ulonglong(size), ctypes.byref(ptr))
        if rc != 0:
            raise MemoryError(f"ma_tracked_alloc({key}, {size}) -> hipError {rc}")
        self.ptr = ptr.value

    def write(self, data: bytes) -> None:
        assert len(data) == self.size
        lib = load_lib(required=True)
        rc = lib.ma_tracked_write(self.key, data)
        if rc != 0:
            raise RuntimeError(f"ma_tracked_write -> hipError {rc}")

    def read(self) -> bytes:
        lib = load_lib(required=True)
        out = ctypes.create_string_buffer(self.size)
        rc = lib.ma_tracked_read(self.key, out)
        if rc != 0:
            raise RuntimeError(f"ma_tracked_read -> hipError {rc}")
        return out.raw

    def free(self) -> None:
        lib = load_lib(required=True)
        lib.ma_tracked_free(self.key)
        self.ptr = None


def _iter_keys(lib) -> list[str]:
    n = lib.ma_tracked_count()
    keys = []
    buf = ctypes.create_string_buffer(256)
    for i in range(n):
        if lib.ma_tracked_key(i, buf, 256) == 0:
            keys.append(buf.value.decode())
    return keys


def snapshot_all() -> dict[str, bytes]:
    """Page every tracked allocation D2H: {key: raw bytes}."""
    lib = load_lib()
    if lib is None:
        return {}
    out: dict[str, bytes] = {}
    for key in _iter_keys(lib):
        ptr = ctypes.c_void_p()
        size = ctypes.c_ulonglong()
        if lib.ma_tracked_lookup(key.encode(), ctypes.byref(ptr), ctypes.byref(size)) != 0:
            continue
        host = ctypes.create_string_buffer(size.value)
        if lib.ma_tracked_read(key.encode(), host) == 0:
            out[key] = host.raw
    return out


def release_all() -> None:
    """Free device memory of every tracked buffer, keeping the registry rows
    (page-out half of scaledown)."""
    lib = load_lib()
    if lib is None:
        return
    for key in _iter_keys(lib):
        lib.ma_tracked_release(key.encode())


def restore_all(saved: dict[str, bytes]) -> None:
    """(Re-)allocate by key and page the saved bytes back H2D. Works both
    in-process (after release_all) and in a FRESH process (keys absent)."""
    lib = load_lib(required=True)
    existing = set(_iter_keys(lib))
    for key, data in saved.items():
        kb = key.encode()
        if key in existing:
            ptr = ctypes.c_void_p()
            rc = lib.ma_tracked_reacquire(kb, ctypes.byref(ptr))
        else:
            ptr = ctypes.c_void_p()
            rc = lib.ma_tracked_alloc(kb, ctypes.c_ulonglong(len(data)), ctypes.byref(ptr))
        if rc != 0:
            raise RuntimeError(f"restore alloc {key} -> hipError {rc}")
        rc = lib.ma_tracked_write(kb, data)
        if rc != 0:
            raise RuntimeError(f"restore write {key} -> hipError {rc}")


def lookup(key: str) -> Optional[tuple[int, int]]:
    """(device_ptr, size) of a tracked buffer, or None."""
    lib = load_lib()
    if lib is None:
        return None
    ptr = ctypes.c_void_p()
    size = ctypes.c_ulonglong()
    if lib.ma_tracked_lookup(key.encode(), ctypes.byref(ptr), ctypes.byref(size)) != 0:
        return None
    return (ptr.value or 0, size.value)


def fidelity_report() -> dict:
    """How much device memory would a snapshot MISS?

    used - torch_reserved - tracked = untracked bytes (allocations made by
    neither torch nor the tracked allocator; a process-grade snapshot of
    them is impossible without driver support, so surface the number)."""
    lib = load_lib()
    report = {"used": 0, "torch_reserved": 0, "tracked": 0, "untracked": 0}
    if lib is None:
        return report
    used = ctypes.c_ulonglong()
    total = ctypes.c_ulonglong()
    if lib.ma_hip_live_bytes(ctypes.byref(used), ctypes.byref(total)) != 0:
        return report
    report["used"] = used.value
    try:
        import torch

        if torch.cuda.is_available():
            report["torch_reserved"] = torch.cuda.memory_reserved()
    except Exception:
        pass
    report["tracked"] = lib.ma_tracked_total_bytes()
    # context/runtime overhead makes `used` a loose upper bound; anything
    # beyond ~1 GiB of slack is a real untracked allocation
    slack = 1 << 30
    report["untracked"] = max(
        0, report["used"] - report["torch_reserved"] - report["tracked"] - slack
    )
    return report


def _bind(lib) -> None:
    """ctypes signatures (called from ops.__init__ after dlopen)."""
    u64 = ctypes.c_ulonglong
    lib.ma_tracked_alloc.argtypes = [ctypes.c_char_p, u64, ctypes.c_void_p]
    lib.ma_tracked_free.argtypes = [ctypes.c_char_p]
    lib.ma_tracked_lookup.argtypes = [ctypes.c_char_p, ctypes.c_void_p, ctypes.c_void_p]
    lib.ma_tracked_key.argtypes = [ctypes.c_int, ctypes.c_char_p, ctypes.c_int]
    lib.ma_tracked_read.argtypes = [ctypes.c_char_p, ctypes.c_char_p]
    lib.ma_tracked_write.argtypes = [ctypes.c_char_p, ctypes.c_char_p]
    lib.ma_tracked_release.argtypes = [ctypes.c_char_p]
    lib.ma_tracked_reacquire.argtypes = [ctypes.c_char_p, ctypes.c_void_p]
    lib.ma_hip_live_bytes.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
    lib.ma_tracked_total_bytes.restype = u64
