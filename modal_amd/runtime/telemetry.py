"""Import-time telemetry: per-module import timing over a Unix socket.

Parity: /root/reference/py/modal/_runtime/telemetry.py:16-100 — a meta-path
hook timestamps every module import and emits length-prefixed JSON messages
over the socket named by ``$MODAL_AMD_TELEMETRY_SOCKET`` (reference:
``$MODAL_TELEMETRY_SOCKET``, installed before any other import,
_container_entrypoint.py:12-16).
"""

from __future__ import annotations

import json
import os
import socket
import struct
import sys
import threading
import time
from typing import Any, Optional

_MESSAGE_HEADER = struct.Struct("<I")


class TelemetryEmitter:
    def __init__(self, socket_path: str):
        self.socket_path = socket_path
        self._sock: Optional[socket.socket] = None
        self._lock = threading.Lock()

    def _connect(self) -> Optional[socket.socket]:
        if self._sock is None:
            try:
                s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
                s.connect(self.socket_path)
                self._sock = s
            except OSError:
                return None
        return self._sock

    def emit(self, message: dict) -> None:
        sock = self._connect()
        if sock is None:
            return
        payload = json.dumps(message).encode()
        try:
            with self._lock:
                sock.sendall(_MESSAGE_HEADER.pack(len(payload)) + payload)
        except OSError:
            self._sock = None


_orig_import = None
_emitter: Optional[TelemetryEmitter] = None


def _timed_import(name: str, *args: Any, **kwargs: Any) -> Any:
    already = name in sys.modules
    t0 = time.monotonic()
    module = _orig_import(name, *args, **kwargs)
    dt = time.monotonic() - t0
    if not already and _emitter is not None and name in sys.modules:
        _emitter.emit(
            {
                "event": "module_load",
                "name": name,
                "latency_us": int(dt * 1e6),
                "timestamp": time.time(),
            }
        )
    return module


def instrument_imports(socket_path: Optional[str] = None) -> None:
    """Install the import-timing hook (idempotent)."""
    global _orig_import, _emitter
    socket_path = socket_path or os.environ.get("MODAL_AMD_TELEMETRY_SOCKET")
    if not socket_path or _orig_import is not None:
        return
    import builtins

    _emitter = TelemetryEmitter(socket_path)
    _orig_import = builtins.__import__
    builtins.__import__ = _timed_import


def uninstrument_imports() -> None:
    global _orig_import, _emitter
    if _orig_import is not None:
        import builtins

        builtins.__import__ = _orig_import
        _orig_import = None
        _emitter = None
