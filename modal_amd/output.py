"""Output management: worker/app log rendering in the client terminal.

Parity: the reference's OutputManager + ``modal.enable_output``
(/root/reference/py/modal/_output/manager.py:23, output.py). Uses ``rich``
when a TTY is present, plain writes otherwise.
"""

from __future__ import annotations

import contextlib
import sys
import threading
from typing import Any, Optional

_manager_lock = threading.Lock()
_manager: Optional["OutputManager"] = None


class OutputManager:
    def __init__(self, show_timestamps: bool = False):
        self.show_timestamps = show_timestamps
        self._console = None
        try:
            if sys.stdout.isatty():
                from rich.console import Console

                self._console = Console()
        except Exception:
            pass

    def print(self, message: str) -> None:
        if self._console is not None:
            self._console.print(message)
        else:
            sys.stdout.write(message + "\n")
            sys.stdout.flush()

    def print_log(self, entry: dict) -> None:
        data = entry.get("data", "")
        stream = sys.stderr if entry.get("fd") == 2 else sys.stdout
        if self.show_timestamps:
            import datetime

            ts = datetime.datetime.fromtimestamp(entry.get("ts", 0)).strftime("%H:%M:%S")
            data = "".join(f"{ts} {line}\n" for line in data.splitlines())
        stream.write(data)
        stream.flush()


def get_output_manager() -> Optional[OutputManager]:
    return _manager


@contextlib.contextmanager
def enable_output(show_progress: bool = True, show_timestamps: bool = False) -> Any:
    """Stream app/worker logs to this terminal (parity: modal.enable_output)."""
    global _manager
    with _manager_lock:
        prev = _manager
        _manager = OutputManager(show_timestamps=show_timestamps)
    try:
        yield _manager
    finally:
        with _manager_lock:
            _manager = prev
