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

    def print_step(self, message: str) -> None:
        """A completed step line (parity: OutputManager.step_completed —
        the reference's green-check progress lines)."""
        if self._console is not None:
            self._console.print(f"[green]\N{HEAVY CHECK MARK}[/green] {message}")
        else:
            sys.stdout.write(f"+ {message}\n")
            sys.stdout.flush()

    def make_map_progress(self, description: str = "Running map") -> "MapProgress":
        """Live completed/submitted progress for Function.map fan-outs
        (parity: the reference's function-call progress display)."""
        return MapProgress(self._console, description)

    def print_log(self, entry: dict) -> None:
        data = entry.get("data", "")
        stream = sys.stderr if entry.get("fd") == 2 else sys.stdout
        if self.show_timestamps:
            import datetime

            ts = datetime.datetime.fromtimestamp(entry.get("ts", 0)).strftime("%H:%M:%S")
            data = "".join(f"{ts} {line}\n" for line in data.splitlines())
        if self._console is not None and entry.get("task_id"):
            # rich TTY: dim per-task prefix, stderr tinted (parity: the
            # reference's colored container-log rendering, _output/rich.py)
            tag = entry["task_id"][-6:]
            style = "red" if entry.get("fd") == 2 else ""
            for line in data.splitlines():
                self._console.print(
                    f"[dim]\\[{tag}][/dim] {line}",
                    style=style,
                    highlight=False,
                    markup=True,
                )
            return
        stream.write(data)
        stream.flush()

    @contextlib.contextmanager
    def status(self, message: str) -> Any:
        """Spinner for a long-running step (image build, object creation);
        plain begin/end lines off-TTY (parity: reference status.py)."""
        cm = None
        if self._console is not None:
            try:
                cm = self._console.status(message)
                cm.__enter__()
            except Exception:
                cm = None
        if cm is None:
            sys.stdout.write(f"... {message}\n")
            sys.stdout.flush()
        try:
            yield self
        finally:
            if cm is not None:
                try:
                    cm.__exit__(None, None, None)
                except Exception:
                    pass


class MapProgress:
    """Progress bar for a map fan-out: rich on a TTY, silent otherwise.
    ``update`` is cheap enough to call per output batch."""

    def __init__(self, console: Any, description: str):
        self._progress = None
        self._task_id = None
        self.completed = 0
        self.submitted = 0
        if console is not None:
            try:
                from rich.progress import (
                    BarColumn, MofNCompleteColumn, Progress, SpinnerColumn,
                    TaskProgressColumn, TimeElapsedColumn,
                )

                self._progress = Progress(
                    SpinnerColumn(), "[progress.description]{task.description}",
                    BarColumn(), MofNCompleteColumn(), TaskProgressColumn(),
                    TimeElapsedColumn(), console=console, transient=True,
                )
                self._progress.start()
                self._task_id = self._progress.add_task(description, total=None)
            except Exception:
                self._progress = None

    def update(self, completed: int, submitted: int, done_submitting: bool = False) -> None:
        self.completed = completed
        self.submitted = submitted
        if self._progress is not None:
            self._progress.update(
                self._task_id, completed=completed,
                total=submitted if done_submitting else None,
            )

    def close(self) -> None:
        if self._progress is not None:
            try:
                self._progress.stop()
            except Exception:
                pass
            self._progress = None


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
