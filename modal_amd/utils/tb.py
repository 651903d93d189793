"""Remote traceback surgery.

Parity: the reference re-synthesizes remote tracebacks locally and suppresses
framework-internal frames (/root/reference/py/modal/_traceback.py:27,61,141).
Here the exception's traceback is reduced to user-code frames before pickling
so the client re-raises with a clean, combined stack.
"""

from __future__ import annotations

import os
import types
from typing import Optional

_PKG_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _is_internal(frame_file: str) -> bool:
    return frame_file.startswith(_PKG_ROOT) or "asyncio" in frame_file.split(os.sep)


def clean_traceback(exc: BaseException) -> None:
    """Drop framework/asyncio frames from exc.__traceback__ in place."""
    tb = exc.__traceback__
    frames = []
    while tb is not None:
        filename = tb.tb_frame.f_code.co_filename
        if not _is_internal(filename):
            frames.append(tb)
        tb = tb.tb_next
    # relink remaining frames
    new_tb: Optional[types.TracebackType] = None
    for tb in reversed(frames):
        new_tb = types.TracebackType(new_tb, tb.tb_frame, tb.tb_lasti, tb.tb_lineno)
    exc.__traceback__ = new_tb
    if exc.__cause__ is not None and exc.__cause__ is not exc:
        clean_traceback(exc.__cause__)
    if exc.__context__ is not None and exc.__context__ is not exc:
        clean_traceback(exc.__context__)
