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


def extract_frames(exc: BaseException) -> list:
    """Summarize user frames of exc.__traceback__ for the wire (pickle
    drops tracebacks; the reference vendors tblib for this —
    _traceback.py:27 extract_traceback)."""
    out = []
    tb = exc.__traceback__
    while tb is not None:
        code = tb.tb_frame.f_code
        out.append((code.co_filename, tb.tb_lineno, code.co_name))
        tb = tb.tb_next
    return out


_frame_cache: dict = {}


def _forge_frame(filename: str, name: str) -> Optional[object]:
    """Manufacture a frame whose code claims (filename, name) — the tblib
    technique: raise inside exec'd code compiled with that filename."""
    key = (filename, name)
    if key in _frame_cache:
        return _frame_cache[key]
    try:
        code = compile("raise ValueError()", filename, "exec")
        if name.isidentifier():
            code = code.replace(co_name=name)
        frame = None
        try:
            exec(code, {})
        except ValueError as e:
            frame = e.__traceback__.tb_next.tb_frame
        _frame_cache[key] = frame
        return frame
    except Exception:
        _frame_cache[key] = None
        return None


def forge_traceback(frames: list) -> Optional[types.TracebackType]:
    """Rebuild a TracebackType chain from extract_frames output. On a
    single node the filenames resolve locally, so linecache shows real
    source lines in the re-synthesized stack."""
    tb: Optional[types.TracebackType] = None
    for filename, lineno, name in reversed(frames):
        frame = _forge_frame(filename, name)
        if frame is None:
            continue
        try:
            tb = types.TracebackType(tb, frame, -1, lineno)
        except Exception:
            continue
    return tb


def attach_remote_frames(exc: BaseException) -> BaseException:
    """Client side: splice the worker-recorded frames back under exc so
    `raise` shows local frames then the remote user stack."""
    frames = getattr(exc, "__modal_amd_tb__", None)
    if frames:
        tb = forge_traceback(frames)
        if tb is not None:
            return exc.with_traceback(tb)
    return exc


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
