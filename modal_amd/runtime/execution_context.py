"""Execution context: contextvars for the current input / function call.

Parity: /root/reference/py/modal/_runtime/execution_context.py:13,41,60,87-110
(``current_input_id``, ``current_function_call_id``, ``is_local``).
"""

from __future__ import annotations

import contextvars
import os
from typing import Optional

_current_input_id: contextvars.ContextVar[Optional[str]] = contextvars.ContextVar(
    "modal_amd_input_id", default=None
)
_current_function_call_id: contextvars.ContextVar[Optional[str]] = contextvars.ContextVar(
    "modal_amd_function_call_id", default=None
)


def current_input_id() -> Optional[str]:
    return _current_input_id.get()


def current_function_call_id() -> Optional[str]:
    return _current_function_call_id.get()


def _set_current_context(input_id: Optional[str], function_call_id: Optional[str]) -> tuple:
    return (
        _current_input_id.set(input_id),
        _current_function_call_id.set(function_call_id),
    )


def _reset_current_context(tokens: tuple) -> None:
    _current_input_id.reset(tokens[0])
    _current_function_call_id.reset(tokens[1])


def is_local() -> bool:
    """True when running in the user's own process (not inside a worker)."""
    return os.environ.get("MODAL_AMD_IS_REMOTE") != "1"


def interact() -> None:
    """No-op locally (reference: flips the container to interactive PTY mode)."""
