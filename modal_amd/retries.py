"""User-facing retry policy.

Parity: modal.Retries (/root/reference/py/modal/retries.py:12) — delays
clamped to [1 s, 24 h] (reference :8-9), exponential backoff.
"""

from __future__ import annotations

from datetime import timedelta
from typing import Union

from .exception import InvalidError

MIN_DELAY_S = 1.0  # parity: retries.py:8
MAX_DELAY_S = 24 * 60 * 60.0  # parity: retries.py:9


def _to_seconds(value: Union[int, float, timedelta]) -> float:
    if isinstance(value, timedelta):
        return value.total_seconds()
    return float(value)


class Retries:
    """Retry policy for function invocations.

    Usage::

        @app.function(retries=Retries(max_retries=3, initial_delay=1.0))
        def flaky(): ...
    """

    def __init__(
        self,
        *,
        max_retries: int,
        backoff_coefficient: float = 2.0,
        initial_delay: Union[int, float, timedelta] = 1.0,
        max_delay: Union[int, float, timedelta] = 60.0,
    ):
        initial_delay_s = _to_seconds(initial_delay)
        max_delay_s = _to_seconds(max_delay)
        if max_retries < 0 or max_retries > 10:
            raise InvalidError(f"max_retries must be between 0 and 10 (got {max_retries})")
        if backoff_coefficient < 1.0 or backoff_coefficient > 10.0:
            raise InvalidError("backoff_coefficient must be between 1.0 and 10.0")
        if not (MIN_DELAY_S <= initial_delay_s <= MAX_DELAY_S):
            raise InvalidError(f"initial_delay must be between {MIN_DELAY_S}s and {MAX_DELAY_S}s")
        if not (MIN_DELAY_S <= max_delay_s <= MAX_DELAY_S):
            raise InvalidError(f"max_delay must be between {MIN_DELAY_S}s and {MAX_DELAY_S}s")
        self.max_retries = max_retries
        self.backoff_coefficient = backoff_coefficient
        self.initial_delay = initial_delay_s
        self.max_delay = max_delay_s

    def _to_policy_dict(self) -> dict:
        return {
            "max_retries": self.max_retries,
            "backoff_coefficient": self.backoff_coefficient,
            "initial_delay_ms": int(self.initial_delay * 1000),
            "max_delay_ms": int(self.max_delay * 1000),
        }

    def __repr__(self) -> str:
        return (
            f"Retries(max_retries={self.max_retries}, "
            f"backoff_coefficient={self.backoff_coefficient}, "
            f"initial_delay={self.initial_delay}, max_delay={self.max_delay})"
        )
