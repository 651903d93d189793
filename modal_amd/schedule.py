"""Schedules: Cron and Period.

Parity: /root/reference/py/modal/schedule.py:12,60 — schedule objects attached
to ``@app.function(schedule=...)``; the local scheduler runs them with a cron
evaluator (see modal_amd/scheduler/cron.py).
"""

from __future__ import annotations

from .exception import InvalidError


class Schedule:
    """Base class for function schedules."""


class Cron(Schedule):
    """Standard 5-field cron schedule, evaluated in UTC."""

    def __init__(self, cron_string: str, timezone: str = "UTC"):
        fields = cron_string.split()
        if len(fields) != 5:
            raise InvalidError(f"Cron string must have 5 fields (got {cron_string!r})")
        self.cron_string = cron_string
        self.timezone = timezone

    def __repr__(self) -> str:
        return f"Cron({self.cron_string!r})"


class Period(Schedule):
    """Fixed-interval schedule."""

    def __init__(
        self,
        years: int = 0,
        months: int = 0,
        weeks: int = 0,
        days: int = 0,
        hours: int = 0,
        minutes: int = 0,
        seconds: float = 0,
    ):
        self.total_seconds = (
            years * 365 * 86400
            + months * 30 * 86400
            + weeks * 7 * 86400
            + days * 86400
            + hours * 3600
            + minutes * 60
            + seconds
        )
        if self.total_seconds <= 0:
            raise InvalidError("Period must be positive")

    def __repr__(self) -> str:
        return f"Period({self.total_seconds}s)"
