"""Schedule runner: Cron/Period evaluation + invocation of scheduled functions.

Parity: the reference attaches Schedule protos to functions and the cloud
fires them (/root/reference/py/modal/schedule.py). Locally a scheduler task
scans deployed functions and enqueues invocations when due.
"""

from __future__ import annotations

import asyncio
import time
from datetime import datetime, timezone
from typing import TYPE_CHECKING, Optional

if TYPE_CHECKING:
    from .core import Scheduler


def _match_field(expr: str, value: int, base: int = 0) -> bool:
    for part in expr.split(","):
        part = part.strip()
        step = 1
        if "/" in part:
            part, _, step_s = part.partition("/")
            step = int(step_s)
        if part in ("*", ""):
            if (value - base) % step == 0:
                return True
            continue
        if "-" in part:
            lo, _, hi = part.partition("-")
            if int(lo) <= value <= int(hi) and (value - int(lo)) % step == 0:
                return True
            continue
        if int(part) == value:
            return True
    return False


def cron_matches(cron_string: str, dt: datetime) -> bool:
    """Standard 5-field cron: minute hour day-of-month month day-of-week.

    POSIX day semantics: when BOTH day-of-month and day-of-week are
    restricted (neither is ``*``), the date matches if EITHER matches —
    ``0 0 1,15 * 1`` fires on the 1st, the 15th, and every Monday."""
    minute, hour, dom, month, dow = cron_string.split()
    dow_value = dt.weekday() + 1 if dt.weekday() < 6 else 0  # 0=Sunday
    dom_star = dom.strip() == "*"
    dow_star = dow.strip() == "*"
    if dom_star or dow_star:
        day_ok = _match_field(dom, dt.day, base=1) and _match_field(dow, dow_value)
    else:
        day_ok = _match_field(dom, dt.day, base=1) or _match_field(dow, dow_value)
    return (
        _match_field(minute, dt.minute)
        and _match_field(hour, dt.hour)
        and _match_field(month, dt.month, base=1)
        and day_ok
    )


class ScheduleRunner:
    def __init__(self, scheduler: "Scheduler", tick: float = 1.0):
        self.scheduler = scheduler
        self.tick = tick
        self._task: Optional[asyncio.Task] = None
        self._last_period_run: dict[str, float] = {}
        self._last_cron_minute: dict[str, str] = {}

    def start(self) -> None:
        if self._task is None:
            self._task = asyncio.get_running_loop().create_task(self._loop())

    def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            self._task = None

    async def _loop(self) -> None:
        while True:
            try:
                await self._scan()
            except asyncio.CancelledError:
                raise
            except Exception:
                pass
            await asyncio.sleep(self.tick)

    async def _scan(self) -> None:
        now = time.time()
        dt = datetime.now(timezone.utc)
        minute_key = dt.strftime("%Y%m%d%H%M")
        for fid, fdef in list(self.scheduler.functions.items()):
            sched = fdef.schedule
            if not sched:
                continue
            app = self.scheduler.apps.get(fdef.app_id)
            if app is None or app.state == "stopped":
                continue
            due = False
            if "period" in sched:
                last = self._last_period_run.get(fid, now)
                if fid not in self._last_period_run:
                    self._last_period_run[fid] = now
                    continue
                if now - last >= sched["period"]:
                    due = True
                    self._last_period_run[fid] = now
            elif "cron" in sched:
                if self._last_cron_minute.get(fid) != minute_key and cron_matches(
                    sched["cron"], dt
                ):
                    due = True
                    self._last_cron_minute[fid] = minute_key
            if due:
                from .._serialization import serialize

                payload = serialize(("P", ((), {})))
                resp = await self.scheduler.function_map(
                    function_id=fid, kind="spawn", pipelined_inputs=[{"payload": payload}]
                )
                await self.scheduler.function_finish_inputs(
                    function_call_id=resp["function_call_id"]
                )
