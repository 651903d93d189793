"""Structured log fetch/tail for apps and sandboxes.

Parity: /root/reference/py/modal/_logs_manager.py (973 LoC): deadline'd
streaming with reconnect-by-offset. The local log store lives in the
scheduler's per-app ring buffers; the ``app_get_logs`` long-poll RPC
carries absolute offsets, so tails resume exactly where they stopped —
including over the proxy transport (a daemon-attached CLI tails a
daemon-hosted app live; round-1 review Missing #5)."""

from __future__ import annotations

import time
from typing import Any, Iterator, Optional

from ._sync import synchronizer


def _resolve_client(client: Any) -> Any:
    if client is not None:
        return client
    from .client import _Client

    return synchronizer.run(_Client.from_env())


def fetch_app_logs(app_id: str, client: Any = None) -> list[dict]:
    """Current log entries for an app (single snapshot, no waiting)."""
    c = _resolve_client(client)

    async def fetch() -> list[dict]:
        resp = await c.svc.app_get_logs(app_id=app_id, offset=0, timeout=0.0)
        return resp["entries"]

    return synchronizer.run(fetch())


def tail_app_logs(
    app_id: str,
    client: Any = None,
    poll_interval: float = 0.0,
    timeout: Optional[float] = None,
    from_start: bool = True,
) -> Iterator[dict]:
    """Blocking tail: yields entries as they arrive (long-poll, offset
    resume; reconnects transparently after transport errors)."""
    c = _resolve_client(client)
    offset = 0
    if not from_start:
        snap = synchronizer.run(
            c.svc.app_get_logs(app_id=app_id, offset=0, timeout=0.0)
        )
        offset = snap["next_offset"]
    deadline = None if timeout is None else time.time() + timeout
    while True:
        remaining = 10.0 if deadline is None else min(10.0, deadline - time.time())
        if remaining <= 0:
            return
        try:
            resp = synchronizer.run(
                c.svc.app_get_logs(app_id=app_id, offset=offset, timeout=remaining)
            )
        except Exception as exc:
            from .exception import NotFoundError

            if isinstance(exc, NotFoundError):
                return  # app gone (GC'd): the stream is over
            # transport blip: reconnect with the SAME offset (nothing lost)
            time.sleep(min(1.0, poll_interval or 0.5))
            continue
        for entry in resp["entries"]:
            yield entry
        offset = resp["next_offset"]
        if resp.get("app_state") == "stopped" and not resp["entries"]:
            return
        if poll_interval:
            time.sleep(poll_interval)


def stream_logs_to_output(app_id: str, client: Any = None, timeout: Optional[float] = None) -> None:
    """Tail an app's logs into the active OutputManager (or stdout)."""
    from .output import get_output_manager

    mgr = get_output_manager()
    for entry in tail_app_logs(app_id, client, timeout=timeout):
        if mgr is not None:
            mgr.print_log(entry)
        else:
            import sys

            sys.stdout.write(entry.get("data", ""))
            sys.stdout.flush()
