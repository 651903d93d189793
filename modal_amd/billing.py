"""Billing/usage accounting.

Parity: /root/reference/py/modal/_billing.py + cli/billing.py — usage
queries. Locally: per-function aggregate runtime seconds and input counts
from the scheduler's call table (the local "bill" is GPU-seconds consumed).
"""

from __future__ import annotations

from typing import Any

from ._sync import dual_function


@dual_function
async def usage_summary(client: Any = None) -> list[dict]:
    from .client import _Client

    client = client or await _Client.from_env()
    svc = client.svc
    if getattr(svc, "is_proxy", False):
        return []
    per_function: dict[str, dict] = {}
    for record in svc.calls.values():
        fdef = svc.functions.get(record.function_id)
        name = fdef.name if fdef else record.function_id
        row = per_function.setdefault(
            name, {"function": name, "inputs": 0, "runtime_seconds": 0.0, "gpu": bool(fdef and fdef.needs_gpu)}
        )
        # range-protocol successes never materialize per-item records, so
        # the call-level completion counter is the authoritative count
        row["inputs"] += max(record.completed, len(record.inputs))
        for rec in record.inputs.values():
            if rec.finished_at and rec.started_at:
                row["runtime_seconds"] += rec.finished_at - rec.started_at
    return sorted(per_function.values(), key=lambda r: -r["runtime_seconds"])
