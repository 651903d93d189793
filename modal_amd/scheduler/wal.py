"""Durable spawned calls: a per-call journal under ``<run_dir>/wal/``.

SURVEY hard part 5: "exactly-once-ish output accounting under worker death
— reproduce the INTERNAL_FAILURE requeue without a cloud's durable queue
(WAL in scheduler)". Worker death was covered in round 1; this closes the
SCHEDULER-death half for detached work: ``spawn``/``spawn_map`` calls
against DEPLOYED functions journal their inputs at intake, journal results
as they finalize, and a restarted daemon reloads unfinished calls —
completed inputs keep their results (no re-execution), unfinished inputs
re-enter the dispatch queue, and ``FunctionCall.from_id`` on the new
scheduler sees the same call id.

Ephemeral apps' calls stay ephemeral (they die with their client, exactly
like the reference's client-scoped calls).
"""

from __future__ import annotations

import logging
import os
import pickle
from typing import Any, Optional

logger = logging.getLogger("modal_amd.scheduler.wal")

WAL_DIR = "wal"


def _wal_dir(scheduler: Any) -> str:
    path = os.path.join(scheduler.run_dir, WAL_DIR)
    os.makedirs(path, exist_ok=True)
    return path


def _call_path(scheduler: Any, call_id: str) -> str:
    return os.path.join(_wal_dir(scheduler), call_id)


def journaled(scheduler: Any, record: Any) -> bool:
    """Spawn-kind calls on deployed (non-ephemeral) apps are durable."""
    if record.kind not in ("spawn", "spawn_map"):
        return False
    fdef = scheduler.functions.get(record.function_id)
    if fdef is None or fdef.cluster_size > 1:
        return False  # gang calls carry live rendezvous state: not journaled
    app = scheduler.apps.get(fdef.app_id)
    return app is not None and not app.ephemeral


def append(scheduler: Any, record: Any, kind: str, payload: dict) -> None:
    """Append one journal entry (atomic-enough: single append write)."""
    try:
        with open(_call_path(scheduler, record.call_id), "ab") as f:
            entry = pickle.dumps((kind, payload), 4)
            f.write(len(entry).to_bytes(4, "little") + entry)
    except OSError:
        logger.warning("WAL append failed for %s", record.call_id, exc_info=True)


def journal_input(scheduler: Any, record: Any, rec: Any) -> None:
    append(
        scheduler, record, "input",
        {
            "idx": rec.idx,
            "payload": rec.payload,
            "payload_blob": rec.payload_blob,
            "method_name": rec.method_name,
        },
    )


def journal_finish(scheduler: Any, record: Any) -> None:
    append(scheduler, record, "finish", {"total": record.num_inputs_final})


def journal_result(
    scheduler: Any, record: Any, rec: Any, output: Optional[bytes] = None,
    output_format: Optional[int] = None,
) -> None:
    """``output`` overrides rec.output for results whose bytes live in a
    shared output chunk (the worker bulk path extracts them per item)."""
    append(
        scheduler, record, "result",
        {
            "idx": rec.idx,
            "status": rec.status,
            "output": rec.output if output is None else output,
            "output_blob": rec.output_blob,
            "output_format": rec.output_format if output_format is None else output_format,
            "exc_repr": rec.exc_repr,
        },
    )


def journal_created(scheduler: Any, record: Any) -> None:
    append(
        scheduler, record, "created",
        {"call_id": record.call_id, "function_id": record.function_id, "kind": record.kind},
    )


def drop(scheduler: Any, call_id: str) -> None:
    try:
        os.unlink(_call_path(scheduler, call_id))
    except OSError:
        pass


def _read_entries(path: str) -> list:
    entries = []
    try:
        with open(path, "rb") as f:
            while True:
                header = f.read(4)
                if len(header) < 4:
                    break
                n = int.from_bytes(header, "little")
                blob = f.read(n)
                if len(blob) < n:
                    break  # torn tail write: ignore
                entries.append(pickle.loads(blob))
    except OSError:
        pass
    return entries


def replay(scheduler: Any) -> int:
    """Rebuild unfinished journaled calls on a fresh scheduler. Returns the
    number of calls restored. Must run after persist.load (functions need
    to exist) and before the pool starts dispatching is NOT required —
    enqueue_pending uses the normal intake path."""
    from .calls import CallRecord

    wal = os.path.join(scheduler.run_dir, WAL_DIR)
    if not os.path.isdir(wal):
        return 0
    restored = 0
    for name in sorted(os.listdir(wal)):
        path = os.path.join(wal, name)
        entries = _read_entries(path)
        created = next((p for k, p in entries if k == "created"), None)
        if created is None or created["function_id"] not in scheduler.functions:
            drop(scheduler, name)
            continue
        record = CallRecord(created["function_id"], created["kind"])
        record.call_id = created["call_id"]  # keep the durable identity
        record.durable = True  # results keep journaling; journal drops on completion
        scheduler.calls[record.call_id] = record
        results = {p["idx"]: p for k, p in entries if k == "result"}
        pending = []
        for k, p in entries:
            if k != "input":
                continue
            rec = record.add_input(
                p["payload"], p.get("method_name", ""), payload_blob=p.get("payload_blob")
            )
            done = results.get(rec.idx)
            if done is not None:
                rec.status = done["status"]
                rec.output = done["output"]
                rec.output_blob = done.get("output_blob")
                rec.output_format = done.get("output_format", 0)
                rec.exc_repr = done.get("exc_repr")
                rec.final = True
                record.completed += 1
                record.output_ready.put_nowait(rec.idx)
            else:
                pending.append(rec)
        finish = next((p for k, p in entries if k == "finish"), None)
        if finish is not None:
            record.num_inputs_final = record.next_idx
        if record.num_inputs_final is not None and record.completed >= record.num_inputs_final:
            record.done_event.set()
            drop(scheduler, name)
            continue
        for rec in pending:
            scheduler.pool.enqueue(rec, function_id=record.function_id)
        restored += 1
        logger.info(
            "WAL: restored %s (%d inputs, %d already complete)",
            record.call_id, record.next_idx, record.completed,
        )
    return restored
