"""Map fan-out engine: pipelined scatter of inputs, streamed gather of outputs.

Parity: /root/reference/py/modal/parallel_map.py — ``_map_invocation`` (:362)
runs four concurrent stages (serialize, pump, poll outputs, order/yield) with
a 1,000-outstanding backpressure semaphore (:79,1522), exactly-once output
accounting by (idx, retry_count) (:1416-1431), and ordered output buffering
(:552-578).

MI355X-native shape: the scheduler is in-process, so the pump stage writes
straight into the call table (batches of up to 512 inputs per call — the
reference's 49-per-request constant exists to amortize gRPC round-trips;
here a "request" is a method call, and 512 matches the reference's
spawn_map batch, parallel_map.py:83). Retries and lost-input requeue are
owned by the scheduler (scheduler/core.py:on_worker_output, workerhost's
worker-death path), which is the single-node equivalent of the reference's
client-side retry manager + input_jwts lost-input protocol.
"""

from __future__ import annotations

import asyncio
import os
from typing import Any, AsyncGenerator

from ..functions import make_payload_item, process_output_item
from ..scheduler.calls import GENERIC_STATUS_SUCCESS

PUT_BATCH_SIZE = 512  # parity: spawn_map batch, reference parallel_map.py:83
OUTPUT_FETCH_MAX = 1024
# items per shared pickle chunk: one C-pickler pass serves ~64 inputs on the
# client AND one unpickle serves them on the worker (SURVEY §2 row 6's
# "tensor-aware fast path" generalized to all small map payloads)
CHUNK_ITEMS = 128  # default; MODAL_AMD_CHUNK_ITEMS overrides per run


def _chunk_items() -> int:
    raw = os.environ.get("MODAL_AMD_CHUNK_ITEMS")
    return int(raw) if raw else CHUNK_ITEMS


class BulkSemaphore:
    """Counting semaphore with O(1) bulk acquire/release (the 1000-outstanding
    backpressure cap acquires per chunk, not per item)."""

    def __init__(self, value: int):
        self._value = value
        self._event = asyncio.Event()
        self._event.set()

    async def acquire(self, n: int = 1) -> None:
        while self._value < n:
            self._event.clear()
            await self._event.wait()
        self._value -= n

    def release(self, n: int = 1) -> None:
        self._value += n
        self._event.set()


async def _iterate_maybe_async(it: Any) -> AsyncGenerator[Any, None]:
    if hasattr(it, "__aiter__"):
        async for item in it:
            yield item
        return
    iterator = iter(it)
    count = 0
    while True:
        try:
            item = next(iterator)
        except StopIteration:
            return
        yield item
        count += 1
        if count % 256 == 0:
            await asyncio.sleep(0)  # let outputs flow while we serialize


async def map_invocation(
    fn: Any,
    input_iter: Any,
    kwargs_common: dict,
    order_outputs: bool,
    return_exceptions: bool,
    wrap_returned_exceptions: bool,
) -> AsyncGenerator[Any, None]:
    """Per-item view over map_invocation_batches (one flat loop, no extra
    async-generator layer per item)."""
    async for batch in map_invocation_batches(
        fn, input_iter, kwargs_common, order_outputs, return_exceptions,
        wrap_returned_exceptions,
    ):
        for value in batch:
            yield value


async def map_invocation_batches(
    fn: Any,
    input_iter: Any,
    kwargs_common: dict,
    order_outputs: bool,
    return_exceptions: bool,
    wrap_returned_exceptions: bool,
    fast_zip_iters: Any = None,
) -> AsyncGenerator[list, None]:
    """The map engine proper. Yields LISTS of decoded output values — the
    natural unit is the ~64-item range-protocol group, so per-item work in
    the engine is a plain list append/extend, not an async-generator frame.
    Callers that need per-item semantics flatten at their own (single)
    layer; the sync bridge crosses threads once per batch instead of once
    per item."""
    if not fn._is_hydrated:
        await fn.hydrate()
    client = fn._client
    svc = client.svc
    resp = await svc.function_map(function_id=fn.object_id, kind="map")
    call_id = resp["function_call_id"]
    max_outstanding = resp.get("max_inputs_outstanding") or 1000
    sem = BulkSemaphore(max_outstanding)
    pump_done = asyncio.Event()
    total_inputs = 0
    pump_error: list[BaseException] = []

    async def pump() -> None:
        nonlocal total_inputs
        from .._serialization import serialize_fast

        # proxied schedulers take chunk puts as ONE-WAY frames (ordered on
        # the socket; put_chunk's intake prefix is await-free, so index
        # bases stay sequential) — a per-chunk RTT would serialize the pump
        if "function_put_chunk_oneway" in type(svc).__dict__:
            put_chunk = svc.function_put_chunk_oneway  # real method, not an
            # RPC stub a __getattr__-based proxy would synthesize
        else:
            put_chunk = svc.function_put_chunk

        chunk_buf: list = []
        chunk_seq = 0

        def _chunk_serialize(buf: list) -> bytes:
            # serialize_fast detects tensors/handles in the stream itself and
            # reroutes to the hook-aware pickler (device staging / mesh export)
            return serialize_fast(("C", buf))

        async def flush_chunk() -> None:
            nonlocal chunk_buf, chunk_seq
            if not chunk_buf:
                return
            payload = _chunk_serialize(chunk_buf)
            chunk_id = f"{call_id}.c{chunk_seq}"
            chunk_seq += 1
            count = len(chunk_buf)
            chunk_buf = []
            if len(payload) > 2 * 1024 * 1024:
                # big-item chunks spill off the wire (parity: the 2 MiB
                # inline payload limit, blob_utils.py:36) — workers read
                # from the shared filesystem instead of the socket
                payload = await asyncio.get_running_loop().run_in_executor(
                    None, _spill, payload, chunk_id
                )
            await put_chunk(
                function_call_id=call_id,
                chunk_id=chunk_id,
                payload=payload,
                count=count,
                method=fn._method_name or "",
            )

        approx_bytes = 0

        def _add(item: tuple) -> None:
            nonlocal approx_bytes, total_inputs
            args, extra_kwargs = item
            kw = {**kwargs_common, **extra_kwargs} if extra_kwargs else kwargs_common
            chunk_buf.append((args, kw))
            for a in args:
                if type(a) in (bytes, bytearray, str):
                    approx_bytes += len(a)
            total_inputs += 1

        def _spill(payload: bytes, chunk_id: str) -> Any:
            """Oversized chunk payloads leave the control socket. Transport
            spill = one-shot file handoff (no hash, no compress — the
            scheduler unlinks it when the chunk completes); CAS fallback
            when no shared run_dir exists."""
            xfer = client.xfer_dir
            if xfer is not None:
                path = os.path.join(xfer, chunk_id)
                with open(path, "wb") as f:
                    f.write(payload)
                return {"xfer": path}
            store = client.blob_store
            if store is not None:
                return {"blob": store.put(payload)}
            return payload

        # Big-payload chunks overlap: serialize+spill run on executor
        # threads in a bounded window, but the put_chunk RPCs DRAIN IN
        # ORDER (chunk index bases are assigned at put time — out-of-order
        # puts would scramble map output ordering).
        prepare_fifo: list = []  # (prepare_task, chunk_id, count)

        def _prepare(argsbatch: list, chunk_id: str) -> Any:
            payload = serialize_fast(("C2", kwargs_common, argsbatch))
            if len(payload) > 2 * 1024 * 1024:
                payload = _spill(payload, chunk_id)
            return payload

        async def _drain_prepared(block: bool) -> None:
            while prepare_fifo:
                task, chunk_id, count = prepare_fifo[0]
                if not block and not task.done():
                    return
                payload = await task
                prepare_fifo.pop(0)
                await put_chunk(
                    function_call_id=call_id,
                    chunk_id=chunk_id,
                    payload=payload,
                    count=count,
                    method=fn._method_name or "",
                )

        async def flush_args_chunk(argsbatch: list) -> None:
            # "C2" wire form: common kwargs factored out, args list built by
            # C-level zip+islice — no per-item Python in the pump at all
            nonlocal chunk_seq, total_inputs
            total_inputs += len(argsbatch)
            await sem.acquire(len(argsbatch))
            chunk_id = f"{call_id}.c{chunk_seq}"
            chunk_seq += 1
            approx = sum(
                len(a) for a in argsbatch[0] if type(a) in (bytes, bytearray, str)
            ) * len(argsbatch)
            loop = asyncio.get_running_loop()
            if approx > 1024 * 1024:
                fut = loop.run_in_executor(None, _prepare, argsbatch, chunk_id)
                prepare_fifo.append((asyncio.ensure_future(fut), chunk_id, len(argsbatch)))
                await _drain_prepared(block=len(prepare_fifo) > 4)
                return
            await _drain_prepared(block=True)  # keep put order across sizes
            payload = serialize_fast(("C2", kwargs_common, argsbatch))
            if len(payload) > 2 * 1024 * 1024:
                payload = await loop.run_in_executor(None, _spill, payload, chunk_id)
            await put_chunk(
                function_call_id=call_id,
                chunk_id=chunk_id,
                payload=payload,
                count=len(argsbatch),
                method=fn._method_name or "",
            )

        try:
            chunk_items = _chunk_items()
            if fast_zip_iters is not None:
                from itertools import islice

                iterator = zip(*fast_zip_iters)
                n_flushed = 0
                while True:
                    argsbatch = list(islice(iterator, chunk_items))
                    if not argsbatch:
                        break
                    # byte-size guard sampled from the first item: split big-
                    # payload chunks so one frame stays well under the CAS
                    # spill threshold
                    est = sum(
                        len(a) for a in argsbatch[0] if type(a) in (bytes, bytearray, str)
                    )
                    if est * len(argsbatch) > 16 * 1024 * 1024 and len(argsbatch) > 1:
                        step = max(1, (4 * 1024 * 1024) // max(est, 1))
                        for s in range(0, len(argsbatch), step):
                            await flush_args_chunk(argsbatch[s : s + step])
                    else:
                        await flush_args_chunk(argsbatch)
                    n_flushed += 1
                await _drain_prepared(block=True)
                await svc.function_finish_inputs(function_call_id=call_id)
                return
            if hasattr(input_iter, "__aiter__"):
                async for item in input_iter:
                    _add(item)
                    if len(chunk_buf) >= chunk_items or approx_bytes > 4 * 1024 * 1024:
                        approx_bytes = 0
                        await sem.acquire(len(chunk_buf))
                        await flush_chunk()
            else:
                # sync iterators (the common case): pull whole chunks via
                # islice — no per-item async generator frames
                from itertools import islice

                iterator = iter(input_iter)
                while True:
                    batch = list(islice(iterator, chunk_items - len(chunk_buf) or chunk_items))
                    if not batch and not chunk_buf:
                        break
                    for item in batch:
                        _add(item)
                    if len(chunk_buf) >= chunk_items or approx_bytes > 4 * 1024 * 1024:
                        approx_bytes = 0
                        await sem.acquire(len(chunk_buf))
                        await flush_chunk()
                    elif not batch:
                        break
            if chunk_buf:
                await sem.acquire(len(chunk_buf))
            await flush_chunk()
            await svc.function_finish_inputs(function_call_id=call_id)
        except BaseException as exc:
            pump_error.append(exc)
            raise
        finally:
            pump_done.set()

    pump_task = asyncio.get_running_loop().create_task(pump())

    received = 0
    next_output_idx = 0
    ordering_buffer: dict[int, Any] = {}

    from ..output import get_output_manager

    _mgr = get_output_manager()
    progress = None
    if _mgr is not None:
        label = getattr(fn, "_info_name", None) or getattr(fn, "_method_name", "") or "map"
        progress = _mgr.make_map_progress(f"Running {label}")

    out_chunk_cache: dict[str, list] = {}

    async def decode(out: dict) -> Any:
        cid = out.get("out_chunk")
        if cid is not None and out["status"] == GENERIC_STATUS_SUCCESS:
            values = out_chunk_cache.get(cid)
            if values is None:
                import pickle as _pickle

                data = out.get("chunk_data")
                if data is None:
                    return await process_output_item(out, client)  # raises missing-chunk
                values = _pickle.loads(data)
                out_chunk_cache[cid] = values
                if len(out_chunk_cache) > 64:
                    out_chunk_cache.pop(next(iter(out_chunk_cache)))
            return values[out.get("out_ci", 0)]
        try:
            return await process_output_item(out, client)
        except BaseException as exc:
            if return_exceptions:
                return exc
            raise

    try:
        while True:
            if pump_done.is_set() and received >= total_inputs:
                break
            outs = await svc.function_get_outputs(
                function_call_id=call_id, max_values=OUTPUT_FETCH_MAX, timeout=0.5
            )
            if pump_error:
                raise pump_error[0]
            if progress is not None:
                progress.update(received, total_inputs, pump_done.is_set())
            batch_out: list = []
            for out in outs:
                if out.get("group"):
                    # range-protocol group: one pickled value list for ~64 idxs
                    import pickle as _pickle

                    values = _pickle.loads(out["chunk_data"])
                    cis = out["cis"]
                    base = out["idx_base"]
                    # val_off: the position in `values` of cis[0] — nonzero
                    # when the scheduler split a group to honor max_values
                    voff = out.get("val_off", 0)
                    n_here = len(values) if cis is None else len(cis)
                    received += n_here
                    sem.release(n_here)
                    if order_outputs:
                        if cis is None:
                            for ci, value in enumerate(values):
                                ordering_buffer[base + ci] = value
                        else:
                            for ci, value in zip(cis, values[voff:]):
                                ordering_buffer[base + ci] = value
                        while next_output_idx in ordering_buffer:
                            batch_out.append(ordering_buffer.pop(next_output_idx))
                            next_output_idx += 1
                    elif cis is None:
                        batch_out.extend(values)
                    else:
                        batch_out.extend(values[voff : voff + len(cis)])
                    continue
                received += 1
                sem.release()
                value = await decode(out)
                if order_outputs:
                    ordering_buffer[out["idx"]] = value
                    while next_output_idx in ordering_buffer:
                        batch_out.append(ordering_buffer.pop(next_output_idx))
                        next_output_idx += 1
                else:
                    batch_out.append(value)
            if batch_out:
                yield batch_out
    finally:
        if progress is not None:
            progress.close()
        if not pump_task.done():
            pump_task.cancel()
        try:
            await pump_task
        except (asyncio.CancelledError, BaseException):
            pass


async def spawn_map_invocation(fn: Any, items: list, kwargs_common: dict) -> str:
    """Enqueue all inputs without consuming outputs (reference spawn_map,
    parallel_map.py:1227-1258; 512-input batches :83)."""
    client = fn._client
    svc = client.svc
    resp = await svc.function_map(function_id=fn.object_id, kind="spawn_map")
    call_id = resp["function_call_id"]
    batch: list[dict] = []
    for args, extra_kwargs in items:
        kw = {**kwargs_common, **extra_kwargs} if extra_kwargs else kwargs_common
        item = make_payload_item(client, args, kw)
        if fn._method_name:
            item["method"] = fn._method_name
        batch.append(item)
        if len(batch) >= PUT_BATCH_SIZE:
            await svc.function_put_inputs(function_call_id=call_id, items=batch)
            batch = []
    if batch:
        await svc.function_put_inputs(function_call_id=call_id, items=batch)
    await svc.function_finish_inputs(function_call_id=call_id)
    return call_id
