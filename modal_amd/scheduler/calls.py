"""Function-call table: the state the control plane keeps per invocation.

This is the in-process re-implementation of the server-side state behind the
reference's Function* RPCs (FunctionMap with pipelined inputs, PutInputs,
GetOutputs with clear_on_success + lost-input signalling, RetryInputs,
FinishInputs — /root/reference/modal_proto/api.proto, mock behavior
/root/reference/py/test/conftest.py:2303-2485), with the statuses of
GenericResult and the retry semantics of the invocation engine
(/root/reference/py/modal/_functions.py:106,286-316).
"""

from __future__ import annotations

import asyncio
import time
from collections import deque
from dataclasses import dataclass, field
from typing import Any, Optional

from ..utils.ids import new_id

# GenericResult.status values (parity: api.proto GenericResult)
GENERIC_STATUS_UNSPECIFIED = 0
GENERIC_STATUS_SUCCESS = 1
GENERIC_STATUS_FAILURE = 2
GENERIC_STATUS_TERMINATED = 3
GENERIC_STATUS_TIMEOUT = 4
GENERIC_STATUS_INTERNAL_FAILURE = 5
GENERIC_STATUS_INIT_FAILURE = 6

# parity: /root/reference/py/modal/_functions.py:106
MAX_INTERNAL_FAILURE_COUNT = 8

# parity: /root/reference/py/modal/parallel_map.py:79
MAX_INPUTS_OUTSTANDING_DEFAULT = 1000


@dataclass
class RetryPolicy:
    """User-facing retry policy (parity: modal.Retries, reference retries.py:12)."""

    max_retries: int = 0
    backoff_coefficient: float = 2.0
    initial_delay_ms: int = 1000
    max_delay_ms: int = 60_000

    def delay_ms(self, retry_count: int) -> float:
        if retry_count <= 0:
            return self.initial_delay_ms
        delay = self.initial_delay_ms * (self.backoff_coefficient ** (retry_count - 1))
        return min(delay, self.max_delay_ms)

    def to_dict(self) -> dict:
        return {
            "max_retries": self.max_retries,
            "backoff_coefficient": self.backoff_coefficient,
            "initial_delay_ms": self.initial_delay_ms,
            "max_delay_ms": self.max_delay_ms,
        }

    @classmethod
    def from_dict(cls, d: Optional[dict]) -> "RetryPolicy":
        return cls(**d) if d else cls()


@dataclass
class FunctionDef:
    """A registered function: the scheduler's row for ``fu-`` objects.

    The serialized definition travels to workers once and is cached there
    (parity: the container entrypoint imports user code once,
    /root/reference/py/modal/_runtime/user_code_imports.py:118).
    """

    function_id: str
    app_id: str
    name: str
    definition: bytes  # cloudpickled callable or import-ref descriptor
    definition_kind: str = "serialized"  # "serialized" | "ref"
    is_generator: bool = False
    needs_gpu: bool = False
    gpu_count: int = 0
    timeout: Optional[float] = None
    retry_policy: RetryPolicy = field(default_factory=RetryPolicy)
    max_concurrent_inputs: int = 1  # @modal.concurrent(max_inputs=...)
    target_concurrent_inputs: int = 0
    batch_max_size: int = 0  # @modal.batched
    batch_linger_ms: int = 0
    is_method: bool = False
    is_class_service: bool = False
    cluster_size: int = 0  # @modal.clustered(size=...)
    min_containers: int = 0
    max_containers: int = 0
    buffer_containers: int = 0
    scaledown_window: float = 60.0
    definition_version: int = 1
    metadata: dict = field(default_factory=dict)
    web_config: Optional[dict] = None
    secret_ids: list = field(default_factory=list)
    volume_mounts: dict = field(default_factory=dict)  # mount path -> volume id
    schedule: Optional[dict] = None  # {"cron": "..."} | {"period": seconds}
    image_id: Optional[str] = None
    placement: Optional[dict] = None  # {"gpu_index": i, "gpu_set": [..]} hints
    proxy_url: Optional[str] = None  # HTTP_PROXY/HTTPS_PROXY for user code

    def placement_tag(self) -> str:
        return "gpu" if self.needs_gpu else "any"

    def public_metadata(self) -> dict:
        return {
            "function_name": self.name,
            "is_generator": self.is_generator,
            "needs_gpu": self.needs_gpu,
            "is_method": self.is_method,
            "cluster_size": self.cluster_size,
            "batch_max_size": self.batch_max_size,
            "definition_id": self.function_id,
            **self.metadata,
        }


@dataclass(slots=True)
class InputRecord:
    call_id: str
    idx: int
    input_id: str
    payload: bytes
    payload_blob: Optional[str] = None  # CAS digest when payload exceeded the inline limit
    method_name: str = ""
    retry_count: int = 0  # user-policy retries consumed
    internal_failures: int = 0
    status: int = GENERIC_STATUS_UNSPECIFIED
    final: bool = False
    output: Optional[bytes] = None
    output_blob: Optional[str] = None  # CAS digest for oversized outputs
    output_format: int = 0
    exc_repr: Optional[str] = None
    worker_id: Optional[int] = None
    enqueued_at: float = 0.0
    started_at: float = 0.0
    finished_at: float = 0.0
    tensors: Optional[list] = None  # tensor sidecar (CUDA-IPC / pinned staging)
    cancelled: bool = False
    cluster: Optional[dict] = None  # gang identity: {rank, size, cluster_id, master_port}
    chunk_id: Optional[str] = None  # shared chunk payload (map fan-out fast path)
    chunk_index: int = 0
    out_chunk: Optional[str] = None  # shared output chunk (worker fast path)
    out_ci: int = 0

    @property
    def token(self) -> str:
        """Stable retry-versioned token — plays the role of the reference's
        per-input JWT used for lost-input detection (parallel_map.py:447-523)."""
        return f"{self.call_id}:{self.idx}:{self.retry_count}"


class ChunkGroup:
    """Scheduling state for one map chunk dispatched as a unit.

    The map fast path never materializes per-item InputRecords while things
    succeed: a chunk of ~64 inputs is one pending descriptor, one wire frame,
    one completion. Items that fail (user exception, worker death past the
    chunk level) materialize real InputRecords and re-enter the per-item FSM,
    so retry/dedup semantics stay per input (parity: parallel_map.py FSM).
    """

    __slots__ = (
        "chunk_id", "base_idx", "count", "method", "state", "worker_id", "done_cis",
        "internal_failures",
    )

    def __init__(self, chunk_id: str, base_idx: int, count: int, method: str = ""):
        self.chunk_id = chunk_id
        self.base_idx = base_idx
        self.count = count
        self.method = method
        self.state = "pending"  # pending | inflight | done
        self.worker_id: Optional[int] = None
        self.done_cis: Optional[set] = None  # populated only on partial success
        self.internal_failures = 0

    @property
    def token(self) -> str:
        return f"g:{self.chunk_id}"


class CallRecord:
    """One function call (``fc-``): unary, spawn, or map fan-out."""

    def __init__(
        self,
        function_id: str,
        kind: str,
        return_exceptions: bool = False,
    ):
        self.call_id = new_id("function_call")
        self.function_id = function_id
        self.kind = kind  # "unary" | "spawn" | "map" | "spawn_map"
        self.return_exceptions = return_exceptions
        self.inputs: dict[int, InputRecord] = {}
        # shared chunk payloads: chunk_id -> {"data": bytes, "refs": int}
        # (one pickled list serves many inputs; freed when all are final)
        self.chunks: dict[str, dict] = {}
        # fast-path chunk scheduling state (range protocol)
        self.chunk_groups: dict[str, ChunkGroup] = {}
        self.next_idx = 0
        self.num_inputs_final: Optional[int] = None
        self.completed: int = 0
        self.cancelled = False
        self.created_at = time.time()
        self.finished_at: Optional[float] = None
        # completion-order queue of idx for streaming GetOutputs
        self.output_ready: asyncio.Queue[int] = asyncio.Queue()
        # entries split/deferred by a bounded GetOutputs response; consumed
        # before the queue so completion order is preserved
        self.output_pushback: deque = deque()
        # per-input completion events for unary waits
        self._waiters: dict[int, asyncio.Future] = {}
        # generator data-out channel, per input idx
        self.gen_queues: dict[int, asyncio.Queue] = {}
        self.done_event = asyncio.Event()


    def _drop_chunk(self, chunk_id: str) -> None:
        """Release a fully-consumed input chunk; one-shot xfer spill files
        (parallel/map.py _spill) are unlinked here."""
        chunk = self.chunks.pop(chunk_id, None)
        if chunk is None:
            return
        data = chunk.get("data")
        if isinstance(data, dict) and data.get("xfer"):
            import os

            try:
                os.unlink(data["xfer"])
            except OSError:
                pass

    # -- input intake ----------------------------------------------------
    def add_input(
        self,
        payload: bytes,
        method_name: str = "",
        tensors: Optional[list] = None,
        payload_blob: Optional[str] = None,
    ) -> InputRecord:
        idx = self.next_idx
        self.next_idx += 1
        rec = InputRecord(
            call_id=self.call_id,
            idx=idx,
            # derived id: unique per runtime without a per-input entropy draw
            input_id=f"in-{self.call_id[3:]}-{idx}",
            payload=payload,
            payload_blob=payload_blob,
            method_name=method_name,
            enqueued_at=time.time(),
            tensors=tensors,
        )
        self.inputs[idx] = rec
        return rec

    def add_chunk(self, chunk_id: str, payload: Any, count: int, method: str = "") -> ChunkGroup:
        """Register a whole chunk without materializing per-item records."""
        group = ChunkGroup(chunk_id, self.next_idx, count, method)
        self.next_idx += count
        self.chunks[chunk_id] = {"data": payload, "refs": count}
        self.chunk_groups[chunk_id] = group
        return group

    def materialize_chunk_item(self, group: ChunkGroup, ci: int) -> InputRecord:
        """Create the real per-item record for a chunk member (failure path)."""
        idx = group.base_idx + ci
        rec = self.inputs.get(idx)
        if rec is not None:
            return rec
        rec = InputRecord(
            call_id=self.call_id,
            idx=idx,
            input_id=f"in-{self.call_id[3:]}-{idx}",
            payload=b"",
            enqueued_at=time.time(),
            method_name=group.method,
        )
        rec.chunk_id = group.chunk_id
        rec.chunk_index = ci
        self.inputs[idx] = rec
        return rec

    def complete_chunk_success(self, group: ChunkGroup, cis: Optional[list], out_chunk: str) -> None:
        """Bulk completion of a chunk's successful members.
        cis=None means the whole chunk succeeded."""
        if group.state == "done":
            return
        n = group.count if cis is None else len(cis)
        chunk = self.chunks.get(group.chunk_id)
        if chunk is not None:
            chunk["refs"] -= n
            if chunk["refs"] <= 0:
                self._drop_chunk(group.chunk_id)
        self.completed += n
        if cis is None:
            group.state = "done"
            self.output_ready.put_nowait(("g", group.base_idx, group.count, out_chunk, None))
        else:
            group.done_cis = set(cis)
            group.state = "done"
            self.output_ready.put_nowait(("g", group.base_idx, group.count, out_chunk, list(cis)))
        self._check_done()

    def finish_inputs(self) -> None:
        self.num_inputs_final = self.next_idx
        self._check_done()

    def _check_done(self) -> None:
        if self.num_inputs_final is not None and self.completed >= self.num_inputs_final:
            if not self.done_event.is_set():
                self.finished_at = time.time()  # call-record GC clock
            self.done_event.set()

    # -- completion ------------------------------------------------------
    def post_output(
        self,
        idx: int,
        status: int,
        output: Optional[bytes],
        output_format: int,
        exc_repr: Optional[str],
        retry_count: int,
        output_blob: Optional[str] = None,
        out_chunk: Optional[str] = None,
        out_ci: int = 0,
    ) -> bool:
        """Record a final output for input idx. Returns False on stale/dup
        delivery (parity: dedup by (idx, retry_count),
        reference parallel_map.py:1416-1431)."""
        rec = self.inputs.get(idx)
        if rec is None or rec.final:
            return False
        if retry_count != rec.retry_count:
            return False  # output from a superseded attempt
        rec.status = status
        rec.output = output
        rec.output_blob = output_blob
        rec.out_chunk = out_chunk
        rec.out_ci = out_ci
        rec.output_format = output_format
        rec.exc_repr = exc_repr
        rec.final = True
        rec.finished_at = time.time()
        if rec.chunk_id:
            chunk = self.chunks.get(rec.chunk_id)
            if chunk is not None:
                chunk["refs"] -= 1
                if chunk["refs"] <= 0:
                    self._drop_chunk(rec.chunk_id)
        self.completed += 1
        self.output_ready.put_nowait(idx)
        waiter = self._waiters.pop(idx, None)
        if waiter is not None and not waiter.done():
            waiter.set_result(rec)
        self._check_done()
        return True

    def post_outputs_bulk(self, triples: list, out_chunk: str) -> int:
        """Bulk success delivery for a shared output chunk:
        triples = [(idx, retry_count, chunk_index), ...]. One queue entry
        covers the whole group. Returns the number actually recorded."""
        ready: list[int] = []
        now = time.time()
        for idx, retry_count, ci in triples:
            rec = self.inputs.get(idx)
            if rec is None or rec.final or retry_count != rec.retry_count:
                continue
            rec.status = GENERIC_STATUS_SUCCESS
            rec.out_chunk = out_chunk
            rec.out_ci = ci
            rec.final = True
            rec.finished_at = now
            if rec.chunk_id:
                chunk = self.chunks.get(rec.chunk_id)
                if chunk is not None:
                    chunk["refs"] -= 1
                    if chunk["refs"] <= 0:
                        self._drop_chunk(rec.chunk_id)
            ready.append(idx)
            waiter = self._waiters.pop(idx, None)
            if waiter is not None and not waiter.done():
                waiter.set_result(rec)
        if ready:
            self.completed += len(ready)
            self.output_ready.put_nowait(ready)
            self._check_done()
        return len(ready)

    async def wait_output(self, idx: int, timeout: Optional[float] = None) -> InputRecord:
        rec = self.inputs[idx]
        if rec.final:
            return rec
        fut = self._waiters.get(idx)
        if fut is None:
            fut = asyncio.get_running_loop().create_future()
            self._waiters[idx] = fut
        if timeout is None:
            return await asyncio.shield(fut)
        return await asyncio.wait_for(asyncio.shield(fut), timeout)

    # -- generator data plane -------------------------------------------
    def gen_queue(self, idx: int) -> asyncio.Queue:
        q = self.gen_queues.get(idx)
        if q is None:
            q = asyncio.Queue()
            self.gen_queues[idx] = q
        return q

    def stats(self) -> dict:
        return {
            "total": self.next_idx,
            "completed": self.completed,
            "pending": self.next_idx - self.completed,
            "final": self.num_inputs_final is not None,
        }
