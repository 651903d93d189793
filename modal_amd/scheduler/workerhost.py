"""Worker pool: per-GPU worker processes and the dispatch plane.

MI355X-native replacement for the reference's container fleet: instead of
cloud-scheduled containers pulling inputs over gRPC long-polls
(/root/reference/py/modal/_runtime/container_io_manager.py:856), the scheduler
*pushes* credit-bounded input batches to worker processes over Unix sockets —
one worker per GPU (``HIP_VISIBLE_DEVICES``-pinned) plus CPU workers, all on
this node. Worker death reproduces the INTERNAL_FAILURE requeue path
(reference _functions.py:106): in-flight inputs are rescheduled up to 8 times.
"""

from __future__ import annotations

import asyncio
import heapq
import os
import subprocess
import sys
import time
from collections import deque
from typing import TYPE_CHECKING, Optional

from ..utils.ids import new_id
from .calls import (
    GENERIC_STATUS_INTERNAL_FAILURE,
    MAX_INTERNAL_FAILURE_COUNT,
    FunctionDef,
    InputRecord,
)
from .transport import Connection

_CPU_COUNT = os.cpu_count()  # cached: _pool_limit runs per dispatch

if TYPE_CHECKING:
    from .core import Scheduler

# deep-enough per-worker pipeline so tiny functions amortize frame overhead;
# the analog of the reference's 49-inputs-per-PutInputs batching
# (/root/reference/py/modal/parallel_map.py:82) tuned for a local socket.
DEFAULT_PIPELINE_DEPTH = 256


class WorkerHandle:
    def __init__(
        self,
        worker_id: int,
        conn: Connection,
        gpu_index: Optional[int],
        proc: Optional[subprocess.Popen] = None,
        external: bool = False,
    ):
        self.worker_id = worker_id
        self.task_id = new_id("task")
        self.conn = conn
        self.gpu_index = gpu_index
        self.proc = proc
        self.external = external
        self.alive = True
        self.last_heartbeat = time.time()
        self.last_active = time.time()  # last input assignment (scaledown clock)
        self.gpu_stats: Optional[dict] = None  # HBM gauges from heartbeats
        self.paged = False  # GPU memory snapshotted to host (scaledown)
        # tokens of inputs currently assigned here, mapped to their records
        self.inflight: dict[str, InputRecord] = {}
        # per-function outstanding count (for credit computation)
        self.outstanding: dict[str, int] = {}
        self.functions_loaded: set[str] = set()
        self.defs_version: dict[str, int] = {}
        self.chunks_sent: set[str] = set()
        self.draining = False

    @property
    def has_gpu(self) -> bool:
        return self.gpu_index is not None

    def credit_for(self, fdef: FunctionDef) -> int:
        # deep enough for >=3 chunk groups in flight so a worker never idles
        # between group completions (map chunks are ~64 items each)
        cap = max(fdef.max_concurrent_inputs * 2, DEFAULT_PIPELINE_DEPTH)
        if fdef.batch_max_size:
            cap = max(cap, fdef.batch_max_size * 2)
        return cap - self.outstanding.get(fdef.function_id, 0)


class WorkerPool:
    def __init__(self, scheduler: "Scheduler"):
        self.scheduler = scheduler
        self.run_dir = scheduler.run_dir
        self.socket_path = os.path.join(self.run_dir, "scheduler.sock")
        self.workers: dict[int, WorkerHandle] = {}
        self._next_worker_id = 0
        self._server: Optional[asyncio.AbstractServer] = None
        self._dispatch_wake = asyncio.Event()
        self._dispatch_task: Optional[asyncio.Task] = None
        self._retry_task: Optional[asyncio.Task] = None
        self._stopping = False
        # pending inputs per function (FIFO) and delayed retries (timestamp heap,
        # parity: TimestampPriorityQueue, reference async_utils.py:656)
        self.pending: dict[str, deque[InputRecord]] = {}
        self.delayed: list[tuple[float, int, str, InputRecord]] = []
        self._delay_seq = 0
        self._spawn_lock = asyncio.Lock()
        self._pending_spawns = 0
        self._worker_ready = asyncio.Event()
        self._procs: list[subprocess.Popen] = []

    # -- lifecycle -------------------------------------------------------
    async def start(self) -> None:
        os.makedirs(self.run_dir, exist_ok=True)
        self._server = await asyncio.start_unix_server(self._on_connect, path=self.socket_path)
        try:
            os.chmod(self.socket_path, 0o600)
        except OSError:
            pass
        self._dispatch_task = asyncio.get_running_loop().create_task(self._dispatch_loop())
        self._retry_task = asyncio.get_running_loop().create_task(self._retry_loop())
        self._health_task = asyncio.get_running_loop().create_task(self._health_loop())

    async def _scaledown_once(self) -> None:
        """Reap idle workers beyond the warm floor (parity: scaledown_window
        autoscaler setting, reference _functions.py:1195-1292)."""
        window = min(
            (f.scaledown_window for f in self.scheduler.functions.values() if f.scaledown_window),
            default=60.0,
        )
        floor = max(
            (f.min_containers + f.buffer_containers for f in self.scheduler.functions.values()),
            default=0,
        )
        now = time.time()
        idle = [
            w
            for w in self.workers.values()
            if w.alive
            and not w.draining
            and not w.inflight
            and not w.external  # torchrun-owned workers are not ours to reap
            and now - w.last_active > window
        ]
        reapable = []
        for w in idle:
            if not w.paged and self._hosts_snapshot_fn(w):
                # enable_memory_snapshot: page GPU memory to host instead of
                # reaping — HBM frees, the warm import/enter state survives
                # (parity: the reference's memory-snapshot cold-start
                # elimination, gpu_memory_snapshot.py:230-300). Reversible,
                # so not subject to the warm floor.
                w.paged = True  # BEFORE the await: a concurrent dispatch
                # must see it and queue a gpu_restore, which the worker's
                # snapshot lock orders strictly after this page-out
                try:
                    await w.conn.call("gpu_snapshot", timeout=120)
                except Exception:
                    w.paged = False
                continue
            if w.paged:
                continue  # already costs ~no HBM; keep it warm
            reapable.append(w)
        alive = sum(1 for w in self.workers.values() if w.alive)
        excess = alive - max(floor, 1)
        for w in reapable[: max(excess, 0)]:
            w.draining = True
            try:
                await w.conn.send({"t": "shutdown"})
            except Exception:
                pass

    def _hosts_snapshot_fn(self, w: WorkerHandle) -> bool:
        for fid in w.functions_loaded:
            fdef = self.scheduler.functions.get(fid)
            if fdef is not None and fdef.metadata.get("enable_memory_snapshot"):
                return True
        return False

    async def _ensure_unpaged(self, w: WorkerHandle) -> None:
        """Restore a paged-out worker's GPU state before giving it work."""
        if w.paged:
            w.paged = False
            try:
                await w.conn.call("gpu_restore", timeout=300)
            except Exception:
                pass

    async def _health_loop(self) -> None:
        """Worker supervision beyond process liveness: a worker that stops
        heartbeating (hung GPU kernel, wedged loop) is declared dead and its
        in-flight inputs requeue (the INTERNAL_FAILURE path). The MI355X
        analog of the reference's server-side container health tracking."""
        while True:
            await asyncio.sleep(15.0)
            try:
                await self._scaledown_once()
            except Exception:
                pass
            now = time.time()
            for w in list(self.workers.values()):
                if not w.alive:
                    continue
                stale = now - w.last_heartbeat
                if stale > 60.0 and w.inflight:
                    self.scheduler.log(
                        f"worker {w.worker_id} missed heartbeats for {stale:.0f}s; recycling"
                    )
                    try:
                        if w.proc is not None:
                            w.proc.kill()
                    except Exception:
                        pass
                    await w.conn.close()  # triggers _watch_worker requeue

    async def stop(self) -> None:
        self._stopping = True
        for task in (self._dispatch_task, self._retry_task, getattr(self, "_health_task", None)):
            if task is not None:
                task.cancel()
        for w in list(self.workers.values()):
            try:
                await w.conn.send({"t": "shutdown"})
            except Exception:
                pass
        # give workers a moment to exit cleanly, then kill
        deadline = time.time() + 3.0
        for w in list(self.workers.values()):
            if w.proc is not None:
                try:
                    w.proc.wait(timeout=max(0.05, deadline - time.time()))
                except subprocess.TimeoutExpired:
                    w.proc.kill()
        for w in list(self.workers.values()):
            await w.conn.close()
        # reap any process that never connected (or is still exiting)
        for proc in self._procs:
            if proc.poll() is None:
                try:
                    proc.terminate()
                    proc.wait(timeout=2)
                except Exception:
                    try:
                        proc.kill()
                    except Exception:
                        pass
        if self._server is not None:
            self._server.close()
            try:
                await self._server.wait_closed()
            except Exception:
                pass
        try:
            os.unlink(self.socket_path)
        except OSError:
            pass

    # -- worker spawning -------------------------------------------------
    def _gpu_count(self) -> int:
        if os.environ.get("MODAL_AMD_FAKE_GPUS"):
            return int(os.environ["MODAL_AMD_FAKE_GPUS"])
        try:
            import torch

            if torch.cuda.is_available():
                return torch.cuda.device_count()
        except Exception:
            pass
        return 0

    async def spawn_worker(self, gpu_index: Optional[int] = None, extra_env: Optional[dict] = None) -> Any:
        """Spawn one worker process; it will connect back to our socket."""
        worker_id = self._next_worker_id
        self._next_worker_id += 1
        env = dict(os.environ)
        env["MODAL_AMD_WORKER_SOCKET"] = self.socket_path
        env["MODAL_AMD_WORKER_ID"] = str(worker_id)
        env["MODAL_AMD_IS_REMOTE"] = "1"
        if gpu_index is not None:
            env["MODAL_AMD_GPU_INDEX"] = str(gpu_index)
            env["HIP_VISIBLE_DEVICES"] = str(gpu_index)
            env["CUDA_VISIBLE_DEVICES"] = str(gpu_index)
        if extra_env:
            env.update(extra_env)
        self._pending_spawns += 1
        proc = subprocess.Popen(
            [sys.executable, "-m", "modal_amd.runtime.worker"],
            env=env,
            cwd=os.getcwd(),
            start_new_session=True,
        )
        self._procs.append(proc)
        return proc

    async def ensure_workers(self, needs_gpu: bool) -> None:
        """Lazily bring up the pool sized to the hardware (288 GB/GPU MI355X:
        one worker per GPU; CPU functions get a small CPU pool)."""
        have = [w for w in self.workers.values() if w.alive and (w.has_gpu or not needs_gpu)]
        if needs_gpu:
            have = [w for w in have if w.has_gpu]
        if have or self._pending_spawns > 0:
            return
        async with self._spawn_lock:
            have = [w for w in self.workers.values() if w.alive and ((w.has_gpu and needs_gpu) or not needs_gpu)]
            if have or self._pending_spawns > 0:
                return
            from ..config import config

            n_gpus = self._gpu_count()
            if needs_gpu or n_gpus > 0:
                count = config.get("worker_count") or max(n_gpus, 1)
                for i in range(count):
                    await self.spawn_worker(gpu_index=i % n_gpus if n_gpus else None)
            else:
                count = config.get("worker_count") or min(max((_CPU_COUNT or 4) // 2, 1), 8)
                for _ in range(count):
                    await self.spawn_worker(gpu_index=None)

    async def wait_for_worker(self, needs_gpu: bool, timeout: float = 120.0) -> None:
        deadline = time.time() + timeout
        while True:
            if any(w.alive and (w.has_gpu or not needs_gpu) for w in self.workers.values()):
                return
            await self.ensure_workers(needs_gpu)
            self._worker_ready.clear()
            remaining = deadline - time.time()
            if remaining <= 0:
                raise TimeoutError("No worker became available")
            try:
                await asyncio.wait_for(self._worker_ready.wait(), min(remaining, 1.0))
            except asyncio.TimeoutError:
                pass

    # -- connections -----------------------------------------------------
    async def _on_connect(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter) -> None:
        conn: Connection = None  # type: ignore[assignment]

        handle_holder: dict = {}

        authed = False

        async def handler(msg: dict) -> None:
            nonlocal authed
            kind = msg.get("t")
            if not authed:
                # handshake gate: nothing (RPCs included) is served until a
                # hello carrying the run_dir token arrives
                if kind != "hello" or msg.get("auth") != self.scheduler.auth_token:
                    await conn.close()
                    return
                authed = True
                conn.rpc_target = self.scheduler.rpc_adapter
            if kind == "hello":
                role = msg.get("role", "worker")
                if role == "worker":
                    worker_id = msg.get("worker_id")
                    if worker_id is None or worker_id < 0:
                        worker_id = self._next_worker_id
                        self._next_worker_id += 1
                    gpu_index = msg.get("gpu_index")
                    handle = WorkerHandle(
                        worker_id,
                        conn,
                        gpu_index,
                        external=bool(msg.get("external")),
                    )
                    handle_holder["h"] = handle
                    self.workers[worker_id] = handle
                    self._pending_spawns = max(0, self._pending_spawns - 1)
                    self._worker_ready.set()
                    self._dispatch_wake.set()
                    # shm ring pair: bulk frames bypass the socket
                    to_worker = os.path.join(self.run_dir, f"ring.{handle.task_id}.in")
                    from_worker = os.path.join(self.run_dir, f"ring.{handle.task_id}.out")
                    rings_ok = conn.attach_rings(to_worker, from_worker, create=True)
                    await conn.send(
                        {
                            "t": "hello_ack",
                            "task_id": handle.task_id,
                            "worker_id": worker_id,
                            "ring_in": to_worker if rings_ok else None,
                            "ring_out": from_worker if rings_ok else None,
                        }
                    )
                    asyncio.get_running_loop().create_task(self._watch_worker(handle))
                else:
                    # tooling/sandbox client connection: RPC only
                    await conn.send({"t": "hello_ack"})
                return
            if kind == "putc":
                # fire-and-forget chunk intake from a proxied client's pump
                asyncio.get_running_loop().create_task(
                    self.scheduler.function_put_chunk(**msg["p"])
                )
                return
            handle = handle_holder.get("h")
            if handle is None:
                return
            if kind == "outputs":
                self._on_outputs(handle, msg)
            elif kind == "outputs_chunk":
                self._on_outputs_chunk(handle, msg)
            elif kind == "chunk_done":
                self._on_chunk_done(handle, msg)
            elif kind == "gen_data":
                self.scheduler.on_generator_data(msg)
            elif kind == "hb":
                handle.last_heartbeat = time.time()
                if msg.get("gpu"):
                    handle.gpu_stats = msg["gpu"]
            elif kind == "mesh_ready":
                self.scheduler.on_mesh_ready(handle, bool(msg.get("ok")))
            elif kind == "log":
                self.scheduler.on_worker_log(handle, msg)

        conn = Connection(reader, writer, handler, rpc_target=None)  # set after auth
        conn.start()

    async def _watch_worker(self, handle: WorkerHandle) -> None:
        await handle.conn.wait_closed()
        handle.alive = False
        self.workers.pop(handle.worker_id, None)
        if self._stopping:
            return
        # requeue in-flight inputs: the INTERNAL_FAILURE path
        for token, entry in list(handle.inflight.items()):
            if type(entry) is tuple:  # chunk group
                _tag, record, group = entry
                if group.state == "done" or record.cancelled:
                    continue
                group.internal_failures += 1
                if group.internal_failures > MAX_INTERNAL_FAILURE_COUNT:
                    for ci in range(group.count):
                        rec = record.materialize_chunk_item(group, ci)
                        self.scheduler.finalize_input(
                            rec,
                            GENERIC_STATUS_INTERNAL_FAILURE,
                            None,
                            0,
                            f"worker died while executing chunk (x{group.internal_failures})",
                            rec.retry_count,
                        )
                    group.state = "done"
                else:
                    group.state = "pending"
                    group.worker_id = None
                    self.enqueue_chunk(record, group, record.function_id, front=True)
                continue
            rec = entry
            if rec.final:
                continue
            rec.internal_failures += 1
            rec.worker_id = None
            if rec.internal_failures > MAX_INTERNAL_FAILURE_COUNT:
                self.scheduler.finalize_input(
                    rec,
                    GENERIC_STATUS_INTERNAL_FAILURE,
                    None,
                    0,
                    f"worker died while executing input (x{rec.internal_failures})",
                    rec.retry_count,
                )
            else:
                self.enqueue(rec, front=True)
        handle.inflight.clear()

    # -- output handling -------------------------------------------------
    def _on_outputs(self, handle: WorkerHandle, msg: dict) -> None:
        for item in msg["items"]:
            token = item["token"]
            rec = handle.inflight.pop(token, None)
            call_id, idx_s, retry_s = token.rsplit(":", 2)
            if rec is not None:
                fid = rec.call_id  # noqa: F841  (token bookkeeping)
            fdef_id = item.get("function_id")
            if fdef_id:
                cnt = handle.outstanding.get(fdef_id, 0)
                if cnt > 0:
                    handle.outstanding[fdef_id] = cnt - 1
            self.scheduler.on_worker_output(
                call_id=call_id,
                idx=int(idx_s),
                retry_count=int(retry_s),
                status=item["status"],
                output=item.get("data"),
                output_format=item.get("format", 0),
                exc_repr=item.get("exc"),
                output_blob=item.get("data_blob"),
            )
        self._dispatch_wake.set()

    def _on_outputs_chunk(self, handle: WorkerHandle, msg: dict) -> None:
        """One pickled value-list covering many outputs (worker fast path);
        bulk bookkeeping grouped per call."""
        tokens = msg["tokens"]
        chunk_id = self.scheduler.register_out_chunk(msg["data"], len(tokens))
        fdef_id = msg.get("function_id")
        if fdef_id:
            cnt = handle.outstanding.get(fdef_id, 0)
            handle.outstanding[fdef_id] = max(cnt - len(tokens), 0)
        inflight_pop = handle.inflight.pop
        by_call: dict[str, list] = {}
        for ci, token in enumerate(tokens):
            inflight_pop(token, None)
            call_id, idx_s, retry_s = token.rsplit(":", 2)
            by_call.setdefault(call_id, []).append((int(idx_s), int(retry_s), ci))
        for call_id, triples in by_call.items():
            record = self.scheduler.calls.get(call_id)
            if record is not None:
                record.post_outputs_bulk(triples, chunk_id)
                if getattr(record, "durable", False):
                    # durable spawn calls journal results with the bytes
                    # EXTRACTED (the shared chunk dies with this process)
                    import pickle as _pickle

                    from . import wal

                    values = _pickle.loads(msg["data"])
                    for idx, _retry, ci in triples:
                        rec = record.inputs.get(idx)
                        if rec is not None and rec.final:
                            wal.journal_result(
                                self.scheduler, record, rec,
                                output=_pickle.dumps(values[ci], 4),
                                output_format=1,
                            )
                    if (
                        record.num_inputs_final is not None
                        and record.completed >= record.num_inputs_final
                    ):
                        wal.drop(self.scheduler, record.call_id)
        self._dispatch_wake.set()

    # -- dispatch --------------------------------------------------------
    def enqueue(self, rec: InputRecord, front: bool = False, function_id: Optional[str] = None) -> None:
        fid = function_id or self._function_id_of(rec)
        q = self.pending.setdefault(fid, deque())
        if front:
            q.appendleft(rec)
        else:
            q.append(rec)
        self._dispatch_wake.set()

    def enqueue_many(self, function_id: str, recs: list) -> None:
        q = self.pending.setdefault(function_id, deque())
        q.extend(recs)
        self._dispatch_wake.set()

    def enqueue_chunk(self, record: Any, group: Any, function_id: str, front: bool = False) -> None:
        """Queue a whole chunk group as one pending descriptor."""
        q = self.pending.setdefault(function_id, deque())
        entry = ("g", record, group)
        if front:
            q.appendleft(entry)
        else:
            q.append(entry)
        self._dispatch_wake.set()

    def enqueue_delayed(self, rec: InputRecord, delay_s: float) -> None:
        self._delay_seq += 1
        heapq.heappush(
            self.delayed, (time.time() + delay_s, self._delay_seq, self._function_id_of(rec), rec)
        )

    def _function_id_of(self, rec: InputRecord) -> str:
        return self.scheduler.calls[rec.call_id].function_id

    async def _retry_loop(self) -> None:
        while True:
            if not self.delayed:
                await asyncio.sleep(0.05)
                continue
            ready_at, _, fid, rec = self.delayed[0]
            now = time.time()
            if ready_at <= now:
                heapq.heappop(self.delayed)
                if not rec.final and not rec.cancelled:
                    self.pending.setdefault(fid, deque()).append(rec)
                    self._dispatch_wake.set()
            else:
                await asyncio.sleep(min(ready_at - now, 0.5))

    async def _dispatch_loop(self) -> None:
        while True:
            await self._dispatch_wake.wait()
            self._dispatch_wake.clear()
            try:
                await self._dispatch_once()
            except asyncio.CancelledError:
                raise
            except Exception as exc:  # pragma: no cover - defensive
                self.scheduler.log(f"dispatch error: {exc!r}")

    async def _dispatch_once(self) -> None:
        for fid, q in list(self.pending.items()):
            if not q:
                continue
            fdef = self.scheduler.functions.get(fid)
            if fdef is None:
                q.clear()
                continue
            await self.ensure_workers(fdef.needs_gpu)
            candidates = [
                w
                for w in self.workers.values()
                if w.alive and not w.draining and (w.has_gpu or not fdef.needs_gpu)
            ]
            if fdef.needs_gpu:
                candidates = [w for w in candidates if w.has_gpu]
            candidates = self._apply_placement(fdef, candidates)
            if not candidates:
                continue
            # round-robin over workers by current outstanding (least-loaded first)
            candidates.sort(key=lambda w: w.outstanding.get(fid, 0))
            for w in candidates:
                if not q:
                    break
                credit = w.credit_for(fdef)
                if credit <= 0:
                    continue
                batch: list[InputRecord] = []
                while q and credit > 0:
                    entry = q.popleft()
                    if type(entry) is tuple:  # ("g", record, group) chunk descriptor
                        _tag, record, group = entry
                        if group.state != "pending" or record.cancelled:
                            continue
                        await self._send_group(w, record, group, fdef)
                        credit -= group.count
                        continue
                    rec = entry
                    if rec.final or rec.cancelled:
                        continue
                    batch.append(rec)
                    credit -= 1
                if batch:
                    await self._send_batch(w, fdef, batch)
            if q:
                # backlog remains with no free credit: autoscale
                # (parity: min/max_containers autoscaler settings,
                # reference _functions.py:1195-1292)
                await self._maybe_scale_up(fdef, backlog=len(q), active=len(candidates))

    def _apply_placement(self, fdef: FunctionDef, candidates: list) -> list:
        """GPU-affinity steering (SchedulerPlacement gpu_index/gpu_set —
        the single-node analog of region/zone constraints)."""
        placement = fdef.placement
        if not placement:
            return candidates
        gpu_index = placement.get("gpu_index")
        if gpu_index is not None:
            # strict pin: wait for that GPU's worker rather than mis-placing
            return [w for w in candidates if w.gpu_index == gpu_index]
        gpu_set = placement.get("gpu_set")
        if gpu_set:
            subset = [w for w in candidates if w.gpu_index in gpu_set]
            if subset:
                return subset
        return candidates

    async def _maybe_scale_up(self, fdef: FunctionDef, backlog: int, active: int) -> None:
        if self._pending_spawns > 0:
            return
        limit = self._pool_limit(fdef)
        if fdef.max_containers:
            limit = min(limit, fdef.max_containers)
        current = sum(
            1 for w in self.workers.values() if w.alive and (w.has_gpu or not fdef.needs_gpu)
        )
        if current >= limit:
            return
        want = min(limit - current, max(1, backlog // DEFAULT_PIPELINE_DEPTH))
        n_gpus = self._gpu_count()
        for i in range(want):
            gpu_index = (current + i) % n_gpus if (fdef.needs_gpu and n_gpus) else (
                (current + i) % n_gpus if n_gpus else None
            )
            await self.spawn_worker(gpu_index=gpu_index)

    async def ensure_min(self, fdef: FunctionDef) -> None:
        """Warm pool: keep min_containers workers alive for this function."""
        want = min(fdef.min_containers + fdef.buffer_containers, self._pool_limit(fdef))
        current = sum(
            1 for w in self.workers.values() if w.alive and (w.has_gpu or not fdef.needs_gpu)
        ) + self._pending_spawns
        n_gpus = self._gpu_count()
        for i in range(max(0, want - current)):
            gpu_index = (current + i) % n_gpus if n_gpus else None
            await self.spawn_worker(gpu_index=gpu_index if fdef.needs_gpu or n_gpus else None)

    def _pool_limit(self, fdef: FunctionDef) -> int:
        from ..config import config

        n_gpus = self._gpu_count()
        configured = config.get("worker_count")
        if fdef.needs_gpu:
            # >1 worker per GPU is a legitimate MI355X shape (288 GB HBM per
            # device easily hosts several concurrent payload processes)
            return max(configured or 0, n_gpus, 1)
        if configured:
            return configured
        return min(max((_CPU_COUNT or 4) // 2, 1), 8)

    async def _send_def(self, w: WorkerHandle, fdef: FunctionDef) -> None:
        await w.conn.send(
            {
                "t": "def",
                "function_id": fdef.function_id,
                "app_id": fdef.app_id,
                "name": fdef.name,
                "definition": fdef.definition,
                "definition_kind": fdef.definition_kind,
                "is_generator": fdef.is_generator,
                "timeout": fdef.timeout,
                "max_concurrent_inputs": fdef.max_concurrent_inputs,
                "batch_max_size": fdef.batch_max_size,
                "batch_linger_ms": fdef.batch_linger_ms,
                "version": fdef.definition_version,
                "app_layout": self.scheduler.app_layout(fdef.app_id),
                "env": self.scheduler.resolve_function_env(fdef),
                "volumes": fdef.volume_mounts,
                "python_paths": self.scheduler.resolve_function_pythonpaths(fdef),
                "web_config": fdef.web_config,
            }
        )
        w.functions_loaded.add(fdef.function_id)
        w.defs_version[fdef.function_id] = fdef.definition_version

    async def _send_batch(self, w: WorkerHandle, fdef: FunctionDef, batch: list[InputRecord]) -> None:
        try:
            await self._ensure_unpaged(w)
            if (
                fdef.function_id not in w.functions_loaded
                or w.defs_version.get(fdef.function_id, 0) != fdef.definition_version
            ):
                await self._send_def(w, fdef)
            w.last_active = time.time()
            items = []
            chunks_needed: dict[str, bytes] = {}
            record = None
            for rec in batch:
                rec.worker_id = w.worker_id
                rec.started_at = time.time()
                w.inflight[rec.token] = rec
                item = {
                    "token": rec.token,
                    "input_id": rec.input_id,
                    "method": rec.method_name,
                    "retry_count": rec.retry_count,
                }
                if rec.chunk_id:
                    item["chunk"] = rec.chunk_id
                    item["ci"] = rec.chunk_index
                    if rec.chunk_id not in w.chunks_sent:
                        if record is None or record.call_id != rec.call_id:
                            record = self.scheduler.calls.get(rec.call_id)
                        chunk = record.chunks.get(rec.chunk_id) if record else None
                        if chunk is not None:
                            chunks_needed[rec.chunk_id] = chunk["data"]
                            w.chunks_sent.add(rec.chunk_id)
                else:
                    item["payload"] = rec.payload
                    if rec.payload_blob:
                        item["payload_blob"] = rec.payload_blob
                if rec.cluster:
                    item["cluster"] = rec.cluster
                items.append(item)
            w.outstanding[fdef.function_id] = w.outstanding.get(fdef.function_id, 0) + len(batch)
            frame = {"t": "inputs", "function_id": fdef.function_id, "items": items}
            if chunks_needed:
                frame["chunks"] = chunks_needed
            await w.conn.send(frame)
        except Exception:
            # connection died mid-send: requeue, the watcher will clean up
            for rec in batch:
                if rec.token in w.inflight:
                    del w.inflight[rec.token]
                if not rec.final:
                    self.enqueue(rec, front=True)

    async def _send_group(self, w: WorkerHandle, record: Any, group: Any, fdef: FunctionDef) -> None:
        """One frame carries a whole chunk (range protocol)."""
        try:
            await self._ensure_unpaged(w)
            if (
                fdef.function_id not in w.functions_loaded
                or w.defs_version.get(fdef.function_id, 0) != fdef.definition_version
            ):
                await self._send_def(w, fdef)
            group.state = "inflight"
            group.worker_id = w.worker_id
            w.last_active = time.time()
            w.inflight[group.token] = ("g", record, group)
            w.outstanding[fdef.function_id] = (
                w.outstanding.get(fdef.function_id, 0) + group.count
            )
            chunk = record.chunks.get(group.chunk_id) or {}
            await w.conn.send(
                {
                    "t": "inputs_chunk",
                    "function_id": fdef.function_id,
                    "call_id": record.call_id,
                    "chunk_id": group.chunk_id,
                    "token": group.token,
                    "count": group.count,
                    "payload": chunk.get("data"),
                    "method": group.method,
                    "max_concurrent": fdef.max_concurrent_inputs,
                }
            )
        except Exception:
            w.inflight.pop(group.token, None)
            group.state = "pending"
            if not record.cancelled:
                self.enqueue_chunk(record, group, fdef.function_id, front=True)

    def _on_chunk_done(self, handle: WorkerHandle, msg: dict) -> None:
        """Completion of a range-protocol chunk."""
        from .calls import GENERIC_STATUS_FAILURE, GENERIC_STATUS_SUCCESS

        handle.inflight.pop(msg["token"], None)
        fid = msg.get("function_id")
        count = msg.get("count", 0)
        if fid:
            handle.outstanding[fid] = max(handle.outstanding.get(fid, 0) - count, 0)
        record = self.scheduler.calls.get(msg["call_id"])
        if record is None:
            return
        group = record.chunk_groups.get(msg["chunk_id"])
        if group is None or group.state == "done":
            return
        if msg.get("failed"):
            # worker-side setup failure (e.g. missing def): internal requeue
            group.internal_failures += 1
            if group.internal_failures <= MAX_INTERNAL_FAILURE_COUNT and not record.cancelled:
                group.state = "pending"
                self.enqueue_chunk(record, group, record.function_id, front=True)
            else:
                from .calls import GENERIC_STATUS_INTERNAL_FAILURE as _GIF

                for ci in range(group.count):
                    rec = record.materialize_chunk_item(group, ci)
                    self.scheduler.finalize_input(
                        rec, _GIF, None, 0, "chunk repeatedly failed to start", rec.retry_count
                    )
                group.state = "done"
            return
        # per-item exceptions materialize real records (the per-item FSM owns
        # retries from here)
        for ci_s, (data, repr_s) in (msg.get("exceptions") or {}).items():
            ci = int(ci_s)
            rec = record.materialize_chunk_item(group, ci)
            self.scheduler.on_worker_output(
                call_id=record.call_id,
                idx=rec.idx,
                retry_count=rec.retry_count,
                status=GENERIC_STATUS_FAILURE,
                output=data,
                output_format=1,
                exc_repr=repr_s,
            )
        # per-item success payloads (tensor/oversized fallback)
        for ci_s, data in (msg.get("items") or {}).items():
            ci = int(ci_s)
            rec = record.materialize_chunk_item(group, ci)
            self.scheduler.on_worker_output(
                call_id=record.call_id,
                idx=rec.idx,
                retry_count=rec.retry_count,
                status=GENERIC_STATUS_SUCCESS,
                output=data,
                output_format=1,
                exc_repr=None,
            )
        cis = msg.get("cis")  # None => all count items succeeded in order
        data = msg.get("data")
        if data is not None:
            slot = self.scheduler.register_out_chunk(data, count if cis is None else len(cis))
            record.complete_chunk_success(group, cis, slot)
            self.scheduler.metrics_counters["outputs_total"] += (
                count if cis is None else len(cis)
            )
        else:
            group.state = "done"
            record._check_done()
        self._dispatch_wake.set()

    async def dispatch_gang(self, fdef: FunctionDef, recs: list) -> None:
        """Place a gang on len(recs) distinct workers at once, bypassing the
        credit queue (gang members block on each other, so partial placement
        would deadlock)."""
        n = len(recs)
        deadline = time.time() + 120
        while True:
            await self.ensure_workers(fdef.needs_gpu)
            candidates = [
                w
                for w in self.workers.values()
                if w.alive and not w.draining and (w.has_gpu or not fdef.needs_gpu)
            ]
            if fdef.needs_gpu:
                candidates = [w for w in candidates if w.has_gpu]
            if len(candidates) >= n:
                break
            if time.time() > deadline:
                raise TimeoutError(
                    f"Gang of {n} workers unavailable (have {len(candidates)})"
                )
            await asyncio.sleep(0.05)
        candidates.sort(key=lambda w: (sum(w.outstanding.values()), w.worker_id))
        for rec, w in zip(recs, candidates):
            await self._send_batch(w, fdef, [rec])

    async def cancel_inputs(self, tokens: list[str], terminate: bool = False) -> None:
        """Propagate cancellation to workers holding these inputs
        (parity: server-pushed cancellation via heartbeats,
        reference container_io_manager.py:645-710)."""
        by_worker: dict[int, list[str]] = {}
        for w in self.workers.values():
            hit = [t for t in tokens if t in w.inflight]
            if hit:
                by_worker[w.worker_id] = hit
        for worker_id, toks in by_worker.items():
            w = self.workers.get(worker_id)
            if w is None:
                continue
            try:
                await w.conn.send({"t": "cancel", "tokens": toks, "terminate": terminate})
            except Exception:
                pass
