"""Worker process: the per-GPU execution runtime.

Re-implementation of the reference's container entrypoint + IO manager
(/root/reference/py/modal/_container_entrypoint.py:477,
/root/reference/py/modal/_runtime/container_io_manager.py:531) as a
single-node worker: one process per MI355X GPU (``HIP_VISIBLE_DEVICES``
pinned by the pool), connected to the scheduler over a Unix socket.

Differences from the reference, by design:
* inputs are *pushed* (credit-bounded) instead of long-polled;
* sync user functions run on a daemonized thread pool, async ones on the
  worker's event loop (parity: _container_entrypoint.py:193-266);
* dynamic batching transposes args and splits outputs per item
  (parity: container_io_manager.py:196-262);
* generator outputs stream over a data channel with a GeneratorDone tail
  (parity: container_io_manager.py:759-846).
"""

from __future__ import annotations

import asyncio
import contextvars
import inspect
import os
import sys
import time
import traceback
from concurrent.futures import ThreadPoolExecutor
from typing import Any, Optional

from .._serialization import (
    DataFormat,
    GeneratorDone,
    deserialize,
    deserialize_payload,
    serialize,
    serialize_data_format,
    set_client_context,
)
from ..scheduler.blobs import INLINE_LIMIT, BlobStore
from ..scheduler.calls import (
    GENERIC_STATUS_FAILURE,
    GENERIC_STATUS_INTERNAL_FAILURE,
    GENERIC_STATUS_SUCCESS,
    GENERIC_STATUS_TERMINATED,
    GENERIC_STATUS_TIMEOUT,
)
from ..exception import InputCancellation
from ..scheduler.transport import Connection
from .execution_context import _reset_current_context, _set_current_context

HEARTBEAT_INTERVAL = 15.0  # parity: reference config.py:318
OUTPUT_FLUSH_INTERVAL = 0.001  # batch outputs ~1 ms (beats the 20/req constant)
OUTPUT_BATCH_MAX = 512

_app_id_var: contextvars.ContextVar[str] = contextvars.ContextVar("modal_amd_app_id", default="")

import threading as _threading

_slot_tls = _threading.local()


def _slot_streams_enabled() -> bool:
    """One-time-per-range guard mirroring _invoke_on_slot_stream's checks."""
    torch = sys.modules.get("torch")
    cuda = getattr(torch, "cuda", None) if torch is not None else None
    try:
        return (
            cuda is not None
            and os.environ.get("MODAL_AMD_SLOT_STREAMS", "0") not in ("0", "false")
            and cuda.is_initialized()
        )
    except AttributeError:
        return False


def _invoke_on_slot_stream(fn: Any, args: tuple, kwargs: dict) -> Any:
    """Run a user callable on this executor thread's own HIP stream.

    Concurrency slots each get a side stream, so one item's device sync
    (``.item()``, ``.cpu()``) waits only its own work instead of convoying
    every slot's kernels on the default stream. OFF by default: measured
    2x SLOWER on the map bench in BOTH variants on MI355X — with a host
    ``stream.synchronize()`` per item (21.5k vs 44.3k items/s) AND with
    device-side event ordering only (20.8k/17.6k vs 41.7k/34.0k, same
    box) — many active HSA queues cost more than the default-stream
    convoy for microsecond ops. Opt in with MODAL_AMD_SLOT_STREAMS=1 for
    workloads with long per-item kernels.
    """
    torch = sys.modules.get("torch")
    # sys.modules can expose a *partially initialized* torch while another
    # thread is mid-import; fall back to a plain call until it is whole.
    cuda = getattr(torch, "cuda", None) if torch is not None else None
    try:
        if (
            cuda is None
            or os.environ.get("MODAL_AMD_SLOT_STREAMS", "0") in ("0", "false")
            or not cuda.is_initialized()
        ):
            return fn(*args, **kwargs)
    except AttributeError:
        return fn(*args, **kwargs)
    stream = getattr(_slot_tls, "stream", None)
    if stream is None:
        stream = torch.cuda.Stream()
        _slot_tls.stream = stream
    with torch.cuda.stream(stream):
        result = fn(*args, **kwargs)
    # order the default stream after this slot's work WITHOUT a host sync
    # (the first attempt host-blocked in stream.synchronize() per item and
    # measured 2x slower; an event wait is device-side only)
    ev = torch.cuda.Event()
    ev.record(stream)
    torch.cuda.default_stream().wait_event(ev)
    return result

#: the live worker runtime of this process (None in client processes);
#: serialization hooks use it to export/fetch device tensors
RUNTIME: Optional["WorkerRuntime"] = None


class WorkerRPCTarget:
    """RPCs the scheduler can invoke ON a worker (reverse direction)."""

    def __init__(self, runtime: "WorkerRuntime"):
        self._runtime = runtime

    async def tensor_pull(self, token: str) -> Optional[bytes]:
        """Host-staged export of a registered tensor (for non-mesh consumers)."""
        tensor = self._runtime.tensor_table.get(token)
        if tensor is None:
            return None
        import pickle as _pickle

        loop = asyncio.get_running_loop()
        return await loop.run_in_executor(
            None, lambda: _pickle.dumps(tensor.detach().cpu(), 4)
        )

    async def ping(self) -> str:
        return self._runtime.task_id

    async def app_stop(self, app_id: str) -> bool:
        """Synchronous teardown: the scheduler awaits @exit hooks and the
        exit-time volume commit before app.run() returns."""
        await self._runtime._app_teardown(app_id)
        return True

    async def gpu_snapshot(self) -> dict:
        """Page this worker's GPU tensors to host memory (the scaledown
        alternative for enable_memory_snapshot functions: HBM is freed,
        the warm process and its loaded user code survive)."""
        rt = self._runtime
        async with rt.snapshot_lock:  # RPCs serve concurrently: a restore
            # racing this page-out must strictly order after it
            if rt.mem_snapshot is not None:
                return {"state": rt.mem_snapshot.state.value}
            from .gpu_snapshot import GPUMemorySnapshot

            snap = GPUMemorySnapshot()
            await asyncio.get_running_loop().run_in_executor(None, snap.checkpoint)
            rt.mem_snapshot = snap
            return {"state": snap.state.value}

    async def exec_command(self, cmd: list, timeout: float = 60.0) -> dict:
        """Run a command in this worker's context (parity: `modal
        container exec`, reference cli/container.py:297 — the local
        analog executes in the worker process's env/cwd)."""
        proc = await asyncio.create_subprocess_exec(
            *cmd,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.STDOUT,
            env=dict(os.environ),
        )
        try:
            out, _ = await asyncio.wait_for(proc.communicate(), timeout)
        except asyncio.TimeoutError:
            proc.kill()
            return {"returncode": -1, "output": "(timed out)"}
        return {"returncode": proc.returncode, "output": out[-65536:].decode(errors="replace")}

    async def gpu_restore(self) -> dict:
        rt = self._runtime
        async with rt.snapshot_lock:
            snap, rt.mem_snapshot = rt.mem_snapshot, None
            if snap is not None:
                await asyncio.get_running_loop().run_in_executor(None, snap.restore)
            return {"state": "running"}

    async def snapshot_state(self) -> bytes:
        """Cross-process snapshot payload: registered tensors, tracked raw
        hipMalloc allocations, RNG state (gpu_snapshot.capture_payload).
        The scheduler stores it and can restore it into a FRESH worker."""
        from .gpu_snapshot import capture_payload

        return await asyncio.get_running_loop().run_in_executor(None, capture_payload)


class FunctionRuntime:
    """Worker-side state for one registered function."""

    def __init__(self, runtime: "WorkerRuntime", msg: dict):
        self.runtime = runtime
        self.function_id = msg["function_id"]
        self.app_id = msg.get("app_id", "")
        self.name = msg.get("name", "")
        self.is_generator = bool(msg.get("is_generator"))
        self.timeout = msg.get("timeout")
        self.max_concurrent = max(int(msg.get("max_concurrent_inputs") or 1), 1)
        self.batch_max_size = int(msg.get("batch_max_size") or 0)
        self.batch_linger_ms = int(msg.get("batch_linger_ms") or 0)
        self.version = msg.get("version", 1)
        self.definition_blob = msg["definition"]
        self.definition_kind = msg.get("definition_kind", "serialized")
        self.env = msg.get("env") or {}
        self.volumes = msg.get("volumes") or {}
        self.python_paths = msg.get("python_paths") or []
        self.web_config = msg.get("web_config")
        self._web_runtime: Any = None
        self.sem = asyncio.Semaphore(self.max_concurrent)
        self._callable: Any = None
        self._service: Any = None
        self._loaded = False
        self._load_error: Optional[BaseException] = None
        self._enter_done = False
        # dynamic batching state
        self._batch: list[tuple[dict, tuple, dict]] = []
        self._batch_flush_handle: Optional[asyncio.TimerHandle] = None

    def load(self) -> Any:
        """Deserialize/import the user function (once per definition version)."""
        if self._loaded:
            if self._load_error is not None:
                raise self._load_error
            return self._callable
        try:
            if self.env:
                os.environ.update(self.env)  # image + secret env bundles
            for p in self.python_paths:
                if p not in sys.path:
                    sys.path.insert(0, p)  # image site-packages
            if self.volumes:
                from .volumes import mount_volumes

                mount_volumes(self.volumes, self.runtime)
            obj = deserialize(self.definition_blob)
            if isinstance(obj, dict) and obj.get("kind") == "cls_service":
                # class-backed service: instantiate, run @enter hooks lazily
                self._service = _ClsService(obj)
                self._callable = None
            else:
                self._callable = obj
            self._loaded = True
            return self._callable
        except BaseException as exc:
            self._load_error = exc
            self._loaded = True
            raise

    def get_callable(self, method_name: str) -> Any:
        fn = self.load()
        if self._service is not None:
            return self._service.get_method(method_name)
        return fn


class _ClsService:
    """Instantiated class service with lifecycle hooks (parity:
    user_code_imports.py:388 ImportedClass; lifecycle hooks
    _partial_function.py:589,617)."""

    def __init__(self, spec: dict):
        self.cls = spec["cls"]
        self.params_args = spec.get("args", ())
        self.params_kwargs = spec.get("kwargs", {})
        self.instance: Any = None
        self._entered = False

    def _ensure_instance(self) -> Any:
        if self.instance is None:
            from ..cls import _Parameter

            if "__init__" in vars(self.cls):
                self.instance = self.cls(*self.params_args, **self.params_kwargs)
            else:
                inst = self.cls.__new__(self.cls)
                for name, value in vars(self.cls).items():
                    if isinstance(value, _Parameter) and value.default is not ...:
                        setattr(inst, name, value.default)
                for key, value in self.params_kwargs.items():
                    setattr(inst, key, value)
                self.instance = inst
        if not self._entered:
            self._entered = True
            # snap=True hooks run first — their state is what a memory
            # snapshot captures; plain enter hooks model post-restore work
            # (parity: enter pre/post-snapshot, _partial_function.py:589)
            for hook_name in _lifecycle_hooks(self.cls, "enter_snap"):
                getattr(self.instance, hook_name)()
            for hook_name in _lifecycle_hooks(self.cls, "enter"):
                getattr(self.instance, hook_name)()
        return self.instance

    def get_method(self, method_name: str) -> Any:
        import functools

        inst = self._ensure_instance()
        raw = inspect.getattr_static(type(inst), method_name)
        raw_fn = getattr(raw, "raw_f", None) or raw
        # functools.partial preserves generator/coroutine detection in inspect
        return functools.partial(raw_fn, inst)

    def exit(self) -> None:
        if self.instance is not None and self._entered:
            for hook_name in _lifecycle_hooks(self.cls, "exit"):
                try:
                    getattr(self.instance, hook_name)()
                except Exception:
                    traceback.print_exc()


def _lifecycle_hooks(cls: type, kind: str) -> list[str]:
    out = []
    for name in dir(cls):
        attr = inspect.getattr_static(cls, name, None)
        if getattr(attr, "_modal_amd_lifecycle", None) == kind:
            out.append(name)
    return out


class _LogForwarder:
    """Tee stdout/stderr lines to the scheduler (parity: container log streaming)."""

    def __init__(self, runtime: "WorkerRuntime", orig: Any, fd: int):
        self.runtime = runtime
        self.orig = orig
        self.fd = fd

    def write(self, data: str) -> int:
        self.orig.write(data)
        if data:
            self.runtime.post_log(self.fd, data)
        return len(data)

    def flush(self) -> None:
        self.orig.flush()

    def isatty(self) -> bool:
        return False

    def fileno(self) -> int:
        return self.orig.fileno()


class WorkerRuntime:
    def __init__(self) -> None:
        self.worker_id = int(os.environ.get("MODAL_AMD_WORKER_ID", "-1"))
        self.gpu_index = (
            int(os.environ["MODAL_AMD_GPU_INDEX"]) if "MODAL_AMD_GPU_INDEX" in os.environ else None
        )
        self.socket_path = os.environ["MODAL_AMD_WORKER_SOCKET"]
        self.external = os.environ.get("MODAL_AMD_EXTERNAL_WORKER") == "1"
        self.conn: Optional[Connection] = None
        self.task_id: str = ""
        self.functions: dict[str, FunctionRuntime] = {}
        self.mem_snapshot: Any = None  # GPUMemorySnapshot while paged out
        self.snapshot_lock = asyncio.Lock()  # orders page-out vs restore RPCs
        # token -> executor-thread idents running that input's sync code
        self._sync_threads: dict[str, set] = {}
        self._sync_threads_lock = _threading.Lock()
        self.executor = ThreadPoolExecutor(
            max_workers=int(os.environ.get("MODAL_AMD_WORKER_THREADS", "16")),
            thread_name_prefix="modal-amd-input",
        )
        run_dir = os.path.dirname(self.socket_path)
        self.blob_store = BlobStore(os.path.join(run_dir, "blobs"))
        from .devicemesh import DeviceMesh, TensorTable

        self.mesh = DeviceMesh()
        self.tensor_table = TensorTable()
        # shared chunk payloads (map fast path): chunk_id -> entry
        self._chunk_cache: dict[str, dict] = {}
        self._chunk_order: list[str] = []
        self._outbox: list[dict] = []
        self._outbox_flush_scheduled = False
        self._running: dict[str, asyncio.Task] = {}
        self._abandoned: set[str] = set()  # tokens whose (sync) execution timed out
        self._shutdown = asyncio.Event()
        self._hello_ack = asyncio.Event()
        self._log_buffer: list[tuple[int, str]] = []
        self._log_flush_scheduled = False
        self.loop: Optional[asyncio.AbstractEventLoop] = None

    # ---- wiring ---------------------------------------------------------
    async def run(self) -> None:
        global RUNTIME
        RUNTIME = self
        self.loop = asyncio.get_running_loop()
        if os.environ.get("MODAL_AMD_RESTORE_STATE_PATH"):
            # restored worker: busy-wait for restore-state.json, rehydrate
            # tracked GPU state, exit 222 on failure (parity: reference
            # task_lifecycle_manager.py:146-193 + sentinel contract)
            from .gpu_snapshot import wait_and_restore_from_state_file

            await asyncio.get_running_loop().run_in_executor(
                None, wait_and_restore_from_state_file
            )
        deadline = time.time() + (60.0 if self.external else 0.0)
        while True:
            try:
                reader, writer = await asyncio.open_unix_connection(self.socket_path)
                break
            except (FileNotFoundError, ConnectionRefusedError):
                # external workers (torchrun ranks) may race the scheduler's
                # socket (re)creation; spawned workers exit quietly
                if time.time() >= deadline:
                    return
                await asyncio.sleep(0.05)
        self.conn = Connection(reader, writer, self._handle, rpc_target=WorkerRPCTarget(self))
        self.conn.start()
        from ..scheduler.core import read_auth_token

        await self.conn.send(
            {
                "t": "hello",
                "role": "worker",
                "worker_id": self.worker_id,
                "gpu_index": self.gpu_index,
                "external": self.external,
                "pid": os.getpid(),
                "auth": read_auth_token(self.socket_path),
            }
        )
        # install the process-default client so user code's handles bind here
        from ..client import _Client
        from .. import client as client_mod  # noqa: F401

        # user code's handles: blocking calls run on a per-thread loop in
        # the calling thread with a per-loop socket (UserCodeProxy), so a
        # handle RPC is one socket round trip with ZERO cross-thread hops
        # (each hop costs ~1-2 ms on contended hosts)
        from .._sync import synchronizer as _synchronizer

        _synchronizer.thread_local_blocking = os.environ.get(
            "MODAL_AMD_TL_BLOCKING", "1"
        ) not in ("0", "false")
        proxy_client = _Client(
            client_mod.UserCodeProxy(self.socket_path), "container",
            run_dir=os.path.dirname(self.socket_path),  # shared CAS for big args
        )
        _Client.set_default(proxy_client)
        set_client_context(proxy_client)

        sys.stdout = _LogForwarder(self, sys.__stdout__, 1)  # type: ignore[assignment]
        sys.stderr = _LogForwarder(self, sys.__stderr__, 2)  # type: ignore[assignment]

        hb_task = asyncio.get_running_loop().create_task(self._heartbeat_loop())
        closed_task = asyncio.get_running_loop().create_task(self.conn.wait_closed())
        stop_task = asyncio.get_running_loop().create_task(self._shutdown.wait())
        await asyncio.wait({closed_task, stop_task}, return_when=asyncio.FIRST_COMPLETED)
        hb_task.cancel()
        for frt in self.functions.values():
            if frt._service is not None:
                frt._service.exit()
        await self._flush_outbox()
        await self.conn.close()

    def _gpu_stats(self) -> Optional[dict]:
        """HBM gauges for the scheduler's per-GPU view."""
        torch = sys.modules.get("torch")
        if torch is None or not torch.cuda.is_initialized():
            return None
        try:
            free, total = torch.cuda.mem_get_info()
            return {
                "hbm_total": total,
                "hbm_free": free,
                "hbm_reserved": torch.cuda.memory_reserved(),
            }
        except Exception:
            return None

    async def _heartbeat_loop(self) -> None:
        while True:
            await asyncio.sleep(HEARTBEAT_INTERVAL)
            try:
                msg: dict = {"t": "hb"}
                stats = self._gpu_stats()
                if stats:
                    msg["gpu"] = stats
                await self.conn.send(msg)
            except Exception:
                return

    async def _handle(self, msg: dict) -> None:
        kind = msg.get("t")
        if kind == "hello_ack":
            self.task_id = msg.get("task_id", "")
            os.environ["MODAL_AMD_TASK_ID"] = self.task_id
            if msg.get("ring_in"):
                # worker's out-ring is the scheduler's in-ring and vice versa
                self.conn.attach_rings(msg.get("ring_out"), msg.get("ring_in"), create=False)
            self._hello_ack.set()
        elif kind == "def":
            frt = FunctionRuntime(self, msg)
            self.functions[frt.function_id] = frt
        elif kind == "inputs":
            for cid, data in (msg.get("chunks") or {}).items():
                self._chunk_cache[cid] = {"raw": data, "decoded": None}
                self._chunk_order.append(cid)
                while len(self._chunk_order) > 512:
                    old = self._chunk_order.pop(0)
                    self._chunk_cache.pop(old, None)
            frt = self.functions.get(msg["function_id"])
            if frt is None:
                for item in msg["items"]:
                    self.post_output(
                        item["token"],
                        msg["function_id"],
                        GENERIC_STATUS_INTERNAL_FAILURE,
                        None,
                        0,
                        "worker missing function definition",
                    )
                return
            if (
                frt.batch_max_size <= 1
                and not frt.is_generator
                and not frt.timeout
                and len(msg["items"]) > 1
            ):
                # fast path: run the frame in max_concurrent executor chunks —
                # the per-item thread round-trip dominates tiny map items
                items = msg["items"]
                n_chunks = min(frt.max_concurrent, max(1, len(items) // 4))
                size = (len(items) + n_chunks - 1) // n_chunks
                for start in range(0, len(items), size):
                    chunk = items[start : start + size]
                    task = asyncio.get_running_loop().create_task(
                        self._run_frame_fast(frt, chunk)
                    )
                    for item in chunk:
                        self._running[item["token"]] = task
                return
            for item in msg["items"]:
                if frt.batch_max_size > 1:
                    self._batch_add(frt, item)
                else:
                    task = asyncio.get_running_loop().create_task(self._run_input(frt, item))
                    self._running[item["token"]] = task
                    task.add_done_callback(lambda _t, tok=item["token"]: self._running.pop(tok, None))
        elif kind == "inputs_chunk":
            frt = self.functions.get(msg["function_id"])
            if frt is None:
                await self.conn.send(
                    {
                        "t": "chunk_done",
                        "token": msg["token"],
                        "call_id": msg["call_id"],
                        "chunk_id": msg["chunk_id"],
                        "function_id": msg["function_id"],
                        "count": msg["count"],
                        "failed": True,
                    }
                )
                return
            task = asyncio.get_running_loop().create_task(self._run_chunk(frt, msg))
            self._running[msg["token"]] = task
            task.add_done_callback(
                lambda _t, tok=msg["token"]: self._running.pop(tok, None)
            )
        elif kind == "cancel":
            for token in msg.get("tokens", []):
                task = self._running.get(token)
                if task is not None:
                    task.cancel()
                else:
                    self._abandoned.add(token)
                # sync user code on executor threads: deliver the
                # cancellation as an async exception at the next bytecode
                # (SURVEY hard part 4: the SIGUSR1 analog for thread pools)
                self._inject_cancel(token)
            if msg.get("terminate"):
                self._shutdown.set()
        elif kind == "mesh_init":
            self.mesh.init(msg["rank"], msg["world"], msg["port"], msg["backend"])

            def wait_and_ack() -> bool:
                return self.mesh.wait_ready()

            ok = await asyncio.get_running_loop().run_in_executor(self.executor, wait_and_ack)
            await self.conn.send({"t": "mesh_ready", "ok": ok, "rank": msg["rank"]})
        elif kind == "dev_send":
            tensor = self.tensor_table.get(msg["token"])
            if tensor is not None:
                self.mesh.submit_send(tensor, msg["dst_rank"])
        elif kind == "dev_recv":
            self.mesh.submit_recv(msg["xfer_id"], msg["meta"], msg["src_rank"])
        elif kind == "app_stop":
            await self._app_teardown(msg.get("app_id"))
        elif kind == "shutdown":
            self._shutdown.set()

    # ---- device tensors --------------------------------------------------
    def fetch_device_tensor(self, owner_task: str, token: str, meta: dict) -> Any:
        """Materialize another worker's exported tensor (executor threads
        only: blocks on the mesh transfer or the host-staged pull)."""
        if owner_task == self.task_id:
            tensor = self.tensor_table.get(token)
            if tensor is not None:
                return tensor
        from .._sync import synchronizer
        from ..client import _Client

        svc = _Client._singleton.svc
        resp = synchronizer.run(
            svc.device_transfer(
                owner_task=owner_task, token=token, meta=meta, dest_task=self.task_id
            )
        )
        if resp and resp.get("xfer_id"):
            return self.mesh.wait_result(resp["xfer_id"])
        raw = synchronizer.run(svc.tensor_pull_relay(owner_task=owner_task, token=token))
        if raw is None:
            raise RuntimeError(f"device tensor {token} no longer available on {owner_task}")
        import pickle as _pickle

        tensor = _pickle.loads(raw)
        if meta.get("device") == "cuda":
            import torch

            if torch.cuda.is_available():
                tensor = tensor.cuda()
        return tensor

    # ---- execution ------------------------------------------------------
    def _resolve_item_args(self, item: dict) -> tuple[tuple, dict]:
        """Thread-context arg resolution: chunk payloads decode once and
        serve every item of the chunk."""
        cid = item.get("chunk")
        if cid is not None:
            pairs, kw_common = self._chunk_pairs(cid)
            if kw_common is None:
                args, kwargs = pairs[item.get("ci", 0)]
                return args, kwargs
            return pairs[item.get("ci", 0)], kw_common
        return self._decode_args(item)

    def _chunk_pairs(self, cid: str) -> tuple[list, Any]:
        """Decoded chunk body: (items, kw_common). kw_common None means each
        item is an (args, kwargs) pair (C form); otherwise items are bare
        args tuples sharing kw_common (C2 form)."""
        entry = self._chunk_cache.get(cid)
        if entry is None:
            raise RuntimeError(f"chunk {cid} not delivered to this worker")
        decoded = entry["decoded"]
        if decoded is None:
            # one decode serves every range-thread of the chunk
            lock = entry.setdefault("lock", __import__("threading").Lock())
            with lock:
                decoded = entry["decoded"]
                if decoded is None:
                    raw = entry["raw"]
                    if isinstance(raw, dict):  # spilled chunk (>2 MiB)
                        if raw.get("xfer"):
                            # one-shot file handoff (scheduler unlinks it
                            # when the chunk completes)
                            with open(raw["xfer"], "rb") as f:
                                raw = f.read()
                        else:
                            raw = self.blob_store.get(raw["blob"])
                    obj = deserialize(raw)
                    if obj[0] == "C":
                        decoded = (obj[1], None)  # list of (args, kwargs)
                    elif obj[0] == "C2":
                        # common kwargs factored out client-side: the
                        # chunk body is a bare args list
                        decoded = (obj[2], obj[1])
                    else:
                        raise RuntimeError(f"unknown chunk form {obj[0]!r}")
                    entry["decoded"] = decoded
                    entry["raw"] = None
        return decoded

    def _decode_args(self, item: dict) -> tuple[tuple, dict]:
        if item.get("payload_blob"):
            payload = self.blob_store.get(item["payload_blob"])
        else:
            payload = item.get("payload") or b""
        return deserialize_payload(payload)

    async def _decode_args_async(self, item: dict) -> tuple[tuple, dict]:
        """Device-tensor markers block on transfers: decode those off-loop."""
        if item.get("chunk") is not None:
            return await asyncio.get_running_loop().run_in_executor(
                self.executor, self._resolve_item_args, item
            )
        if item.get("payload_blob"):
            payload = self.blob_store.get(item["payload_blob"])
        else:
            payload = item.get("payload") or b""
        if b"modal-amd-devtensor" in payload:
            return await asyncio.get_running_loop().run_in_executor(
                self.executor, deserialize_payload, payload
            )
        return deserialize_payload(payload)

    async def _run_input(self, frt: FunctionRuntime, item: dict) -> None:
        token = item["token"]
        call_id = token.rsplit(":", 2)[0]
        async with frt.sem:
            if token in self._abandoned:
                self._abandoned.discard(token)
                return
            ctx_tokens = _set_current_context(item.get("input_id"), call_id)
            app_tok = _app_id_var.set(frt.app_id)
            started = time.monotonic()
            cluster = item.get("cluster")
            if cluster is not None:
                self._setup_cluster(cluster)
            try:
                if frt.web_config and item.get("method") == "__web__":
                    if frt._web_runtime is None:
                        from .web import WebEndpointRuntime

                        frt._web_runtime = WebEndpointRuntime(frt.web_config, frt.load())
                    args, kwargs = await self._decode_args_async(item)
                    # stream head + body chunks over the generator data
                    # channel (parity: asgi.py:140 streams via data_out)
                    index = 0

                    async def emit(obj: Any) -> None:
                        nonlocal index
                        await self._send_gen_item(token, index, obj)
                        index += 1

                    await frt._web_runtime.handle_streaming(args[0], emit)
                    await self.conn.send({"t": "gen_data", "token": token, "index": index, "done": True})
                    self.post_output(
                        token, frt.function_id, GENERIC_STATUS_SUCCESS,
                        serialize_data_format(GeneratorDone(items_total=index), DataFormat.GENERATOR_DONE),
                        DataFormat.GENERATOR_DONE,
                    )
                    return
                fn = frt.get_callable(item.get("method", ""))
                args, kwargs = await self._decode_args_async(item)
                is_gen = (
                    frt.is_generator
                    or inspect.isgeneratorfunction(fn)
                    or inspect.isasyncgenfunction(fn)
                )
                if is_gen:
                    await self._run_generator(frt, fn, token, args, kwargs)
                    return
                result = await self._execute(frt, fn, args, kwargs, token)
                data = serialize(result)
                self.post_output(token, frt.function_id, GENERIC_STATUS_SUCCESS, data, DataFormat.PICKLE)
            except asyncio.TimeoutError:
                self.post_output(
                    token,
                    frt.function_id,
                    GENERIC_STATUS_TIMEOUT,
                    None,
                    0,
                    f"Function exceeded timeout of {frt.timeout}s "
                    f"(ran {time.monotonic() - started:.1f}s)",
                )
            except (asyncio.CancelledError, InputCancellation):
                self.post_output(
                    token, frt.function_id, GENERIC_STATUS_TERMINATED, None, 0, "input cancelled"
                )
            except BaseException as exc:
                self.post_output(
                    token,
                    frt.function_id,
                    GENERIC_STATUS_FAILURE,
                    self._serialize_exception(exc),
                    DataFormat.PICKLE,
                    "".join(traceback.format_exception_only(type(exc), exc)).strip(),
                )
            finally:
                if cluster is not None:
                    from ..experimental import _set_cluster_info

                    _set_cluster_info(None)
                _app_id_var.reset(app_tok)
                _reset_current_context(ctx_tokens)

    async def _app_teardown(self, app_id: str) -> None:
        """Drop the app's services: @exit hooks run BEFORE the exit-time
        volume commit so files they write are included (parity: lifecycle
        finalization then task_lifecycle_manager.py:117-120)."""
        volume_ids: set = set()
        for fid in [f for f, frt in self.functions.items() if frt.app_id == app_id]:
            frt = self.functions.pop(fid)
            from .volumes import mount_spec

            for spec in frt.volumes.values():
                vid, ro, _sub = mount_spec(spec)
                if not ro:
                    volume_ids.add(vid)
            if frt._service is not None:
                await asyncio.get_running_loop().run_in_executor(
                    self.executor, frt._service.exit
                )
        if volume_ids:
            await asyncio.get_running_loop().run_in_executor(None, os.sync)
            for volume_id in volume_ids:
                try:
                    await self.conn.call("volume_commit", {"volume_id": volume_id}, timeout=10)
                except Exception:
                    pass

    def _setup_cluster(self, cluster: dict) -> None:
        """Rank/world bootstrap for @clustered gangs: env for
        torch.distributed env:// rendezvous (RCCL over xGMI on GPU workers)
        plus ClusterInfo (parity: reference _clustered_functions.py:42-94)."""
        from ..experimental import ClusterInfo, _set_cluster_info

        rank, size = cluster["rank"], cluster["size"]
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(size)
        os.environ["LOCAL_RANK"] = "0"  # each worker sees exactly its own GPU
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(cluster["master_port"])
        _set_cluster_info(
            ClusterInfo(
                rank=rank,
                cluster_id=cluster["cluster_id"],
                container_ips=["127.0.0.1"] * size,
                fabric_ids=[0] * size,  # one xGMI hive on a single node
            )
        )

    async def _run_chunk(self, frt: FunctionRuntime, msg: dict) -> None:
        """Range-protocol execution: one frame = one chunk of ~64 inputs.
        Splits into max_concurrent executor sub-ranges; one serialized
        value-chunk goes back (per-item bytes only for failures/tensors)."""
        count = msg["count"]
        token = msg["token"]
        call_id = msg["call_id"]
        self._chunk_cache[msg["chunk_id"]] = {"raw": msg.get("payload"), "decoded": None}
        self._chunk_order.append(msg["chunk_id"])
        while len(self._chunk_order) > 512:  # bound the payload cache
            old = self._chunk_order.pop(0)
            self._chunk_cache.pop(old, None)
        loop = asyncio.get_running_loop()

        values: dict[int, Any] = {}
        errors: dict[int, tuple] = {}

        def run_range(start: int, end: int) -> None:
            try:
                fn = frt.get_callable(msg.get("method", ""))
                pairs, kw_common = self._chunk_pairs(msg["chunk_id"])
            except BaseException as exc:
                data = self._serialize_exception(exc)
                err = "".join(traceback.format_exception_only(type(exc), exc)).strip()
                for ci in range(start, end):
                    errors[ci] = (data, err)
                return
            # tight loop: chunk decode + slot-stream decision hoisted out;
            # per item only the contextvar, the call, and the result store
            prefix = f"in-{call_id[3:]}-"
            plain_call = not _slot_streams_enabled()
            tok_c = _current_function_call_id.set(call_id)
            # executor threads have their OWN context: the app id must be
            # set here for user prints to route to the right app's logs
            tok_a = _app_id_var.set(frt.app_id)
            self._register_sync_thread(msg["token"])
            try:
                for ci in range(start, end):
                    tok_i = _current_input_id.set(prefix + str(ci))
                    try:
                        if kw_common is None:
                            args, kwargs = pairs[ci]
                        else:
                            args, kwargs = pairs[ci], kw_common
                        if plain_call:
                            values[ci] = fn(*args, **kwargs)
                        else:
                            values[ci] = _invoke_on_slot_stream(fn, args, kwargs)
                    except InputCancellation:
                        raise  # injected cancel: abort the whole range
                    except BaseException as exc:
                        errors[ci] = (
                            self._serialize_exception(exc),
                            "".join(traceback.format_exception_only(type(exc), exc)).strip(),
                        )
                    finally:
                        _current_input_id.reset(tok_i)
            finally:
                self._unregister_sync_thread(msg["token"])
                _current_function_call_id.reset(tok_c)
                _app_id_var.reset(tok_a)

        from .execution_context import _current_function_call_id, _current_input_id

        try:
            fn0 = await loop.run_in_executor(self.executor, frt.load)
            if frt._service is not None:
                fn0 = frt.get_callable(msg.get("method", ""))
        except BaseException as exc:
            data = self._serialize_exception(exc)
            err = "".join(traceback.format_exception_only(type(exc), exc)).strip()
            for ci in range(count):
                errors[ci] = (data, err)
            fn0 = None

        app_tok = _app_id_var.set(frt.app_id)
        try:
            if fn0 is not None and frt.batch_max_size > 1:
                # @modal.batched riding the chunk path: the chunk IS the
                # accumulated batch (no linger needed) — transpose args per
                # batch_max_size slice, one user call per slice, per-item
                # outputs flow through the chunk_done aggregation
                is_async = inspect.iscoroutinefunction(fn0)

                def _transpose(pairs: Any, kw_common: Any, idxs: range) -> tuple:
                    rows = [
                        (pairs[ci], kw_common) if kw_common is not None else pairs[ci]
                        for ci in idxs
                    ]
                    if kw_common is not None:
                        # C2 form: pairs[ci] is the positional tuple
                        n_args = len(rows[0][0]) if rows else 0
                        arglists = [[row[0][i] for row in rows] for i in range(n_args)]
                        kwlists = {k: [v] * len(rows) for k, v in kw_common.items()}
                    else:
                        n_args = max((len(a) for a, _ in rows), default=0)
                        arglists = [
                            [a[i] if i < len(a) else None for a, _ in rows]
                            for i in range(n_args)
                        ]
                        keys: set = set()
                        for _, kw in rows:
                            keys.update(kw)
                        kwlists = {k: [kw.get(k) for _, kw in rows] for k in keys}
                    return tuple(arglists), kwlists

                def _store(idxs: range, results: Any) -> None:
                    if not isinstance(results, (list, tuple)) or len(results) != len(idxs):
                        raise ValueError(
                            f"Batched function {frt.name} must return a list of "
                            f"{len(idxs)} results, got {type(results).__name__}"
                        )
                    for ci, r in zip(idxs, results):
                        values[ci] = r

                try:
                    pairs, kw_common = await loop.run_in_executor(
                        self.executor, self._chunk_pairs, msg["chunk_id"]
                    )
                    bsz = frt.batch_max_size
                    for start in range(0, count, bsz):
                        idxs = range(start, min(start + bsz, count))
                        try:
                            arglists, kwlists = _transpose(pairs, kw_common, idxs)
                            if is_async:
                                results = await self._execute(frt, fn0, arglists, kwlists)
                            else:

                                def _call_batch(
                                    arglists: tuple = arglists, kwlists: dict = kwlists
                                ) -> Any:
                                    # executor threads carry their own context
                                    tok_c = _current_function_call_id.set(call_id)
                                    tok_a2 = _app_id_var.set(frt.app_id)
                                    try:
                                        return fn0(*arglists, **kwlists)
                                    finally:
                                        _current_function_call_id.reset(tok_c)
                                        _app_id_var.reset(tok_a2)

                                results = await loop.run_in_executor(
                                    self.executor, _call_batch
                                )
                            _store(idxs, results)
                        except (asyncio.CancelledError, InputCancellation):
                            raise  # cancel: abort the chunk, don't mark failed
                        except BaseException as exc:
                            data = self._serialize_exception(exc)
                            err = "".join(
                                traceback.format_exception_only(type(exc), exc)
                            ).strip()
                            for ci in idxs:
                                errors[ci] = (data, err)
                except (asyncio.CancelledError, InputCancellation):
                    raise
                except BaseException as exc:
                    data = self._serialize_exception(exc)
                    err = "".join(traceback.format_exception_only(type(exc), exc)).strip()
                    for ci in range(count):
                        errors.setdefault(ci, (data, err))
            elif fn0 is not None and (
                inspect.iscoroutinefunction(fn0) or inspect.isasyncgenfunction(fn0)
            ):
                # async user function: per-item awaits on this loop
                item = {"chunk": msg["chunk_id"], "ci": 0}
                for ci in range(count):
                    item["ci"] = ci
                    try:
                        args, kwargs = await loop.run_in_executor(
                            self.executor, self._resolve_item_args, dict(item)
                        )
                        values[ci] = await self._execute(frt, fn0, args, kwargs)
                    except BaseException as exc:
                        errors[ci] = (
                            self._serialize_exception(exc),
                            "".join(traceback.format_exception_only(type(exc), exc)).strip(),
                        )
            elif fn0 is not None:
                if count <= frt.max_concurrent:
                    n_ranges = count  # small chunks: full per-item overlap
                else:
                    # >=16 items per executor hop, up to max_concurrent ranges
                    n_ranges = min(frt.max_concurrent, max(1, count // 16))
                size = (count + n_ranges - 1) // n_ranges

                async def run_one(start: int, end: int) -> None:
                    async with frt.sem:
                        await loop.run_in_executor(self.executor, run_range, start, end)

                await asyncio.gather(
                    *(run_one(s, min(s + size, count)) for s in range(0, count, size))
                )
        except (asyncio.CancelledError, InputCancellation):
            return
        finally:
            _app_id_var.reset(app_tok)

        reply: dict[str, Any] = {
            "t": "chunk_done",
            "token": token,
            "call_id": call_id,
            "chunk_id": msg["chunk_id"],
            "function_id": frt.function_id,
            "count": count,
        }
        if errors:
            reply["exceptions"] = {str(ci): [d, r] for ci, (d, r) in errors.items()}
        if values:
            ordered_cis = sorted(values)
            from ._serialize_chunk import serialize_value_chunk

            def pack() -> tuple:
                return serialize_value_chunk([values[ci] for ci in ordered_cis])

            data, per_item = await loop.run_in_executor(self.executor, pack)
            if data is not None:
                reply["data"] = data
                reply["cis"] = None if len(ordered_cis) == count else ordered_cis
            else:
                reply["items"] = {str(ci): b for ci, b in zip(ordered_cis, per_item)}
        try:
            await self.conn.send(reply)
        except Exception:
            pass

    async def _run_frame_fast(self, frt: FunctionRuntime, items: list[dict]) -> None:
        """Sequentially execute a frame of inputs in one worker thread,
        posting all outputs in bulk (amortizes the executor hop and the
        output-flush wakeups across the frame)."""
        loop = asyncio.get_running_loop()
        from .execution_context import _current_function_call_id, _current_input_id

        def run_all() -> Any:
            results: list[dict] = []
            try:
                fn = frt.get_callable(items[0].get("method", ""))
            except BaseException as exc:
                err = "".join(traceback.format_exception_only(type(exc), exc)).strip()
                data = self._serialize_exception(exc)
                return [
                    self._make_output(
                        item["token"], frt.function_id, GENERIC_STATUS_FAILURE, data,
                        DataFormat.PICKLE, err,
                    )
                    for item in items
                ]
            if inspect.iscoroutinefunction(fn) or inspect.isasyncgenfunction(fn) or inspect.isgeneratorfunction(fn):
                return None  # signal: fall back to the general path
            # happy path: collect raw values and pickle the whole frame once
            # (mirrors the input chunking; one serialize per ~64 outputs)
            values: list = []
            value_tokens: list[str] = []
            for item in items:
                if item["token"] in self._abandoned:
                    self._abandoned.discard(item["token"])
                    continue
                call_id = item["token"].rsplit(":", 2)[0]
                tok_i = _current_input_id.set(item.get("input_id"))
                tok_c = _current_function_call_id.set(call_id)
                try:
                    args, kwargs = self._resolve_item_args(item)
                    values.append(fn(*args, **kwargs))
                    value_tokens.append(item["token"])
                except BaseException as exc:
                    results.append(
                        self._make_output(
                            item["token"], frt.function_id, GENERIC_STATUS_FAILURE,
                            self._serialize_exception(exc), DataFormat.PICKLE,
                            "".join(traceback.format_exception_only(type(exc), exc)).strip(),
                        )
                    )
                finally:
                    _current_input_id.reset(tok_i)
                    _current_function_call_id.reset(tok_c)
            if value_tokens:
                from ._serialize_chunk import serialize_value_chunk

                chunk_data, inline = serialize_value_chunk(values)
                if inline is not None:
                    # tiny frames or unpicklable-in-bulk: per-item fallback
                    for token, data in zip(value_tokens, inline):
                        results.append(
                            self._make_output(
                                token, frt.function_id, GENERIC_STATUS_SUCCESS,
                                data, DataFormat.PICKLE,
                            )
                        )
                else:
                    return (results, value_tokens, chunk_data)
            return results

        app_tok = _app_id_var.set(frt.app_id)
        try:
            async with frt.sem:  # one slot per chunk-thread => <= max_concurrent items running
                ret = await loop.run_in_executor(self.executor, run_all)
        finally:
            _app_id_var.reset(app_tok)
        if ret is None:
            # not a plain sync function after all: general path per item
            for item in items:
                task = asyncio.get_running_loop().create_task(self._run_input(frt, item))
                self._running[item["token"]] = task
                task.add_done_callback(lambda _t, tok=item["token"]: self._running.pop(tok, None))
            return
        for item in items:
            self._running.pop(item["token"], None)
        if isinstance(ret, tuple):
            results, value_tokens, chunk_data = ret
            if results:
                self._outbox.extend(results)
            try:
                await self.conn.send(
                    {
                        "t": "outputs_chunk",
                        "function_id": frt.function_id,
                        "tokens": value_tokens,
                        "data": chunk_data,
                    }
                )
            except Exception:
                return
            if self._outbox and not self._outbox_flush_scheduled:
                await self._flush_outbox()
            return
        self._outbox.extend(ret)
        if not self._outbox_flush_scheduled:
            await self._flush_outbox()

    def _register_sync_thread(self, token: str) -> None:
        if not token:
            return
        with self._sync_threads_lock:
            self._sync_threads.setdefault(token, set()).add(_threading.get_ident())

    def _unregister_sync_thread(self, token: str) -> None:
        if not token:
            return
        with self._sync_threads_lock:
            idents = self._sync_threads.get(token)
            if idents is not None:
                idents.discard(_threading.get_ident())
                if not idents:
                    self._sync_threads.pop(token, None)

    def _inject_cancel(self, token: str) -> None:
        """Raise InputCancellation inside executor threads running this
        input's sync user code (delivered at the next Python bytecode;
        blocking C calls finish first). The thread-pool analog of the
        reference's SIGUSR1 cancellation (container_io_manager.py:890)."""
        import ctypes

        with self._sync_threads_lock:
            idents = list(self._sync_threads.get(token, ()))
        for ident in idents:
            ctypes.pythonapi.PyThreadState_SetAsyncExc(
                ctypes.c_ulong(ident), ctypes.py_object(InputCancellation)
            )

    async def _execute(
        self, frt: FunctionRuntime, fn: Any, args: tuple, kwargs: dict, token: str = ""
    ) -> Any:
        from ..utils.tracing import enabled as trace_enabled, trace_range

        if inspect.iscoroutinefunction(fn):
            coro = fn(*args, **kwargs)
            if frt.timeout:
                return await asyncio.wait_for(coro, frt.timeout)
            return await coro
        ctx = contextvars.copy_context()

        def invoke() -> Any:
            self._register_sync_thread(token)
            try:
                if trace_enabled():
                    with trace_range(f"modal_amd::{frt.name}"):
                        return ctx.run(_invoke_on_slot_stream, fn, args, kwargs)
                return ctx.run(_invoke_on_slot_stream, fn, args, kwargs)
            finally:
                self._unregister_sync_thread(token)

        fut = asyncio.get_running_loop().run_in_executor(self.executor, invoke)
        if frt.timeout:
            return await asyncio.wait_for(fut, frt.timeout)
        return await fut

    async def _run_generator(self, frt: FunctionRuntime, fn: Any, token: str, args: tuple, kwargs: dict) -> None:
        """Stream generator items over the data channel, then a GeneratorDone
        output (parity: generator_output_sender, container_io_manager.py:802)."""
        index = 0
        if inspect.isasyncgenfunction(fn):
            agen = fn(*args, **kwargs)
            async for value in agen:
                await self._send_gen_item(token, index, value)
                index += 1
        else:
            loop = asyncio.get_running_loop()
            queue: asyncio.Queue = asyncio.Queue(maxsize=256)
            SENTINEL = object()

            ctx = contextvars.copy_context()

            def pump() -> Any:
                try:
                    for value in ctx.run(fn, *args, **kwargs):
                        fut = asyncio.run_coroutine_threadsafe(queue.put(value), loop)
                        fut.result()
                    return None
                finally:
                    asyncio.run_coroutine_threadsafe(queue.put(SENTINEL), loop).result()

            pump_fut = loop.run_in_executor(self.executor, pump)
            while True:
                value = await queue.get()
                if value is SENTINEL:
                    break
                await self._send_gen_item(token, index, value)
                index += 1
            await pump_fut  # surface exceptions
        await self.conn.send({"t": "gen_data", "token": token, "index": index, "done": True})
        done = GeneratorDone(items_total=index)
        self.post_output(
            token,
            frt.function_id,
            GENERIC_STATUS_SUCCESS,
            serialize_data_format(done, DataFormat.GENERATOR_DONE),
            DataFormat.GENERATOR_DONE,
        )

    async def _send_gen_item(self, token: str, index: int, value: Any) -> None:
        from .._serialization import serialize_fast

        # generator/web items are usually primitives; serialize_fast falls
        # back to the hook-aware pickler for tensors/handles itself
        data = serialize_fast(value)
        await self.conn.send(
            {"t": "gen_data", "token": token, "index": index, "data": data, "format": int(DataFormat.PICKLE)}
        )

    # ---- dynamic batching ------------------------------------------------
    def _batch_add(self, frt: FunctionRuntime, item: dict) -> None:
        """Accumulate inputs; flush at batch_max_size or after batch_linger_ms
        (parity: @modal.batched, container_io_manager.py:135-283)."""
        frt._batch.append((item, (), {}))
        if len(frt._batch) >= frt.batch_max_size:
            self._batch_flush(frt)
        elif frt._batch_flush_handle is None:
            frt._batch_flush_handle = asyncio.get_running_loop().call_later(
                frt.batch_linger_ms / 1000.0, self._batch_flush, frt
            )

    def _batch_flush(self, frt: FunctionRuntime) -> None:
        if frt._batch_flush_handle is not None:
            frt._batch_flush_handle.cancel()
            frt._batch_flush_handle = None
        batch = frt._batch
        frt._batch = []
        if batch:
            task = asyncio.get_running_loop().create_task(self._run_batch(frt, [b[0] for b in batch]))
            for item, _, _ in batch:
                self._running[item["token"]] = task

    async def _run_batch(self, frt: FunctionRuntime, items: list[dict]) -> None:
        async with frt.sem:
            try:
                fn = frt.get_callable(items[0].get("method", ""))
                decoded = [await self._decode_args_async(item) for item in items]
                # transpose: positional args and kwargs become per-arg lists
                n_args = max((len(a) for a, _ in decoded), default=0)
                arg_lists = [[d[0][i] if i < len(d[0]) else None for d in decoded] for i in range(n_args)]
                kwarg_keys: set = set()
                for _, kw in decoded:
                    kwarg_keys.update(kw)
                kw_lists = {k: [d[1].get(k) for d in decoded] for k in kwarg_keys}
                results = await self._execute(frt, fn, tuple(arg_lists), kw_lists)
                if not isinstance(results, (list, tuple)) or len(results) != len(items):
                    raise ValueError(
                        f"Batched function {frt.name} must return a list of {len(items)} results, "
                        f"got {type(results).__name__}"
                    )
                for item, result in zip(items, results):
                    self.post_output(
                        item["token"], frt.function_id, GENERIC_STATUS_SUCCESS,
                        serialize(result), DataFormat.PICKLE,
                    )
            except asyncio.TimeoutError:
                for item in items:
                    self.post_output(
                        item["token"], frt.function_id, GENERIC_STATUS_TIMEOUT, None, 0,
                        f"Batched call exceeded timeout of {frt.timeout}s",
                    )
            except BaseException as exc:
                err = "".join(traceback.format_exception_only(type(exc), exc)).strip()
                data = self._serialize_exception(exc)
                for item in items:
                    self.post_output(
                        item["token"], frt.function_id, GENERIC_STATUS_FAILURE, data,
                        DataFormat.PICKLE, err,
                    )

    # ---- outputs ---------------------------------------------------------
    def _serialize_exception(self, exc: BaseException) -> bytes:
        from ..utils.tb import clean_traceback, extract_frames

        clean_traceback(exc)
        try:
            # pickle drops __traceback__: carry the user frames explicitly
            # so the client re-synthesizes them (utils/tb.py forge)
            exc.__modal_amd_tb__ = extract_frames(exc)
        except Exception:
            pass
        try:
            return serialize(exc)
        except BaseException:
            try:
                stripped = type(exc)(*[repr(a) for a in exc.args])
                return serialize(stripped)
            except BaseException:
                return serialize(RuntimeError(repr(exc)))

    def _make_output(
        self,
        token: str,
        function_id: str,
        status: int,
        data: Optional[bytes],
        data_format: int,
        exc_repr: Optional[str] = None,
    ) -> dict:
        item: dict[str, Any] = {
            "token": token,
            "function_id": function_id,
            "status": int(status),
            "format": int(data_format),
        }
        if data is not None:
            if len(data) > INLINE_LIMIT:
                item["data"] = None
                item["data_blob"] = self.blob_store.put(data)
            else:
                item["data"] = data
        if exc_repr:
            item["exc"] = exc_repr
        return item

    def post_output(
        self,
        token: str,
        function_id: str,
        status: int,
        data: Optional[bytes],
        data_format: int,
        exc_repr: Optional[str] = None,
    ) -> None:
        item = self._make_output(token, function_id, status, data, data_format, exc_repr)
        self._outbox.append(item)
        if not self._outbox_flush_scheduled:
            # flush NOW (latency path: a lone unary result should not wait on
            # a coalescing timer); the flag suppresses re-entrant sends so
            # bursts posted in the same tick still batch into one frame
            self._outbox_flush_scheduled = True
            asyncio.get_running_loop().call_soon(self._flush_outbox_cb)

    def _flush_outbox_cb(self) -> None:
        self._outbox_flush_scheduled = False
        asyncio.get_running_loop().create_task(self._flush_outbox())

    async def _flush_outbox(self) -> None:
        while self._outbox:
            chunk, self._outbox = self._outbox[:OUTPUT_BATCH_MAX], self._outbox[OUTPUT_BATCH_MAX:]
            try:
                await self.conn.send({"t": "outputs", "items": chunk})
            except Exception:
                return

    def post_log(self, fd: int, data: str) -> None:
        if self.loop is None or self.conn is None or self.conn.closed:
            return
        # capture the app id HERE (the writer's context — executor threads
        # carry it); the flush task's context would read an empty default
        self._log_buffer.append((fd, data, _app_id_var.get()))
        if not self._log_flush_scheduled:
            self._log_flush_scheduled = True
            try:
                self.loop.call_soon_threadsafe(self._schedule_log_flush)
            except RuntimeError:
                pass

    def _schedule_log_flush(self) -> None:
        self.loop.call_later(0.02, self._flush_logs)

    def _flush_logs(self) -> None:
        self._log_flush_scheduled = False
        buf, self._log_buffer = self._log_buffer, []
        if not buf:
            return
        by_key: dict[tuple[int, str], list[str]] = {}
        for fd, data, app_id in buf:
            by_key.setdefault((fd, app_id), []).append(data)
        for (fd, app_id), chunks in by_key.items():
            asyncio.get_running_loop().create_task(
                self._send_log(fd, "".join(chunks), app_id)
            )

    async def _send_log(self, fd: int, data: str, app_id: str) -> None:
        try:
            await self.conn.send({"t": "log", "fd": fd, "data": data, "app_id": app_id})
        except Exception:
            pass


def main() -> None:
    os.environ["MODAL_AMD_IS_REMOTE"] = "1"
    runtime = WorkerRuntime()
    asyncio.run(runtime.run())


if __name__ == "__main__":
    main()
