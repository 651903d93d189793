"""Device mesh: worker-to-worker tensor transfers over RCCL/xGMI.

The MI355X-native device data plane (BASELINE config 5):
GPU-resident tensors put into Queues/Dicts or returned between functions move
directly between the per-GPU worker processes with torch.distributed
point-to-point send/recv — backend "nccl" IS RCCL on ROCm, so on an 8-GPU
node the bytes ride the xGMI links (7 p2p links x ~153 GB/s per GPU), never
bouncing through host memory. On CPU-only pools the same coordination runs
over gloo, which is how the protocol is tested without hardware.

Flow:
  producer worker A: serialize(tensor) -> registers it in its TensorTable,
    pickles a ("devtensor", owner_task, token, meta) marker
  consumer worker B: deserialize -> fetch_device_tensor() -> RPC
    device_transfer to the scheduler, which tells A "send to rank(B)" and B
    "recv from rank(A)"; the comm thread pair executes the RCCL p2p op
  non-mesh consumers (the client process): tensor_pull RPC -> host-staged
    copy (scheduler relays to A).
"""

from __future__ import annotations

import queue as queue_mod
import threading
import uuid
from typing import Any, Optional

TENSOR_TABLE_CAP = 4096


class TensorTable:
    """Exported tensors kept alive until consumed (FIFO-capped)."""

    def __init__(self) -> None:
        self._table: dict[str, Any] = {}
        self._order: list[str] = []
        self._lock = threading.Lock()

    def register(self, tensor: Any) -> str:
        token = uuid.uuid4().hex
        with self._lock:
            self._table[token] = tensor
            self._order.append(token)
            while len(self._order) > TENSOR_TABLE_CAP:
                old = self._order.pop(0)
                self._table.pop(old, None)
        return token

    def get(self, token: str) -> Any:
        with self._lock:
            return self._table.get(token)

    def release(self, token: str) -> None:
        with self._lock:
            self._table.pop(token, None)


def meta_of(tensor: Any) -> dict:
    return {
        "shape": list(tensor.shape),
        "dtype": str(tensor.dtype).replace("torch.", ""),
        "device": "cuda" if tensor.is_cuda else "cpu",
    }


def empty_like_meta(meta: dict) -> Any:
    import torch

    dtype = getattr(torch, meta["dtype"])
    device = "cuda" if (meta["device"] == "cuda" and torch.cuda.is_available()) else "cpu"
    return torch.empty(meta["shape"], dtype=dtype, device=device)


class DeviceMesh:
    """One rank of the worker collective plane; owns the comm thread."""

    def __init__(self) -> None:
        self.rank: Optional[int] = None
        self.world_size: int = 0
        self.backend: str = "gloo"
        self._ops: queue_mod.Queue = queue_mod.Queue()
        self._thread: Optional[threading.Thread] = None
        self._ready = threading.Event()
        self._init_error: Optional[str] = None
        self._results: dict[str, tuple[threading.Event, Any]] = {}
        self._results_lock = threading.Lock()

    @property
    def active(self) -> bool:
        return self._ready.is_set() and self._init_error is None

    def init(self, rank: int, world_size: int, port: int, backend: str) -> None:
        if self._thread is not None:
            return
        self.rank = rank
        self.world_size = world_size
        self.backend = backend
        self._thread = threading.Thread(
            target=self._run, args=(rank, world_size, port, backend), daemon=True,
            name="modal-amd-mesh",
        )
        self._thread.start()

    def wait_ready(self, timeout: float = 120.0) -> bool:
        ok = self._ready.wait(timeout)
        return ok and self._init_error is None

    def _run(self, rank: int, world_size: int, port: int, backend: str) -> None:
        try:
            import torch
            import torch.distributed as dist

            # dedicated process group for the data plane (independent of any
            # group user code forms inside @clustered functions)
            self.group = dist.init_process_group(
                backend=backend,
                init_method=f"tcp://127.0.0.1:{port}",
                rank=rank,
                world_size=world_size,
                group_name="modal_amd_mesh",
            )
            self._dist = dist
            if backend == "nccl" and torch.cuda.is_available():
                torch.cuda.set_device(0)  # each worker sees exactly its GPU
        except BaseException as exc:
            self._init_error = repr(exc)
            self._ready.set()
            return
        self._ready.set()
        while True:
            op = self._ops.get()
            if op is None:
                return
            kind, payload = op
            try:
                if kind == "send":
                    tensor, dst = payload
                    self._dist.send(tensor.contiguous(), dst)
                elif kind == "recv":
                    xfer_id, meta, src = payload
                    tensor = empty_like_meta(meta)
                    self._dist.recv(tensor, src)
                    self._complete(xfer_id, tensor)
            except BaseException as exc:  # deliver the failure to the waiter
                if kind == "recv":
                    self._complete(payload[0], exc)

    def submit_send(self, tensor: Any, dst_rank: int) -> None:
        self._ops.put(("send", (tensor, dst_rank)))

    def submit_recv(self, xfer_id: str, meta: dict, src_rank: int) -> None:
        with self._results_lock:
            self._results.setdefault(xfer_id, (threading.Event(), None))
        self._ops.put(("recv", (xfer_id, meta, src_rank)))

    def _complete(self, xfer_id: str, value: Any) -> None:
        with self._results_lock:
            event, _ = self._results.setdefault(xfer_id, (threading.Event(), None))
            self._results[xfer_id] = (event, value)
        event.set()

    def wait_result(self, xfer_id: str, timeout: float = 120.0) -> Any:
        with self._results_lock:
            event, _ = self._results.setdefault(xfer_id, (threading.Event(), None))
        if not event.wait(timeout):
            raise TimeoutError(f"device transfer {xfer_id} timed out")
        with self._results_lock:
            _, value = self._results.pop(xfer_id)
        if isinstance(value, BaseException):
            raise value
        return value

    def shutdown(self) -> None:
        if self._thread is not None:
            self._ops.put(None)
