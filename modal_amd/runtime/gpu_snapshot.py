"""GPU memory snapshot/restore for ROCm (the cuda-checkpoint replacement).

The reference drives NVIDIA's external ``cuda-checkpoint`` binary to page a
PID's GPU memory to host RAM and back (/root/reference/py/modal/_runtime/
gpu_memory_snapshot.py:158-300, four-state machine :35-43, sentinel exit 222
:20-23). No such tool exists for ROCm, so the MI355X-native design snapshots
at the framework level: every live torch CUDA tensor in the process is paged
D2H into pinned host buffers (one hipMemcpyAsync per allocation on a side
stream), device memory is released back to the HIP runtime, and restore
re-pages H2D and re-links the tensors in place. The state machine and the
exit-222 degraded-fallback contract are preserved.
"""

from __future__ import annotations

import enum
import gc
import sys
from typing import Any

# parity: the runtime retries without snapshot when it sees this exit code
CUDA_CHECKPOINT_SENTINEL_EXIT = 222


class CudaCheckpointState(enum.Enum):
    """Parity: reference gpu_memory_snapshot.py:35-43."""

    RUNNING = "running"
    CHECKPOINTED = "checkpointed"
    FAILED = "failed"
    LOCKED = "locked"


class GPUMemorySnapshot:
    """Snapshot all torch CUDA state of this process to host memory."""

    def __init__(self) -> None:
        self.state = CudaCheckpointState.RUNNING
        self._saved: list[tuple[Any, Any, Any]] = []  # (tensor, host_copy, device)

    def checkpoint(self) -> None:
        if self.state is not CudaCheckpointState.RUNNING:
            raise RuntimeError(f"checkpoint() in state {self.state}")
        self.state = CudaCheckpointState.LOCKED
        try:
            import torch

            if not torch.cuda.is_available():
                self.state = CudaCheckpointState.CHECKPOINTED
                return
            stream = torch.cuda.Stream()
            seen_storages: set[int] = set()
            with torch.cuda.stream(stream):
                for obj in gc.get_objects():
                    try:
                        if not isinstance(obj, torch.Tensor) or not obj.is_cuda:
                            continue
                    except ReferenceError:
                        continue
                    storage_key = obj.untyped_storage().data_ptr()
                    if storage_key in seen_storages:
                        continue
                    seen_storages.add(storage_key)
                    host = torch.empty_like(obj, device="cpu", pin_memory=True)
                    host.copy_(obj, non_blocking=True)
                    self._saved.append((obj, host, obj.device))
            stream.synchronize()
            # release device memory: re-point tensors at empty storage
            for tensor, _host, _device in self._saved:
                tensor.data = torch.empty(0, dtype=tensor.dtype, device="cpu")
            torch.cuda.empty_cache()
            self.state = CudaCheckpointState.CHECKPOINTED
        except BaseException:
            self.state = CudaCheckpointState.FAILED
            raise

    def restore(self) -> None:
        if self.state is not CudaCheckpointState.CHECKPOINTED:
            raise RuntimeError(f"restore() in state {self.state}")
        self.state = CudaCheckpointState.LOCKED
        try:
            import torch

            stream = torch.cuda.Stream() if torch.cuda.is_available() else None
            for tensor, host, device in self._saved:
                if stream is not None:
                    with torch.cuda.stream(stream):
                        tensor.data = host.to(device, non_blocking=True)
                else:
                    tensor.data = host
            if stream is not None:
                stream.synchronize()
            self._saved.clear()
            self.state = CudaCheckpointState.RUNNING
        except BaseException:
            self.state = CudaCheckpointState.FAILED
            raise

    def fail_with_sentinel(self) -> None:
        """Exit so the supervisor retries without snapshotting
        (parity: exit-222 contract, reference :20-23,191)."""
        sys.exit(CUDA_CHECKPOINT_SENTINEL_EXIT)
