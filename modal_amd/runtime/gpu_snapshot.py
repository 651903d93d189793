"""GPU memory snapshot/restore for ROCm (the cuda-checkpoint replacement).

The reference drives NVIDIA's external ``cuda-checkpoint`` binary to page a
PID's GPU memory to host RAM and back (/root/reference/py/modal/_runtime/
gpu_memory_snapshot.py:158-300, four-state machine :35-43, sentinel exit 222
:20-23), and restores a process via ``/__modal/restore-state.json``
(task_lifecycle_manager.py:146-215). ROCm has no cuda-checkpoint, so the
MI355X-native design snapshots at the framework level, in three layers:

1. torch tensors — every live CUDA tensor is paged D2H into pinned host
   buffers and re-paged on restore (in-process page-out for scaledown).
2. raw hipMalloc allocations — anything allocated through the tracked
   allocator (ops/rawmem.py, csrc/rawmem.hip) is paged by stable key and
   survives into a FRESH process.
3. RNG state — torch CPU+CUDA generators, python ``random``, numpy.

Cross-process restore follows the reference's restore-state contract: the
new worker busy-waits for a restore-state JSON (path in
$MODAL_AMD_RESTORE_STATE_PATH, defaulting to the reference's
``/__modal/restore-state.json``), applies env overrides, loads the snapshot
payload, and exits with the sentinel code 222 on failure so the supervisor
retries without snapshot.
"""

from __future__ import annotations

import enum
import gc
import os
import pickle
import sys
import time
import warnings
from typing import Any, Optional

# parity: the runtime retries without snapshot when it sees this exit code
CUDA_CHECKPOINT_SENTINEL_EXIT = 222

RESTORE_STATE_PATH_DEFAULT = "/__modal/restore-state.json"


class CudaCheckpointState(enum.Enum):
    """Parity: reference gpu_memory_snapshot.py:35-43."""

    RUNNING = "running"
    CHECKPOINTED = "checkpointed"
    FAILED = "failed"
    LOCKED = "locked"


# ---------------------------------------------------------------------------
# snapshot-visible registries (cross-process survivors)
# ---------------------------------------------------------------------------

#: tensors user/framework code explicitly registered to survive a
#: cross-process restore: key -> tensor
_registered_tensors: dict[str, Any] = {}
#: state restored from a snapshot payload in THIS process
_restored_tensors: dict[str, Any] = {}


def register_tensor(key: str, tensor: Any) -> None:
    """Mark a tensor as snapshot-visible: it will be serialized into
    cross-process snapshots under this key."""
    _registered_tensors[key] = tensor


def restored_tensor(key: str) -> Any:
    """Fetch a tensor restored from a snapshot (falls back to the live
    registry so code is identical on first run and after restore)."""
    if key in _restored_tensors:
        return _restored_tensors[key]
    return _registered_tensors.get(key)


def _rng_capture() -> dict:
    state: dict[str, Any] = {}
    try:
        import random

        state["py"] = random.getstate()
    except Exception:
        pass
    try:
        import numpy as np

        state["np"] = np.random.get_state()
    except Exception:
        pass
    try:
        import torch

        state["torch_cpu"] = torch.get_rng_state()
        if torch.cuda.is_available():
            state["torch_cuda"] = torch.cuda.get_rng_state_all()
    except Exception:
        pass
    return state


def _rng_restore(state: dict) -> None:
    try:
        if "py" in state:
            import random

            random.setstate(state["py"])
    except Exception:
        pass
    try:
        if "np" in state:
            import numpy as np

            np.random.set_state(state["np"])
    except Exception:
        pass
    try:
        import torch

        if "torch_cpu" in state:
            torch.set_rng_state(state["torch_cpu"])
        if "torch_cuda" in state and torch.cuda.is_available():
            saved = state["torch_cuda"]
            n = min(len(saved), torch.cuda.device_count())
            for i in range(n):
                torch.cuda.set_rng_state(saved[i], i)
    except Exception:
        pass


# ---------------------------------------------------------------------------
# in-process page-out (scaledown) — tensors stay object-identical
# ---------------------------------------------------------------------------


class GPUMemorySnapshot:
    """Page all torch CUDA state + tracked raw allocations of this process
    to host memory, reversibly (the scaledown page-out)."""

    def __init__(self) -> None:
        self.state = CudaCheckpointState.RUNNING
        self._saved: list[tuple[Any, Any, Any]] = []  # (tensor, host_copy, device)
        self._raw: dict[str, bytes] = {}
        self._rng: dict = {}

    def checkpoint(self) -> None:
        if self.state is not CudaCheckpointState.RUNNING:
            raise RuntimeError(f"checkpoint() in state {self.state}")
        self.state = CudaCheckpointState.LOCKED
        try:
            import torch

            if not torch.cuda.is_available():
                self.state = CudaCheckpointState.CHECKPOINTED
                return
            self._rng = _rng_capture()
            stream = torch.cuda.Stream()
            seen_storages: set[int] = set()
            with torch.cuda.stream(stream), warnings.catch_warnings():
                # scanning gc objects trips deprecation warnings in
                # libraries' lazy attributes — silence the scan only
                warnings.simplefilter("ignore")
                for obj in gc.get_objects():
                    try:
                        if not isinstance(obj, torch.Tensor) or not obj.is_cuda:
                            continue
                    except Exception:
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
            # tracked raw hipMalloc allocations page out by key
            from ..ops import rawmem

            self._raw = rawmem.snapshot_all()
            rawmem.release_all()
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
            if self._raw:
                from ..ops import rawmem

                rawmem.restore_all(self._raw)
                self._raw = {}
            if self._rng:
                _rng_restore(self._rng)
                self._rng = {}
            self._saved.clear()
            self.state = CudaCheckpointState.RUNNING
        except BaseException:
            self.state = CudaCheckpointState.FAILED
            raise

    def fail_with_sentinel(self) -> None:
        """Exit so the supervisor retries without snapshotting
        (parity: exit-222 contract, reference :20-23,191)."""
        sys.exit(CUDA_CHECKPOINT_SENTINEL_EXIT)


# ---------------------------------------------------------------------------
# cross-process snapshot payloads (worker death / warm restore)
# ---------------------------------------------------------------------------


def capture_payload() -> bytes:
    """Serialize the snapshot-visible state of this process: registered
    tensors (paged D2H), tracked raw allocations, RNG. The result restores
    into a FRESH process (pointers re-allocated, keys preserved)."""
    from ..ops import rawmem

    tensors: dict[str, tuple] = {}
    if _registered_tensors:
        import torch

        for key, t in _registered_tensors.items():
            host = t.detach().cpu().contiguous()
            # byte-level extraction: numpy() would reject bf16/fp8 dtypes
            raw = host.reshape(-1).view(torch.uint8).numpy().tobytes() if host.numel() else b""
            tensors[key] = (
                raw,
                str(t.dtype).removeprefix("torch."),
                tuple(t.shape),
                t.is_cuda,
            )
    payload = {
        "version": 1,
        "tensors": tensors,
        "raw": rawmem.snapshot_all(),
        "rng": _rng_capture(),
        "fidelity": rawmem.fidelity_report(),
    }
    return pickle.dumps(payload, 4)


def restore_payload(data: bytes) -> dict:
    """Rehydrate a capture_payload() blob into THIS process. Returns the
    fidelity report recorded at capture time."""
    payload = pickle.loads(data)
    from ..ops import rawmem

    if payload.get("raw"):
        rawmem.restore_all(payload["raw"])
    try:
        import torch

        for key, (raw, dtype_s, shape, was_cuda) in payload.get("tensors", {}).items():
            dtype = getattr(torch, dtype_s)
            if raw:
                t = torch.frombuffer(bytearray(raw), dtype=torch.uint8).view(dtype)
                t = t.reshape(shape) if shape else t.reshape(())
            else:
                t = torch.empty(shape or (0,), dtype=dtype)
            if was_cuda and torch.cuda.is_available():
                t = t.cuda()
            _restored_tensors[key] = t
            _registered_tensors[key] = t
    except Exception:
        if payload.get("tensors"):
            raise
    _rng_restore(payload.get("rng", {}))
    return payload.get("fidelity", {})


def wait_and_restore_from_state_file(path: Optional[str] = None, timeout: float = 60.0) -> Optional[dict]:
    """The restored-process half of the reference's restore contract
    (task_lifecycle_manager.py:146-193): busy-wait for the restore-state
    JSON, apply env overrides, load the snapshot payload. Returns the state
    dict, or None when no restore is configured. Exits 222 on failure."""
    path = path or os.environ.get("MODAL_AMD_RESTORE_STATE_PATH")
    if not path:
        if os.path.exists(RESTORE_STATE_PATH_DEFAULT):
            path = RESTORE_STATE_PATH_DEFAULT
        else:
            return None
    import json

    deadline = time.time() + timeout
    while not os.path.exists(path):
        if time.time() > deadline:
            sys.exit(CUDA_CHECKPOINT_SENTINEL_EXIT)
        time.sleep(0.05)
    try:
        with open(path) as f:
            state = json.load(f)
        for key, value in (state.get("env") or {}).items():
            os.environ[key] = str(value)
        blob_path = state.get("snapshot_path")
        if blob_path:
            with open(blob_path, "rb") as f:
                restore_payload(f.read())
        return state
    except BaseException:
        sys.exit(CUDA_CHECKPOINT_SENTINEL_EXIT)
