"""Pipelined hash+compress for block uploads (the config-4 hot path).

The round-1 wall: the LZ4 kernel runs at ~11 GB/s but the end-to-end
compress path did stage → digest → compress → pack → D2H strictly in
series, capping volume upload at ~1.9 GiB/s. This module overlaps the
stages across WINDOWS of blocks:

  thread pool : memcpy window i+1 into the ping-pong pinned arena
  h2d stream  : DMA window i+1 to device
  main stream : sha256 leaves + lz4 + pack for window i
  d2h stream  : read back window i-1's digests/containers
  thread pool : assemble window i-1's container bytes + CAS writes

Window state ping-pongs across two slots, so the host memcpy (the single
largest serial cost) hides behind the previous window's kernels.

Everything is bit-identical to the serial ops (hashing.content_digest,
compress.compress_buffer): same tree digests, same MALZ41 containers.
"""

from __future__ import annotations

import struct
from concurrent.futures import ThreadPoolExecutor
from typing import Any, Callable, Optional

from .compress import MAGIC, MIN_GAIN, OUT_STRIDE, SEG_SIZE
from .hashing import GPU_MIN_BYTES, LEAF_SIZE, _root_digest

WINDOW_BYTES = 64 * 1024 * 1024

# pipeline state is expensive (2 x ~200 MiB of pinned+device buffers):
# allocate once, reuse across calls, guard with a lock
import threading as _threading

_state_lock = _threading.Lock()
_cached: dict = {}


def _get_state(cap: int, torch: Any) -> tuple:
    st = _cached.get("state")
    if st is None or st[0][0].cap < cap:
        slots = [_Slot(cap, torch), _Slot(cap, torch)]
        h2d = torch.cuda.Stream()
        d2h = torch.cuda.Stream()
        st = (slots, h2d, d2h)
        _cached["state"] = st
    return st


class _Slot:
    """One ping-pong pipeline slot: pinned arena + device buffers + events."""

    def __init__(self, cap: int, torch: Any):
        self.cap = cap
        self.pin = torch.empty(cap, dtype=torch.uint8, pin_memory=True)
        self.pin_np = self.pin.numpy()
        self.dev = torch.empty(cap, dtype=torch.uint8, device="cuda")
        max_seg = cap // SEG_SIZE + 64
        max_leaf = cap // LEAF_SIZE + 64
        self.stride = torch.empty(max_seg * OUT_STRIDE, dtype=torch.uint8, device="cuda")
        self.comp_lens = torch.zeros(max_seg, dtype=torch.int32, device="cuda")
        self.digests = torch.empty(max_leaf * 32, dtype=torch.uint8, device="cuda")
        self.off_leaf = torch.empty(max_leaf, dtype=torch.int64, device="cuda")
        self.len_leaf = torch.empty(max_leaf, dtype=torch.int64, device="cuda")
        # host-side pinned landing zones for the readbacks
        self.out_digests = torch.empty(max_leaf * 32, dtype=torch.uint8, pin_memory=True)
        self.out_lens = torch.zeros(max_seg, dtype=torch.int32, pin_memory=True)
        self.out_payload = torch.empty(
            max_seg * OUT_STRIDE, dtype=torch.uint8, pin_memory=True
        )
        self.ev_h2d = torch.cuda.Event()
        self.ev_comp = torch.cuda.Event()
        self.ev_d2h = torch.cuda.Event()
        self.ev_pin_free = torch.cuda.Event()  # H2D done: pin reusable
        self.meta: Any = None  # per-window bookkeeping


def hash_compress_blocks(
    blocks: list,
    compress: bool = True,
    store: Optional[Callable[[int, str, bytes, bool], None]] = None,
) -> tuple[list, list]:
    """(digests, containers) for a list of blocks, pipelined.

    ``containers[i]`` is MALZ41 bytes or None (incompressible / tiny).
    ``store(i, digest, payload, compressed)``, when given, is called from a
    worker thread per block AS WINDOWS COMPLETE — CAS writes overlap the
    next window's GPU work.
    """
    import numpy as np
    import torch

    lib = _lib()
    n_blocks = len(blocks)
    if n_blocks == 0:
        return [], []
    digests: list = [None] * n_blocks
    containers: list = [None] * n_blocks

    # window partition: consecutive blocks up to WINDOW_BYTES, every block
    # offset SEG/LEAF aligned (8 MiB volume blocks are exact multiples)
    windows: list[list[int]] = []
    cur: list[int] = []
    cur_bytes = 0
    for i, b in enumerate(blocks):
        nb = _aligned(len(b))
        if cur and cur_bytes + nb > WINDOW_BYTES:
            windows.append(cur)
            cur, cur_bytes = [], 0
        cur.append(i)
        cur_bytes += nb
    if cur:
        windows.append(cur)

    cap = max(
        sum(_aligned(len(blocks[i])) for i in w) for w in windows
    )
    _state_lock.acquire()
    slots, h2d, d2h = _get_state(max(cap, WINDOW_BYTES + (8 << 20)), torch)
    pool = ThreadPoolExecutor(max_workers=4)
    slot_assemble: dict[int, Any] = {}  # slot index -> pending assemble future

    def fill_pin(slot: _Slot, win: list[int]) -> list[int]:
        # host memcpy into the pinned arena (worker thread; the big serial
        # cost we're hiding). Returns per-block offsets.
        offs = []
        off = 0
        for i in win:
            b = blocks[i]
            slot.pin_np[off : off + len(b)] = np.frombuffer(b, dtype=np.uint8)
            offs.append(off)
            off += _aligned(len(b))
        return offs

    def launch_window(slot: _Slot, win: list[int], offs: list[int]) -> None:
        """H2D + digest + compress kernels + readbacks for one window."""
        total = sum(_aligned(len(blocks[i])) for i in win)
        with torch.cuda.stream(h2d):
            slot.dev[:total].copy_(slot.pin[:total], non_blocking=True)
            slot.ev_h2d.record(h2d)
            slot.ev_pin_free.record(h2d)
        main = torch.cuda.current_stream()
        main.wait_event(slot.ev_h2d)
        # --- digest: one sha256 dispatch over every leaf of the window ---
        leaf_offs: list[int] = []
        leaf_lens: list[int] = []
        spans = []  # (first_leaf, n_leaves) per block
        for i, off in zip(win, offs):
            n = len(blocks[i])
            n_leaves = (n + LEAF_SIZE - 1) // LEAF_SIZE
            spans.append((len(leaf_offs), n_leaves))
            for leaf in range(n_leaves):
                leaf_offs.append(off + leaf * LEAF_SIZE)
                leaf_lens.append(min(LEAF_SIZE, n - leaf * LEAF_SIZE))
        n_leaf = len(leaf_offs)
        slot.off_leaf[:n_leaf].copy_(
            torch.tensor(leaf_offs, dtype=torch.int64), non_blocking=True
        )
        slot.len_leaf[:n_leaf].copy_(
            torch.tensor(leaf_lens, dtype=torch.int64), non_blocking=True
        )
        rc = lib.ma_sha256_many(
            slot.dev.data_ptr(), slot.off_leaf.data_ptr(), slot.len_leaf.data_ptr(),
            slot.digests.data_ptr(), n_leaf, main.cuda_stream,
        )
        if rc != 0:
            raise RuntimeError(f"sha256 kernel failed: hipError {rc}")
        # --- compress: one lz4 dispatch per block over the shared source ---
        seg_meta = []  # (seg_base, n_seg) per block
        seg_base = 0
        if compress:
            for i, off in zip(win, offs):
                n = len(blocks[i])
                n_seg = (n + SEG_SIZE - 1) // SEG_SIZE
                rc = lib.ma_lz4_compress(
                    slot.dev.data_ptr() + off, n,
                    slot.stride.data_ptr() + seg_base * OUT_STRIDE,
                    slot.comp_lens.data_ptr() + seg_base * 4,
                    OUT_STRIDE, n_seg, main.cuda_stream,
                )
                if rc != 0:
                    raise RuntimeError(f"lz4 kernel failed: hipError {rc}")
                seg_meta.append((seg_base, n_seg))
                seg_base += n_seg
        slot.ev_comp.record(main)
        # --- readbacks on the d2h stream ---
        with torch.cuda.stream(d2h):
            d2h.wait_event(slot.ev_comp)
            slot.out_digests[: n_leaf * 32].copy_(
                slot.digests[: n_leaf * 32], non_blocking=True
            )
            if compress and seg_base:
                slot.out_lens[:seg_base].copy_(
                    slot.comp_lens[:seg_base], non_blocking=True
                )
                # payload D2H: the full stride buffer region (compaction
                # happens host-side per segment — stride slots are already
                # densely indexable, and D2H bandwidth is not the wall)
                slot.out_payload[: seg_base * OUT_STRIDE].copy_(
                    slot.stride[: seg_base * OUT_STRIDE], non_blocking=True
                )
            slot.ev_d2h.record(d2h)
        slot.meta = (win, offs, spans, seg_meta, n_leaf)

    def assemble(slot: _Slot) -> None:
        """After ev_d2h: build digests + containers on a worker thread."""
        win, offs, spans, seg_meta, n_leaf = slot.meta
        leaf_bytes = slot.out_digests[: n_leaf * 32].numpy().tobytes()
        lens_np = slot.out_lens.numpy()
        payload_np = slot.out_payload.numpy()
        for k, (i, off) in enumerate(zip(win, offs)):
            n = len(blocks[i])
            first, n_leaves = spans[k]
            if n < GPU_MIN_BYTES:
                import hashlib

                digest = hashlib.sha256(blocks[i]).hexdigest()
            else:
                digest = _root_digest(
                    n, leaf_bytes[first * 32 : (first + n_leaves) * 32]
                ).hex()
            digests[i] = digest
            container = None
            if compress and seg_meta:
                seg_base, n_seg = seg_meta[k]
                comp_lens = lens_np[seg_base : seg_base + n_seg].tolist()
                eff_total = 0
                for si, clen in enumerate(comp_lens):
                    seg_raw = min(SEG_SIZE, n - si * SEG_SIZE)
                    eff_total += clen if clen else seg_raw
                if eff_total < n * MIN_GAIN:
                    parts = [
                        MAGIC,
                        struct.pack("<QI", n, n_seg),
                        struct.pack(f"<{n_seg}I", *comp_lens),
                    ]
                    blk = blocks[i]
                    for si, clen in enumerate(comp_lens):
                        if clen:
                            s0 = (seg_base + si) * OUT_STRIDE
                            parts.append(payload_np[s0 : s0 + clen].tobytes())
                        else:  # raw segment comes from the HOST copy
                            parts.append(
                                bytes(blk[si * SEG_SIZE : si * SEG_SIZE + min(
                                    SEG_SIZE, n - si * SEG_SIZE
                                )])
                            )
                    container = b"".join(parts)
            containers[i] = container
            if store is not None:
                store(i, digest, container if container is not None else blocks[i],
                      container is not None)

    # --- drive the pipeline -------------------------------------------------
    fill_futs: dict[int, Any] = {}
    assemble_futs: list = []
    try:
        fill_futs[0] = pool.submit(fill_pin, slots[0], windows[0])
        for w in range(len(windows)):
            slot = slots[w % 2]
            if w >= 2:
                # slot reuse: the previous window on this slot must be fully
                # read back AND assembled (assemble reads the out_* pinned
                # buffers this launch would overwrite)
                prev = slot_assemble.pop(w % 2, None)
                if prev is not None:
                    prev.result()
            offs = fill_futs.pop(w).result()
            launch_window(slot, windows[w], offs)
            # prefetch next window into the other slot once ITS pin is free
            if w + 1 < len(windows):
                nxt = slots[(w + 1) % 2]

                def _fill_next(nxt=nxt, wn=windows[w + 1], prev_ev=nxt.ev_pin_free, first=(w + 1 < 2)):
                    if not first:
                        prev_ev.synchronize()  # pin arena still DMA-ing its old window
                    return fill_pin(nxt, wn)

                fill_futs[w + 1] = pool.submit(_fill_next)
            # assembly of THIS window once its D2H completes
            def _assemble(slot=slot):
                slot.ev_d2h.synchronize()
                assemble(slot)

            fut = pool.submit(_assemble)
            assemble_futs.append(fut)
            slot_assemble[w % 2] = fut
        for fut in assemble_futs:
            fut.result()
    finally:
        pool.shutdown(wait=True)
        _state_lock.release()
    return digests, containers


def _aligned(n: int) -> int:
    align = max(SEG_SIZE, LEAF_SIZE)
    return -(-n // align) * align


def _lib():
    from . import load_lib

    return load_lib(required=True)
