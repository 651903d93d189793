"""Chunk (de)compression: container format over the HIP LZ4 kernels.

Container layout (little-endian):
  magic  b"MALZ41"             (6 bytes)
  raw_len  u64
  n_segments  u32              (raw split into 4 KiB segments)
  comp_len[n_segments]  u32    (0 => segment stored raw)
  payload                      (concatenated per-segment LZ4 blocks / raw)

Each segment is a standard LZ4 block, so any LZ4 block decoder can read the
payload; the CPU reference codec (utils/lz4ref.py) serves GPU-less readers.
"""

from __future__ import annotations

import struct
from typing import Optional

from . import gpu_available, load_lib

MAGIC = b"MALZ41"
SEG_SIZE = 4096
OUT_STRIDE = SEG_SIZE + 256
MIN_GAIN = 0.95  # store raw unless compression saves >=5%


def _header(raw_len: int, comp_lens: list[int]) -> bytes:
    return (
        MAGIC
        + struct.pack("<QI", raw_len, len(comp_lens))
        + struct.pack(f"<{len(comp_lens)}I", *comp_lens)
    )


def parse_header(blob: bytes) -> tuple[int, list[int], int]:
    """Returns (raw_len, comp_lens, payload_offset)."""
    if not blob.startswith(MAGIC):
        raise ValueError("not a MALZ41 container")
    raw_len, n_seg = struct.unpack_from("<QI", blob, len(MAGIC))
    off = len(MAGIC) + 12
    comp_lens = list(struct.unpack_from(f"<{n_seg}I", blob, off))
    return raw_len, comp_lens, off + 4 * n_seg


def is_compressed(blob: bytes) -> bool:
    return blob.startswith(MAGIC)


# ---------------------------------------------------------------------------
# CPU reference paths
# ---------------------------------------------------------------------------


def _native_core() -> Optional[object]:
    try:
        from .. import _core

        return _core if hasattr(_core, "malz_compress") else None
    except ImportError:
        return None


def compress_buffer_cpu(data: bytes) -> Optional[bytes]:
    core = _native_core()
    if core is not None:
        # C++ segment-parallel codec (csrc/core.cpp): same container, same
        # compress/raw decisions, ~1000x the pure-Python reference
        return core.malz_compress(bytes(data), MIN_GAIN)
    from ..utils import lz4ref

    n = len(data)
    comp_lens: list[int] = []
    payload = bytearray()
    for start in range(0, n, SEG_SIZE):
        seg = data[start : start + SEG_SIZE]
        comp = lz4ref.compress_block(seg)
        if len(comp) < len(seg):
            comp_lens.append(len(comp))
            payload += comp
        else:
            comp_lens.append(0)
            payload += seg
    if not comp_lens:
        return None
    total = len(payload)
    if total >= n * MIN_GAIN:
        return None
    return _header(n, comp_lens) + bytes(payload)


def decompress_buffer_cpu(blob: bytes) -> bytes:
    core = _native_core()
    if core is not None:
        return core.malz_decompress(bytes(blob))
    from ..utils import lz4ref

    raw_len, comp_lens, off = parse_header(blob)
    out = bytearray()
    pos = off
    for i, clen in enumerate(comp_lens):
        seg_raw = min(SEG_SIZE, raw_len - i * SEG_SIZE)
        if clen == 0:
            out += blob[pos : pos + seg_raw]
            pos += seg_raw
        else:
            out += lz4ref.decompress_block(blob[pos : pos + clen], seg_raw)
            pos += clen
    return bytes(out)


# ---------------------------------------------------------------------------
# GPU paths
# ---------------------------------------------------------------------------


def compress_buffer_gpu(data: bytes) -> Optional[bytes]:
    lib = load_lib(required=True)
    import torch

    n = len(data)
    n_seg = (n + SEG_SIZE - 1) // SEG_SIZE
    if n_seg == 0:
        return None
    from .staging import stage_to_gpu

    src = stage_to_gpu(data)
    stride_buf = torch.empty(n_seg * OUT_STRIDE, dtype=torch.uint8, device="cuda")
    comp_lens_d = torch.zeros(n_seg, dtype=torch.int32, device="cuda")
    rc = lib.ma_lz4_compress(
        src.data_ptr(), n, stride_buf.data_ptr(), comp_lens_d.data_ptr(),
        OUT_STRIDE, n_seg, torch.cuda.current_stream().cuda_stream,
    )
    if rc != 0:
        raise RuntimeError(f"lz4 compress kernel failed: hipError {rc}")
    torch.cuda.synchronize()
    comp_lens = comp_lens_d.cpu().tolist()
    eff_lens = []
    total = 0
    for i, clen in enumerate(comp_lens):
        seg_raw = min(SEG_SIZE, n - i * SEG_SIZE)
        eff = clen if clen else seg_raw
        eff_lens.append(eff)
        total += eff
    if total >= n * MIN_GAIN:
        return None
    # stage raw segments into their stride slots, then compact with the pack kernel
    for i, clen in enumerate(comp_lens):
        if clen == 0:
            seg_raw = min(SEG_SIZE, n - i * SEG_SIZE)
            stride_buf[i * OUT_STRIDE : i * OUT_STRIDE + seg_raw] = src[
                i * SEG_SIZE : i * SEG_SIZE + seg_raw
            ]
    from .packing import pack_gpu

    offsets = torch.arange(n_seg, dtype=torch.int64) * OUT_STRIDE
    packed, _ = pack_gpu(stride_buf, offsets, torch.tensor(eff_lens, dtype=torch.int64))
    torch.cuda.synchronize()
    from .staging import fetch_from_gpu

    payload = fetch_from_gpu(packed)
    return _header(n, comp_lens) + payload


def compress_buffers_gpu(buffers: list, staged: "object" = None, only: "object" = None) -> list:
    """Batched compression: ONE host->device transfer and ONE sync for a
    whole upload's blocks (volume v2 8 MiB blocks are exact multiples of
    SEG_SIZE, so per-buffer kernel launches share the staged source at
    4 KiB-aligned offsets). Per-block round trips previously dominated
    config-4 upload wall time. Returns per-buffer container bytes or None
    (incompressible / empty)."""
    lib = load_lib(required=True)
    import torch

    pick = set(range(len(buffers))) if only is None else set(only)
    if staged is not None:
        src, src_offsets = staged  # shared with the digest pass
    else:
        from .staging import stage_many_to_gpu

        # one pinned H2D for all buffers, 4 KiB-aligned (no concat bytearray)
        src, src_offsets = stage_many_to_gpu(buffers, align=SEG_SIZE)
    metas = []  # (src_off, n, n_seg, seg_base); n_seg=0 for skipped buffers
    seg_base = 0
    for bi, (data, src_off) in enumerate(zip(buffers, src_offsets)):
        n = len(data)
        n_seg = (n + SEG_SIZE - 1) // SEG_SIZE if bi in pick else 0
        metas.append((src_off, n, n_seg, seg_base))
        seg_base += n_seg
    total_seg = seg_base
    if total_seg == 0:
        return [None] * len(buffers)
    stride_buf = torch.empty(total_seg * OUT_STRIDE, dtype=torch.uint8, device="cuda")
    comp_lens_d = torch.zeros(total_seg, dtype=torch.int32, device="cuda")
    stream = torch.cuda.current_stream().cuda_stream
    for src_off, n, n_seg, sb in metas:
        if n_seg == 0:
            continue
        rc = lib.ma_lz4_compress(
            src.data_ptr() + src_off, n,
            stride_buf.data_ptr() + sb * OUT_STRIDE,
            comp_lens_d.data_ptr() + sb * 4,
            OUT_STRIDE, n_seg, stream,
        )
        if rc != 0:
            raise RuntimeError(f"lz4 compress kernel failed: hipError {rc}")
    torch.cuda.synchronize()
    comp_lens_all = comp_lens_d.cpu().tolist()  # one D2H

    from .packing import pack_gpu
    from .staging import fetch_from_gpu

    # decide per buffer, then compact EVERY kept buffer's segments with ONE
    # pack launch and ONE pinned D2H (a per-buffer pack+sync+readback loop
    # was 6x the kernel time in the config-4 trace)
    kept: list = []  # (meta, comp_lens, eff_lens, total)
    out: list = [None] * len(buffers)
    for bi, (src_off, n, n_seg, sb) in enumerate(metas):
        if n_seg == 0:
            continue
        comp_lens = comp_lens_all[sb : sb + n_seg]
        eff_lens, total = [], 0
        for i, clen in enumerate(comp_lens):
            seg_raw = min(SEG_SIZE, n - i * SEG_SIZE)
            eff = clen if clen else seg_raw
            eff_lens.append(eff)
            total += eff
        if total >= n * MIN_GAIN:
            continue
        for i, clen in enumerate(comp_lens):
            if clen == 0:
                seg_raw = min(SEG_SIZE, n - i * SEG_SIZE)
                dst0 = (sb + i) * OUT_STRIDE
                src0 = src_off + i * SEG_SIZE
                stride_buf[dst0 : dst0 + seg_raw] = src[src0 : src0 + seg_raw]
        kept.append((bi, n, sb, n_seg, comp_lens, eff_lens, total))
    if not kept:
        return out
    all_offsets = torch.cat(
        [(torch.arange(n_seg, dtype=torch.int64) + sb) * OUT_STRIDE
         for _bi, _n, sb, n_seg, _cl, _el, _t in kept]
    )
    all_eff = torch.tensor(
        [e for _bi, _n, _sb, _ns, _cl, eff_lens, _t in kept for e in eff_lens],
        dtype=torch.int64,
    )
    packed, _ = pack_gpu(stride_buf, all_offsets, all_eff)
    torch.cuda.synchronize()
    payload_all = fetch_from_gpu(packed)
    pos = 0
    for bi, n, _sb, _ns, comp_lens, _eff_lens, total in kept:
        out[bi] = _header(n, comp_lens) + payload_all[pos : pos + total]
        pos += total
    return out


def compress_buffers(buffers: list, staged: "object" = None, only: "object" = None) -> list:
    """GPU-batched when present; None entries otherwise (CPU compression
    is not worth its cost on the storage path)."""
    if gpu_available():
        return compress_buffers_gpu(buffers, staged=staged, only=only)
    return [None] * len(buffers)


def decompress_buffer_gpu(blob: bytes) -> bytes:
    lib = load_lib(required=True)
    import torch

    raw_len, comp_lens, off = parse_header(blob)
    n_seg = len(comp_lens)
    from .staging import stage_to_gpu

    payload = stage_to_gpu(blob[off:])
    offs = []
    pos = 0
    for i, clen in enumerate(comp_lens):
        offs.append(pos)
        pos += clen if clen else min(SEG_SIZE, raw_len - i * SEG_SIZE)
    comp_offs = torch.tensor(offs, dtype=torch.int64).cuda()
    comp_lens_d = torch.tensor(comp_lens, dtype=torch.int32).cuda()
    dst = torch.empty(max(raw_len, 1), dtype=torch.uint8, device="cuda")
    status = torch.zeros(1, dtype=torch.int32, device="cuda")
    rc = lib.ma_lz4_decompress(
        payload.data_ptr(), comp_offs.data_ptr(), comp_lens_d.data_ptr(),
        dst.data_ptr(), raw_len, status.data_ptr(), n_seg,
        torch.cuda.current_stream().cuda_stream,
    )
    if rc != 0:
        raise RuntimeError(f"lz4 decompress kernel failed: hipError {rc}")
    torch.cuda.synchronize()
    bad = int(status.item())
    if bad:
        raise ValueError(f"LZ4 container corrupt at segment {bad - 1}")
    from .staging import fetch_from_gpu

    return fetch_from_gpu(dst[:raw_len])


# ---------------------------------------------------------------------------
# dispatching front door (used by the CAS)
# ---------------------------------------------------------------------------


def compress_buffer(data: bytes) -> Optional[bytes]:
    """GPU when present; else the native C++ segment-parallel codec (the
    pure-Python fallback alone was not worth its cost — the C++ one is)."""
    if gpu_available():
        return compress_buffer_gpu(data)
    core = _native_core()
    if core is not None:
        return core.malz_compress(bytes(data), MIN_GAIN)
    return None


def decompress_buffer(blob: bytes) -> bytes:
    if gpu_available():
        try:
            return decompress_buffer_gpu(blob)
        except RuntimeError:
            pass
    return decompress_buffer_cpu(blob)
