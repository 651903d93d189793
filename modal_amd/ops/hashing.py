"""Content hashing: plain SHA-256 for small payloads, GPU tree-SHA-256 above.

The reference hashes every blob/Volume block with streaming SHA-256 on the
CPU (/root/reference/py/modal/_utils/hash_utils.py:68; 8 MiB Volume blocks
blob_utils.py:63). SHA-256 of one message is inherently serial, so the GPU
design re-shapes the problem: a buffer is split into 64 KiB leaves, every
leaf hashed by one CDNA4 lane (csrc/sha256.hip), and the content digest is
SHA-256 over the concatenated leaf digests with a domain separator. The CPU
reference implementation below produces bit-identical digests, so the CAS is
consistent across GPU-ful and GPU-less processes, and kernel correctness is
testable against hashlib.

Small payloads bypass the GPU entirely (SURVEY.md §7 hard part 7: keep
``.remote()`` p50 low — the crossover is config ``gpu_hash_threshold``).
"""

from __future__ import annotations

import hashlib
import struct
from typing import Optional, Union

from . import gpu_available, load_lib

# Leaf size trades tree overhead against GPU parallelism: SHA-256 is serial
# within a leaf, so leaves are the unit of parallelism. Swept 4/8/16/32 KiB
# on MI355X (profiles/README.md): 8 KiB wins (911 GiB/s at 512 MiB) — enough
# waves to hide the round-dependency chain, padding overhead still small.
LEAF_SIZE = 8 * 1024
TREE_DOMAIN = b"modal-amd-tree-v1"
GPU_MIN_BYTES = 8 * 1024 * 1024  # below this, CPU wins (kernel+copy overhead)

Buffer = Union[bytes, bytearray, memoryview]


def _root_digest(total_len: int, leaf_digests: bytes) -> bytes:
    h = hashlib.sha256()
    h.update(TREE_DOMAIN)
    h.update(struct.pack("<Q", total_len))
    h.update(leaf_digests)
    return h.digest()


def tree_sha256_cpu(data: Buffer) -> bytes:
    data = memoryview(data)
    digests = bytearray()
    for off in range(0, len(data), LEAF_SIZE):
        digests += hashlib.sha256(data[off : off + LEAF_SIZE]).digest()
    return _root_digest(len(data), bytes(digests))


def _tree_sha256_gpu(data: Buffer) -> Optional[bytes]:
    lib = load_lib(required=True)
    import torch

    n = len(data)
    n_leaves = (n + LEAF_SIZE - 1) // LEAF_SIZE
    from .staging import stage_to_gpu

    src = stage_to_gpu(data)  # pinned arena + side-stream hipMemcpyAsync
    offsets = torch.arange(0, n_leaves, dtype=torch.int64) * LEAF_SIZE
    lengths = torch.full((n_leaves,), LEAF_SIZE, dtype=torch.int64)
    if n % LEAF_SIZE:
        lengths[-1] = n % LEAF_SIZE
    offsets_d = offsets.cuda()
    lengths_d = lengths.cuda()
    out = torch.empty((n_leaves, 32), dtype=torch.uint8, device="cuda")
    stream = torch.cuda.current_stream().cuda_stream
    rc = lib.ma_sha256_many(
        src.data_ptr(), offsets_d.data_ptr(), lengths_d.data_ptr(), out.data_ptr(),
        n_leaves, stream,
    )
    if rc != 0:
        raise RuntimeError(f"sha256 kernel failed: hipError {rc}")
    torch.cuda.synchronize()
    return _root_digest(n, out.cpu().numpy().tobytes())


def sha256_many_gpu(
    buf: "object", offsets: "object", lengths: "object", ilp: int = 1
) -> "object":
    """Hash arbitrary (offset, length) slices of a GPU-resident uint8 tensor.

    Returns an [n, 32] uint8 CUDA tensor of standard SHA-256 digests.
    ilp=2 uses the dual-chain kernel (two messages per lane, interleaved
    rounds) — wins when occupancy is low (few messages).
    """
    lib = load_lib(required=True)
    import torch

    assert buf.dtype == torch.uint8 and buf.is_cuda
    n = len(offsets)
    offsets_d = offsets.to(device="cuda", dtype=torch.int64)
    lengths_d = lengths.to(device="cuda", dtype=torch.int64)
    out = torch.empty((n, 32), dtype=torch.uint8, device="cuda")
    fn = lib.ma_sha256_many2 if ilp == 2 else lib.ma_sha256_many
    rc = fn(
        buf.data_ptr(), offsets_d.data_ptr(), lengths_d.data_ptr(), out.data_ptr(),
        n, torch.cuda.current_stream().cuda_stream,
    )
    if rc != 0:
        raise RuntimeError(f"sha256 kernel failed: hipError {rc}")
    return out


def tree_sha256(data: Buffer) -> bytes:
    """GPU when it pays, CPU otherwise; identical digests either way."""
    if len(data) >= GPU_MIN_BYTES and gpu_available():
        return _tree_sha256_gpu(data)
    return tree_sha256_cpu(data)


def content_digests_batch(
    buffers: list, gpu_threshold: int = GPU_MIN_BYTES, staged: "object" = None
) -> list[str]:
    """CAS keys for many buffers with ONE kernel launch over all leaves.

    This is the volume-upload hot path: a multi-GiB file's 8 MiB blocks hash
    in a single batched dispatch (thousands of leaves -> full chip), instead
    of one under-occupied launch per block.
    """
    total = sum(len(b) for b in buffers)
    if total < gpu_threshold or not gpu_available():
        return [content_digest(b, gpu_threshold) for b in buffers]
    lib = load_lib(required=True)
    import torch

    spans: list[tuple[int, int]] = []  # (first_leaf_index, n_leaves) per buffer
    offsets: list[int] = []
    lengths: list[int] = []
    small_idx: dict[int, str] = {}
    big_bufs: list = []
    big_meta: list[int] = []  # index into spans to fix up with real offsets
    for i, data in enumerate(buffers):
        if len(data) < gpu_threshold:
            # digest form must depend only on the buffer's own size so every
            # path (single, batch, CPU, GPU) computes the same CAS key
            import hashlib as _hl

            small_idx[i] = _hl.sha256(bytes(data)).hexdigest()
            spans.append((len(offsets), 0))
            continue
        n_leaves = (len(data) + LEAF_SIZE - 1) // LEAF_SIZE
        spans.append((len(offsets), n_leaves))
        big_meta.append((len(big_bufs), i, len(offsets), n_leaves))
        big_bufs.append(data)
        offsets.extend([0] * n_leaves)  # placeholders until staged
        for leaf in range(n_leaves):
            lengths.append(min(LEAF_SIZE, len(data) - leaf * LEAF_SIZE))
    if not lengths:
        return [small_idx[i] for i in range(len(buffers))]
    if staged is not None:
        # caller pre-staged ALL buffers (one pinned H2D shared with the
        # compression pass, blobs.put_many); map big-buffer offsets from it
        src, all_offsets = staged
        base_offsets = [all_offsets[i] for _bk, i, _f, _n in big_meta]
    else:
        from .staging import stage_many_to_gpu

        # one pinned H2D for all large buffers (no concat bytearray pass)
        src, base_offsets = stage_many_to_gpu(big_bufs, align=LEAF_SIZE)
    for (bk, _i, first, n_leaves), base in zip(big_meta, base_offsets):
        for leaf in range(n_leaves):
            offsets[first + leaf] = base + leaf * LEAF_SIZE
    out = torch.empty((len(offsets), 32), dtype=torch.uint8, device="cuda")
    off_d = torch.tensor(offsets, dtype=torch.int64).cuda()
    len_d = torch.tensor(lengths, dtype=torch.int64).cuda()
    rc = lib.ma_sha256_many(
        src.data_ptr(), off_d.data_ptr(), len_d.data_ptr(), out.data_ptr(),
        len(offsets), torch.cuda.current_stream().cuda_stream,
    )
    if rc != 0:
        raise RuntimeError(f"sha256 kernel failed: hipError {rc}")
    torch.cuda.synchronize()
    leaf_digests = out.cpu().numpy().tobytes()
    results: list[str] = []
    for i, data in enumerate(buffers):
        if i in small_idx:
            results.append(small_idx[i])
        else:
            first, n_leaves = spans[i]
            results.append(
                _root_digest(len(data), leaf_digests[first * 32 : (first + n_leaves) * 32]).hex()
            )
    return results


def content_digest(data: Buffer, gpu_threshold: int = GPU_MIN_BYTES) -> str:
    """The CAS key: plain SHA-256 below the threshold, tree digest above.

    Deterministic for a given payload size, so every process computes the
    same key regardless of whether it has a GPU.
    """
    if len(data) < gpu_threshold:
        return hashlib.sha256(data).hexdigest()
    return tree_sha256(data).hex()
