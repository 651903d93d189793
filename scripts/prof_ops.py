"""Exercise the HIP data-plane kernels for rocprofv3 capture.

Run under rocprofv3 --kernel-trace --stats; summaries land in profiles/.
"""

from __future__ import annotations

import time

import torch

from modal_amd.ops.hashing import sha256_many_gpu, tree_sha256_cpu, _tree_sha256_gpu
from modal_amd.ops.packing import pack_gpu, unpack_gpu

assert torch.cuda.is_available()


def bench_sha(total_mb: int = 512) -> None:
    n = total_mb * 1024 * 1024
    from modal_amd.ops.hashing import LEAF_SIZE as leaf
    n_leaves = n // leaf
    buf = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
    offsets = (torch.arange(n_leaves, dtype=torch.int64) * leaf)
    lengths = torch.full((n_leaves,), leaf, dtype=torch.int64)
    # warmup
    sha256_many_gpu(buf, offsets, lengths)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 5
    for _ in range(iters):
        out = sha256_many_gpu(buf, offsets, lengths)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"sha256_many: {total_mb} MiB in {dt*1000:.2f} ms -> {total_mb/1024/dt:.2f} GiB/s")


def bench_pack(total_mb: int = 512, n_seg: int = 4096) -> None:
    n = total_mb * 1024 * 1024
    seg = n // n_seg
    src = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
    offsets = torch.arange(n_seg, dtype=torch.int64) * seg
    lengths = torch.full((n_seg,), seg, dtype=torch.int64)
    packed, dst_off = pack_gpu(src, offsets, lengths)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 5
    for _ in range(iters):
        packed, dst_off = pack_gpu(src, offsets, lengths)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"pack: {total_mb} MiB in {dt*1000:.2f} ms -> {2*total_mb/1024/dt:.2f} GiB/s r+w")


def verify() -> None:
    data = bytes(bytearray(range(256)) * (64 * 1024))  # 16 MiB
    assert _tree_sha256_gpu(data) == tree_sha256_cpu(data)
    print("sha256 verify ok")





def bench_sha_leaf_sweep(total_mb: int = 512) -> None:
    """Leaf-size sweep: occupancy/latency tradeoff for the tree hash."""
    n = total_mb * 1024 * 1024
    buf = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
    for leaf_kb in (4, 8, 16, 32):
        leaf = leaf_kb * 1024
        n_leaves = n // leaf
        offsets = torch.arange(n_leaves, dtype=torch.int64) * leaf
        lengths = torch.full((n_leaves,), leaf, dtype=torch.int64)
        for ilp in (1, 2):
            sha256_many_gpu(buf, offsets, lengths, ilp=ilp)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(5):
                sha256_many_gpu(buf, offsets, lengths, ilp=ilp)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / 5
            print(f"leaf={leaf_kb:3d}KiB ilp={ilp}: {total_mb} MiB in {dt*1000:.2f} ms -> {total_mb/1024/dt:.2f} GiB/s")


if __name__ == "__main__":
    verify()
    bench_sha_leaf_sweep()
    bench_sha()
    bench_pack()
