"""Decompose the per-item worker cost of the map bench GPU op.

Measures, per 64-item chunk on one MI355X:
  A. the bare GPU op (3 launches/item), no readback
  B. op + per-item .item() sync             (bench ITEM_SYNC variant)
  C. op + batched cat+pinned D2H readback   (runtime _serialize_chunk path)
  D. pickle cost of 64 small HOST tensors vs 64 ints vs _FastTensor wrapper
Run on a GPU box: python scripts/worker_microbench.py
"""

from __future__ import annotations

import pickle
import time

import torch


def timeit(label: str, fn, iters: int = 50) -> float:
    fn()  # warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    per_item = dt / 64 * 1e6
    print(f"{label:44s} {dt * 1e3:8.3f} ms/chunk  {per_item:7.2f} us/item")
    return dt


def main() -> None:
    cache = torch.ones(4096, device="cuda", dtype=torch.bfloat16)

    def op(x: int):
        t = cache * float(x % 7 + 1)
        return t[:4].float().sum()

    def chunk_op_only():
        vals = [op(x) for x in range(64)]
        torch.cuda.synchronize()
        return vals

    def chunk_item_sync():
        return [int(op(x).item()) for x in range(64)]

    from modal_amd.runtime._serialize_chunk import _batch_cuda_to_host

    def chunk_batched_readback():
        vals = [op(x) for x in range(64)]
        _batch_cuda_to_host(vals, torch)
        return vals

    def chunk_batched_plus_pickle():
        vals = [op(x) for x in range(64)]
        _batch_cuda_to_host(vals, torch)
        return pickle.dumps(vals)

    timeit("A op only (3 launches/item) + sync", chunk_op_only)
    timeit("B op + per-item .item()", chunk_item_sync)
    timeit("C op + batched readback", chunk_batched_readback)
    timeit("C' op + batched readback + pickle", chunk_batched_plus_pickle)

    # host-side pickle costs in isolation
    host_tensors = chunk_batched_readback()
    ints = list(range(64))
    t0 = time.perf_counter()
    for _ in range(200):
        pickle.dumps(host_tensors)
    print(f"pickle 64 host tensors: {(time.perf_counter() - t0) / 200 * 1e6:.1f} us/chunk")
    t0 = time.perf_counter()
    for _ in range(200):
        pickle.dumps(ints)
    print(f"pickle 64 ints:         {(time.perf_counter() - t0) / 200 * 1e6:.1f} us/chunk")

    # one-kernel-per-item op variant (single fused launch count)
    def op1(x: int):
        return (cache * float(x % 7 + 1)).sum()

    def chunk_op1():
        vals = [op1(x) for x in range(64)]
        torch.cuda.synchronize()
        return vals

    timeit("E 1-launch... (2 kernels: mul+sum)", chunk_op1)

    # launch overhead floor: 1 trivial kernel per item
    def chunk_single_launch():
        vals = [cache.sum() for _ in range(64)]
        torch.cuda.synchronize()
        return vals

    timeit("F floor: one sum launch per item", chunk_single_launch)


if __name__ == "__main__":
    main()
