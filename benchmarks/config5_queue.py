"""BASELINE config 5: modal.Queue producer/consumer with GPU worker
Functions (put/get through the scheduler; device payloads touch the GPU
in each consumer).

Measures queue put+get throughput (items/s) with W consumer workers each
running a small bf16 op per item on the GPU.

Run: python benchmarks/config5_queue.py [--items 20000] [--consumers 8]
"""

from __future__ import annotations

import argparse
import sys as _sys
from os.path import abspath, dirname

_sys.path.insert(0, dirname(dirname(abspath(__file__))))  # repo root

import json
import time


def consume(q: object, n: int) -> int:
    import torch

    have_gpu = torch.cuda.is_available()
    if have_gpu:
        cache = torch.ones(2048, device="cuda", dtype=torch.bfloat16)
    total = 0
    for _ in range(n):
        item = q.get(timeout=60)
        if have_gpu:
            total += int((cache * float(item % 5 + 1))[:2].float().sum().item()) and item or item
        else:
            total += item
    return total


def consume_batched(q: object, n: int) -> int:
    """get_many(64) variant: amortizes the per-get RPC round trip
    (reference consumers batch exactly this way with get_many)."""
    import torch

    have_gpu = torch.cuda.is_available()
    if have_gpu:
        cache = torch.ones(2048, device="cuda", dtype=torch.bfloat16)
    total = 0
    got = 0
    while got < n:
        items = q.get_many(min(64, n - got), timeout=60)
        if not items:
            continue
        got += len(items)
        for item in items:
            if have_gpu:
                total += int((cache * float(item % 5 + 1))[:2].float().sum().item()) and item or item
            else:
                total += item
    return total


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--items", type=int, default=20_000)
    parser.add_argument("--consumers", type=int, default=8)
    parser.add_argument("--batched", action="store_true", help="consumers use get_many(64)")
    args = parser.parse_args()

    import torch

    import modal_amd as modal

    app = modal.App("bench-queue")
    gpu = 1 if torch.cuda.is_available() else None
    consumer = app.function(gpu=gpu)(consume_batched if args.batched else consume)

    per = args.items // args.consumers
    with app.run():
        with modal.Queue.ephemeral() as q:
            # warm the workers (definition load + first GPU touch)
            q.put_many([0] * args.consumers)
            for fc in [consumer.spawn(q, 1) for _ in range(args.consumers)]:
                fc.get(timeout=120)

            t0 = time.perf_counter()
            calls = [consumer.spawn(q, per) for _ in range(args.consumers)]
            put = 0
            BATCH = 512
            while put < per * args.consumers:
                n = min(BATCH, per * args.consumers - put)
                q.put_many(list(range(put, put + n)))
                put += n
            totals = [fc.get(timeout=600) for fc in calls]
            elapsed = time.perf_counter() - t0

    n_done = per * args.consumers
    assert sum(totals) == sum(range(n_done))
    print(json.dumps({
        "config": 5,
        "queue_items_per_sec": round(n_done / elapsed, 1),
        "items": n_done,
        "consumers": args.consumers,
        "batched": args.batched,
        "gpu": bool(gpu),
    }))


if __name__ == "__main__":
    main()
