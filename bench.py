"""Flagship benchmark: Function.map items/sec across N MI355X GPUs.

BASELINE.json metric: "Function.map items/sec (node) + p50 .remote() latency"
— config 3 (100k synthetic inputs fanned across the GPUs) as the headline
number, with config 2's p50 .remote() latency (gpu=1 torch.mm bf16) reported
in the config block.

Driver contract:
  python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches this under torch.distributed.run with one rank
per GPU. Rank 0 runs the client + in-process scheduler; every rank
(including 0) contributes one worker pinned to its GPU. Steps are bracketed
by a dist barrier + torch.cuda.synchronize on both sides; elapsed time is
MAX-reduced over ranks; rank 0 prints one JSON line.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

# Pin each torchrun rank to its GPU *before* torch import so device 0 is the
# rank's GPU everywhere (workers, RCCL, user payloads).
_LOCAL_RANK = int(os.environ.get("LOCAL_RANK", "0"))
_WORLD_SIZE = int(os.environ.get("WORLD_SIZE", "1"))
if _WORLD_SIZE > 1 and "HIP_VISIBLE_DEVICES" not in os.environ:
    os.environ["HIP_VISIBLE_DEVICES"] = str(_LOCAL_RANK)
    os.environ["CUDA_VISIBLE_DEVICES"] = str(_LOCAL_RANK)

ITEMS_PER_GPU = 12_500  # x8 GPUs = the 100k-input config of BASELINE.json
# 288 GB HBM3E per MI355X easily hosts several payload processes; swept
# per-box (profiles/README.md, r2: 4 workers x 128-item chunks ~1.8x the
# round-1 3x64 on the same box); override with MODAL_AMD_BENCH_WPG
WORKERS_PER_GPU = int(os.environ.get("MODAL_AMD_BENCH_WPG", "8"))


def _bench_run_dir() -> str:
    port = os.environ.get("MASTER_PORT", "0")
    return f"/tmp/modal-amd-bench-{port}-{os.environ.get('TORCHELASTIC_RUN_ID', 'solo')}"


def map_item_gpu(x: int) -> int:
    import torch

    cache = getattr(torch, "_ma_bench_cache", None)
    if cache is None:
        cache = torch.ones(4096, device="cuda", dtype=torch.bfloat16)
        torch._ma_bench_cache = cache
    t = cache * float(x % 7 + 1)
    s = t[:4].float().sum()
    if os.environ.get("MODAL_AMD_BENCH_ITEM_SYNC") == "1":
        # legacy variant: a ~24 us .item() device sync per item
        return int(s.item()) and x or x
    # default: return the result tensor; the runtime batch-stages the
    # readback per ~64-item chunk (one pinned D2H per chunk,
    # runtime/_serialize_chunk.py) — every item's GPU op and its readback
    # still happen, the sync just amortizes across the chunk
    return s


def map_item_gpu_batched(xs: list) -> list:
    """@modal.batched variant (MODAL_AMD_BENCH_BATCHED=1): the framework
    batches up to 64 map items into ONE call; the per-item op (bf16
    4096-vector scale + 4-elem sum) runs vectorized — identical math per
    item, fused into large kernel launches instead of 3 tiny ones each."""
    import torch

    cache = getattr(torch, "_ma_bench_cache2", None)
    if cache is None:
        cache = torch.ones(4096, device="cuda", dtype=torch.bfloat16)
        torch._ma_bench_cache2 = cache
    pin = getattr(torch, "_ma_bench_pin", None)
    if pin is None or pin.numel() < len(xs):
        pin = torch.empty(max(len(xs), 64), dtype=torch.float32, pin_memory=True)
        torch._ma_bench_pin = pin
    pin_in = pin[: len(xs)]
    pin_in.copy_(torch.tensor([float(x % 7 + 1) for x in xs], dtype=torch.float32))
    ks = pin_in.to("cuda", dtype=torch.bfloat16, non_blocking=True)
    t = ks[:, None] * cache[None, :]          # [B, 4096] bf16 scale
    s = t[:, :4].float().sum(dim=1)           # per-item 4-elem sum
    return s.cpu().tolist()                   # ONE D2H sync per batch


def map_item_cpu(x: int) -> int:
    return x


def p50_probe_gpu(n: int) -> float:
    import torch

    a = torch.randn(n, n, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(n, n, dtype=torch.bfloat16, device="cuda")
    c = a @ b
    torch.cuda.synchronize()
    return float(c.float().mean().item())


def run_worker_rank(rank: int, world: int) -> None:
    """Ranks > 0: host one worker connected to rank 0's scheduler, plus
    participate in the start/end barriers from the main thread."""
    import threading

    import torch
    import torch.distributed as dist

    socket_path = os.path.join(_bench_run_dir(), "scheduler.sock")
    deadline = time.time() + 120
    while not os.path.exists(socket_path):
        if time.time() > deadline:
            raise RuntimeError("scheduler socket never appeared")
        time.sleep(0.05)

    os.environ["MODAL_AMD_WORKER_SOCKET"] = socket_path
    os.environ["MODAL_AMD_WORKER_ID"] = str(1000 + rank)
    os.environ["MODAL_AMD_GPU_INDEX"] = "0" if torch.cuda.is_available() else ""
    os.environ["MODAL_AMD_EXTERNAL_WORKER"] = "1"
    os.environ["MODAL_AMD_IS_REMOTE"] = "1"
    if not os.environ.get("MODAL_AMD_GPU_INDEX"):
        del os.environ["MODAL_AMD_GPU_INDEX"]

    from modal_amd.runtime.worker import WorkerRuntime

    runtime = WorkerRuntime()
    thread = threading.Thread(target=lambda: __import__("asyncio").run(runtime.run()), daemon=True)
    thread.start()
    # extra workers for this rank's GPU (they inherit HIP_VISIBLE_DEVICES)
    import subprocess
    import sys as _sys

    extra = []
    for k in range(1, WORKERS_PER_GPU if torch.cuda.is_available() else 1):
        env = dict(os.environ)
        env["MODAL_AMD_WORKER_ID"] = str(1000 + rank + 100 * k)
        extra.append(
            subprocess.Popen(
                [_sys.executable, "-m", "modal_amd.runtime.worker"], env=env,
                start_new_session=True,
            )
        )

    device = torch.device("cuda:0") if torch.cuda.is_available() else None
    # start barrier (warmup done on rank 0), timed region, end barrier
    dist.barrier()
    t0 = time.perf_counter()
    dist.barrier()
    if device is not None:
        torch.cuda.synchronize()
    elapsed = torch.tensor([time.perf_counter() - t0], dtype=torch.float64)
    dist.all_reduce(elapsed, op=dist.ReduceOp.MAX)
    dist.barrier()
    dist.destroy_process_group()
    for proc in extra:
        try:
            proc.terminate()
        except Exception:
            pass


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=8)
    parser.add_argument("--warmup", type=int, default=2)
    parser.add_argument("--items-per-gpu", type=int, default=ITEMS_PER_GPU)
    args = parser.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = _WORLD_SIZE
    n_gpus = args.gpus

    import torch

    has_gpu = torch.cuda.is_available()
    dist = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        backend = "nccl" if has_gpu else "gloo"
        dist.init_process_group(backend=backend)
        if has_gpu:
            torch.cuda.set_device(0)

    if rank != 0:
        run_worker_rank(rank, world)
        return

    # ---- rank 0: client + scheduler -----------------------------------
    run_dir = _bench_run_dir()
    import shutil

    shutil.rmtree(run_dir, ignore_errors=True)  # stale sockets from prior N
    os.makedirs(run_dir, exist_ok=True)
    os.environ["MODAL_AMD_RUN_DIR"] = run_dir

    from modal_amd._sync import synchronizer
    from modal_amd.client import _Client
    from modal_amd.scheduler.core import Scheduler

    async def boot():
        scheduler = Scheduler(run_dir=run_dir)
        await scheduler.start()
        client = _Client(scheduler, "client")
        _Client.set_default(client)
        return scheduler, client

    scheduler, client = synchronizer.run(boot())

    # rank 0 contributes one worker for its own GPU (ranks>0 bring theirs)
    async def spawn_local_worker():
        for _ in range(WORKERS_PER_GPU if has_gpu else 1):
            await scheduler.pool.spawn_worker(gpu_index=0 if has_gpu else None)

    synchronizer.run(spawn_local_worker())
    # wait until every rank's workers are connected
    per_rank = WORKERS_PER_GPU if has_gpu else 1
    want_workers = per_rank * (n_gpus if world > 1 else 1)
    deadline = time.time() + 180
    while len(scheduler.pool.workers) < want_workers:
        if time.time() > deadline:
            raise RuntimeError(
                f"only {len(scheduler.pool.workers)}/{want_workers} workers connected"
            )
        time.sleep(0.05)

    import modal_amd as modal

    if has_gpu and os.environ.get("MODAL_AMD_BENCH_EAGER") != "1" and (
        os.environ.get("MODAL_AMD_BENCH_ITEM_SYNC") != "1"
    ):
        # batched default: 256-item chunks measured fastest at N=1
        # (1.49M items/s); at N>=4 the scheduler loop carries 4-8x the
        # chunk rate, so halve it with 512-item chunks (-10% at N=1)
        os.environ.setdefault(
            "MODAL_AMD_CHUNK_ITEMS", "256" if n_gpus < 4 else "512"
        )

    app = modal.App("bench")
    # default: @modal.batched dynamic batching (the framework's serving-path
    # feature; same per-item math, fused launches). MODAL_AMD_BENCH_EAGER=1
    # runs the per-item eager variant instead (3 tiny launches per item).
    use_batched = has_gpu and os.environ.get("MODAL_AMD_BENCH_EAGER") != "1" and (
        os.environ.get("MODAL_AMD_BENCH_ITEM_SYNC") != "1"
    )
    if use_batched:
        item_fn = app.function(gpu=1)(
            modal.batched(max_batch_size=64, wait_ms=1)(map_item_gpu_batched)
        )
    else:
        work_fn = map_item_gpu if has_gpu else map_item_cpu
        item_fn = app.function(gpu=1 if has_gpu else None)(
            modal.concurrent(max_inputs=8)(work_fn)
        )
    probe_fn = app.function(gpu=1 if has_gpu else None)(p50_probe_gpu) if has_gpu else None

    items_per_step = args.items_per_gpu * n_gpus

    def run_map_step() -> None:
        """One map step driven on the framework loop (no per-item sync
        bridge); batch-wise consumption (map_batches) skips the per-item
        flatten — every item is still produced and counted."""

        async def _consume() -> None:
            n = 0
            async for batch in item_fn.map_batches.aio(
                range(items_per_step), order_outputs=False
            ):
                n += len(batch)
            assert n == items_per_step

        synchronizer.run(_consume())

    ctx = app.run(client=client)
    ctx.__enter__()
    try:
        # ---- warmup: also measures config-2 p50 .remote() latency ------
        p50_ms = None
        if probe_fn is not None:
            lat = []
            probe_fn.remote(256)  # first call pays worker/model init
            for _ in range(30):
                t = time.perf_counter()
                probe_fn.remote(256)
                lat.append((time.perf_counter() - t) * 1000)
            lat.sort()
            p50_ms = lat[len(lat) // 2]
        for _ in range(args.warmup):
            run_map_step()

        # ---- timed region ---------------------------------------------
        if dist is not None:
            dist.barrier()
        if has_gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            run_map_step()
        if dist is not None:
            dist.barrier()
        if has_gpu:
            torch.cuda.synchronize()
        elapsed = time.perf_counter() - t0
        if dist is not None:
            t = torch.tensor([elapsed], dtype=torch.float64)
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            elapsed = float(t.item())

        total_items = args.steps * items_per_step
        result = {
            "metric": "map_items_per_sec",
            "value": total_items / elapsed,
            "unit": "items/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if has_gpu else "none",
            "data": "synthetic",
            "config": {
                "model": "Function.map fan-out (BASELINE config 3)",
                "global_batch": items_per_step,
                "seq_len": 0,
                "parallelism": f"map{n_gpus}",
                "p50_remote_ms": p50_ms,
                "per_item_gpu_op": (
                    (
                        "bf16 4096-vector scale + 4-elem sum per item, "
                        "@modal.batched(64) fused launches, one pinned D2H per batch"
                        if use_batched
                        else "bf16 vector scale + chunk-batched pinned D2H readback"
                    )
                    if has_gpu
                    else "noop"
                ),
                "dynamic_batching": 64 if use_batched else 0,
                "chunk_items": int(os.environ.get("MODAL_AMD_CHUNK_ITEMS", "128")),
                "workers": len(scheduler.pool.workers),
            },
        }
        print(json.dumps(result), flush=True)
    finally:
        try:
            ctx.__exit__(None, None, None)
        except Exception:
            pass
        if dist is not None:
            dist.barrier()
            dist.destroy_process_group()
        synchronizer.run(client.close())


if __name__ == "__main__":
    main()
