"""Data-parallel training hosted by the framework: a gang-scheduled
@clustered function runs torch DistributedDataParallel over RCCL/xGMI
(CPU fallback: gloo), one rank per MI355X GPU.

This is SURVEY §2.3's last row in practice — the framework does not
implement TP/DP itself (neither does the reference); it *hosts* user
payloads that do, providing rank/world bootstrap, gang placement and
result gather.

Run on a multi-GPU node:  modal-amd run examples/train_ddp.py::app.main
"""

import os

import modal_amd as modal
from modal_amd.experimental import clustered, get_cluster_info

app = modal.App("example-train-ddp")

# MODAL_AMD_FORCE_CPU=1 runs the same gang on CPU workers (gloo) — used by
# the CPU test suite; on an MI355X node each rank gets its own GPU.
_GPU = None if os.environ.get("MODAL_AMD_FORCE_CPU") == "1" else 1


@app.function(gpu=_GPU, timeout=600)
@clustered(size=2)
def train(steps: int = 20, hidden: int = 1024, batch: int = 64) -> dict:
    import time

    import torch
    import torch.distributed as dist
    import torch.nn as nn
    from torch.nn.parallel import DistributedDataParallel as DDP

    info = get_cluster_info()
    use_gpu = torch.cuda.is_available()
    dist.init_process_group("nccl" if use_gpu else "gloo")
    device = torch.device("cuda:0") if use_gpu else torch.device("cpu")
    dtype = torch.bfloat16 if use_gpu else torch.float32

    torch.manual_seed(1234)  # identical init on every rank
    model = nn.Sequential(
        nn.Linear(hidden, 4 * hidden), nn.GELU(), nn.Linear(4 * hidden, hidden)
    ).to(device)
    ddp = DDP(model)  # bucketed all-reduce over xGMI during backward
    opt = torch.optim.AdamW(ddp.parameters(), lr=1e-3)

    # synthetic per-rank shard (data-parallel: different data, same model)
    gen = torch.Generator().manual_seed(1000 + info.rank)
    x = torch.randn(batch, hidden, generator=gen).to(device, dtype)
    y = torch.randn(batch, hidden, generator=gen).to(device, dtype)

    try:
        losses = []
        t0 = time.perf_counter()
        for _ in range(steps):
            opt.zero_grad(set_to_none=True)
            loss = torch.nn.functional.mse_loss(ddp(x).float(), y.float())
            loss.backward()
            opt.step()
            losses.append(loss.item())
        if use_gpu:
            torch.cuda.synchronize()
        elapsed = time.perf_counter() - t0

        # gradient sync check: parameters must be bit-identical across ranks
        flat = torch.cat([p.detach().float().flatten() for p in model.parameters()])
        digest = flat.sum().reshape(1)
        world = dist.get_world_size()
        gathered = [torch.zeros_like(digest) for _ in range(world)]
        dist.all_gather(gathered, digest)
        in_sync = all(torch.allclose(g, gathered[0]) for g in gathered)
    finally:
        dist.destroy_process_group()
    return {
        "rank": info.rank,
        "world": len(info.container_ips) or 2,
        "first_loss": losses[0],
        "last_loss": losses[-1],
        "steps_per_sec": steps / elapsed,
        "params_in_sync": in_sync,
    }


@app.local_entrypoint()
def main(steps: int = 20):
    result = train.remote(steps)
    print(
        f"rank0/{result['world']}: loss {result['first_loss']:.4f} -> "
        f"{result['last_loss']:.4f} at {result['steps_per_sec']:.1f} steps/s, "
        f"ranks in sync: {result['params_in_sync']}"
    )
