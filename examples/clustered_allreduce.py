"""Gang-scheduled RCCL collective over xGMI (the @clustered substrate).

Run on a multi-GPU node:  modal-amd run examples/clustered_allreduce.py::app.main
"""

import modal_amd as modal
from modal_amd.experimental import clustered, get_cluster_info

app = modal.App("example-clustered")


@app.function(gpu=1)
@clustered(size=2)
def allreduce_bandwidth(mb: int = 256) -> float:
    import time

    import torch
    import torch.distributed as dist

    info = get_cluster_info()
    dist.init_process_group("nccl")  # RCCL over xGMI on ROCm
    t = torch.ones(mb * 1024 * 1024 // 4, device="cuda")
    dist.all_reduce(t)  # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        dist.all_reduce(t)
    torch.cuda.synchronize()
    gbps = 10 * 2 * (t.numel() * 4) / (time.perf_counter() - t0) / 1e9
    dist.destroy_process_group()
    return gbps if info.rank == 0 else gbps


@app.local_entrypoint()
def main():
    print(f"allreduce bus bandwidth: {allreduce_bandwidth.remote():.1f} GB/s")
