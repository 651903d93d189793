"""GPU function: BASELINE config 2 shape (torch.mm bf16 on one MI355X).

Run:  modal-amd run examples/gpu_matmul.py::app.main
"""

import modal_amd as modal

app = modal.App("example-gpu")


@app.function(gpu=1)
def matmul_flops(n: int = 4096, iters: int = 10) -> float:
    import time

    import torch

    a = torch.randn(n, n, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(n, n, dtype=torch.bfloat16, device="cuda")
    (a @ b).sum().item()  # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        c = a @ b
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return 2 * n**3 / dt / 1e12  # TFLOP/s


@app.local_entrypoint()
def main():
    print(f"bf16 GEMM: {matmul_flops.remote():.0f} TFLOP/s")
