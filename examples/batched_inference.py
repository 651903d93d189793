"""Dynamic batching: many .remote/.map calls fuse into one GPU call.

Run on a GPU node:  modal-amd run examples/batched_inference.py::app.main
"""

import modal_amd as modal

app = modal.App("example-batched")


@app.function(gpu=1)
@modal.batched(max_batch_size=64, wait_ms=20)
def embed(values: list) -> list:
    """One fused bf16 GEMM serves the whole accumulated batch."""
    import torch

    weights = getattr(torch, "_embed_w", None)
    if weights is None:
        weights = torch.randn(256, 256, dtype=torch.bfloat16, device="cuda")
        torch._embed_w = weights
    x = torch.tensor(values, dtype=torch.bfloat16, device="cuda")[:, None]
    batch = x * torch.ones(len(values), 256, dtype=torch.bfloat16, device="cuda")
    out = batch @ weights  # ONE kernel for up to 64 logical calls
    return out.float().sum(dim=1).cpu().tolist()


@app.local_entrypoint()
def main(n: int = 512):
    results = list(embed.map(range(n)))
    print(f"{len(results)} items served through fused batches; first={results[0]:.1f}")
