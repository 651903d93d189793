"""A model-serving class service: @enter loads weights once per worker,
requests share the warm instance; @modal.concurrent overlaps requests.

Run on a GPU node:  modal-amd run examples/llm_server.py::app.main
(random-init weights — no network on this node)
"""

import modal_amd as modal

app = modal.App("example-llm-server")


@app.cls(gpu=1)
class TinyLM:
    dim = modal.parameter(default=1024)

    @modal.enter()
    def load(self):
        import torch

        torch.manual_seed(0)
        self.w1 = torch.randn(self.dim, 4 * self.dim, dtype=torch.bfloat16, device="cuda")
        self.w2 = torch.randn(4 * self.dim, self.dim, dtype=torch.bfloat16, device="cuda")

    @modal.method()
    def forward_tokens(self, n_tokens: int) -> float:
        """One MLP block over n_tokens (the serving hot path shape)."""
        import torch

        x = torch.randn(n_tokens, self.dim, dtype=torch.bfloat16, device="cuda")
        y = torch.nn.functional.gelu(x @ self.w1) @ self.w2
        torch.cuda.synchronize()
        return float(y.float().abs().mean().item())


@app.local_entrypoint()
def main(requests: int = 32, tokens: int = 512):
    import time

    lm = TinyLM()
    lm.forward_tokens.remote(tokens)  # warm: @enter runs once
    t0 = time.perf_counter()
    for _ in range(requests):
        lm.forward_tokens.remote(tokens)
    dt = time.perf_counter() - t0
    print(f"{requests} requests x {tokens} tokens: {requests/dt:.1f} req/s "
          f"({dt/requests*1000:.2f} ms/req incl. runtime round-trip)")
