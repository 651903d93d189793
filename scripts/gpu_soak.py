"""GPU robustness soak: worker death mid-map on a real GPU + serving loop."""
import os
import sys
import time

sys.path.insert(0, "/root/repo")
import modal_amd as modal
from modal_amd._sync import synchronizer
from modal_amd.client import _Client
from modal_amd.scheduler.core import Scheduler


def map_item(x: int) -> int:
    import torch

    cache = getattr(torch, "_soak_cache", None)
    if cache is None:
        cache = torch.ones(4096, device="cuda", dtype=torch.bfloat16)
        torch._soak_cache = cache
    t = cache * float(x % 5 + 1)
    return int(t[:2].float().sum().item()) and x or x


async def boot():
    s = Scheduler()
    await s.start()
    c = _Client(s, "client")
    _Client.set_default(c)
    for _ in range(2):
        await s.pool.spawn_worker(gpu_index=0)
    return s, c


s, c = synchronizer.run(boot())
app = modal.App("gpu-soak")
fn = app.function(gpu=1)(modal.concurrent(max_inputs=8)(map_item))

N = 60_000
with app.run(client=c):
    async def consume():
        got = 0
        killed = False
        t0 = time.perf_counter()
        async for _ in fn.map.aio(range(N), order_outputs=False):
            got += 1
            if not killed and got > N // 4:
                killed = True
                # kill one worker by exact PID mid-stream
                for w in list(s.pool.workers.values()):
                    if w.proc is not None:
                        print(f"killing worker pid={w.proc.pid} at item {got}", flush=True)
                        w.proc.kill()
                        break
        return got, time.perf_counter() - t0

    got, dt = synchronizer.run(consume())
    assert got == N, f"lost items: {got}/{N}"
    print(f"soak map: {got} items in {dt:.1f}s = {got/dt:.0f} items/s "
          f"(1 of 2 workers killed mid-stream; all items recovered)")

    # p50 serving probe after the carnage
    @app.function(gpu=1)
    def probe(n):
        import torch

        a = torch.randn(n, n, dtype=torch.bfloat16, device="cuda")
        return float((a @ a).float().mean().item())

    probe.remote(256)
    lat = []
    for _ in range(30):
        t0 = time.perf_counter()
        probe.remote(256)
        lat.append((time.perf_counter() - t0) * 1e3)
    lat.sort()
    print(f"post-soak p50 .remote(): {lat[15]:.3f} ms")
print("SOAK OK")
