"""Mixed-workload GPU soak: every subsystem at once on one box."""
import io
import os
import sys
import threading
import time
import urllib.request

sys.path.insert(0, "/root/repo")
import modal_amd as modal
from modal_amd._sync import synchronizer
from modal_amd.client import _Client
from modal_amd.scheduler.core import Scheduler


def gpu_item(x: int) -> int:
    import torch

    c = getattr(torch, "_mx_cache", None)
    if c is None:
        c = torch.ones(4096, device="cuda", dtype=torch.bfloat16)
        torch._mx_cache = c
    return int((c * float(x % 3 + 1))[:2].float().sum().item()) and x or x


def consume_queue(q, n):
    return sum(q.get(timeout=60) for _ in range(n))


def counter(n: int):
    for i in range(n):
        yield i * i


async def boot():
    s = Scheduler()
    await s.start()
    c = _Client(s, "client")
    _Client.set_default(c)
    for _ in range(2):
        await s.pool.spawn_worker(gpu_index=0)
    return s, c


s, c = synchronizer.run(boot())
app = modal.App("mixed-soak")
map_fn = app.function(gpu=1)(modal.concurrent(max_inputs=8)(gpu_item))
qc_fn = app.function()(consume_queue)
gen_fn = app.function()(counter)


@app.function()
@modal.fastapi_endpoint()
def ping():
    return {"pong": True}


errors: list = []
results: dict = {}


def run(name, fn):
    def _wrapped():
        try:
            results[name] = fn()
        except BaseException as e:
            errors.append((name, repr(e)))
    t = threading.Thread(target=_wrapped, daemon=True)
    t.start()
    return t


with app.run(client=c):
    N = 40_000
    t0 = time.perf_counter()

    def do_map():
        async def consume():
            got = 0
            async for _ in map_fn.map.aio(range(N), order_outputs=False):
                got += 1
            return got
        return synchronizer.run(consume())

    def do_sandbox_volume():
        vol = modal.Volume.from_name("mx-vol", create_if_missing=True)
        blob = (os.urandom(1024) + b"\x00" * 3072) * (64 * 256)
        with vol.batch_upload(force=True) as b:
            b.put_file(io.BytesIO(blob), "/m.bin")
        assert b"".join(vol.read_file("m.bin")) == blob
        sb = modal.Sandbox.create("bash", "-c", "wc -c < d/m.bin", volumes={"d": vol})
        sb.wait(raise_on_termination=False)
        return int(sb.stdout.read().strip())

    def do_queue():
        with modal.Queue.ephemeral() as q:
            fcs = [qc_fn.spawn(q, 2000) for _ in range(2)]
            q.put_many(list(range(4000)))
            return sum(fc.get(timeout=240) for fc in fcs)

    def do_gen():
        return sum(gen_fn.remote_gen(500))

    def do_web():
        url = ping.web_url
        ok = 0
        for _ in range(200):
            with urllib.request.urlopen(url, timeout=30) as r:
                ok += r.status == 200
        return ok

    threads = [run("map", do_map), run("sbvol", do_sandbox_volume),
               run("queue", do_queue), run("gen", do_gen), run("web", do_web)]
    for t in threads:
        t.join(timeout=400)
    el = time.perf_counter() - t0
    assert not errors, errors
    assert results["map"] == N
    assert results["sbvol"] == 64 * 1024 * 1024
    assert results["queue"] == sum(range(4000))
    assert results["gen"] == sum(i * i for i in range(500))
    assert results["web"] == 200
    print(f"MIXED SOAK OK in {el:.1f}s: map {N} items, 64MiB volume+sandbox, "
          f"4k queue items/2 consumers, 500-item generator, 200 web hits — concurrently")
