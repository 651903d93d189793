"""Large-payload map on GPU: 1 MiB args force chunk CAS spill + blob path."""
import hashlib
import os
import sys
import time

sys.path.insert(0, "/root/repo")
import modal_amd as modal
from modal_amd._sync import synchronizer
from modal_amd.client import _Client
from modal_amd.scheduler.core import Scheduler


def digest_item(blob: bytes) -> str:
    import hashlib as h

    return h.sha256(blob).hexdigest()[:16]


async def boot():
    s = Scheduler()
    await s.start()
    c = _Client(s, "client")
    _Client.set_default(c)
    for _ in range(int(os.environ.get("SOAK_WORKERS", "4"))):
        await s.pool.spawn_worker(gpu_index=0)
    return s, c


s, c = synchronizer.run(boot())
app = modal.App("bigpayload")
fn = app.function(gpu=1)(modal.concurrent(max_inputs=4)(digest_item))

N = 1000
payloads = [os.urandom(1024 * 1024 - 16) + i.to_bytes(16, "little") for i in range(N)]
expected = sorted(hashlib.sha256(p).hexdigest()[:16] for p in payloads)

with app.run(client=c):
    t0 = time.perf_counter()
    outs = sorted(fn.map(payloads, order_outputs=False))
    el = time.perf_counter() - t0
assert outs == expected, "payload corruption through the CAS spill path!"
print(f"BIGPAYLOAD OK: {N} x 1 MiB args in {el:.1f}s = "
      f"{N/el:.0f} items/s, {N/1024/el:.2f} GiB/s arg throughput")
