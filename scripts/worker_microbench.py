"""Isolate per-item worker-side costs for the map hot path (GPU box)."""
import pickle
import sys
import time

sys.path.insert(0, "/root/repo")
import torch

assert torch.cuda.is_available()
N = 20_000
cache = torch.ones(4096, device="cuda", dtype=torch.bfloat16)

def op(x: int) -> int:
    t = cache * float(x % 7 + 1)
    return int(t[:4].float().sum().item()) and x or x

# warm
for i in range(200): op(i)
torch.cuda.synchronize()

t0 = time.perf_counter()
for i in range(N): op(i)
el = time.perf_counter() - t0
print(f"gpu op alone: {el/N*1e6:.1f} us/item")

# kernel-only (no readback)
t0 = time.perf_counter()
for i in range(N):
    t = cache * float(i % 7 + 1)
torch.cuda.synchronize()
el = time.perf_counter() - t0
print(f"gpu op no readback: {el/N*1e6:.1f} us/item")

# readback alone
t = cache * 2.0
t0 = time.perf_counter()
for i in range(N):
    t[:4].float().sum().item()
el = time.perf_counter() - t0
print(f"readback alone: {el/N*1e6:.1f} us/item")

# worker-pipeline mimicry without GPU: chunk decode + dispatch + result chunk
chunk = pickle.dumps(("C", [((i,), {}) for i in range(64)]))
t0 = time.perf_counter()
reps = N // 64
for r in range(reps):
    _tag, items = pickle.loads(chunk)
    values = [None] * 64
    for ci, (args, kwargs) in enumerate(items):
        values[ci] = args[0]
    out = pickle.dumps(values)
el = time.perf_counter() - t0
print(f"chunk decode+loop+encode (no op): {el/(reps*64)*1e6:.2f} us/item")
