"""GPU numerics tests for the HIP/CDNA4 kernels (vs CPU references)."""

from __future__ import annotations

import hashlib
import random

import pytest

pytestmark = pytest.mark.gpu


def _require_gpu():
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    return torch


def test_sha256_kernel_vs_hashlib():
    torch = _require_gpu()
    from modal_amd.ops.hashing import sha256_many_gpu

    rng = random.Random(7)
    lengths = [0, 1, 55, 56, 63, 64, 65, 119, 120, 127, 128, 1000, 64 * 1024, 100_000]
    lengths += [rng.randrange(0, 70_000) for _ in range(50)]
    blob = bytes(rng.randrange(256) for _ in range(sum(lengths)))
    offsets, off = [], 0
    for ln in lengths:
        offsets.append(off)
        off += ln

    buf = torch.frombuffer(bytearray(blob), dtype=torch.uint8).cuda()
    out = sha256_many_gpu(
        buf,
        torch.tensor(offsets, dtype=torch.int64),
        torch.tensor(lengths, dtype=torch.int64),
    )
    torch.cuda.synchronize()
    got = out.cpu().numpy().tobytes()
    for i, (o, ln) in enumerate(zip(offsets, lengths)):
        expect = hashlib.sha256(blob[o : o + ln]).digest()
        assert got[i * 32 : (i + 1) * 32] == expect, f"mismatch at segment {i} (len {ln})"


def test_tree_sha256_gpu_matches_cpu():
    _require_gpu()
    from modal_amd.ops.hashing import _tree_sha256_gpu, tree_sha256_cpu

    rng = random.Random(3)
    for size in [8 * 1024 * 1024, 8 * 1024 * 1024 + 12345, 32 * 1024 * 1024 + 7]:
        data = bytes(rng.randrange(256) for _ in range(1024)) * (size // 1024)
        data = data[:size]
        assert _tree_sha256_gpu(data) == tree_sha256_cpu(data)


def test_pack_unpack_roundtrip():
    torch = _require_gpu()
    from modal_amd.ops.packing import pack_gpu, unpack_gpu

    rng = random.Random(11)
    lengths = [rng.randrange(1, 200_000) for _ in range(64)]
    total = sum(lengths)
    src = torch.randint(0, 256, (total + 1024,), dtype=torch.uint8, device="cuda")
    offsets, off = [], 0
    for ln in lengths:
        offsets.append(off)
        off += ln

    packed, dst_off = pack_gpu(
        src, torch.tensor(offsets, dtype=torch.int64), torch.tensor(lengths, dtype=torch.int64)
    )
    torch.cuda.synchronize()
    # reference: concatenation of slices
    expect = torch.cat([src[o : o + l] for o, l in zip(offsets, lengths)])
    assert torch.equal(packed, expect)

    # scatter back into a fresh buffer at the same offsets
    dst = torch.zeros_like(src)
    unpack_gpu(
        packed,
        dst,
        torch.tensor(offsets, dtype=torch.int64),
        torch.tensor(lengths, dtype=torch.int64),
    )
    torch.cuda.synchronize()
    for o, l in zip(offsets, lengths):
        assert torch.equal(dst[o : o + l], src[o : o + l])


def test_gpu_function_remote(client):
    """BASELINE config 2: gpu=1 function running torch.mm bf16 via .remote()."""
    torch = _require_gpu()
    import modal_amd as modal

    app = modal.App("gpu-smoke")

    @app.function(gpu=1)
    def mm(n):
        import torch

        a = torch.randn(n, n, dtype=torch.bfloat16, device="cuda")
        b = torch.randn(n, n, dtype=torch.bfloat16, device="cuda")
        c = (a @ b).float().mean()
        torch.cuda.synchronize()
        return float(c.item())

    with app.run(client=client):
        r = mm.remote(512)
        assert isinstance(r, float)


def test_gpu_map_small(client):
    torch = _require_gpu()
    import modal_amd as modal

    app = modal.App("gpu-map")

    @app.function(gpu=1)
    def work(x):
        import torch

        t = torch.full((64,), float(x), device="cuda")
        return float(t.sum().item())

    with app.run(client=client):
        out = list(work.map(range(32)))
        assert out == [x * 64.0 for x in range(32)]


def test_blobstore_uses_gpu_hash(tmp_path):
    _require_gpu()
    from modal_amd.ops.hashing import content_digest, tree_sha256_cpu
    from modal_amd.scheduler.blobs import BlobStore

    store = BlobStore(str(tmp_path / "blobs"))
    data = bytes(bytearray(range(256)) * (40 * 1024))  # 10 MiB -> tree/GPU path
    digest = store.put(data)
    assert digest == tree_sha256_cpu(data).hex()
    assert store.get(digest) == data


def test_content_digests_batch_gpu_matches_cpu():
    _require_gpu()
    from modal_amd.ops.hashing import content_digest, content_digests_batch

    import os as _os

    buffers = [_os.urandom(9 * 1024 * 1024), b"small", _os.urandom(8 * 1024 * 1024), b""]
    got = content_digests_batch(buffers)
    expect = [content_digest(b) for b in buffers]
    assert got == expect


def test_sha256_ilp2_matches_hashlib():
    torch = _require_gpu()
    from modal_amd.ops.hashing import sha256_many_gpu

    rng = random.Random(13)
    lengths = [0, 1, 63, 64, 65, 4096, 16384, 16385] + [rng.randrange(0, 40_000) for _ in range(21)]
    blob = bytes(rng.randrange(256) for _ in range(sum(lengths)))
    offsets, off = [], 0
    for ln in lengths:
        offsets.append(off)
        off += ln
    buf = torch.frombuffer(bytearray(blob), dtype=torch.uint8).cuda()
    out = sha256_many_gpu(
        buf,
        torch.tensor(offsets, dtype=torch.int64),
        torch.tensor(lengths, dtype=torch.int64),
        ilp=2,
    )
    torch.cuda.synchronize()
    got = out.cpu().numpy().tobytes()
    for i, (o, ln) in enumerate(zip(offsets, lengths)):
        assert got[i * 32 : (i + 1) * 32] == hashlib.sha256(blob[o : o + ln]).digest(), i


def test_sandbox_gpu_visible(client):
    _require_gpu()
    import modal_amd as modal

    sb = modal.Sandbox.create(
        "python3", "-c", "import torch; print(torch.cuda.is_available())", gpu=1
    )
    rc = sb.wait(raise_on_termination=False)
    assert rc == 0, sb.stderr.read()
    assert sb.stdout.read().strip() == "True"


def test_volume_gpu_hash_upload(client):
    """Config 4: volume batch_upload of a large file rides the GPU hash."""
    _require_gpu()
    import io
    import os as _os

    import modal_amd as modal

    payload = _os.urandom(1 << 20) * 24  # 24 MiB -> 3 blocks, GPU-batched hash
    with modal.Volume.ephemeral() as vol:
        with vol.batch_upload() as batch:
            batch.put_file(io.BytesIO(payload), "model.bin")
        data = b"".join(chunk for chunk in vol.read_file("model.bin"))
        assert data == payload


@pytest.mark.gpu
def test_map_returning_gpu_tensors_batched_readback(client):
    """Map items returning CUDA tensors arrive as host tensors via ONE
    batched D2H per chunk (not a per-item sync); values are exact."""
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import modal_amd as modal

    app = modal.App("tensor-map-app")

    @app.function(gpu=1)
    def make(i):
        import torch

        return torch.full((8,), float(i), device="cuda", dtype=torch.bfloat16)

    with app.run(client=client):
        outs = list(make.map(range(200), order_outputs=True))
    assert len(outs) == 200
    for i, t in enumerate(outs):
        assert not t.is_cuda          # host tensor on arrival
        assert t.dtype == torch.bfloat16 and t.shape == (8,)
        assert float(t[0]) == float(i)


@pytest.mark.gpu
def test_hipgraph_captured_call_numerics_and_speed():
    """GraphedCall: identical numerics to eager fp32 reference; one replay
    replaces the k eager dispatches of a launch-bound chain."""
    import time

    import torch

    from modal_amd.ops.hipgraph import GraphedCall

    w = torch.randn(4096, device="cuda", dtype=torch.float32)

    def chain(x, scale):
        # 20 small elementwise kernels: the launch-bound shape graphs exist for
        y = x * scale
        for _ in range(18):
            y = y + w * 0.001
        return y.sum().reshape(())

    g = GraphedCall(chain, ring_depth=8)
    x0 = torch.randn(4096, device="cuda", dtype=torch.float32)

    # numerics: captured path == eager fp32 reference
    for k in (1.0, 2.5, -3.0):
        got = g(x0, k)
        want = chain(x0, torch.tensor(k, device="cuda"))
        torch.cuda.synchronize()
        assert torch.allclose(got, want, rtol=1e-5, atol=1e-5), (got, want)

    # ring holds results across ring_depth-1 further calls
    first = g(x0, 7.0).clone()
    for i in range(6):
        g(x0, float(i))
    torch.cuda.synchronize()
    assert torch.allclose(first, chain(x0, torch.tensor(7.0, device="cuda")), rtol=1e-5, atol=1e-5)

    # dispatch cost: graphed replay must beat eager for the 20-kernel chain
    n = 300
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(n):
        chain(x0, float(i % 5 + 1))
    torch.cuda.synchronize()
    eager_s = time.perf_counter() - t0

    t0 = time.perf_counter()
    for i in range(n):
        g(x0, float(i % 5 + 1))
    torch.cuda.synchronize()
    graph_s = time.perf_counter() - t0
    assert g.replays >= n
    # require a real win, with margin for box variance
    assert graph_s < eager_s * 0.7, (graph_s, eager_s)
