"""LZ4 chunk compression: CPU reference codec + container + CAS integration."""

from __future__ import annotations

import os
import random

import pytest

from modal_amd.ops import compress as C
from modal_amd.utils import lz4ref


def test_block_roundtrip_assorted():
    rng = random.Random(9)
    cases = [
        b"",
        b"x",
        b"abcd" * 2000,
        os.urandom(4096),
        b"the quick brown fox jumps over the lazy dog " * 100,
        bytes(rng.randrange(3) for _ in range(4096)),
        b"\x00" * 4096,
    ]
    for data in cases:
        comp = lz4ref.compress_block(data)
        assert lz4ref.decompress_block(comp, len(data)) == data


def test_container_roundtrip_cpu():
    text = (b"log line: request served in 3ms path=/api/v1\n" * 5000)[:180_000]
    blob = C.compress_buffer_cpu(text)
    assert blob is not None
    assert len(blob) < len(text) // 3
    assert C.decompress_buffer_cpu(blob) == text
    raw_len, comp_lens, _ = C.parse_header(blob)
    assert raw_len == len(text)
    assert len(comp_lens) == (len(text) + C.SEG_SIZE - 1) // C.SEG_SIZE


def test_incompressible_bails():
    assert C.compress_buffer_cpu(os.urandom(50_000)) is None


def test_cas_transparent_compression(tmp_path, monkeypatch):
    """Compressed storage is invisible to CAS users (round trip + key)."""
    from modal_amd.scheduler.blobs import BlobStore

    store = BlobStore(str(tmp_path))
    # force the CPU compressor in for the test (GPU-only by default)
    monkeypatch.setattr("modal_amd.ops.compress.compress_buffer", C.compress_buffer_cpu)
    data = b"A repetitive volume block. " * 60_000  # ~1.6 MiB, compressible
    digest = store.put(data)
    stored = open(store._zpath(digest), "rb").read()  # stored under the .z name
    assert stored.startswith(b"MALZ41")
    assert len(stored) < len(data) // 3
    assert store.get(digest) == data
    # materialize() writes RAW bytes for volume/mount trees
    dest = str(tmp_path / "materialized")
    store.materialize(digest, dest)
    assert open(dest, "rb").read() == data
    # open_path always yields the RAW form
    assert open(store.open_path(digest), "rb").read() == data


def test_raw_payload_with_magic_prefix_roundtrips(tmp_path):
    """A raw-stored payload that itself begins with the MALZ41 magic must not
    be mis-decompressed: the compressed bit is out-of-band (.z filename), not
    sniffed from content (advisor finding, round 1)."""
    from modal_amd.scheduler.blobs import BlobStore

    store = BlobStore(str(tmp_path))
    for data in (b"MALZ41", b"MALZ41" + os.urandom(4096), b"MALZ41\x00\x00\x00\x00garbage"):
        digest = store.put(data)
        assert store.get(digest) == data
        assert open(store.open_path(digest), "rb").read() == data
        dest = str(tmp_path / "m")
        store.materialize(digest, dest)
        assert open(dest, "rb").read() == data


def test_put_file_matches_put_digest(tmp_path):
    """put_file and put() compute the same CAS key for identical content on
    both sides of the tree-digest threshold (advisor finding, round 1)."""
    from modal_amd.ops.hashing import GPU_MIN_BYTES
    from modal_amd.scheduler.blobs import BlobStore

    store = BlobStore(str(tmp_path / "cas"))
    for size in (4096, GPU_MIN_BYTES + 12345):
        data = (b"dedup-check-%d " % size) * (size // 16 + 1)
        data = data[:size]
        p = tmp_path / f"f{size}"
        p.write_bytes(data)
        assert store.put_file(str(p)) == store.put(data)


@pytest.mark.gpu
def test_gpu_compress_matches_cpu_decompress():
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    text = (b"GPU compression parity line with some repetition. " * 40_000)[:1_900_000]
    blob = C.compress_buffer_gpu(text)
    assert blob is not None and len(blob) < len(text) // 2
    assert C.decompress_buffer_cpu(blob) == text
    assert C.decompress_buffer_gpu(blob) == text


@pytest.mark.gpu
def test_gpu_decompress_cpu_container():
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    mixed = ("compressible部分 ".encode() * 30_000) + os.urandom(300_000)
    blob = C.compress_buffer_cpu(mixed)
    assert blob is not None
    assert C.decompress_buffer_gpu(blob) == mixed


@pytest.mark.gpu
def test_gpu_batched_compress_matches_single():
    """compress_buffers_gpu (one H2D + one sync for all blocks) produces
    containers byte-decodable to the originals, agreeing with the
    single-buffer path's compress/raw decisions."""
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    blocks = [
        (b"repetitive volume block content " * 90_000)[: 8 * 1024 * 1024],  # full block
        os.urandom(8 * 1024 * 1024),                                        # incompressible
        (b"tail block data " * 50_000)[: 3 * 1024 * 1024 + 1234],           # short odd tail
        b"x" * 4096,                                                        # single segment
    ]
    batched = C.compress_buffers_gpu(blocks)
    assert batched[1] is None  # random data: stored raw
    for data, blob in zip(blocks, batched):
        single = C.compress_buffer_gpu(data)
        assert (blob is None) == (single is None)
        if blob is not None:
            assert C.decompress_buffer_cpu(blob) == data
            assert C.decompress_buffer_gpu(blob) == data


@pytest.mark.gpu
def test_put_many_batched_compression_roundtrip():
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import tempfile

    from modal_amd.scheduler.blobs import BlobStore

    store = BlobStore(tempfile.mkdtemp())
    blocks = [
        (b"volume block %d " % i) * 600_000 for i in range(4)
    ] + [os.urandom(2 * 1024 * 1024)]
    digests = store.put_many(blocks)
    for digest, data in zip(digests, blocks):
        assert store.get(digest) == data
    # compressible blocks got stored as MALZ41 (smaller on disk)
    assert store.size(digests[0]) < len(blocks[0]) // 2


@pytest.mark.gpu
def test_pipelined_put_many_bit_identical():
    """The pipelined window path (ops/pipeline.py) produces the same CAS
    keys and round-trips the same bytes as the serial path."""
    import tempfile

    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from modal_amd.scheduler.blobs import BlobStore

    BLOCK = 8 * 1024 * 1024
    blocks = []
    for i in range(20):  # 160 MiB > PIPELINE_MIN_BYTES
        if i % 3 == 0:
            blocks.append(os.urandom(BLOCK))  # incompressible
        else:
            blocks.append((b"compressible %d " % i) * (BLOCK // 16))
    blocks.append(b"tail-block" * 1000)  # small odd-size tail

    store_a = BlobStore(tempfile.mkdtemp())
    # call the pipelined path DIRECTLY so a failure can't silently fall
    # back to the serial path
    digests_pipelined = store_a._put_many_pipelined(blocks)

    store_b = BlobStore(tempfile.mkdtemp())
    store_b.PIPELINE_MIN_BYTES = 1 << 60  # force the serial path
    digests_serial = store_b.put_many(blocks)

    assert digests_pipelined == digests_serial
    for digest, block in zip(digests_pipelined, blocks):
        assert store_a.get(digest) == block


def test_native_codec_cross_compatible_with_reference_codec():
    """csrc/core.cpp's segment-parallel LZ4 codec and the pure-Python
    reference (utils/lz4ref.py) decode each other's containers and agree on
    compress/raw decisions."""
    from modal_amd.ops.compress import _native_core
    from modal_amd.utils import lz4ref

    core = _native_core()
    if core is None:
        pytest.skip("native core not built")

    def py_container(data):
        # the pre-native pure-python container builder
        n = len(data)
        comp_lens, payload = [], bytearray()
        for start in range(0, n, C.SEG_SIZE):
            seg = data[start : start + C.SEG_SIZE]
            comp = lz4ref.compress_block(seg)
            if len(comp) < len(seg):
                comp_lens.append(len(comp))
                payload += comp
            else:
                comp_lens.append(0)
                payload += seg
        if not comp_lens or len(payload) >= n * C.MIN_GAIN:
            return None
        return C._header(n, comp_lens) + bytes(payload)

    cases = [
        b"",
        b"abcd" * 3000,
        os.urandom(60_000),
        (b"mixed \x00\x01 payload " * 10_000) + os.urandom(9000),
        bytes(range(256)) * 40,
    ]
    for data in cases:
        native = core.malz_compress(data, C.MIN_GAIN)
        ref = py_container(data)
        assert (native is None) == (ref is None)
        if native is not None:
            assert core.malz_decompress(bytes(ref)) == data   # C++ reads py
        if ref is not None:
            # py reference decoder reads the C++ container
            raw_len, comp_lens, off = C.parse_header(native)
            out = bytearray()
            pos = off
            for i, clen in enumerate(comp_lens):
                seg_raw = min(C.SEG_SIZE, raw_len - i * C.SEG_SIZE)
                if clen == 0:
                    out += native[pos : pos + seg_raw]
                    pos += seg_raw
                else:
                    out += lz4ref.decompress_block(native[pos : pos + clen], seg_raw)
                    pos += clen
            assert bytes(out) == data
