"""CPU-side hashing invariants (the GPU kernel is tested against these)."""

from __future__ import annotations

import hashlib
import os

from modal_amd.ops.hashing import (
    GPU_MIN_BYTES,
    LEAF_SIZE,
    content_digest,
    content_digests_batch,
    tree_sha256_cpu,
)


def test_content_digest_small_is_plain_sha256():
    data = b"hello world"
    assert content_digest(data) == hashlib.sha256(data).hexdigest()


def test_content_digest_large_is_tree():
    data = os.urandom(1024) * (GPU_MIN_BYTES // 1024)
    assert content_digest(data) == tree_sha256_cpu(data).hex()


def test_tree_digest_deterministic_and_length_sensitive():
    a = b"x" * (LEAF_SIZE * 3 + 17)
    assert tree_sha256_cpu(a) == tree_sha256_cpu(bytes(a))
    assert tree_sha256_cpu(a) != tree_sha256_cpu(a + b"y")
    assert tree_sha256_cpu(a[:-1]) != tree_sha256_cpu(a)


def test_batch_matches_single_cpu():
    buffers = [b"tiny", os.urandom(100_000), b"", os.urandom(GPU_MIN_BYTES + 5)]
    assert content_digests_batch(buffers) == [content_digest(b) for b in buffers]


def test_blobstore_put_many(tmp_path):
    from modal_amd.scheduler.blobs import BlobStore

    store = BlobStore(str(tmp_path))
    bufs = [b"aaa", b"bbb", b"aaa"]
    digests = store.put_many(bufs)
    assert digests[0] == digests[2]
    for d, b in zip(digests, bufs):
        assert store.get(d) == b
