"""Property-based fuzzing of the wire/data codecs (hypothesis).

Parity with the reference's random proto fuzzing
(/root/reference/py/test/../_utils/rand_pb_testing.py:96): generated
values round-trip through each codec layer byte-exactly.
"""

from __future__ import annotations

from hypothesis import given, settings
from hypothesis import strategies as st

from modal_amd.utils import cbor, lz4ref

json_like = st.recursive(
    st.none()
    | st.booleans()
    | st.integers(min_value=-(2**63), max_value=2**64 - 1)
    | st.floats(allow_nan=False)
    | st.text(max_size=40)
    | st.binary(max_size=40),
    lambda children: st.lists(children, max_size=6)
    | st.dictionaries(st.text(max_size=12), children, max_size=6),
    max_leaves=24,
)


@settings(max_examples=200, deadline=None)
@given(json_like)
def test_cbor_roundtrip(value):
    assert cbor.loads(cbor.dumps(value)) == value


@settings(max_examples=150, deadline=None)
@given(st.binary(max_size=20_000))
def test_lz4_block_roundtrip(data):
    comp = lz4ref.compress_block(data)
    assert lz4ref.decompress_block(comp, len(data)) == data


@settings(max_examples=100, deadline=None)
@given(st.binary(min_size=0, max_size=5_000))
def test_malz41_container_roundtrip(data):
    from modal_amd.ops import compress as C

    blob = C.compress_buffer_cpu(data)
    if blob is not None:  # incompressible data is stored raw by callers
        assert C.decompress_buffer_cpu(blob) == data


@settings(max_examples=100, deadline=None)
@given(st.binary(max_size=64_000))
def test_tree_digest_structure(data):
    """Tree digest is deterministic and distinct for distinct payloads of
    the same length (probabilistically)."""
    from modal_amd.ops import hashing as H

    d1 = H.tree_sha256_cpu(data)
    assert d1 == H.tree_sha256_cpu(data)
    if data:
        flipped = bytes([data[0] ^ 1]) + data[1:]
        assert H.tree_sha256_cpu(flipped) != d1


@settings(max_examples=80, deadline=None)
@given(st.lists(st.binary(max_size=3_000), max_size=8))
def test_pack_payloads_roundtrip(payloads):
    """The C++ pack/unpack payload codec (csrc/core.cpp) is exact."""
    try:
        from modal_amd import _core
    except ImportError:
        import pytest

        pytest.skip("_core extension not built")
    packed = _core.pack_payloads(payloads)
    assert _core.unpack_payloads(packed) == payloads
