"""Minimal CBOR (RFC 8949) codec.

The reference supports a ``cbor`` payload format next to pickle
(/root/reference/py/modal/_serialization.py:365,393; config ``payload_format``).
No CBOR library ships in this environment, so this is a small self-contained
implementation of the subset needed for function payloads: ints, floats,
bytes, str, bool, None, lists, dicts, and tagged bignums. Encodings follow the
RFC so outputs are interoperable with any standard CBOR decoder.
"""

from __future__ import annotations

import struct
from io import BytesIO
from typing import Any

_MT_UINT = 0
_MT_NINT = 1
_MT_BYTES = 2
_MT_TEXT = 3
_MT_ARRAY = 4
_MT_MAP = 5
_MT_TAG = 6
_MT_SIMPLE = 7


def _encode_head(out: BytesIO, major: int, value: int) -> None:
    mt = major << 5
    if value < 24:
        out.write(bytes([mt | value]))
    elif value < 0x100:
        out.write(bytes([mt | 24, value]))
    elif value < 0x10000:
        out.write(bytes([mt | 25]) + struct.pack(">H", value))
    elif value < 0x100000000:
        out.write(bytes([mt | 26]) + struct.pack(">I", value))
    else:
        out.write(bytes([mt | 27]) + struct.pack(">Q", value))


def _encode(out: BytesIO, obj: Any) -> None:
    if obj is None:
        out.write(b"\xf6")
    elif obj is True:
        out.write(b"\xf5")
    elif obj is False:
        out.write(b"\xf4")
    elif isinstance(obj, int):
        if obj >= 0:
            if obj >= 1 << 64:
                payload = obj.to_bytes((obj.bit_length() + 7) // 8, "big")
                _encode_head(out, _MT_TAG, 2)
                _encode_head(out, _MT_BYTES, len(payload))
                out.write(payload)
            else:
                _encode_head(out, _MT_UINT, obj)
        else:
            n = -1 - obj
            if n >= 1 << 64:
                payload = n.to_bytes((n.bit_length() + 7) // 8, "big")
                _encode_head(out, _MT_TAG, 3)
                _encode_head(out, _MT_BYTES, len(payload))
                out.write(payload)
            else:
                _encode_head(out, _MT_NINT, n)
    elif isinstance(obj, float):
        out.write(b"\xfb" + struct.pack(">d", obj))
    elif isinstance(obj, (bytes, bytearray, memoryview)):
        data = bytes(obj)
        _encode_head(out, _MT_BYTES, len(data))
        out.write(data)
    elif isinstance(obj, str):
        data = obj.encode("utf-8")
        _encode_head(out, _MT_TEXT, len(data))
        out.write(data)
    elif isinstance(obj, (list, tuple)):
        _encode_head(out, _MT_ARRAY, len(obj))
        for item in obj:
            _encode(out, item)
    elif isinstance(obj, dict):
        _encode_head(out, _MT_MAP, len(obj))
        for k, v in obj.items():
            _encode(out, k)
            _encode(out, v)
    else:
        raise TypeError(f"Object of type {type(obj).__name__} is not CBOR-serializable")


def dumps(obj: Any) -> bytes:
    out = BytesIO()
    _encode(out, obj)
    return out.getvalue()


class _Decoder:
    def __init__(self, data: bytes):
        self.data = data
        self.pos = 0

    def _read(self, n: int) -> bytes:
        if self.pos + n > len(self.data):
            raise ValueError("CBOR: truncated input")
        chunk = self.data[self.pos : self.pos + n]
        self.pos += n
        return chunk

    def _read_uint(self, info: int) -> int:
        if info < 24:
            return info
        if info == 24:
            return self._read(1)[0]
        if info == 25:
            return struct.unpack(">H", self._read(2))[0]
        if info == 26:
            return struct.unpack(">I", self._read(4))[0]
        if info == 27:
            return struct.unpack(">Q", self._read(8))[0]
        raise ValueError(f"CBOR: unsupported additional info {info}")

    def decode(self) -> Any:
        initial = self._read(1)[0]
        major, info = initial >> 5, initial & 0x1F
        if major == _MT_UINT:
            return self._read_uint(info)
        if major == _MT_NINT:
            return -1 - self._read_uint(info)
        if major == _MT_BYTES:
            return self._read(self._read_uint(info))
        if major == _MT_TEXT:
            return self._read(self._read_uint(info)).decode("utf-8")
        if major == _MT_ARRAY:
            return [self.decode() for _ in range(self._read_uint(info))]
        if major == _MT_MAP:
            return {self.decode(): self.decode() for _ in range(self._read_uint(info))}
        if major == _MT_TAG:
            tag = self._read_uint(info)
            value = self.decode()
            if tag == 2:
                return int.from_bytes(value, "big")
            if tag == 3:
                return -1 - int.from_bytes(value, "big")
            return value  # unknown tags: pass the inner value through
        # simple / float
        if info == 20:
            return False
        if info == 21:
            return True
        if info in (22, 23):
            return None
        if info == 25:  # half float
            h = struct.unpack(">H", self._read(2))[0]
            sign = (h >> 15) & 1
            exp = (h >> 10) & 0x1F
            frac = h & 0x3FF
            if exp == 0:
                val = frac * 2.0**-24
            elif exp == 31:
                val = float("inf") if frac == 0 else float("nan")
            else:
                val = (frac + 1024) * 2.0 ** (exp - 25)
            return -val if sign else val
        if info == 26:
            return struct.unpack(">f", self._read(4))[0]
        if info == 27:
            return struct.unpack(">d", self._read(8))[0]
        raise ValueError(f"CBOR: unsupported simple value {info}")


def loads(data: bytes) -> Any:
    decoder = _Decoder(bytes(data))
    value = decoder.decode()
    if decoder.pos != len(decoder.data):
        raise ValueError("CBOR: trailing bytes")
    return value
