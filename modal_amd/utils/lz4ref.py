"""Pure-Python LZ4 block codec (reference implementation for the HIP kernel).

Standard LZ4 block format (token | literals | 2-byte LE offset | extended
lengths). Used to cross-check ops/csrc/lz4.hip on CPU and to decompress
CAS entries on GPU-less machines. Not fast — the GPU kernel is the hot path.
"""

from __future__ import annotations

MIN_MATCH = 4
MFLIMIT = 12
LASTLITERALS = 5
HASH_BITS = 7


def _hash(v: int) -> int:
    return ((v * 2654435761) & 0xFFFFFFFF) >> (32 - HASH_BITS)


def compress_block(src: bytes) -> bytes:
    """Greedy LZ4 block compression (mirrors the kernel's parser exactly)."""
    n = len(src)
    out = bytearray()
    table: dict[int, int] = {}
    ip = 0
    anchor = 0
    mflimit = n - MFLIMIT

    def emit_literals(lit_start: int, lit_len: int, match_len: int = -1) -> None:
        ml = match_len - MIN_MATCH if match_len >= 0 else 0
        token = (min(lit_len, 15) << 4) | (min(ml, 15) if match_len >= 0 else 0)
        out.append(token)
        if lit_len >= 15:
            rest = lit_len - 15
            while rest >= 255:
                out.append(255)
                rest -= 255
            out.append(rest)
        out.extend(src[lit_start : lit_start + lit_len])

    if n >= MIN_MATCH + LASTLITERALS:
        while ip < mflimit:
            seq = int.from_bytes(src[ip : ip + 4], "little")
            h = _hash(seq)
            cand = table.get(h, -1)
            table[h] = ip
            if (
                cand >= 0
                and cand < ip
                and ip - cand <= 0xFFFF
                and src[cand : cand + 4] == src[ip : ip + 4]
            ):
                mlen = MIN_MATCH
                maxm = n - LASTLITERALS - ip
                while mlen < maxm and src[cand + mlen] == src[ip + mlen]:
                    mlen += 1
                emit_literals(anchor, ip - anchor, mlen)
                off = ip - cand
                out.append(off & 0xFF)
                out.append(off >> 8)
                ml = mlen - MIN_MATCH
                if ml >= 15:
                    rest = ml - 15
                    while rest >= 255:
                        out.append(255)
                        rest -= 255
                    out.append(rest)
                ip += mlen
                anchor = ip
            else:
                ip += 1
    emit_literals(anchor, n - anchor)
    return bytes(out)


def decompress_block(comp: bytes, raw_len: int) -> bytes:
    """Standard LZ4 block decompression."""
    out = bytearray()
    ip = 0
    n = len(comp)
    while ip < n:
        token = comp[ip]
        ip += 1
        lit = token >> 4
        if lit == 15:
            while True:
                b = comp[ip]
                ip += 1
                lit += b
                if b != 255:
                    break
        out += comp[ip : ip + lit]
        ip += lit
        if ip >= n:
            break
        off = comp[ip] | (comp[ip + 1] << 8)
        ip += 2
        mlen = token & 0xF
        if mlen == 15:
            while True:
                b = comp[ip]
                ip += 1
                mlen += b
                if b != 255:
                    break
        mlen += MIN_MATCH
        start = len(out) - off
        if start < 0:
            raise ValueError("LZ4: bad offset")
        for i in range(mlen):  # overlap-safe forward copy
            out.append(out[start + i])
    if len(out) != raw_len:
        raise ValueError(f"LZ4: expected {raw_len} bytes, got {len(out)}")
    return bytes(out)
