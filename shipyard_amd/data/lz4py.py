"""Pure-Python LZ4 block codec (reference implementation).

Used (a) to author compressed shards/layers in the cascade-analogue and
stager on CPU-only hosts, and (b) as the numerics reference the GPU
decoder (shipyard_amd/ops/csrc/lz4_decode.hip) is tested against.  The
``lz4`` wheel is not available in this image, so the codec is local.

Produces spec-compliant LZ4 *block* format: greedy hash-table matcher,
min match 4, offsets <= 65535, last 5 bytes literal, no match within the
final 12 bytes.  Correctness over speed — the hot path is the GPU.
"""
from __future__ import annotations

from typing import List, Tuple

MIN_MATCH = 4
MAX_OFFSET = 0xFFFF
LAST_LITERALS = 5
MFLIMIT = 12  # no match may start within the last 12 bytes


def _write_lsic(out: bytearray, value: int) -> None:
    while value >= 255:
        out.append(255)
        value -= 255
    out.append(value)


def compress_block(data: bytes) -> bytes:
    n = len(data)
    out = bytearray()
    if n == 0:
        return bytes(out)
    table: dict = {}
    anchor = 0
    i = 0
    end = n - MFLIMIT
    while i < end:
        key = data[i:i + 4]
        j = table.get(key)
        table[key] = i
        if (j is not None and i - j <= MAX_OFFSET
                and data[j:j + 4] == key):
            # extend the match, respecting the last-5-literals rule
            mlen = 4
            maxm = n - LAST_LITERALS - i
            while mlen < maxm and data[j + mlen] == data[i + mlen]:
                mlen += 1
            lit = i - anchor
            token = (min(lit, 15) << 4) | min(mlen - MIN_MATCH, 15)
            out.append(token)
            if lit >= 15:
                _write_lsic(out, lit - 15)
            out += data[anchor:i]
            offset = i - j
            out.append(offset & 0xFF)
            out.append(offset >> 8)
            if mlen - MIN_MATCH >= 15:
                _write_lsic(out, mlen - MIN_MATCH - 15)
            i += mlen
            anchor = i
        else:
            i += 1
    # final literal run
    lit = n - anchor
    token = min(lit, 15) << 4
    out.append(token)
    if lit >= 15:
        _write_lsic(out, lit - 15)
    out += data[anchor:]
    return bytes(out)


def decompress_block(comp: bytes, raw_len: int) -> bytes:
    """Reference decoder; raises ValueError on malformed input."""
    out = bytearray()
    n = len(comp)
    pos = 0
    while pos < n:
        token = comp[pos]
        pos += 1
        lit = token >> 4
        if lit == 15:
            while True:
                if pos >= n:
                    raise ValueError("truncated literal length")
                b = comp[pos]
                pos += 1
                lit += b
                if b != 255:
                    break
        if pos + lit > n:
            raise ValueError("literal overrun")
        out += comp[pos:pos + lit]
        pos += lit
        if pos == n:
            break  # last sequence: literals only
        if pos + 2 > n:
            raise ValueError("truncated offset")
        offset = comp[pos] | (comp[pos + 1] << 8)
        pos += 2
        if offset == 0 or offset > len(out):
            raise ValueError("bad offset")
        mlen = (token & 0xF) + MIN_MATCH
        if (token & 0xF) == 15:
            while True:
                if pos >= n:
                    raise ValueError("truncated match length")
                b = comp[pos]
                pos += 1
                mlen += b
                if b != 255:
                    break
        for _ in range(mlen):
            out.append(out[-offset])
    if len(out) != raw_len:
        raise ValueError(f"decoded {len(out)} != declared {raw_len}")
    return bytes(out)


def compress_buffer(data: bytes, block_raw: int = 64 * 1024
                    ) -> Tuple[bytes, List[Tuple[int, int, int, int]]]:
    """Split ``data`` into independent blocks and compress each.

    Returns (concatenated compressed bytes, block table) where each
    table row is (in_off, in_len, out_off, out_len) matching the GPU
    decoder's argument arrays.
    """
    comp = bytearray()
    table: List[Tuple[int, int, int, int]] = []
    for off in range(0, len(data), block_raw):
        raw = data[off:off + block_raw]
        c = compress_block(raw)
        table.append((len(comp), len(c), off, len(raw)))
        comp += c
    return bytes(comp), table
