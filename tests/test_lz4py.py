"""CPU tests for the pure-Python LZ4 block codec (the GPU decoder's
reference; see shipyard_amd/ops/csrc/lz4_decode.hip)."""
import os
import random

import pytest

from shipyard_amd.data import lz4py


CASES = [
    b"",
    b"a",
    b"abcd" * 5,
    b"hello world! " * 500,
    os.urandom(1000),
    os.urandom(64 * 1024),
]


@pytest.mark.parametrize("data", CASES, ids=range(len(CASES)))
def test_roundtrip(data):
    comp = lz4py.compress_block(data)
    assert lz4py.decompress_block(comp, len(data)) == data


def test_roundtrip_low_entropy():
    random.seed(7)
    data = bytes(random.choices(b"ab", k=100_000))
    comp, table = lz4py.compress_buffer(data, block_raw=16 * 1024)
    assert len(comp) < len(data) * 3 // 4  # actually compresses
    out = bytearray(len(data))
    for (io, il, oo, ol) in table:
        out[oo:oo + ol] = lz4py.decompress_block(comp[io:io + il], ol)
    assert bytes(out) == data


def test_decoder_rejects_bad_offset():
    # token with match but offset 0
    bad = bytes([0x10, ord("x"), 0x00, 0x00])
    with pytest.raises(ValueError):
        lz4py.decompress_block(bad, 100)


def test_decoder_rejects_truncated():
    with pytest.raises(ValueError):
        lz4py.decompress_block(bytes([0xF0]), 100)
