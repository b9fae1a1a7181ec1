"""CPU tests for the GF(2) CRC32C algebra used by the crc32c.hip combine
tree and the host-side finish/combine."""
import os

from shipyard_amd.ops import gf2


def test_crc32c_known_vectors():
    # RFC 3720 test vectors for CRC32C
    assert gf2.crc32c(b"") == 0x00000000
    assert gf2.crc32c(b"123456789") == 0xE3069283
    assert gf2.crc32c(bytes(32)) == 0x8A9136AA
    assert gf2.crc32c(bytes([0xFF] * 32)) == 0x62A8AB43


def test_combine_matches_direct():
    data = os.urandom(10_000)
    for split in (0, 1, 137, 5000, 9999, 10_000):
        a, b = data[:split], data[split:]
        raw = gf2.combine_raw(gf2.crc32c_raw(a), gf2.crc32c_raw(b), len(b))
        assert raw == gf2.crc32c_raw(data)
        assert gf2.finish(raw, len(data)) == gf2.crc32c(data)


def test_leading_zeros_identity():
    # raw CRC (init 0) ignores leading zero bytes — the property the
    # kernel's front-padded ragged-chunk handling relies on
    data = os.urandom(512)
    assert gf2.crc32c_raw(bytes(100) + data) == gf2.crc32c_raw(data)


def test_level_matrices_tree():
    chunk = 256 * 16
    data = os.urandom(chunk)
    mats = gf2.level_matrices(chunk)
    seg = chunk // 256
    lane = [gf2.crc32c_raw(data[t * seg:(t + 1) * seg]) for t in range(256)]
    for k in range(8):
        stride = 1 << k
        m = mats[k * 32:(k + 1) * 32]
        for t in range(256 >> (k + 1)):
            idx = t * (stride << 1)
            lane[idx] = gf2.matvec(m, lane[idx]) ^ lane[idx + stride]
    assert lane[0] == gf2.crc32c_raw(data)
