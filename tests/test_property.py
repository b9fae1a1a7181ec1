"""Property-based tests (hypothesis): LZ4 codec roundtrip, GF(2) CRC
combine algebra, SYSHARD pack/unpack."""
from hypothesis import given, settings, strategies as st

from shipyard_amd.data import lz4py, shardfmt
from shipyard_amd.ops import gf2


@settings(max_examples=60, deadline=None)
@given(st.binary(max_size=20000))
def test_lz4_roundtrip_property(data):
    comp = lz4py.compress_block(data)
    assert lz4py.decompress_block(comp, len(data)) == data


@settings(max_examples=60, deadline=None)
@given(st.binary(max_size=4000), st.binary(max_size=4000))
def test_crc_combine_property(a, b):
    raw = gf2.combine_raw(gf2.crc32c_raw(a), gf2.crc32c_raw(b), len(b))
    assert raw == gf2.crc32c_raw(a + b)
    assert gf2.finish(raw, len(a) + len(b)) == gf2.crc32c(a + b)


@settings(max_examples=30, deadline=None)
@given(st.binary(max_size=40000),
       st.sampled_from([4096, 8192, 16384]))
def test_syshard_roundtrip_property(data, block):
    packed = shardfmt.pack(data, block_raw=block)
    assert shardfmt.unpack_cpu(packed) == data


@settings(max_examples=40, deadline=None)
@given(st.lists(st.binary(min_size=1, max_size=300), min_size=1,
                max_size=6))
def test_runs_compress_repetitive(parts):
    # repetition-heavy inputs still roundtrip (match-copy paths)
    data = b"".join(p * 7 for p in parts)
    comp = lz4py.compress_block(data)
    assert lz4py.decompress_block(comp, len(data)) == data


@given(st.binary(max_size=30000),
       st.sampled_from([4096, 8192, 65536]))
@settings(max_examples=60, deadline=None)
def test_native_compressor_property(data, block):
    """Property: every native-compressed block decodes to its source
    with the independent python decoder."""
    from shipyard_amd import ops

    if not ops.native_compress_available():
        return
    comps = ops.lz4_compress_blocks(data, block)
    for i, c in enumerate(comps):
        raw = data[i * block:(i + 1) * block]
        if c is None:
            continue
        assert len(c) < len(raw)
        assert lz4py.decompress_block(c, len(raw)) == raw


@given(st.binary(max_size=30000))
@settings(max_examples=40, deadline=None)
def test_pack_roundtrip_with_native_property(data):
    """pack (native compressor path) -> unpack_cpu is identity, with
    CRC verification on."""
    assert shardfmt.unpack_cpu(shardfmt.pack(data)) == data
