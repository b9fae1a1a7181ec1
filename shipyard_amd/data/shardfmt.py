"""SYSHARD — the framework's packed shard/layer format.

Container layers and data shards are stored as sequences of
independently-compressed LZ4 blocks with a per-block CRC32C manifest, so
the GPU can decode and verify a whole shard in one pass
(shipyard_amd/ops: lz4_decode_blocks + crc32c_chunks).  This replaces
the reference's reliance on dockerd gzip inflate + CPU MD5
(reference cascade/cascade.py:500-571, convoy/util.py:461-508).

Layout (little-endian):
  8s  magic  b"SYSHARD1"
  u32 flags            (bit0: blocks are LZ4; 0 = all stored)
  u32 block_raw        (max raw bytes per block; 64 KiB default)
  u64 raw_size
  u32 n_blocks
  n_blocks * { u64 comp_off, u32 comp_len, u32 raw_len, u32 crc32c }
  comp bytes...

A block with comp_len == raw_len is STORED (incompressible); the GPU
decoder copies it through.  comp offsets are relative to the payload
start and 16 B aligned so uint4 paths stay aligned.
"""
from __future__ import annotations

import io
import struct
from dataclasses import dataclass
from pathlib import Path
from typing import List

from shipyard_amd.data import lz4py
from shipyard_amd.ops import gf2

MAGIC = b"SYSHARD1"
HEADER = struct.Struct("<8sIIQI")
ENTRY = struct.Struct("<QIII")
# 8 KiB: the GPU decoder is serial-latency-bound per block; smaller
# blocks raise workgroups/CU (measured: 64K=4.9, 16K=35, 8K=94 GB/s).
# CRC chunk math requires >= 4 KiB and the block size to be 4 KiB-
# aligned.  Compression-window loss vs 64 KiB is a few percent.
DEFAULT_BLOCK_RAW = 8 * 1024


@dataclass
class BlockEntry:
    comp_off: int
    comp_len: int
    raw_len: int
    crc32c: int

    @property
    def stored(self) -> bool:
        return self.comp_len == self.raw_len


@dataclass
class ShardIndex:
    block_raw: int
    raw_size: int
    blocks: List[BlockEntry]
    payload_off: int  # byte offset of payload within the file


def _align16(n: int) -> int:
    return (n + 15) & ~15


def pack(data: bytes, block_raw: int = DEFAULT_BLOCK_RAW,
         compress: bool = True) -> bytes:
    """Pack raw bytes into SYSHARD format (CPU writer)."""
    blocks: List[BlockEntry] = []
    payload = io.BytesIO()
    off = 0
    crcs = gf2.crc32c_chunks_numpy(data, block_raw)
    for bi, boff in enumerate(range(0, len(data), block_raw) or [0]):
        raw = data[boff:boff + block_raw]
        comp = lz4py.compress_block(raw) if compress else raw
        if not compress or len(comp) >= len(raw):
            comp = raw  # stored
        crc = crcs[bi] if bi < len(crcs) else gf2.crc32c(raw)
        blocks.append(BlockEntry(off, len(comp), len(raw), crc))
        payload.write(comp)
        pad = _align16(len(comp)) - len(comp)
        payload.write(b"\x00" * pad)
        off += len(comp) + pad
    if len(data) == 0:
        blocks = []
    hdr = HEADER.pack(MAGIC, 1 if compress else 0, block_raw, len(data),
                      len(blocks))
    table = b"".join(ENTRY.pack(b.comp_off, b.comp_len, b.raw_len, b.crc32c)
                     for b in blocks)
    return hdr + table + payload.getvalue()


def read_index(buf: bytes) -> ShardIndex:
    magic, flags, block_raw, raw_size, n_blocks = HEADER.unpack_from(buf, 0)
    if magic != MAGIC:
        raise ValueError("not a SYSHARD file")
    blocks = []
    pos = HEADER.size
    for _ in range(n_blocks):
        comp_off, comp_len, raw_len, crc = ENTRY.unpack_from(buf, pos)
        pos += ENTRY.size
        blocks.append(BlockEntry(comp_off, comp_len, raw_len, crc))
    return ShardIndex(block_raw=block_raw, raw_size=raw_size, blocks=blocks,
                      payload_off=pos)


def unpack_cpu(buf: bytes, verify: bool = True) -> bytes:
    """CPU reference reader (tests + CPU-only hosts)."""
    idx = read_index(buf)
    out = io.BytesIO()
    for b in idx.blocks:
        comp = buf[idx.payload_off + b.comp_off:
                   idx.payload_off + b.comp_off + b.comp_len]
        raw = comp if b.stored else lz4py.decompress_block(comp, b.raw_len)
        if verify and gf2.crc32c(raw) != b.crc32c:
            raise ValueError("CRC mismatch in shard block")
        out.write(raw)
    data = out.getvalue()
    if len(data) != idx.raw_size:
        raise ValueError("shard size mismatch")
    return data



def _stored_contiguous(idx) -> bool:
    """True when every block is stored AND the payload layout equals the
    raw layout (comp_off == cumulative raw offset) — then the payload IS
    the decoded data and no per-block copies are needed.  This is the
    common case for incompressible shards packed with compress=False
    (block sizes that are 16 B multiples keep offsets aligned)."""
    acc = 0
    for b in idx.blocks:
        if not b.stored or b.comp_off != acc:
            return False
        acc += b.raw_len + (-b.raw_len) % 16
        if b.raw_len % 16 and b is not idx.blocks[-1]:
            return False
    return True


def unpack_gpu(buf: bytes, device=None, verify: bool = True):
    """GPU reader: upload payload once, LZ4-decode all blocks on the
    MI355X, CRC32C-verify the decoded bytes, return a uint8 CUDA tensor.

    This is the cascade-analogue/stager hot path: NVMe -> pinned host ->
    HBM (hipMemcpyAsync under torch) -> wave-cooperative decode -> chunk
    CRC — no CPU inflate, no CPU hash.
    """
    import torch

    from shipyard_amd import ops

    dev = device or torch.device("cuda", torch.cuda.current_device())
    idx = read_index(buf)
    payload = buf[idx.payload_off:]
    d_comp = torch.frombuffer(bytearray(payload), dtype=torch.uint8).to(
        dev, non_blocking=True)
    out = torch.empty(max(idx.raw_size, 1), dtype=torch.uint8, device=dev)
    if not idx.blocks:
        return out[:0]

    if _stored_contiguous(idx):
        out = d_comp[:idx.raw_size]
        if verify:
            crcs = ops.crc32c_chunks(out.contiguous(),
                                     chunk_size=idx.block_raw)
            want = [b.crc32c for b in idx.blocks]
            if [int(x) for x in crcs.tolist()] != want:
                raise ValueError("GPU CRC mismatch in stored shard")
        return out

    lz4_blocks = [(i, b) for i, b in enumerate(idx.blocks) if not b.stored]
    stored_blocks = [(i, b) for i, b in enumerate(idx.blocks) if b.stored]

    # raw offsets are cumulative; precompute once
    raw_offs = []
    acc = 0
    for b in idx.blocks:
        raw_offs.append(acc)
        acc += b.raw_len
    for i, b in stored_blocks:
        out[raw_offs[i]:raw_offs[i] + b.raw_len] = \
            d_comp[b.comp_off:b.comp_off + b.comp_len]

    if lz4_blocks:
        mk64 = lambda v: torch.tensor(v, dtype=torch.int64, device=dev)
        mk32 = lambda v: torch.tensor(v, dtype=torch.int64).to(
            torch.uint32).to(dev)
        status = ops.lz4_decode_blocks(
            d_comp,
            mk64([b.comp_off for _, b in lz4_blocks]),
            mk32([b.comp_len for _, b in lz4_blocks]),
            out,
            mk64([raw_offs[i] for i, _ in lz4_blocks]),
            mk32([b.raw_len for _, b in lz4_blocks]),
            raw_cap=idx.block_raw)
        if not ops.lz4_all_ok(status):
            raise ValueError(
                f"GPU LZ4 decode failed: status={status.cpu().tolist()}")

    if verify:
        # block_raw-aligned chunks == block boundaries (last may be short)
        crcs = ops.crc32c_chunks(out[:idx.raw_size], chunk_size=idx.block_raw)
        want = [b.crc32c for b in idx.blocks]
        got = [int(x) for x in crcs.tolist()]
        if got != want:
            bad = [i for i, (a, c) in enumerate(zip(got, want)) if a != c]
            raise ValueError(f"GPU CRC mismatch in blocks {bad[:8]}")
    return out[:idx.raw_size]


def pack_file(src: Path, dst: Path, block_raw: int = DEFAULT_BLOCK_RAW,
              compress: bool = True) -> ShardIndex:
    data = Path(src).read_bytes()
    packed = pack(data, block_raw=block_raw, compress=compress)
    Path(dst).write_bytes(packed)
    return read_index(packed)
