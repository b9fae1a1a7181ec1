#!/usr/bin/env python3
"""LZ4 decode benchmark on realistic compressible data (source text),
at several block sizes — reports both throughput AND compression ratio
so the block-size tradeoff is visible on real-ish content (the
synthetic random-letter corpus overstates ratio loss)."""
from __future__ import annotations

import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from shipyard_amd.data import shardfmt


def corpus(target: int = 8 << 20) -> bytes:
    """Concatenate ROCm headers (text) to target size, tiled."""
    buf = bytearray()
    for p in sorted(Path("/opt/rocm/include").rglob("*.h")):
        try:
            buf += p.read_bytes()
        except OSError:
            continue
        if len(buf) >= target:
            break
    if not buf:
        buf = bytearray(b"fallback text corpus " * 100000)
    while len(buf) < target:
        buf += buf[:target - len(buf)]
    return bytes(buf[:target])


def main():
    assert torch.cuda.is_available()
    data = corpus()
    for block_raw in (4096, 8192, 16384, 65536):
        packed = shardfmt.pack(data, block_raw=block_raw)
        idx = shardfmt.read_index(packed)
        ratio = sum(b.comp_len for b in idx.blocks) / max(idx.raw_size, 1)
        # warm
        out = shardfmt.unpack_gpu(packed, verify=False)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        iters = 5
        for _ in range(iters):
            out = shardfmt.unpack_gpu(packed, verify=False)
            torch.cuda.synchronize()
        sec = (time.perf_counter() - t0) / iters
        assert bytes(out[:4096].cpu().numpy().tobytes()) == data[:4096]
        print(json.dumps({
            "block_raw": block_raw,
            "comp_ratio": round(ratio, 3),
            "decode_GBps": round(len(data) / sec / 1e9, 2),
            "mb": len(data) >> 20,
        }), flush=True)


if __name__ == "__main__":
    main()
