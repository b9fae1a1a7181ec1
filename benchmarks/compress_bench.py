#!/usr/bin/env python3
"""Native LZ4 block compressor benchmark (CPU-only; the SYSHARD
authoring path).  Reports MB/s and ratio per block size on a
container-layer-like mix."""
from __future__ import annotations

import os
import random
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from shipyard_amd import ops  # noqa: E402


def main(mb: int = 256) -> None:
    random.seed(4)
    unit = (b"text line that compresses well\n" * 2000 +
            os.urandom(60_000) +
            bytes(random.choices(b"abcdefgh", k=60_000)))
    data = (unit * (mb * 1_000_000 // len(unit) + 1))[:mb * 1_000_000]
    for blk in (4096, 8192, 65536):
        t0 = time.perf_counter()
        comps = ops.lz4_compress_blocks(data, blk)
        dt = time.perf_counter() - t0
        comp_bytes = sum(len(c) if c else min(blk, len(data) - i * blk)
                         for i, c in enumerate(comps))
        print(f"blk {blk//1024:>2}K: {len(data)/1e6/dt:7.0f} MB/s  "
              f"ratio {comp_bytes/len(data):.3f}")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 256)
