#!/usr/bin/env python3
"""SYSHARD block-size sweep, end to end (compressible data).

The decode kernel is fastest at small blocks (occupancy), but the
stager pipeline is H2D-bound, so SMALLER compressed payloads (bigger
blocks, better LZ4 windows) can win end-to-end as long as decode rate
stays above the bus feed.  This measures raw-GB/s of
NVMe/page-cache -> pinned -> HBM -> GPU decode per block size on a
container-layer-like corpus (ELF + text), to pick DEFAULT_BLOCK_RAW.
"""
from __future__ import annotations

import io
import os
import sys
import tarfile
import tempfile
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from shipyard_amd.data import shardfmt
from shipyard_amd.data.stager import ShardStager


def corpus(target_mb: int = 512) -> bytes:
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w") as tf:
        for root in ("/opt/rocm/bin", "/opt/rocm/lib", "/usr/share/doc",
                     "/etc"):
            for dirpath, _, files in os.walk(root):
                for f in files:
                    p = os.path.join(dirpath, f)
                    try:
                        if os.path.isfile(p) and \
                                os.path.getsize(p) < 16e6:
                            tf.add(p, arcname=p)
                    except (PermissionError, FileNotFoundError):
                        pass
                if buf.tell() > target_mb * 1e6:
                    return buf.getvalue()[:int(target_mb * 1e6)]
    return buf.getvalue()


def main(mb: int = 512, sizes=(4096, 8192, 16384, 32768)) -> None:
    assert torch.cuda.is_available()
    torch.zeros(1, device="cuda")
    torch.cuda.synchronize()
    data = corpus(mb)
    n = len(data)
    print(f"corpus {n/1e6:.0f} MB")
    td = Path(tempfile.mkdtemp(prefix="blk-bench-"))
    out = {}
    for blk in sizes:
        t0 = time.perf_counter()
        packed = shardfmt.pack(data, block_raw=blk)
        pack_s = time.perf_counter() - t0
        path = td / f"b{blk}.syshard"
        path.write_bytes(packed)
        st = ShardStager(staging_mb=128, verify=True, native=True)
        best = 0.0
        for rep in range(4):  # rep 0 warms page cache + DVFS
            t0 = time.perf_counter()
            tensor, _ = st.stage_file(path)
            torch.cuda.synchronize()
            sec = time.perf_counter() - t0
            if rep:
                best = max(best, n / sec / 1e9)
            got = bytes(tensor[:4096].cpu().numpy().tobytes())
            assert got == data[:4096]
            del tensor
        out[blk] = (len(packed) / n, best, pack_s)
        print(f"block {blk//1024:>2}K: ratio {len(packed)/n:.4f}  "
              f"stage+decode {best:6.2f} GB/s raw  "
              f"(pack {n/1e6/pack_s:.0f} MB/s)", flush=True)
    os.makedirs("gpurun_out", exist_ok=True)
    import json

    with open("gpurun_out/block_size_bench.json", "w") as f:
        json.dump({str(k): {"ratio": round(v[0], 4),
                            "raw_GBps": round(v[1], 2)}
                   for k, v in out.items()}, f, indent=1)


if __name__ == "__main__":
    import sys as _sys

    mb = int(_sys.argv[1]) if len(_sys.argv) > 1 else 512
    sizes = tuple(int(x) for x in _sys.argv[2].split(",")) \
        if len(_sys.argv) > 2 else (4096, 8192, 16384, 32768)
    main(mb, sizes)
