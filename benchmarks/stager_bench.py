#!/usr/bin/env python3
"""Shard-stager benchmark at scale: a multi-GiB SYSHARD staged
NVMe/page-cache -> pinned -> HBM with GPU CRC verification
(BASELINE config #4's data path, measured end to end).

Stored-mode shards (incompressible synthetic) isolate the transport +
verify path; pack time is bytearray-rate.  Reports GB/s for the
native C++ pipeline and the python double-buffer, with and without
verification.
"""
from __future__ import annotations

import json
import os
import sys
import tempfile
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from shipyard_amd.data import shardfmt
from shipyard_amd.data.stager import ShardStager


def main(gib: float = 2.0) -> None:
    assert torch.cuda.is_available()
    torch.zeros(1, device="cuda")
    torch.cuda.synchronize()

    n = int(gib * (1 << 30))
    data = os.urandom(n)  # incompressible -> stored blocks
    td = Path(tempfile.mkdtemp(prefix="stager-bench-"))
    path = td / "big.syshard"
    t0 = time.perf_counter()
    path.write_bytes(shardfmt.pack(data, compress=False))
    pack_s = time.perf_counter() - t0

    results = {"shard_gib": round(n / (1 << 30), 2),
               "pack_seconds": round(pack_s, 2)}
    for native in (True, False):
        for verify in (True, False):
            st = ShardStager(staging_mb=128, verify=verify, native=native)
            # warm once (page cache + pools)
            tensor, _ = st.stage_file(path)
            del tensor
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            tensor, res = st.stage_file(path)
            torch.cuda.synchronize()
            sec = time.perf_counter() - t0
            key = f"{'native' if native else 'python'}" \
                  f"{'_verified' if verify else ''}"
            results[key + "_GBps"] = round(n / sec / 1e9, 2)
            # spot-check content
            assert bytes(tensor[:4096].cpu().numpy().tobytes()) == \
                data[:4096]
            del tensor
    print(json.dumps(results), flush=True)
    path.unlink()


if __name__ == "__main__":
    main(float(sys.argv[1]) if len(sys.argv) > 1 else 2.0)
