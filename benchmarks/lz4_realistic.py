#!/usr/bin/env python3
"""LZ4 decode KERNEL benchmark on realistic compressible data.

Compresses an 8 MiB text corpus (ROCm headers) once per block size,
then tiles the block table to 512 MiB of decoded output and times the
decode kernel alone (payload + tables resident on device) — the same
methodology as data_plane_bench, but with real-content sequence
statistics (ratio ~0.2-0.35, long matches) instead of the synthetic
worst case.  Reports throughput and ratio per block size.
"""
from __future__ import annotations

import ctypes
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from shipyard_amd import ops
from shipyard_amd.data import shardfmt


def corpus(target: int = 8 << 20) -> bytes:
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


def bench_block_size(data: bytes, block_raw: int, total_raw: int) -> dict:
    packed = shardfmt.pack(data, block_raw=block_raw)
    idx = shardfmt.read_index(packed)
    lz4_blocks = [b for b in idx.blocks if not b.stored]
    ratio = sum(b.comp_len for b in idx.blocks) / max(idx.raw_size, 1)
    dev = torch.device("cuda")
    payload = packed[idx.payload_off:]
    d_comp = torch.frombuffer(bytearray(payload), dtype=torch.uint8).to(dev)

    # tile the lz4 block table to total_raw decoded bytes
    reps = max(total_raw // (len(lz4_blocks) * block_raw), 1)
    in_off, in_len, out_off, out_len = [], [], [], []
    pos = 0
    for _ in range(reps):
        for b in lz4_blocks:
            in_off.append(b.comp_off)
            in_len.append(b.comp_len)
            out_off.append(pos)
            out_len.append(b.raw_len)
            pos += block_raw  # keep 16B-aligned slots
    n_blocks = len(in_off)
    t64 = lambda v: torch.tensor(v, dtype=torch.int64, device=dev)
    t32 = lambda v: torch.tensor(v, dtype=torch.int64).to(
        torch.uint32).to(dev)
    d_in_off, d_in_len = t64(in_off), t32(in_len)
    d_out_off, d_out_len = t64(out_off), t32(out_len)
    d_out = torch.empty(pos, dtype=torch.uint8, device=dev)
    d_status = torch.empty(n_blocks, dtype=torch.uint32, device=dev)
    lib = ops._load()

    def run():
        lib.sy_lz4_decode_blocks(
            ctypes.c_void_p(d_comp.data_ptr()),
            ctypes.c_void_p(d_in_off.data_ptr()),
            ctypes.c_void_p(d_in_len.data_ptr()),
            ctypes.c_void_p(d_out.data_ptr()),
            ctypes.c_void_p(d_out_off.data_ptr()),
            ctypes.c_void_p(d_out_len.data_ptr()),
            ctypes.c_void_p(d_status.data_ptr()),
            ctypes.c_uint32(n_blocks), ctypes.c_uint32(block_raw),
            ops._stream())

    run()
    torch.cuda.synchronize()
    assert ops.lz4_all_ok(d_status)
    got = bytes(d_out[:min(block_raw, 4096)].cpu().numpy().tobytes())
    assert got == data[:len(got)], "decode mismatch"
    # DVFS warm-up: the CPU-side compression above leaves the GPU idle
    # and clocks low; short timed bursts then under-read by 2x.  Spin
    # the kernel ~1s before timing.
    t_warm = time.perf_counter()
    while time.perf_counter() - t_warm < 1.0:
        run()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 20
    for _ in range(iters):
        run()
    torch.cuda.synchronize()
    sec = (time.perf_counter() - t0) / iters
    decoded = sum(out_len)
    return {"block_raw": block_raw, "comp_ratio": round(ratio, 3),
            "decode_GBps": round(decoded / sec / 1e9, 2),
            "decoded_mb": decoded >> 20, "n_blocks": n_blocks}


def main():
    assert torch.cuda.is_available()
    data = corpus()
    for block_raw in (4096, 8192, 16384, 65536):
        print(json.dumps(bench_block_size(data, block_raw, 512 << 20)),
              flush=True)


if __name__ == "__main__":
    main()
