#!/usr/bin/env python3
"""Cascade-analogue staging benchmark: pack a synthetic container image
(text corpus layers) into the object store, then stage it into a pool
image cache with GPU LZ4 decode + CRC verify — the end-to-end
image-replication hot path of the north star (reference
cascade/cascade.py pull+load, dockerd gzip inflate)."""
from __future__ import annotations

import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from benchmarks.lz4_realistic import corpus  # noqa: E402
from shipyard_amd.cascade.replicator import Replicator  # noqa: E402
from shipyard_amd.data.storage import ObjectStore  # noqa: E402


def main(layer_mb: int = 8, n_layers: int = 2) -> None:
    import tempfile

    td = Path(tempfile.mkdtemp(prefix="cascade-bench-"))
    src = td / "imgsrc"
    src.mkdir()
    base = corpus(layer_mb << 20)
    for i in range(n_layers):
        # vary layers slightly so they are distinct objects
        (src / f"layer{i}.bin").write_bytes(bytes([i]) * 4096 + base)

    # exclude one-time CUDA context creation from the staging time
    try:
        import torch

        if torch.cuda.is_available():
            torch.zeros(1, device="cuda")
            torch.cuda.synchronize()
    except Exception:
        pass

    store = ObjectStore(td / "store")
    events = []
    rep = Replicator(store, td / "cache",
                     perf_cb=lambda s, e, p: events.append((s, e, p)))
    t0 = time.perf_counter()
    rep.pack_image("bench-image", src)
    pack_s = time.perf_counter() - t0

    t0 = time.perf_counter()
    res = rep.stage_image("bench-image")  # GPU decode when available
    stage_s = time.perf_counter() - t0

    t0 = time.perf_counter()
    cached = rep.stage_image("bench-image")
    cached_s = time.perf_counter() - t0

    print(json.dumps({
        "layers": n_layers,
        "raw_mb": res["raw_bytes"] >> 20,
        "comp_mb": res["comp_bytes"] >> 20,
        "pack_seconds_cpu": round(pack_s, 2),
        "stage_seconds": round(stage_s, 3),
        "stage_MBps": round(res["raw_bytes"] / stage_s / 1e6, 1),
        "gpu_decode": res["gpu_decode"],
        "cached_hit_seconds": round(cached_s, 4),
        "perf_events": [e[1] for e in events],
    }), flush=True)


if __name__ == "__main__":
    main()
