import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
from benchmarks.data_plane_bench import bench_lz4  # noqa: E402

for br in (8192, 16384):
    for pc in (False, True):
        r = bench_lz4(total_raw=1 << 30, block_raw=br, pc=pc)
        v = round(r["GBps"], 2)
        print(f"block={br} pc={pc}: {v} GB/s", flush=True)
