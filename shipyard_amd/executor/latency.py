"""Submit->launch latency probe (BASELINE.md metric #2).

Creates a throwaway pool + job in a temp store, submits tiny process
tasks one at a time, and measures submit -> first 'launched' transition.
The reference's analogous path is Azure Batch task submission ->
scheduling on a node (seconds to minutes); here it is the local
scheduler's assignment latency (sub-millisecond store ops + process
spawn).
"""
from __future__ import annotations

import statistics
import tempfile
import time


def measure_submit_launch(samples: int = 10) -> float:
    """Returns p50 submit->launch latency in milliseconds."""
    from shipyard_amd.executor.service import LocalExecutor

    with tempfile.TemporaryDirectory(prefix="sy-lat-") as td:
        ex = LocalExecutor(td, detect_gpus=False)
        ex.pool_add({"pool_specification": {
            "id": "latpool",
            "gpus": {"dedicated": 0, "low_priority": 0},
            "cpu_slots": 1,
            "node_configuration": {"rocm": {"verify": False}},
        }})
        lat_ms = []
        for i in range(samples):
            jid = f"latjob{i}"
            t0 = time.perf_counter()
            ex.jobs_add({"job_specifications": [{
                "id": jid,
                "tasks": [{"id": "t", "command": "true"}],
            }]}, pool_id="latpool")
            # schedule until the task launches
            while True:
                ex.schedule_once()
                row = ex.store.query_one(
                    "SELECT start_time FROM tasks WHERE job_id=? AND id='t'",
                    (jid,))
                if row and row["start_time"]:
                    break
            lat_ms.append((time.perf_counter() - t0) * 1e3)
            ex.run_until_idle(timeout=30)
        ex.store.close()
        return round(statistics.median(lat_ms), 3)
