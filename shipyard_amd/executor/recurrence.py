"""Job schedules (recurrences).

Analogue of the reference's job-schedule + job-manager-task machinery
(reference convoy/batch.py:5390-5536 JobSchedule construction,
cargo/recurrent_job_manager.py re-submitting the pickled task map each
recurrence): the local scheduler materializes a fresh job instance
(`<id>-NNN`) from the stored jobspec each interval.
"""
from __future__ import annotations

import datetime
import json
import time
from typing import Optional

from shipyard_amd.config import settings as cfg


def _parse_when(s: Optional[str]) -> Optional[float]:
    if not s:
        return None
    return datetime.datetime.fromisoformat(s).timestamp()


class JobScheduleRunner:
    """Tracks one recurring jobspec; spawns instance jobs on schedule."""

    def __init__(self, executor, pool_id: str, jobspec: dict,
                 sid: Optional[str] = None,
                 state: Optional[dict] = None):
        self.ex = executor
        self.pool_id = pool_id
        self.jobspec = jobspec
        self.sid = sid
        self.js = cfg.job_settings(jobspec)
        if self.js.recurrence is None:
            raise ValueError("jobspec has no recurrence")
        self.interval = self.js.recurrence.interval.total_seconds()
        self.not_until = _parse_when(self.js.recurrence.do_not_run_until)
        self.not_after = _parse_when(self.js.recurrence.do_not_run_after)
        self.next_run = self.not_until or 0.0  # first run immediately
        self.instance = 0
        self.done = False
        # durable progress: survive daemon restarts (the reference
        # pickles the task map + relies on the Batch service's schedule
        # state; here the kv record carries it)
        if state:
            self.instance = state.get("instance", 0)
            self.next_run = state.get("next_run", self.next_run)
            self.done = state.get("done", False)

    def _persist(self) -> None:
        if self.sid is None:
            return
        self.ex.store.kv_set(f"schedule:{self.sid}", json.dumps({
            "pool": self.pool_id, "jobspec": self.jobspec,
            "state": {"instance": self.instance,
                      "next_run": self.next_run,
                      "done": self.done}}))

    def maybe_spawn(self, now: Optional[float] = None) -> Optional[str]:
        now = now if now is not None else time.time()
        if self.done or now < self.next_run:
            return None
        if self.not_after and now > self.not_after:
            self.done = True
            self._persist()
            return None
        if self.js.recurrence.monitor_task_completion and self.instance:
            prev = f"{self.js.id}-{self.instance - 1:03d}"
            row = self.ex.store.query_one(
                "SELECT COUNT(*) n FROM tasks WHERE job_id=? AND state IN "
                "('pending','ready','running')", (prev,))
            if row and row["n"]:
                return None  # previous recurrence still running
        inst_id = f"{self.js.id}-{self.instance:03d}"
        spec = dict(self.jobspec)
        spec = json.loads(json.dumps(spec))  # deep copy
        spec["id"] = inst_id
        spec.pop("recurrence", None)
        self.ex.jobs_add({"job_specifications": [spec]}, self.pool_id)
        self.instance += 1
        self.next_run = now + self.interval
        self._persist()
        return inst_id
