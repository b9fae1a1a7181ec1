"""Autoscale: scenario evaluation against local pool/task state.

The reference generates Azure Batch autoscale formula TEXT evaluated by
the service (reference convoy/autoscale.py:57-371 `_formula_tasks`,
`_formula_day_of_week`, scenarios at 351-358).  Locally there is no
formula VM — the same scenarios are evaluated directly against the
store every evaluation interval, returning target GPU slot counts which
the scheduler applies via pool_resize.

Scenarios (parity with `_AUTOSCALE_SCENARIOS`):
  active_tasks / pending_tasks — scale to task backlog;
  workday, weekday, weekend, workday_with_offpeak_max_low_priority —
  time-gated min/max occupancy.
"""
from __future__ import annotations

import datetime
from dataclasses import dataclass
from typing import Optional

from shipyard_amd.config.settings import (AutoscaleScenarioSettings,
                                          AutoscaleSettings)


@dataclass
class ScaleDecision:
    dedicated: int
    low_priority: int
    reason: str


def _clamp_increment(current: int, target: int,
                     max_inc: Optional[int]) -> int:
    if max_inc is None or max_inc < 0:
        return target
    if target > current:
        return min(target, current + max_inc)
    return target


def _in_range(day_or_hour: int, lo: int, hi: int) -> bool:
    if lo <= hi:
        return lo <= day_or_hour <= hi
    return day_or_hour >= lo or day_or_hour <= hi


def evaluate(scenario: AutoscaleScenarioSettings,
             active_tasks: int, pending_tasks: int,
             current_dedicated: int, current_low_priority: int,
             now: Optional[datetime.datetime] = None) -> ScaleDecision:
    now = now or datetime.datetime.now()
    name = scenario.name
    max_d = scenario.maximum_gpu_count_dedicated
    max_l = scenario.maximum_gpu_count_low_priority

    if name in ("active_tasks", "pending_tasks"):
        backlog = active_tasks if name == "active_tasks" else \
            active_tasks + pending_tasks
        bias = scenario.bias_node_type
        if bias == "dedicated" or (bias == "auto" and max_d >= max_l):
            want_d = min(backlog, max_d)
            want_l = min(max(backlog - want_d, 0), max_l)
        else:
            want_l = min(backlog, max_l)
            want_d = min(max(backlog - want_l, 0), max_d)
        want_d = _clamp_increment(current_dedicated, want_d,
                                  scenario.maximum_increment_dedicated)
        want_l = _clamp_increment(current_low_priority, want_l,
                                  scenario.maximum_increment_low_priority)
        return ScaleDecision(want_d, want_l,
                             f"{name}: backlog={backlog}")

    # time-gated scenarios (reference autoscale.py:211 _formula_day_of_week)
    weekday = now.isoweekday() % 7  # 0=Sunday..6=Saturday like WeekDay
    hour = now.hour
    if name == "workday":
        on = (_in_range(weekday, *scenario.weekdays)
              and _in_range(hour, *scenario.work_hours))
    elif name == "weekday":
        on = _in_range(weekday, *scenario.weekdays)
    elif name == "weekend":
        on = not _in_range(weekday, *scenario.weekdays)
    elif name == "workday_with_offpeak_max_low_priority":
        peak = (_in_range(weekday, *scenario.weekdays)
                and _in_range(hour, *scenario.work_hours))
        if peak:
            return ScaleDecision(max_d, 0, "workday peak: dedicated")
        return ScaleDecision(0, max_l, "offpeak: low priority")
    else:
        raise ValueError(f"unknown autoscale scenario {name}")
    if on:
        return ScaleDecision(max_d, max_l, f"{name}: on-window")
    return ScaleDecision(0, 0, f"{name}: off-window")


class AutoscaleController:
    """Drives pool_resize from scenario evaluation on an interval."""

    def __init__(self, executor, pool_id: str,
                 settings: AutoscaleSettings):
        self.ex = executor
        self.pool_id = pool_id
        self.settings = settings
        self.last_eval = 0.0

    def maybe_evaluate(self, now_ts: float) -> Optional[ScaleDecision]:
        if not self.settings.enabled or self.settings.scenario is None:
            return None
        if now_ts - self.last_eval < \
                self.settings.evaluation_interval.total_seconds():
            return None
        # runtime kill switch (reference `pool autoscale disable`)
        if self.ex.store.kv_get(f"autoscale_disabled:{self.pool_id}"):
            return None
        self.last_eval = now_ts
        self.ex.store.kv_set(f"autoscale_lastexec:{self.pool_id}",
                             str(now_ts))
        row = self.ex.store.query_one(
            "SELECT COUNT(*) n FROM tasks t JOIN jobs j ON t.job_id=j.id "
            "WHERE j.pool_id=? AND t.state IN ('ready','running')",
            (self.pool_id,))
        active = row["n"]
        row = self.ex.store.query_one(
            "SELECT COUNT(*) n FROM tasks t JOIN jobs j ON t.job_id=j.id "
            "WHERE j.pool_id=? AND t.state='pending'", (self.pool_id,))
        pending = row["n"]
        prow = self.ex.store.query_one(
            "SELECT gpus_dedicated, gpus_low_priority FROM pools "
            "WHERE id=?", (self.pool_id,))
        dec = evaluate(self.settings.scenario, active, pending,
                       prow["gpus_dedicated"], prow["gpus_low_priority"])
        if (dec.dedicated != prow["gpus_dedicated"]
                or dec.low_priority != prow["gpus_low_priority"]):
            self.ex.pool_resize(self.pool_id, dedicated=dec.dedicated,
                                low_priority=dec.low_priority)
            self.ex.store.add_event(f"pool:{self.pool_id}", "autoscale",
                                    {"dedicated": dec.dedicated,
                                     "low_priority": dec.low_priority,
                                     "reason": dec.reason})
        return dec
