"""Autoscale scenario evaluation (reference convoy/autoscale.py parity)."""
import datetime


from shipyard_amd.config.settings import pool_settings
from shipyard_amd.executor import LocalExecutor
from shipyard_amd.executor.autoscale import (AutoscaleController, evaluate)


def scen(name="active_tasks", **kw):
    doc = {"pool_specification": {
        "id": "p", "gpus": {"dedicated": 0, "low_priority": 0},
        "autoscale": {"evaluation_interval": "00:00:00",
                      "scenario": dict(name=name, **kw)},
    }}
    return pool_settings(doc).autoscale


def test_active_tasks_scales_to_backlog():
    s = scen(maximum_gpu_count={"dedicated": 8, "low_priority": 0})
    d = evaluate(s.scenario, active_tasks=3, pending_tasks=10,
                 current_dedicated=0, current_low_priority=0)
    assert d.dedicated == 3 and d.low_priority == 0


def test_pending_tasks_includes_pending():
    s = scen("pending_tasks",
             maximum_gpu_count={"dedicated": 8, "low_priority": 0})
    d = evaluate(s.scenario, active_tasks=2, pending_tasks=4,
                 current_dedicated=0, current_low_priority=0)
    assert d.dedicated == 6


def test_max_cap_applies():
    s = scen(maximum_gpu_count={"dedicated": 4, "low_priority": 2})
    d = evaluate(s.scenario, active_tasks=100, pending_tasks=0,
                 current_dedicated=0, current_low_priority=0)
    assert d.dedicated == 4 and d.low_priority == 2


def test_increment_clamp():
    s = scen(maximum_gpu_count={"dedicated": 8},
             maximum_increment_per_evaluation={"dedicated": 2})
    d = evaluate(s.scenario, active_tasks=8, pending_tasks=0,
                 current_dedicated=1, current_low_priority=0)
    assert d.dedicated == 3


def test_workday_window():
    s = scen("workday", maximum_gpu_count={"dedicated": 8})
    monday_noon = datetime.datetime(2026, 9, 7, 12, 0)
    sunday = datetime.datetime(2026, 9, 6, 12, 0)
    on = evaluate(s.scenario, 0, 0, 0, 0, now=monday_noon)
    off = evaluate(s.scenario, 0, 0, 0, 0, now=sunday)
    assert on.dedicated == 8 and off.dedicated == 0


def test_offpeak_low_priority():
    s = scen("workday_with_offpeak_max_low_priority",
             maximum_gpu_count={"dedicated": 4, "low_priority": 8})
    monday_noon = datetime.datetime(2026, 9, 7, 12, 0)
    night = datetime.datetime(2026, 9, 7, 23, 0)
    peak = evaluate(s.scenario, 0, 0, 0, 0, now=monday_noon)
    off = evaluate(s.scenario, 0, 0, 0, 0, now=night)
    assert (peak.dedicated, peak.low_priority) == (4, 0)
    assert (off.dedicated, off.low_priority) == (0, 8)


def test_controller_resizes_pool(tmp_path):
    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    ex.pool_add({"pool_specification": {
        "id": "pa", "gpus": {"dedicated": 0}, "cpu_slots": 1,
        "node_configuration": {"rocm": {"verify": False}},
        "autoscale": {"evaluation_interval": "00:00:00",
                      "scenario": {"name": "active_tasks",
                                   "maximum_gpu_count": {"dedicated": 4}}},
    }})
    ex.jobs_add({"job_specifications": [{
        "id": "jb", "tasks": [{"id": "t", "command": "true", "gpus": 1}]}]},
        "pa")
    ex.schedule_once()
    ps = ex._pool_settings("pa")
    ctl = AutoscaleController(ex, "pa", ps.autoscale)
    dec = ctl.maybe_evaluate(now_ts=1e12)
    assert dec is not None and dec.dedicated >= 1
    row = ex.store.query_one("SELECT gpus_dedicated FROM pools WHERE id=?",
                             ("pa",))
    assert row["gpus_dedicated"] == dec.dedicated
    ex.store.close()
