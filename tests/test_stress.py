"""Scheduler stress: large task collections through few slots."""
import time

from shipyard_amd.executor import LocalExecutor


def test_200_tasks_through_4_slots(tmp_path):
    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    ex.pool_add({"pool_specification": {
        "id": "sp", "gpus": {"dedicated": 0}, "cpu_slots": 4,
        "node_configuration": {"rocm": {"verify": False}}}})
    n = 200
    ex.jobs_add({"job_specifications": [{
        "id": "big",
        "tasks": [{"command": "true"} for _ in range(n)],
    }]}, "sp")
    t0 = time.monotonic()
    ex.run_until_idle(timeout=180)
    dt = time.monotonic() - t0
    tasks = ex.tasks_list("big")
    assert len(tasks) == n
    assert all(t["state"] == "completed" for t in tasks)
    # throughput sanity: > 10 tasks/s through the scheduler
    assert dt < n / 10, f"{dt:.1f}s for {n} tasks"
    st = ex.job_stats("big")
    assert st["tasks"]["completed"] == n
    ex.store.close()


def test_parallel_jobs_fair_progress(tmp_path):
    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    ex.pool_add({"pool_specification": {
        "id": "sp", "gpus": {"dedicated": 0}, "cpu_slots": 2,
        "node_configuration": {"rocm": {"verify": False}}}})
    for j in range(5):
        ex.jobs_add({"job_specifications": [{
            "id": f"par{j}",
            "tasks": [{"command": "true"} for _ in range(10)],
        }]}, "sp")
    ex.run_until_idle(timeout=120)
    for j in range(5):
        assert all(t["state"] == "completed"
                   for t in ex.tasks_list(f"par{j}"))
    ex.store.close()


def test_deep_dependency_chain(tmp_path):
    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    ex.pool_add({"pool_specification": {
        "id": "sp", "gpus": {"dedicated": 0}, "cpu_slots": 2,
        "node_configuration": {"rocm": {"verify": False}}}})
    depth = 40
    tasks = [{"id": "t0", "command": "true"}]
    for i in range(1, depth):
        tasks.append({"id": f"t{i}", "command": "true",
                      "depends_on": [f"t{i-1}"]})
    ex.jobs_add({"job_specifications": [{"id": "chain",
                                         "tasks": tasks}]}, "sp")
    ex.run_until_idle(timeout=180)
    assert all(t["state"] == "completed"
               for t in ex.tasks_list("chain"))
    ex.store.close()
