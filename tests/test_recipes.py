"""Recipes: every shipped config validates; the CPU-runnable ones
execute end-to-end through the executor."""
from pathlib import Path

import pytest
import yaml

from shipyard_amd.config import ConfigType, validate_config
from shipyard_amd.executor import LocalExecutor

RECIPES = Path(__file__).parents[1] / "recipes"
FAMS = {"pool.yaml": ConfigType.pool, "jobs.yaml": ConfigType.jobs,
        "config.yaml": ConfigType.config,
        "credentials.yaml": ConfigType.credentials,
        "federation.yaml": ConfigType.federation}


def _recipe_files():
    return sorted(p for p in RECIPES.rglob("*.yaml") if p.name in FAMS)


@pytest.mark.parametrize("path", _recipe_files(),
                         ids=lambda p: f"{p.parent.name}/{p.name}")
def test_recipe_validates(path):
    validate_config(FAMS[path.name], yaml.safe_load(path.read_text()),
                    str(path))


def _cpu_pool(ex, pid="rp", slots=2):
    ex.pool_add({"pool_specification": {
        "id": pid, "gpus": {"dedicated": 0}, "cpu_slots": slots,
        "node_configuration": {"rocm": {"verify": False}}}})


def test_merge_reduce_recipe_runs(tmp_path):
    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    _cpu_pool(ex)
    jobs = yaml.safe_load((RECIPES / "merge-reduce" / "jobs.yaml")
                          .read_text())
    ex.jobs_add(jobs, "rp")
    ex.run_until_idle(timeout=60)
    tasks = {t["id"]: t["state"] for t in ex.tasks_list("map-reduce")}
    assert tasks == {"map-0": "completed", "map-1": "completed",
                     "reduce": "completed"}
    out = ex.task_file("rp", "map-reduce", "reduce").read_text()
    assert out.strip() == "3"
    ex.store.close()


def test_factory_random_recipe_runs(tmp_path):
    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    _cpu_pool(ex)
    jobs = yaml.safe_load((RECIPES / "factory-random" / "jobs.yaml")
                          .read_text())
    ex.jobs_add(jobs, "rp")
    ex.run_until_idle(timeout=60)
    tasks = ex.tasks_list("random-sweep")
    assert len(tasks) == 8
    assert all(t["state"] == "completed" for t in tasks)
    ex.store.close()


def test_factory_file_recipe_runs(tmp_path):
    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    _cpu_pool(ex)
    for i in range(3):
        ex.stores["default"].upload_bytes(f"datasets/in/f{i}.bin", b"x")
    ex.stores["default"].upload_bytes("datasets/in/skip.txt", b"x")
    jobs = yaml.safe_load((RECIPES / "factory-file" / "jobs.yaml")
                          .read_text())
    # the executor expands the file factory against the named store
    ex.jobs_add(jobs, "rp")
    ex.run_until_idle(timeout=60)
    tasks = ex.tasks_list("per-file")
    assert len(tasks) == 3
    assert all(t["state"] == "completed" for t in tasks)
    ex.store.close()
