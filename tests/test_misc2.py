"""Tests: auto-scratch, crypto helpers, dashboard json validity."""
import json
import shutil
from pathlib import Path

import pytest

from shipyard_amd.executor import LocalExecutor
from shipyard_amd.utils import crypto


def test_auto_scratch_shared_and_cleaned(tmp_path):
    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    ex.pool_add({"pool_specification": {
        "id": "ps", "gpus": {"dedicated": 0}, "cpu_slots": 2,
        "per_job_auto_scratch": True,
        "node_configuration": {"rocm": {"verify": False}}}})
    ex.jobs_add({"job_specifications": [{
        "id": "jscr",
        "tasks": [
            {"id": "w", "command":
             "echo data > $SHIPYARD_AUTO_SCRATCH_DIR/f"},
            {"id": "r", "command":
             "grep data $SHIPYARD_AUTO_SCRATCH_DIR/f",
             "depends_on": ["w"]},
        ]}]}, "ps")
    ex.run_until_idle(timeout=30)
    states = {t["id"]: t["state"] for t in ex.tasks_list("jscr")}
    assert states == {"w": "completed", "r": "completed"}
    scratch = ex.pool_root("ps") / "scratch" / "jscr"
    assert scratch.exists()
    ex.job_del("jscr")
    assert not scratch.exists()
    ex.store.close()


@pytest.mark.skipif(shutil.which("ssh-keygen") is None,
                    reason="no ssh-keygen")
def test_ssh_keypair(tmp_path):
    priv, pub = crypto.generate_ssh_keypair(tmp_path)
    assert priv.exists() and pub.exists()
    assert (priv.stat().st_mode & 0o777) == 0o600
    cmd = crypto.ssh_command("node1", "hostname", username="u",
                             private_key=str(priv))
    assert cmd[-2] == "u@node1" and cmd[-1] == "hostname"


def test_dashboard_json_valid():
    p = Path(__file__).parents[1] / "shipyard_amd" / "monitor" / \
        "dashboard.json"
    doc = json.loads(p.read_text())
    assert doc["title"] and len(doc["panels"]) >= 6
    exprs = [t["expr"] for panel in doc["panels"]
             for t in panel.get("targets", [])]
    assert any("shipyard_gpu_metric" in e for e in exprs)
    assert any("shipyard_executor_metric" in e for e in exprs)
