"""Multi-process distributed gang test: world_size 2 over gloo on CPU.

Validates the exact path a multi-instance (RCCL) task takes on the GPU
node: the executor gang-launches N rank processes with RANK/WORLD_SIZE/
MASTER_* env, each builds a GangComm and all-reduces.  On the MI355X box
the same path runs with backend nccl (=RCCL) — covered by the gpu-marked
test below.
"""
import sys
import textwrap

import pytest

from shipyard_amd.executor import LocalExecutor

GANG_SCRIPT = textwrap.dedent("""
    import os, torch
    from shipyard_amd.comm import GangComm
    comm = GangComm()
    assert comm.world == 2, comm.world
    t = torch.ones(1000) * (comm.rank + 1)
    comm.all_reduce_(t)
    assert torch.allclose(t, torch.full((1000,), 3.0)), t[0]
    print(f"rank {comm.rank} allreduce ok sum={t[0].item()}")
    comm.barrier()
    comm.shutdown()
""").strip()


def _run_gang(tmp_path, backend: str, gpus_per_rank: int, world: int,
              timeout: float = 120.0):
    ex = LocalExecutor(tmp_path / "root")
    ex.pool_add({"pool_specification": {
        "id": "gp",
        "gpus": {"dedicated": world if gpus_per_rank else 0},
        "cpu_slots": 0 if gpus_per_rank else world,
        "node_configuration": {"rocm": {"verify": bool(gpus_per_rank)}},
        "inter_node_communication_enabled": True,
    }})
    script = tmp_path / "gang.py"
    script.write_text(GANG_SCRIPT)
    ex.jobs_add({"job_specifications": [{
        "id": "gj",
        "tasks": [{
            "id": "gang",
            "command": f"{sys.executable} {script}",
            "max_task_retries": 0,
            "multi_instance": {
                "num_instances": world,
                "gang": {"backend": backend,
                         "gpus_per_rank": gpus_per_rank},
            },
        }],
    }]}, "gp")
    ex.run_until_idle(timeout=timeout)
    t = ex.tasks_list("gj")[0]
    base = ex.pool_root("gp") / "jobs" / "gj" / "tasks" / "gang"
    outs = [
        (base / f"rank{r:03d}" / "stdout.txt").read_text()
        for r in range(world)
    ]
    errs = [
        (base / f"rank{r:03d}" / "stderr.txt").read_text()
        for r in range(world)
    ]
    ex.store.close()
    return t, outs, errs


def test_gloo_gang_allreduce(tmp_path):
    t, outs, errs = _run_gang(tmp_path, backend="gloo", gpus_per_rank=0,
                              world=2)
    assert t["state"] == "completed", errs
    for r, out in enumerate(outs):
        assert f"rank {r} allreduce ok sum=3.0" in out


@pytest.mark.gpu
def test_rccl_gang_allreduce_1gpu(tmp_path):
    """Degenerate 1-rank RCCL gang on the single-GPU box: validates the
    nccl backend init path end-to-end through the executor."""
    import torch

    assert torch.cuda.is_available()
    script = textwrap.dedent("""
        import torch
        from shipyard_amd.comm import GangComm
        comm = GangComm()
        assert comm.device.type == "cuda"
        x = torch.ones(1 << 20, device=comm.device, dtype=torch.bfloat16)
        comm.all_reduce_(x)   # no-op at world 1
        y = torch.zeros_like(x)
        y.add_(x)
        torch.cuda.synchronize()
        print("rccl-1gpu ok", float(y.sum().item()))
        comm.shutdown()
    """).strip()
    ex = LocalExecutor(tmp_path / "root")
    ex.pool_add({"pool_specification": {
        "id": "gp1", "gpus": {"dedicated": 1},
        "inter_node_communication_enabled": True,
    }})
    sp = tmp_path / "g1.py"
    sp.write_text(script)
    import sys as _sys

    ex.jobs_add({"job_specifications": [{
        "id": "gj1",
        "tasks": [{"id": "t", "command": f"{_sys.executable} {sp}",
                   "gpus": 1, "max_task_retries": 0}],
    }]}, "gp1")
    ex.run_until_idle(timeout=300)
    t = ex.tasks_list("gj1")[0]
    err = (ex.pool_root("gp1") / "jobs" / "gj1" / "tasks" / "t" /
           "stderr.txt").read_text()
    assert t["state"] == "completed", err
    out = (ex.pool_root("gp1") / "jobs" / "gj1" / "tasks" / "t" /
           "stdout.txt").read_text()
    assert "rccl-1gpu ok" in out
    ex.store.close()


@pytest.mark.gpu
def test_gpu_slot_cycling(tmp_path):
    """8 sequential GPU tasks through 1 slot: allocation/release cycling
    on real hardware."""
    import sys as _sys

    import torch

    assert torch.cuda.is_available()
    ex = LocalExecutor(tmp_path / "root")
    ex.pool_add({"pool_specification": {
        "id": "cyc", "gpus": {"dedicated": 1}}})
    script = tmp_path / "g.py"
    script.write_text(
        "import os, torch\n"
        "assert os.environ['HIP_VISIBLE_DEVICES'] == '0'\n"
        "x = torch.ones(1024, device='cuda')\n"
        "assert float(x.sum().item()) == 1024.0\n"
        "print('cycle ok')\n")
    ex.jobs_add({"job_specifications": [{
        "id": "cj",
        "tasks": [{"id": f"t{i}", "command": f"{_sys.executable} {script}",
                   "gpus": 1, "max_task_retries": 0} for i in range(8)],
    }]}, "cyc")
    ex.run_until_idle(timeout=600)
    tasks = ex.tasks_list("cj")
    assert all(t["state"] == "completed" for t in tasks), tasks
    ex.store.close()
