"""Collective wrappers: world-2 gloo via the executor gang path."""
import sys
import textwrap

from shipyard_amd.executor import LocalExecutor

SCRIPT = textwrap.dedent("""
    import torch
    from shipyard_amd.comm import GangComm
    comm = GangComm()
    r, w = comm.rank, comm.world
    assert w == 2

    # bucketed all-reduce over mixed sizes
    ts = [torch.full((n,), float(r + 1)) for n in (10, 5000, 3)]
    comm.all_reduce_bucketed_(ts, bucket_bytes=4096)
    for t in ts:
        assert torch.allclose(t, torch.full_like(t, 3.0)), t

    # all_gather
    g = comm.all_gather(torch.full((4,), float(r)))
    assert [int(x[0].item()) for x in g] == [0, 1]

    # reduce_scatter
    out = torch.zeros(8)
    shards = [torch.full((8,), float(r + 1)), torch.full((8,), float(r + 1))]
    comm.reduce_scatter_(out, shards)
    assert torch.allclose(out, torch.full((8,), 3.0))

    # all_to_all
    shards = [torch.full((2,), float(r * 10 + i)) for i in range(2)]
    got = comm.all_to_all(shards)
    # rank r receives shard r from every rank: [0*10+r, 1*10+r]
    assert [int(x[0].item()) for x in got] == [r, 10 + r]

    # broadcast
    b = torch.full((3,), float(r))
    comm.broadcast_(b, src=1)
    assert int(b[0].item()) == 1

    print(f"rank {r} collectives ok")
    comm.barrier(); comm.shutdown()
""").strip()


def test_gloo_collectives_world2(tmp_path):
    ex = LocalExecutor(tmp_path / "r", detect_gpus=False)
    ex.pool_add({"pool_specification": {
        "id": "cp", "gpus": {"dedicated": 0}, "cpu_slots": 2,
        "node_configuration": {"rocm": {"verify": False}},
        "inter_node_communication_enabled": True}})
    script = tmp_path / "coll.py"
    script.write_text(SCRIPT)
    ex.jobs_add({"job_specifications": [{
        "id": "cj",
        "tasks": [{"id": "t", "command": f"{sys.executable} {script}",
                   "max_task_retries": 0,
                   "multi_instance": {"num_instances": 2,
                                      "gang": {"backend": "gloo",
                                               "gpus_per_rank": 0}}}],
    }]}, "cp")
    ex.run_until_idle(timeout=120)
    t = ex.tasks_list("cj")[0]
    base = ex.pool_root("cp") / "jobs" / "cj" / "tasks" / "t"
    errs = [(base / f"rank{r:03d}" / "stderr.txt").read_text()
            for r in range(2)]
    assert t["state"] == "completed", errs
    for r in range(2):
        out = (base / f"rank{r:03d}" / "stdout.txt").read_text()
        assert f"rank {r} collectives ok" in out
    ex.store.close()
