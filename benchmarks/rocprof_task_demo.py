"""Hardware demo of per-task rocprofv3 tracing (jobs.yaml `rocprof`
field): submit the tiny-gpt2 training-step task with tracing enabled
through the LocalExecutor and report the top kernels from the emitted
stats.  Evidence for profiles/ (analogue: the reference's cascade perf
events, but for the compute plane)."""
import csv
import glob
import os
import sys
import tempfile
from pathlib import Path

from shipyard_amd.executor import LocalExecutor

os.environ.setdefault("TMPDIR", "/tmp")
root = Path(tempfile.mkdtemp(prefix="rocprof_demo_"))
ex = LocalExecutor(root, detect_gpus=True)
ex.pool_add({"pool_specification": {
    "id": "p", "gpus": {"dedicated": 1},
    "node_configuration": {"rocm": {"verify": False}}}})
repo = Path(__file__).parents[1]
ex.jobs_add({"job_specifications": [{
    "id": "prof-demo",
    "tasks": [{
        "id": "gpt2",
        "gpus": 1,
        "rocprof": {"enabled": True},
        "command": f"python3 {repo}/recipes/transformers-tiny-gpt2/"
                   "train_step.py",
    }],
}]}, "p")
ex.run_until_idle(timeout=600)
t = ex.tasks_list("prof-demo")[0]
print("task state:", t["state"], "exit:", t["exit_code"])
stdout = ex.task_file("p", "prof-demo", "gpt2", "stdout.txt").read_text()
print("task stdout:", stdout.strip()[-200:])
prof_dir = root / "pools" / "p" / "jobs" / "prof-demo" / "tasks" / \
    "gpt2" / "prof"
stats = sorted(glob.glob(str(prof_dir / "**" / "*kernel_stats.csv"),
                         recursive=True))
if not stats:
    stats = sorted(glob.glob(str(prof_dir / "**" / "*.csv"),
                             recursive=True))
print("stats files:", [Path(s).name for s in stats])
if not stats:
    print("prof dir listing:")
    for f in sorted(prof_dir.rglob("*")):
        print("  ", f.relative_to(prof_dir))
    err = ex.task_file("p", "prof-demo", "gpt2", "stderr.txt")
    print("task stderr tail:", err.read_text()[-500:])
if stats:
    with open(stats[-1]) as f:
        rows = list(csv.DictReader(f))
    key = ("TotalDurationNs" if rows and "TotalDurationNs" in rows[0]
           else ("DurationNs" if rows and "DurationNs" in rows[0]
                 else None))
    if key:
        rows.sort(key=lambda r: -float(r[key]))
    print("top kernels by total duration:")
    for r in rows[:10]:
        name = (r.get("Name") or r.get("Kernel_Name") or "?")[:80]
        print(f"  {name}  {r.get(key, '?')}")
    out = repo / "gpurun_out"
    out.mkdir(exist_ok=True)
    import shutil
    shutil.copy(stats[-1], out / "task_rocprof_kernel_stats.csv")
sys.exit(0 if t["state"] == "completed" and stats else 1)
