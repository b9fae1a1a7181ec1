"""Runtime option synthesis tests (the docker/singularity/process
command compilers + ROCm binder; reference settings.py:3727-4305
semantics, no docker daemon needed)."""
from shipyard_amd.runner import runtime as rt


def test_process_command_pipefail():
    cmd = rt.process_run_command("echo a | grep a")
    assert cmd[0] == "/bin/bash" and "pipefail" in cmd[2]


def test_docker_rocm_binder():
    cmd = rt.docker_run_command(
        image="rocm/pytorch", command="python train.py", name="t1",
        device_ids=[2, 5], shm_size=256 * 10 ** 6,
        volumes=["/data:/data:ro"], working_dir="/work")
    s = " ".join(cmd)
    assert "--device=/dev/kfd" in s
    assert "--device=/dev/dri/renderD130" in s  # device 2
    assert "--device=/dev/dri/renderD133" in s  # device 5
    assert "--group-add video" in s and "--group-add render" in s
    # inside the container the granted GPUs are renumbered 0..k-1
    assert "HIP_VISIBLE_DEVICES=0,1" in s
    assert "--shm-size=256000000" in s
    assert "-v /data:/data:ro" in s
    assert cmd[-3:] == ["rocm/pytorch", "python", "train.py"]


def test_docker_no_gpu_no_binder():
    cmd = rt.docker_run_command(image="busybox", command="true",
                                name="x", device_ids=[])
    s = " ".join(cmd)
    assert "/dev/kfd" not in s and "HIP_VISIBLE_DEVICES" not in s


def test_singularity_rocm_flag():
    cmd = rt.singularity_run_command("img.sif", "run.sh", device_ids=[0])
    assert "--rocm" in cmd
    cmd = rt.singularity_run_command("img.sif", "run.sh", device_ids=[])
    assert "--rocm" not in cmd


def test_gpu_env_hides_gpus_for_cpu_tasks():
    env = rt.gpu_env([])
    assert env["HIP_VISIBLE_DEVICES"] == ""
    env = rt.gpu_env([3, 4])
    assert env["HIP_VISIBLE_DEVICES"] == "3,4"
    assert env["ROCR_VISIBLE_DEVICES"] == "3,4"
    assert env["HSA_ENABLE_IPC_MODE_LEGACY"] == "0"


def test_singularity_exec_and_options_passthrough():
    cmd = rt.singularity_run_command(
        "img.sif", "python run.py", device_ids=[0], exec_cmd="run",
        extra_options=["--contain"])
    assert cmd[:2] == ["singularity", "run"]
    assert "--contain" in cmd and "--rocm" in cmd
    assert cmd[-3:] == ["img.sif", "python", "run.py"]
