"""Task runner: runtime synthesis (process/docker/singularity with the
ROCm binder) and gang launching with wedged-rank teardown."""

from .task_runner import LaunchSpec, TaskHandle, launch  # noqa: F401
