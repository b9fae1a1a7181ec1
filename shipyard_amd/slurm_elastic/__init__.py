"""Slurm elastic-cloud adapter.

Re-implementation of the reference's elastic Slurm integration
(reference slurm/slurm.py:969 `process_resume_action`, 1044
`process_suspend_action`, convoy/slurm.py:63
`_apply_slurm_config_to_batch_pools`, slurm/slurm.conf:101-103
ResumeProgram/SuspendProgram contract) for the local executor:

  * elastic partitions map to pools; a "Slurm node" is a GPU slot;
  * resume(hostlist) grows the backing pool and records host->slot
    assignments; suspend(hostlist) shrinks it;
  * `generate_slurm_conf` emits the partition/node fragment plus
    resume.sh / suspend.sh that call `python -m
    shipyard_amd.slurm_elastic resume|suspend <hosts>` — the same
    contract slurmctld uses against the reference's controller.
"""

from .adapter import (SlurmAdapter, expand_hostlist,  # noqa: F401
                      generate_slurm_conf)
