"""CLI entry for slurmctld Resume/Suspend programs:
``python -m shipyard_amd.slurm_elastic resume|suspend|resume_failed
<hostlist>`` with SHIPYARD_ROOT + SHIPYARD_SLURM_CONF env."""
from __future__ import annotations

import os
import sys

import yaml

from shipyard_amd.executor import LocalExecutor
from shipyard_amd.slurm_elastic import SlurmAdapter


def main() -> int:
    if len(sys.argv) < 3:
        print("usage: ... resume|suspend|resume_failed <hostlist>",
              file=sys.stderr)
        return 2
    action, hostlist = sys.argv[1], sys.argv[2]
    root = os.environ.get("SHIPYARD_ROOT",
                          os.path.expanduser("~/.shipyard_amd"))
    conf_path = os.environ.get("SHIPYARD_SLURM_CONF")
    if not conf_path:
        print("SHIPYARD_SLURM_CONF not set", file=sys.stderr)
        return 2
    with open(conf_path) as f:
        conf = yaml.safe_load(f)
    ex = LocalExecutor(root)
    ad = SlurmAdapter(ex, conf)
    fn = {"resume": ad.resume, "suspend": ad.suspend,
          "resume_failed": ad.resume_failed}[action]
    hosts = fn(hostlist)
    print("\n".join(hosts))
    return 0


if __name__ == "__main__":
    sys.exit(main())
