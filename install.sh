#!/usr/bin/env bash
# Install shipyard-amd into the current python env and build the HIP
# ops for gfx950 (the reference's install.sh analogue; no network use).
set -euo pipefail
cd "$(dirname "$0")"
python3 -m pip install -e . --no-build-isolation --no-index 2>/dev/null || \
  echo "pip install skipped (offline); using in-tree package via PYTHONPATH"
python3 -m shipyard_amd.ops.build
python3 -m shipyard_amd.comm.build_native || \
  echo "rccl bench binary build failed (non-fatal)"
echo "shipyard-amd ready: try 'python3 -m shipyard_amd.cli --help'"
