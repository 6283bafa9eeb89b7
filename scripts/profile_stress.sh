#!/usr/bin/env bash
# rocprofv3 kernel profile of the stress bench (run via gpurun).
set -euo pipefail
REPO="$(cd "$(dirname "$0")/.." && pwd)"
export TMPDIR=/tmp
mkdir -p "$REPO/gpurun_out/profstress2"
cd /tmp
rocprofv3 --kernel-trace --stats -d "$REPO/gpurun_out/profstress2" -o stressprof -- \
  bash -c "cd '$REPO' && python bench.py --config stress --steps 5 --warmup 1 --selfcheck 0"
