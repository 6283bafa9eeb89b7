#!/usr/bin/env bash
set -euo pipefail
REPO="$(cd "$(dirname "$0")/.." && pwd)"
export TMPDIR=/tmp
mkdir -p "$REPO/gpurun_out/profmlp"
cd /tmp
rocprofv3 --kernel-trace --stats -d "$REPO/gpurun_out/profmlp" -o mlpprof -- \
  bash -c "cd '$REPO' && python bench.py --config mlp --steps 3 --warmup 1 --dtype bf16x2 --selfcheck 0"
