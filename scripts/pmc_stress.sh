#!/usr/bin/env bash
# PMC counter collection for the stress kernels (own run: --pmc must not be
# combined with sys/runtime trace domains).
set -euo pipefail
REPO="$(cd "$(dirname "$0")/.." && pwd)"
export TMPDIR=/tmp
mkdir -p "$REPO/gpurun_out/pmcstress"
cd /tmp
rocprofv3 --pmc SQ_INSTS_MFMA SQ_INSTS_VALU SQ_INSTS_LDS SQ_LDS_BANK_CONFLICT \
  --output-format csv -d "$REPO/gpurun_out/pmcstress" -o pmcstress -- \
  bash -c "cd '$REPO' && python bench.py --config stress --steps 2 --warmup 1 --selfcheck 0"
