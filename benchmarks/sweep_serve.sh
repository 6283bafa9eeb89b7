#!/usr/bin/env bash
# Serve sweep (reference benchmarks/k8s_benchmark_serve.sh:1-19 analogue).
# Usage: sweep_serve.sh START END [NRUNS]
set -euo pipefail
START=${1:-1}; END=${2:-4}; NRUNS=${3:-3}
BATCH_SIZE=(1 5 10)
cd "$(dirname "$0")/.."
for (( r=START; r<=END; r++ )); do
  for b in "${BATCH_SIZE[@]}"; do
    echo "=== replicas=$r max_batch=$b ==="
    python benchmarks/serve_explanations.py --replicas "$r" --max-batch-size "$b" --nruns "$NRUNS"
  done
done
