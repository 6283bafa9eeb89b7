#!/usr/bin/env bash
# Worker-count sweep (reference benchmarks/k8s_benchmark_pool.sh:1-13 analogue:
# there a fresh ray cluster per point; here a fresh process pool per point).
# Usage: sweep_pool.sh START END [NRUNS]
set -euo pipefail
START=${1:-1}; END=${2:-8}; NRUNS=${3:-5}
cd "$(dirname "$0")/.."
for (( w=START; w<=END; w++ )); do
  echo "=== workers=$w ==="
  python benchmarks/pool.py --workers "$w" --batch 1 5 10 --nruns "$NRUNS"
done
