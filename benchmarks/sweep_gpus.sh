#!/usr/bin/env bash
# 1/2/4/8-GPU scaling sweep of the flagship bench (the driver runs the same
# command shape at round end; reference analogue: the k8s worker-count sweep).
set -euo pipefail
STEPS=${1:-20}; WARMUP=${2:-5}
cd "$(dirname "$0")/.."
for n in 1 2 4 8; do
  echo "=== gpus=$n ==="
  if [ "$n" -eq 1 ]; then
    python bench.py --gpus 1 --steps "$STEPS" --warmup "$WARMUP"
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$n" \
      --master-addr 127.0.0.1 --master-port 29517 \
      bench.py --gpus "$n" --steps "$STEPS" --warmup "$WARMUP"
  fi
done
