#!/usr/bin/env bash
# Standalone 1/2/4/8-GPU weak-scaling curve for the flagship benchmark
# (the driver runs the same shape itself; this is the hand-run variant).
#
#   bash bench/scaling_curve.sh [steps] [warmup]
set -euo pipefail
STEPS="${1:-300}"
WARMUP="${2:-30}"
cd "$(dirname "$0")/.."
for N in 1 2 4 8; do
  if [ "$N" -gt "$(python -c 'import torch;print(torch.cuda.device_count())')" ]; then
    echo "skipping N=$N (not enough GPUs)"
    continue
  fi
  echo "== N=$N"
  python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
    --master-addr 127.0.0.1 --master-port 29600 \
    bench.py --gpus "$N" --steps "$STEPS" --warmup "$WARMUP"
done
