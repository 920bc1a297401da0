#!/bin/bash
# 8-GPU scaling bench on one MI355X node — the same launch shape the driver
# uses for SCALE_rNN.json. Run on an 8-GPU box:
#   bash scripts/bench8.sh [N] [STEPS] [WARMUP]
# Produces one JSON line per N on stdout (rank 0).
set -e
N=${1:-8}
STEPS=${2:-10}
WARMUP=${3:-3}
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}
if [ "$N" = 1 ]; then
  exec python bench.py --gpus 1 --steps "$STEPS" --warmup "$WARMUP"
fi
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
    --master-addr 127.0.0.1 --master-port 29371 \
    bench.py --gpus "$N" --steps "$STEPS" --warmup "$WARMUP"
