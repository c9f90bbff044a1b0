#!/usr/bin/env bash
# Scaling sweep: bench.py at N=1,2,4,8 ranks (one per GPU), the exact
# launch form the driver uses for SCALE_rNN.json.  Run on an 8-GPU node:
#   bash tools/scale_sweep.sh [steps] [warmup]
# Prints one JSON line per N; scaling efficiency is computed by the
# consumer from the per-N values (weak scaling: per-GPU batch fixed).
set -euo pipefail
cd "$(dirname "$0")/.."
STEPS="${1:-200}"
WARMUP="${2:-20}"
PORT="${MASTER_PORT:-29561}"
export HSA_ENABLE_IPC_MODE_LEGACY="${HSA_ENABLE_IPC_MODE_LEGACY:-0}"
NGPU=$(python -c 'import torch; print(torch.cuda.device_count())')
for N in 1 2 4 8; do
  if [ "$N" -gt "$NGPU" ]; then
    echo "# skipping N=$N (only $NGPU GPUs visible)" >&2
    continue
  fi
  if [ "$N" -eq 1 ]; then
    python bench.py --gpus 1 --steps "$STEPS" --warmup "$WARMUP"
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
      --master-addr 127.0.0.1 --master-port "$PORT" \
      bench.py --gpus "$N" --steps "$STEPS" --warmup "$WARMUP"
  fi
done
