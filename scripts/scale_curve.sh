#!/bin/bash
# Reproduce the rank-sharded scaling curve (profiles/PROFILE.md):
#   bash scripts/scale_curve.sh [steps] [warmup]
# One rank per GPU when GPUs are present; on a single-GPU box all ranks
# sample device 0 (CPU-bound preview of the multi-GPU run).
STEPS=${1:-12}
WARMUP=${2:-3}
for n in 1 2 4 8; do
  python -m torch.distributed.run --nnodes=1 --nproc-per-node "$n" \
      --master-addr 127.0.0.1 --master-port $((29620 + n)) \
      bench.py --gpus "$n" --steps "$STEPS" --warmup "$WARMUP" 2>/dev/null
done
