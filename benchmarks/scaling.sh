#!/usr/bin/env bash
# DDP scaling curve on one node: images/sec at 1/2/4/8 GPUs
# (the round driver runs the same contract).
set -e
STEPS=${STEPS:-30}; WARMUP=${WARMUP:-10}
EXTRA=${EXTRA:-}
for N in 1 2 4 8; do
  if [ "$N" = 1 ]; then
    python bench.py --gpus 1 --steps "$STEPS" --warmup "$WARMUP" $EXTRA
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
      --master-addr 127.0.0.1 --master-port 29531 \
      bench.py --gpus "$N" --steps "$STEPS" --warmup "$WARMUP" $EXTRA
  fi
done
