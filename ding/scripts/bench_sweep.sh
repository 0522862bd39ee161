#!/usr/bin/env bash
# Weak-scaling bench sweep on one node (the round driver runs the same shape).
for N in 1 2 4 8; do
  python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
    --master-addr 127.0.0.1 --master-port 29500 \
    bench.py --gpus "$N" --steps 8 --warmup 2 | tail -1
done
