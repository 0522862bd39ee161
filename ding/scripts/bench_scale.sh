#!/bin/bash
# One-command 1/2/4/8-GPU weak-scaling sweep of the flagship bench on a
# single MI355X node (one rank per GPU over RCCL/xGMI).
#
#   bash ding/scripts/bench_scale.sh [workload] [steps] [warmup]
#
# Writes SCALE_<workload>.jsonl (one bench JSON line per N) in the CWD.
# The driver computes scaling efficiency itself from the per-N values.
set -u
WORKLOAD=${1:-ppo}
STEPS=${2:-20}
WARMUP=${3:-5}
REPO=$(cd "$(dirname "$0")/../.." && pwd)
OUT=SCALE_${WORKLOAD}.jsonl
: > "$OUT"

# RCCL/xGMI environment for this pool: dmabuf IPC only (the host driver
# rejects legacy IPC handles), bucketed all-reduce sized in grad_bucket.py
export HSA_ENABLE_IPC_MODE_LEGACY=0
export NCCL_DEBUG=${NCCL_DEBUG:-WARN}
export MASTER_ADDR=127.0.0.1

NGPUS=$(python -c 'import torch; print(torch.cuda.device_count())')
for N in 1 2 4 8; do
    if [ "$N" -gt "$NGPUS" ]; then
        echo "[bench_scale] skipping N=$N (only $NGPUS GPUs visible)"
        continue
    fi
    PORT=$((29400 + RANDOM % 200))
    echo "[bench_scale] N=$N ..."
    if [ "$N" -eq 1 ]; then
        python "$REPO/bench.py" --gpus 1 --steps "$STEPS" --warmup "$WARMUP" \
            --workload "$WORKLOAD" | tail -1 >> "$OUT"
    else
        python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
            --master-addr 127.0.0.1 --master-port "$PORT" \
            "$REPO/bench.py" --gpus "$N" --steps "$STEPS" --warmup "$WARMUP" \
            --workload "$WORKLOAD" | tail -1 >> "$OUT"
    fi
done
echo "[bench_scale] wrote $OUT:"
cat "$OUT"
