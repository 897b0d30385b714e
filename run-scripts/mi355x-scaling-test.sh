#!/bin/bash
# Weak/strong scaling sweep on one MI355X node (the reference's
# run-scripts/HydraGNN-scaling-test.sh shape, MI355X-native): runs
# bench.py at N = 1, 2, 4, 8 ranks over RCCL/xGMI and prints one JSON
# line per point.  Weak scaling keeps the per-GPU batch fixed (the
# bench default); strong scaling divides a fixed global batch.
#
#   ./run-scripts/mi355x-scaling-test.sh [weak|strong] [steps] [warmup]
set -euo pipefail

MODE=${1:-weak}
STEPS=${2:-20}
WARMUP=${3:-5}
GLOBAL_BATCH=${HYDRAGNN_BENCH_BATCH:-8192}

export HSA_ENABLE_IPC_MODE_LEGACY=0
export MASTER_ADDR=127.0.0.1
export NCCL_PROTO=Simple

for N in 1 2 4 8; do
    if [ "$MODE" = "strong" ]; then
        export HYDRAGNN_BENCH_BATCH=$((GLOBAL_BATCH / N))
        export HYDRAGNN_BENCH_SCALING=strong
    else
        export HYDRAGNN_BENCH_BATCH=$GLOBAL_BATCH
    fi
    echo "== $MODE scaling, N=$N, per-rank batch $HYDRAGNN_BENCH_BATCH =="
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
        --master-addr 127.0.0.1 --master-port 29517 \
        bench.py --gpus "$N" --steps "$STEPS" --warmup "$WARMUP"
done
