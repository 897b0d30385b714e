#!/bin/bash
# Single-node 8x MI355X training launch (one rank per GPU over RCCL).
# Mirrors the reference's Frontier job scripts (run-scripts/ there) with
# MI355X-native settings: RCCL over the 7-link xGMI clique.
set -euo pipefail

export HSA_ENABLE_IPC_MODE_LEGACY=0      # dmabuf IPC on this driver
export MASTER_ADDR=127.0.0.1
export NCCL_PROTO=Simple                 # small-message latency
# export NCCL_MIN_NCHANNELS=32           # more rings for large buckets
# export HYDRAGNN_USE_FSDP=1 HYDRAGNN_FSDP_VERSION=2
# export HYDRAGNN_CUSTOM_DATALOADER=1 HYDRAGNN_NUM_WORKERS=2
# export HYDRAGNN_AFFINITY=1 HYDRAGNN_AFFINITY_WIDTH=12

SCRIPT=${1:-bench.py}
shift || true
python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
    --master-addr 127.0.0.1 "${SCRIPT}" --gpus 8 "$@"
