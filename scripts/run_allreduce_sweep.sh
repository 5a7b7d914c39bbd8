#!/bin/bash
# run_allreduce_sweep.sh — RCCL all-reduce bandwidth sweep at 1/2/4/8 GPUs
# (BASELINE.json config[4]): 4 MB..4 GB per algorithm, CSV to stdout.
set -u
cd "$(dirname "$0")/.."
GPUS=${1:-$(rocm-smi --showid 2>/dev/null | grep -c '^GPU' || echo 1)}
shift || true

exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$GPUS" \
    --master-addr 127.0.0.1 --master-port 29617 \
    -m hpc_patterns_amd.parallel.sweep "$@"
