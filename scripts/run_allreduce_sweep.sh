#!/bin/bash
# run_allreduce_sweep.sh — RCCL all-reduce bandwidth sweep at 1/2/4/8 GPUs
# (BASELINE.json config[4]): 4 MB..4 GB per algorithm, CSV to stdout.
#
#   run_allreduce_sweep.sh [GPUS] [--channels "1 2 4 7 14"] [sweep args...]
#
# --channels runs the whole sweep once per NCCL_MIN/MAX_NCHANNELS setting
# (RCCL reads the env at communicator creation, so each setting needs its
# own launch) — the channels-vs-busBW table for the xGMI link study
# (one ring is single-link-bound at ~153 GB/s; channel
# spreading across the 7 links is where the headroom is).
set -u
cd "$(dirname "$0")/.."
GPUS=${1:-$(rocm-smi --showid 2>/dev/null | grep -c '^GPU' || echo 1)}
shift || true

CHANNELS=""
if [ "${1:-}" = "--channels" ]; then
  shift
  CHANNELS="$1"
  shift
fi

run_one() {
  python -m torch.distributed.run --nnodes=1 --nproc-per-node "$GPUS" \
      --master-addr 127.0.0.1 --master-port 29617 \
      -m hpc_patterns_amd.parallel.sweep "$@"
}

if [ -z "$CHANNELS" ]; then
  run_one "$@"
else
  for c in $CHANNELS; do
    echo "export NCCL_MIN_NCHANNELS=$c NCCL_MAX_NCHANNELS=$c"
    NCCL_MIN_NCHANNELS=$c NCCL_MAX_NCHANNELS=$c run_one "$@"
  done
fi
