#!/bin/bash
# run_p2p.sh — P2P bandwidth sweep (the reference p2p/run.sh matrix on
# MI355X): engines x payload sizes, plus the topology dump that drives
# placement. The reference swept {compact,spread,compact_plan} x {ZAM,ODS} x
# {isend,win} x {2,12 ranks}; here the engines are the native xGMI paths.
set -u
cd "$(dirname "$0")/.."

LOG=${1:-p2p.log}
rm -f "$LOG"

./bin/hpk_topology 2>&1 | tee -a "$LOG"

for engine in peer ipc rccl; do
  for floats in 47185920 4718592; do
    echo "export HPK_P2P_ENGINE=$engine FLOATS=$floats" | tee -a "$LOG"
    ./bin/hpk_p2p --engine "$engine" --floats "$floats" 2>&1 | tee -a "$LOG"
  done
done

# oversubscribed all-pairs RMA (runs the full fence-epoch protocol even on
# a 1-GPU box)
echo "export HPK_P2P_ENGINE=ipc RANKS=4" | tee -a "$LOG"
./bin/hpk_p2p --engine ipc --floats 4718592 --ranks 4 2>&1 | tee -a "$LOG"

# placement-policy x measured-bandwidth matrix (reference run.sh:9-21
# {compact,spread,compact_plan} sweep): one torchrun per policy — device
# binding happens at process-group init. Needs >=2 GPUs for RCCL.
NDEV=$(./bin/hpk_topology 2>/dev/null | sed -n 's/^# \([0-9]*\) HIP.*/\1/p')
NDEV=${NDEV:-0}
CSV=policy_sweep.csv
rm -f "$CSV"
if [ "$NDEV" -ge 2 ]; then
  for policy in compact spread topo; do
    echo "export HPK_PLACEMENT_POLICY=$policy" | tee -a "$LOG"
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NDEV" \
        --master-addr 127.0.0.1 --master-port 29631 \
        -m hpc_patterns_amd.parallel.policy_sweep \
        --policy "$policy" --csv "$CSV" 2>&1 | tee -a "$LOG"
  done
  echo "# policy sweep CSV:" | tee -a "$LOG"
  cat "$CSV" | tee -a "$LOG"
else
  echo "# policy sweep skipped: $NDEV GPU(s) visible (needs >=2)" | tee -a "$LOG"
fi
