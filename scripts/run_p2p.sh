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
