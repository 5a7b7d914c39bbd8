#!/bin/bash
# run_hip.sh — concurrency-bench sweep matrix (the reference run_sycl.sh /
# run_omp.sh merged into one ROCm-native sweep).
#
# Sweeps ROCm tuning knobs x modes x command lists, tees hip.log, parses it
# into a table. Env knobs replace the reference's Level-Zero ones:
#   GPU_MAX_HW_QUEUES      -> hardware-queue concurrency (ref: BATCH_SIZE /
#                             immediate command lists)
#   HSA_ENABLE_SDMA        -> SDMA copy engines on/off (ref: USE_COPY_ENGINE)
#   HIP_VISIBLE_DEVICES    -> device selection (ref: ZE_AFFINITY_MASK)
set -u
cd "$(dirname "$0")/.."

BIN=./bin/hpk_conc
[ -x "$BIN" ] || { echo "build first: make bins"; exit 1; }

LOG=${1:-hip.log}
rm -f "$LOG"

LCOMMANDS=("C C" "C M2D" "C D2M" "M2D D2M" "H2D D2H" "C D2D" "D2D D2D")

for envs in "HIP_VISIBLE_DEVICES=0" \
            "HIP_VISIBLE_DEVICES=0 GPU_MAX_HW_QUEUES=8" \
            "HIP_VISIBLE_DEVICES=0 HSA_ENABLE_SDMA=0" \
            "HIP_VISIBLE_DEVICES=0 GPU_MAX_HW_QUEUES=1"
do
    (
    export $envs
    for engine in auto sdma; do
    echo "export $envs HPK_COPY_ENGINE=$engine"
    for mode in in_order graph graph_explicit host_threads; do
        args=""
        for c in "${LCOMMANDS[@]}"; do args+=" --commands $c"; done
        $BIN "$mode" --copy_engine "$engine" --repetitions 5 $args
    done
    done
    ) 2>&1 | tee -a "$LOG"
done

./scripts/parse.py "$LOG"
