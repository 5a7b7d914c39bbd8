#!/bin/bash
# gpu_mapping.sh — per-rank GPU affinity wrapper (the reference
# tile_mapping.sh for MI355X: ZE_AFFINITY_MASK -> HIP_VISIBLE_DEVICES,
# PALS_LOCAL_RANKID -> LOCAL_RANK/torchrun).
#
# Usage: gpu_mapping.sh {compact|spread|topo} <cmd> [args...]
#   compact: rank r -> GPU r % N (consecutive ranks on consecutive GPUs)
#   spread:  ranks spaced across the node (NUMA/memory headroom first)
#   topo:    ask hpk_topology for the topology-sorted GPU order so
#            neighbouring ranks share direct xGMI links (reference
#            compact_plan policy)
set -u
POLICY=${1:?policy: compact|spread|topo}; shift

RANK=${LOCAL_RANK:-${OMPI_COMM_WORLD_LOCAL_RANK:-${SLURM_LOCALID:-0}}}
WORLD=${LOCAL_WORLD_SIZE:-${WORLD_SIZE:-1}}
NGPUS=$(ls /sys/class/kfd/kfd/topology/nodes 2>/dev/null | wc -l)
NGPUS=${HPK_NGPUS:-$(rocm-smi --showid 2>/dev/null | grep -c "^GPU" || echo 8)}

case "$POLICY" in
  compact)
    DEV=$(( RANK % NGPUS ));;
  spread)
    if [ "$WORLD" -ge "$NGPUS" ]; then DEV=$(( RANK % NGPUS ));
    else STRIDE=$(( NGPUS / WORLD )); [ "$STRIDE" -lt 1 ] && STRIDE=1
         DEV=$(( (RANK * STRIDE) % NGPUS )); fi;;
  topo)
    BIN="$(dirname "$0")/../bin/hpk_topology"
    if [ -x "$BIN" ]; then DEV=$("$BIN" "$RANK"); else DEV=$(( RANK % NGPUS )); fi;;
  *) echo "unknown policy $POLICY" >&2; exit 1;;
esac

export HIP_VISIBLE_DEVICES=$DEV
echo "# rank $RANK -> GPU $DEV (policy $POLICY)" >&2

# NUMA binding: GPU i's nearest NUMA node (MI355X OAM pairs per socket) —
# the reference left this commented out (tile_mapping.sh:32-35); do it.
if command -v numactl >/dev/null 2>&1 && [ -n "${HPK_NUMA_BIND:-}" ]; then
  NODE=$(( DEV / 2 ))
  exec numactl --cpunodebind=$NODE --membind=$NODE "$@"
fi
exec "$@"
