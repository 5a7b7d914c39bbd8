"""Rank -> GPU placement policies.

MI355X-native replacement of the reference's tile-affinity wrapper
(reference p2p/tile_mapping.sh:9-29: compact / spread / compact_plan over
ZE_AFFINITY_MASK). On MI355X there is no tile fission; the unit is the GPU
(optionally a CPX partition), and the "plan" policy consults the xGMI
topology (topology.py / hpk_topology) so consecutive ranks land on
directly-connected, highest-weight pairs.
"""

from __future__ import annotations

POLICIES = ("compact", "spread", "topo")


def map_rank_to_gpu(local_rank: int, world_size: int, n_gpus: int,
                    policy: str = "compact",
                    topo_order: list[int] | None = None) -> int:
    """Pure mapping function (CPU-testable).

    compact: consecutive ranks on consecutive GPUs (rank i -> GPU i%n).
    spread:  ranks spread as far apart as possible first (rank i ->
             GPU (i*n)//world stride walk) — useful when world < n_gpus and
             each rank wants its own memory/NUMA domain.
    topo:    walk GPUs in topology order (from the xGMI link matrix) so
             neighbouring ranks share direct high-weight links (the
             reference's compact_plan, tile_mapping.sh:17-19).
    """
    if n_gpus <= 0:
        raise ValueError("n_gpus must be positive")
    if policy == "compact":
        return local_rank % n_gpus
    if policy == "spread":
        if world_size >= n_gpus:
            return local_rank % n_gpus
        stride = max(n_gpus // world_size, 1)
        return (local_rank * stride) % n_gpus
    if policy == "topo":
        order = topo_order
        if order is None:
            from . import topology

            order = topology.topology_order(n_gpus)
        return order[local_rank % len(order)]
    raise ValueError(f"unknown policy '{policy}' (choose from {POLICIES})")
