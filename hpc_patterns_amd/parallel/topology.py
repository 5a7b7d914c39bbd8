"""xGMI topology discovery (python face).

Wraps the native link matrix (native/topo.hip: hipDeviceCanAccessPeer +
hipExtGetLinkTypeAndHopCount + rocm_smi weights/bandwidths) — the MI355X
replacement of the reference Level-Zero-Sysman fabric enumerator
(reference p2p/topology.cpp). Pure helpers (plane merge, topology order)
are CPU-testable on synthetic matrices.
"""

from __future__ import annotations

LINK_TYPE_XGMI = 2  # hipExtGetLinkTypeAndHopCount: 2 == HSA_AMD_LINK_INFO_TYPE_XGMI


def link_matrix() -> list[list[dict]]:
    """NxN link matrix from the native core (requires GPU)."""
    from .._native import native

    return native().link_matrix()


def planes_from_matrix(matrix: list[list[dict]]) -> list[list[int]]:
    """Connected components under direct-P2P reachability (pure python;
    the reference's connectivity-plane merge, topology.cpp:76-89)."""
    n = len(matrix)
    parent = list(range(n))

    def find(x: int) -> int:
        while parent[x] != x:
            parent[x] = parent[parent[x]]
            x = parent[x]
        return x

    for i in range(n):
        for j in range(n):
            if i != j and matrix[i][j].get("p2p"):
                a, b = find(i), find(j)
                if a != b:
                    parent[a] = b
    planes: dict[int, list[int]] = {}
    for i in range(n):
        planes.setdefault(find(i), []).append(i)
    return list(planes.values())


def topology_order_from_matrix(matrix: list[list[dict]]) -> list[int]:
    """Greedy walk: start at GPU 0, repeatedly hop to the unvisited neighbour
    with the highest link weight (fewest hops as tiebreak), so consecutive
    positions are directly connected. Falls back to numeric order for
    disconnected leftovers."""
    n = len(matrix)
    if n == 0:
        return []
    order = [0]
    visited = {0}
    while len(order) < n:
        cur = order[-1]
        best, best_key = None, None
        for j in range(n):
            if j in visited:
                continue
            li = matrix[cur][j]
            if not li.get("p2p"):
                continue
            # higher weight better; fewer hops better; stable by index
            key = (-(li.get("weight") or 0), li.get("hops") or 0, j)
            if best_key is None or key < best_key:
                best, best_key = j, key
        if best is None:  # disconnected: take smallest unvisited
            best = min(set(range(n)) - visited)
        order.append(best)
        visited.add(best)
    return order


def oam_groups(matrix: list[list[dict]],
               partitions: list[dict] | None = None) -> list[list[int]]:
    """Group HIP devices by physical OAM package (pure, CPU-testable).

    In CPX/DPX compute-partition modes one MI355X OAM enumerates as several
    HIP devices (the MI355X analog of the reference's tile fission,
    devices.hpp:30-34). Partitions of the SAME package report hop count 0
    (or link weight 0) to each other in the link matrix — physically they
    share the package, there is no xGMI hop between them — while distinct
    OAMs are >=1 hop apart. When every device is SPX (or no partition info),
    each device is its own group.
    """
    n = len(matrix)
    partitioned = bool(partitions) and any(
        p.get("compute") not in ("", None, "SPX") for p in partitions)
    parent = list(range(n))

    def find(x: int) -> int:
        while parent[x] != x:
            parent[x] = parent[parent[x]]
            x = parent[x]
        return x

    if partitioned:
        for i in range(n):
            for j in range(n):
                if i == j:
                    continue
                li = matrix[i][j]
                same_pkg = (li.get("hops") == 0 or li.get("weight") == 0)
                if li.get("p2p") and same_pkg:
                    a, b = find(i), find(j)
                    if a != b:
                        parent[a] = b
    groups: dict[int, list[int]] = {}
    for i in range(n):
        groups.setdefault(find(i), []).append(i)
    return sorted(groups.values())


def partition_aware_order(matrix: list[list[dict]],
                          partitions: list[dict] | None = None,
                          mode: str = "compact") -> list[int]:
    """Device visit order that understands CPX partition grouping (pure).

    compact: walk OAM groups one after another (consecutive ranks land on
             partitions of the same package — zero-hop neighbours, the
             reference's compact tile policy, tile_mapping.sh:10-12).
    spread:  round-robin across the groups (consecutive ranks on DIFFERENT
             packages — each gets its own HBM/NUMA domain, the reference's
             spread policy, tile_mapping.sh:13-15).
    Groups themselves are ordered along the greedy xGMI walk so adjacent
    groups are well-connected.
    """
    groups = oam_groups(matrix, partitions)
    if len(groups) > 1:
        walk = topology_order_from_matrix(matrix)
        first_pos = {id(g): min(walk.index(d) for d in g) for g in groups}
        groups = sorted(groups, key=lambda g: first_pos[id(g)])
    if mode == "compact":
        return [d for g in groups for d in g]
    if mode == "spread":
        order: list[int] = []
        idx = 0
        while len(order) < len(matrix):
            for g in groups:
                if idx < len(g):
                    order.append(g[idx])
            idx += 1
        return order
    raise ValueError(f"unknown mode '{mode}'")


def topology_order(n_gpus: int) -> list[int]:
    """Topology order from live hardware (partition-aware when the node is
    in CPX/DPX mode); numeric order if discovery fails."""
    try:
        m = link_matrix()
        if len(m) >= n_gpus:
            try:
                from .._native import native

                parts = native().partition_info()
            except Exception:
                parts = None
            if parts and any(p.get("compute") not in ("", None, "SPX")
                             for p in parts):
                return partition_aware_order(m, parts, "compact")[:n_gpus]
            return topology_order_from_matrix(m)[:n_gpus]
    except Exception:
        pass
    return list(range(n_gpus))


def _main() -> int:  # python -m hpc_patterns_amd.parallel.topology
    try:
        m = link_matrix()
    except Exception as e:
        print(f"no GPU / native core: {e}")
        return 1
    n = len(m)
    print(f"# {n} HIP device(s)")
    for i in range(n):
        row = " ".join(
            "-" if i == j else
            f"p2p={m[i][j]['p2p']},t={m[i][j]['link_type']},w={m[i][j]['weight']}"
            for j in range(n))
        print(f"gpu{i}: {row}")
    print("planes:", planes_from_matrix(m))
    print("order:", topology_order_from_matrix(m))
    try:
        from .._native import native

        print("partitions:", native().partition_info())
    except Exception:
        pass
    return 0


if __name__ == "__main__":
    raise SystemExit(_main())
