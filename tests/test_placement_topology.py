"""CPU tests for rank->GPU placement policies and topology pure helpers."""

import pytest

from hpc_patterns_amd.parallel.placement import POLICIES, map_rank_to_gpu
from hpc_patterns_amd.parallel.topology import (
    planes_from_matrix,
    topology_order_from_matrix,
)


def fully_connected(n, weight=15):
    return [[{"p2p": int(i != j), "weight": weight, "hops": 1,
              "link_type": 2} for j in range(n)] for i in range(n)]


def test_compact():
    assert [map_rank_to_gpu(r, 8, 8, "compact") for r in range(8)] == list(range(8))
    assert map_rank_to_gpu(9, 16, 8, "compact") == 1


def test_spread_fewer_ranks_than_gpus():
    got = [map_rank_to_gpu(r, 2, 8, "spread") for r in range(2)]
    assert got == [0, 4]
    got4 = [map_rank_to_gpu(r, 4, 8, "spread") for r in range(4)]
    assert got4 == [0, 2, 4, 6]


def test_spread_full():
    assert [map_rank_to_gpu(r, 8, 8, "spread") for r in range(8)] == list(range(8))


def test_topo_policy_with_explicit_order():
    order = [0, 3, 1, 2]
    got = [map_rank_to_gpu(r, 4, 4, "topo", topo_order=order) for r in range(4)]
    assert got == order


def test_bad_policy():
    with pytest.raises(ValueError):
        map_rank_to_gpu(0, 1, 1, "bogus")
    assert set(POLICIES) == {"compact", "spread", "topo"}


def test_planes_fully_connected():
    m = fully_connected(8)
    planes = planes_from_matrix(m)
    assert len(planes) == 1 and sorted(planes[0]) == list(range(8))


def test_planes_disconnected():
    # two islands: {0,1}, {2,3}
    m = [[{"p2p": 0} for _ in range(4)] for _ in range(4)]
    m[0][1]["p2p"] = m[1][0]["p2p"] = 1
    m[2][3]["p2p"] = m[3][2]["p2p"] = 1
    planes = planes_from_matrix(m)
    assert sorted(sorted(p) for p in planes) == [[0, 1], [2, 3]]


def test_topology_order_prefers_weight():
    # 0-2 heavy link, 2-1 heavy, others light: order should walk 0,2,1,3
    n = 4
    m = fully_connected(n, weight=1)
    m[0][2]["weight"] = 100
    m[2][1]["weight"] = 100
    order = topology_order_from_matrix(m)
    assert order[:3] == [0, 2, 1]
    assert sorted(order) == list(range(n))


def test_topology_order_handles_disconnect():
    m = [[{"p2p": 0} for _ in range(3)] for _ in range(3)]
    order = topology_order_from_matrix(m)
    assert sorted(order) == [0, 1, 2]


# ---------------------------------------------------------------------------
# Partition-aware placement (CPX-shaped synthetic matrices — VERDICT r1 #10)
# ---------------------------------------------------------------------------

def cpx_matrix(n_oam=2, parts_per_oam=2):
    """Synthetic link matrix of a node in CPX mode: partitions of one OAM
    are 0 hops / weight 0 apart (same package), distinct OAMs 1 hop."""
    n = n_oam * parts_per_oam
    m = [[{"p2p": int(i != j), "weight": 15, "hops": 1, "link_type": 2}
          for j in range(n)] for i in range(n)]
    for i in range(n):
        for j in range(n):
            if i != j and i // parts_per_oam == j // parts_per_oam:
                m[i][j] = {"p2p": 1, "weight": 0, "hops": 0, "link_type": 2}
    return m


def cpx_partitions(n, mode="CPX"):
    return [{"compute": mode, "memory": "NPS1"} for _ in range(n)]


def test_oam_groups_cpx():
    from hpc_patterns_amd.parallel.topology import oam_groups

    m = cpx_matrix(n_oam=2, parts_per_oam=4)
    groups = oam_groups(m, cpx_partitions(8))
    assert groups == [[0, 1, 2, 3], [4, 5, 6, 7]]


def test_oam_groups_spx_is_identity():
    from hpc_patterns_amd.parallel.topology import oam_groups

    m = fully_connected(4)
    assert oam_groups(m, cpx_partitions(4, "SPX")) == [[0], [1], [2], [3]]
    # zero-hop links without partition info must NOT merge (SPX nodes can
    # report odd hop counts; grouping only applies in partition modes)
    m2 = cpx_matrix(2, 2)
    assert oam_groups(m2, None) == [[0], [1], [2], [3]]


def test_partition_aware_order_compact_and_spread():
    from hpc_patterns_amd.parallel.topology import partition_aware_order

    m = cpx_matrix(n_oam=2, parts_per_oam=2)
    parts = cpx_partitions(4)
    # compact: same-package partitions adjacent
    assert partition_aware_order(m, parts, "compact") == [0, 1, 2, 3]
    # spread: alternate packages
    assert partition_aware_order(m, parts, "spread") == [0, 2, 1, 3]
    with pytest.raises(ValueError):
        partition_aware_order(m, parts, "bogus")


def test_topo_policy_with_partition_order():
    from hpc_patterns_amd.parallel.topology import partition_aware_order

    m = cpx_matrix(n_oam=4, parts_per_oam=2)
    order = partition_aware_order(m, cpx_partitions(8), "compact")
    # consecutive rank pairs land on the same OAM package
    got = [map_rank_to_gpu(r, 8, 8, "topo", topo_order=order)
           for r in range(8)]
    for a, b in zip(got[0::2], got[1::2]):
        assert a // 2 == b // 2, got
