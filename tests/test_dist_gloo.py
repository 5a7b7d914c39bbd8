"""Multi-process CPU tests of the distributed patterns (gloo backend,
world_size 2 and 4) — the reference's mpirun -np 4 CTest matrix
 without MPI: same analytic oracles.
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.timeout(300)

N = 4096  # divisible by 4 for rsag


def _worker(rank, world, port, fn_name, out_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        result = globals()[fn_name](rank, world)
        out_q.put((rank, "ok", result))
    except Exception as e:  # surfaced by the parent
        out_q.put((rank, "err", f"{type(e).__name__}: {e}"))
    finally:
        dist.destroy_process_group()


def run_dist(world, fn_name, port):
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, fn_name, out_q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, status, payload = out_q.get(timeout=240)
        assert status == "ok", f"rank {rank} failed: {payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=60)
    return results


# ---- worker bodies (module-level for spawn pickling) ----

def body_ring(rank, world):
    from hpc_patterns_amd.parallel import ring_allreduce

    t = torch.full((N,), float(rank), dtype=torch.float32)
    ring_allreduce(t)
    expected = world * (world - 1) / 2.0
    assert torch.all(t == expected), (t[:4], expected)
    return True


def body_ring_pipelined(rank, world):
    from hpc_patterns_amd.parallel import ring_allreduce_pipelined

    t = torch.full((N,), float(rank), dtype=torch.float32)
    ring_allreduce_pipelined(t, n_chunks=4)
    expected = world * (world - 1) / 2.0
    assert torch.all(t == expected)
    return True


def body_ring_rsag(rank, world):
    from hpc_patterns_amd.parallel.ring import ring_allreduce_rsag

    t = torch.full((N,), float(rank), dtype=torch.float32)
    ring_allreduce_rsag(t)
    expected = world * (world - 1) / 2.0
    assert torch.all(t == expected)
    return True


def body_ring_matches_native_allreduce(rank, world):
    from hpc_patterns_amd.parallel import ring_allreduce

    g = torch.Generator().manual_seed(1234 + rank)
    t = torch.rand(N, generator=g)
    ref = t.clone()
    dist.all_reduce(ref)
    ring_allreduce(t)
    assert torch.allclose(t, ref, atol=1e-5), (t[:3], ref[:3])
    return True


def body_pairwise(rank, world):
    from hpc_patterns_amd.parallel import pairwise_exchange
    from hpc_patterns_amd.parallel.p2p import my_pair_peer

    peer = my_pair_peer(rank, world)
    assert peer is not None
    send = torch.full((N,), float(rank))
    recv = torch.empty(N)
    pairwise_exchange(send, recv, peer)
    assert torch.all(recv == float(peer))
    return True


def body_pairwise_bandwidth(rank, world):
    from hpc_patterns_amd.parallel.p2p import pairwise_bandwidth

    res = pairwise_bandwidth(nbytes=4 * N, iters=3, bidirectional=False,
                             device=torch.device("cpu"))
    assert res["checksum_ok"], res
    assert res["gbps"] > 0
    res2 = pairwise_bandwidth(nbytes=4 * N, iters=3, bidirectional=True,
                              device=torch.device("cpu"))
    assert res2["checksum_ok"], res2
    return True


def body_pingpong(rank, world):
    from hpc_patterns_amd.parallel.p2p import pingpong

    res = pingpong(nbytes=8, iters=20, device=torch.device("cpu"))
    assert res["oneway_us"] > 0
    return res["oneway_us"]


def body_flagship_cpu_accounting(rank, world):
    # byte accounting is device-independent logic; check the distributed
    # branch arithmetic without a GPU
    from hpc_patterns_amd.models.flagship import FlagshipPatternStep

    cfg = dict(d2d_floats=100, h2d_bytes=400, d2h_bytes=400, p2p_floats=50,
               allreduce_floats=64, tripcount=1, compute_globalsize=1)
    obj = FlagshipPatternStep.__new__(FlagshipPatternStep)
    obj.config = cfg
    obj.world_size = world
    obj.rank = rank
    obj.distributed = world > 1
    obj.peer = rank + 1 if rank % 2 == 0 else rank - 1
    b = obj.bytes_per_step_per_rank()
    expected = 100 * 4 + 400 + 400 + 2 * 50 * 4 + int(
        2 * (world - 1) / world * 64 * 4)
    assert b == expected, (b, expected)
    return True


# ---- tests ----

@pytest.mark.parametrize("world,fn", [
    (2, "body_ring"),
    (4, "body_ring"),
    (2, "body_ring_pipelined"),
    (4, "body_ring_pipelined"),
    (2, "body_ring_rsag"),
    (4, "body_ring_rsag"),
    (2, "body_ring_matches_native_allreduce"),
    (2, "body_pairwise"),
    (4, "body_pairwise"),
    (2, "body_pairwise_bandwidth"),
    (2, "body_pingpong"),
    (2, "body_flagship_cpu_accounting"),
    # the driver's 8-rank node shape, rehearsed on CPU
    (8, "body_ring"),
    (8, "body_ring_rsag"),
    (8, "body_pairwise"),
])
def test_dist_pattern(world, fn, dist_env):
    run_dist(world, fn, int(dist_env["MASTER_PORT"]))


def body_sweep(rank, world):
    from hpc_patterns_amd.parallel import sweep

    t = sweep.bench_algo("ring", 4096, iters=2,
                         device=torch.device("cpu"))
    assert t > 0
    t2 = sweep.bench_algo("rsag", 4096, iters=2, device=torch.device("cpu"))
    assert t2 > 0
    return True


def test_sweep_bench_algo(dist_env):
    run_dist(2, "body_sweep", int(dist_env["MASTER_PORT"]))


def body_policy_sweep_row(rank, world):
    """The policy-sweep measurement path on gloo: the CSV
    row must carry the policy, the world size, and one device per rank."""
    from hpc_patterns_amd.parallel.policy_sweep import measure_policy_row

    row = measure_policy_row("compact", 4096 * 4, iters=2,
                             device=torch.device("cpu"))
    parts = row.split(",")
    assert parts[0] == "compact" and parts[1] == str(world)
    assert parts[2] == "gloo"
    assert len(parts[7].split("+")) == world
    return True


def test_policy_sweep_row(dist_env):
    run_dist(2, "body_policy_sweep_row", int(dist_env["MASTER_PORT"]))


def test_policy_sweep_row_world4(dist_env):
    run_dist(4, "body_policy_sweep_row", int(dist_env["MASTER_PORT"]))
