"""Pairwise P2P exchange + device-buffer ping-pong over torch.distributed.

MI355X-native re-design of the reference P2P bandwidth bench's transfer
engines (reference p2p/peer2pear.cpp:19-66 two-sided, 104-156 pairing and
bandwidth math): even rank i pairs with i+1, phase 1 unidirectional, phase 2
bidirectional, aggregate GB/s = payload bytes x pairs / min-time over
iterations with the MIN(start)/MAX(end) global interval realized as a
barrier + MAX-allreduce of per-rank durations.

The third engines (hipMemcpyPeerAsync over xGMI, HIP-IPC one-sided put) live
in the native core and the hpk_p2p binary; this module is the
process-per-GPU two-sided path (ncclSend/Recv on GPU, gloo on CPU tests).
"""

from __future__ import annotations

import time

import torch
import torch.distributed as dist

from ..utils.timing import dist_interval_max


def my_pair_peer(rank: int, size: int) -> int | None:
    """Even<->odd pairing (reference peer2pear.cpp:126-131)."""
    peer = rank + 1 if rank % 2 == 0 else rank - 1
    return peer if peer < size else None


def pairwise_exchange(send: torch.Tensor, recv: torch.Tensor, peer: int,
                      send_first: bool = True, group=None) -> None:
    """Concurrent send+recv with one peer (both directions in flight)."""
    ops = [
        dist.P2POp(dist.isend, send, peer, group),
        dist.P2POp(dist.irecv, recv, peer, group),
    ]
    if not send_first:
        ops.reverse()
    for req in dist.batch_isend_irecv(ops):
        req.wait()


def pairwise_bandwidth(nbytes: int, iters: int = 10, bidirectional: bool = False,
                       device: torch.device | None = None,
                       group=None) -> dict:
    """Reference-style pairwise bandwidth phase. Returns aggregate GB/s.

    Every rank allocates one payload buffer; senders are even ranks in the
    unidirectional phase, everyone in the bidirectional phase."""
    rank = dist.get_rank(group)
    size = dist.get_world_size(group)
    peer = my_pair_peer(rank, size)
    n = nbytes // 4
    dev = device if device is not None else (
        torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu"))
    send = torch.empty(n, dtype=torch.float32, device=dev)
    recv = torch.empty_like(send)
    # shuffled-iota checksum payload (reference fill_randomly) — seeded per
    # rank for reproducibility (the reference's default-constructed
    # minstd_rand had no seed control, SURVEY.md §4 gap)
    g = torch.Generator().manual_seed(0x5EED + rank)
    perm = torch.randperm(n, generator=g)
    send.copy_(perm.to(torch.float32))
    expected = float(perm.to(torch.float32).to(torch.float64).sum())

    sender = (rank % 2 == 0) or bidirectional
    receiver = (rank % 2 == 1) or bidirectional

    best = float("inf")
    for _ in range(iters):
        dist.barrier(group)
        if dev.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        if peer is not None:
            ops = []
            if sender:
                ops.append(dist.P2POp(dist.isend, send, peer, group))
            if receiver:
                ops.append(dist.P2POp(dist.irecv, recv, peer, group))
            for req in dist.batch_isend_irecv(ops):
                req.wait()
            if dev.type == "cuda":
                torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        best = min(best, dist_interval_max(dt, group))

    ok = True
    # exchange expected checksums: all_gather of one double (collective —
    # every rank participates even if it verifies nothing)
    sums = [torch.zeros(1, dtype=torch.float64) for _ in range(size)]
    local = torch.tensor([expected], dtype=torch.float64)
    if dist.get_backend(group) == "nccl":
        sums = [s.to(dev) for s in sums]
        local = local.to(dev)
    dist.all_gather(sums, local, group)
    if peer is not None and receiver:
        want = float(sums[peer].item())
        if dev.type == "cuda":
            from .. import ops as hops

            got = hops.reduce_sum(recv)
            if got != want:
                # second opinion via host: some ROCm 7.2 pods show a
                # partially-visible buffer to the first cross-stream
                # reduction even after sync (profiles/README.md r37-r39);
                # the host readback is authoritative
                torch.cuda.synchronize()
                got = float(recv.cpu().to(torch.float64).sum())
        else:
            got = float(recv.to(torch.float64).sum())
        ok = got == want

    n_pairs = size // 2
    gb = nbytes * max(n_pairs, 1) * (2 if bidirectional else 1) / 1e9
    return {
        "gbps": gb / best if best > 0 else 0.0,
        "min_time_s": best,
        "pairs": n_pairs,
        "bytes_per_pair": nbytes,
        "bidirectional": bidirectional,
        "checksum_ok": ok,
    }


def pingpong(nbytes: int = 8, iters: int = 100,
             device: torch.device | None = None, group=None) -> dict:
    """Device-buffer ping-pong latency between ranks 0 and 1 (µs one-way).

    The BASELINE.json "MPI ping-pong µs (device buffers)" metric, measured
    the RCCL way."""
    rank = dist.get_rank(group)
    size = dist.get_world_size(group)
    if size < 2:
        return {"oneway_us": 0.0, "iters": 0}
    dev = device if device is not None else (
        torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu"))
    buf = torch.zeros(max(nbytes // 4, 1), dtype=torch.float32, device=dev)

    active = rank in (0, 1)
    peer = 1 - rank if active else None
    # warmup
    if active:
        for _ in range(5):
            if rank == 0:
                dist.send(buf, peer, group)
                dist.recv(buf, peer, group)
            else:
                dist.recv(buf, peer, group)
                dist.send(buf, peer, group)
    dist.barrier(group)
    t0 = time.perf_counter()
    if active:
        for _ in range(iters):
            if rank == 0:
                dist.send(buf, peer, group)
                dist.recv(buf, peer, group)
            else:
                dist.recv(buf, peer, group)
                dist.send(buf, peer, group)
        if dev.type == "cuda":
            torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    dt = dist_interval_max(dt if active else 0.0, group)
    return {
        "oneway_us": dt / (2 * iters) * 1e6,
        "iters": iters,
        "nbytes": nbytes,
    }


def _main(argv=None) -> int:
    """Latency/bandwidth sweep CLI (BASELINE: "ping-pong µs vs message size"):

        python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \\
            --master-addr 127.0.0.1 -m hpc_patterns_amd.parallel.p2p

    Prints one-way ping-pong latency from 8 B to 8 MB (doubling) and the
    pairwise bandwidth phases at the reference 188.7 MB size.
    """
    import argparse

    from .init import init_distributed

    ap = argparse.ArgumentParser()
    ap.add_argument("--max-bytes", type=int, default=8 << 20)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--pair-floats", type=int, default=47_185_920)
    args = ap.parse_args(argv)

    rank, _, world = init_distributed()
    dev = (torch.device("cuda", torch.cuda.current_device())
           if torch.cuda.is_available() else torch.device("cpu"))
    if rank == 0:
        print(f"# ping-pong one-way latency, world={world}, device={dev.type}")
        print("bytes,oneway_us")
    nb = 8
    while nb <= args.max_bytes:
        r = pingpong(nbytes=nb, iters=args.iters, device=dev)
        if rank == 0:
            print(f"{nb},{r['oneway_us']:.2f}", flush=True)
        nb *= 2
    for bidir in (False, True):
        r = pairwise_bandwidth(args.pair_floats * 4, iters=10,
                               bidirectional=bidir, device=dev)
        if rank == 0:
            d = "bidirectional" if bidir else "unidirectional"
            print(f"# pairwise {d}: {r['gbps']:.2f} GB/s "
                  f"(pairs={r['pairs']}, checksum_ok={r['checksum_ok']})")
    dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    import sys

    sys.exit(_main())
