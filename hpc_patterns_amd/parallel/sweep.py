"""RCCL all-reduce bandwidth sweep (BASELINE.json config[4]):

    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 -m hpc_patterns_amd.parallel.sweep \
        [--min-mb 4] [--max-mb 4096] [--iters 10] [--algos rccl,ring,...]

Sweeps message sizes 4 MB..4 GB (doubling) across the chosen algorithms and
prints one CSV block per algorithm: size, time, algorithm bandwidth
(bytes/time) and bus bandwidth (2(n-1)/n * bytes / time — comparable to the
per-link xGMI ceiling of ~153 GB/s and the 7-link aggregate ~1 TB/s).
Runs on gloo/CPU too (world_size tests), just slower.

RCCL channel study: NCCL reads NCCL_MIN/MAX_NCHANNELS at
communicator creation, so one launch measures ONE channel setting — the
outer loop lives in scripts/run_allreduce_sweep.sh (--channels N). The CSV
carries a `channels` column from the env so the N-launch table assembles
into channels-vs-busBW directly. --xgmi-sample additionally wraps the
largest size of each algorithm in an amd-smi per-link counter delta
(utils/xgmi.py) to show WHICH xGMI links carried the traffic.
"""

from __future__ import annotations

import argparse
import sys
import time

import torch
import torch.distributed as dist

from .init import init_distributed
from .ring import ring_allreduce, ring_allreduce_pipelined, ring_allreduce_rsag
from ..utils.timing import dist_interval_max

ALGOS = {
    "rccl": lambda t: dist.all_reduce(t),
    "ring": ring_allreduce,
    "pipeline": ring_allreduce_pipelined,
    "rsag": ring_allreduce_rsag,
}


def bench_algo(algo: str, nbytes: int, iters: int, device) -> float:
    n = max(nbytes // 4, 1)
    world = dist.get_world_size()
    if algo == "rsag" and n % world != 0:
        n = (n // world) * world or world
    t = torch.ones(n, dtype=torch.float32, device=device)
    fn = ALGOS[algo]
    fn(t)  # warmup + lazily build comms
    best = float("inf")
    for _ in range(iters):
        dist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        fn(t)
        if device.type == "cuda":
            torch.cuda.synchronize()
        best = min(best, dist_interval_max(time.perf_counter() - t0))
    return best


def main(argv=None) -> int:
    import os

    ap = argparse.ArgumentParser()
    ap.add_argument("--min-mb", type=float, default=4)
    ap.add_argument("--max-mb", type=float, default=4096)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--algos", default="rccl,ring,pipeline,rsag")
    ap.add_argument("--xgmi-sample", action="store_true",
                    help="per-link counter delta around the largest size of "
                         "each algorithm (amd-smi; silently skipped if "
                         "counters are unavailable)")
    args = ap.parse_args(argv)

    rank, _, world = init_distributed()
    device = (torch.device("cuda", torch.cuda.current_device())
              if torch.cuda.is_available() else torch.device("cpu"))

    algos = [a.strip() for a in args.algos.split(",") if a.strip()]
    for a in algos:
        if a not in ALGOS:
            raise SystemExit(f"unknown algo '{a}' (choose from {list(ALGOS)})")

    # channel setting travels with the data: scripts/run_allreduce_sweep.sh
    # sets NCCL_MIN/MAX_NCHANNELS per launch
    channels = os.environ.get("NCCL_MIN_NCHANNELS", "default")
    max_bytes = int(args.max_mb * 1e6)
    if rank == 0:
        print(f"# allreduce sweep: world={world} device={device.type} "
              f"iters={args.iters} channels={channels}")
        print("algo,channels,bytes,time_s,alg_GBps,bus_GBps")
    for algo in algos:
        nbytes = int(args.min_mb * 1e6)
        while nbytes <= max_bytes:
            largest = nbytes * 2 > max_bytes
            sampling = args.xgmi_sample and largest and rank == 0
            xgmi_before = None
            if sampling:
                from ..utils import xgmi

                xgmi_before = xgmi.sample_xgmi()
            t = bench_algo(algo, nbytes, args.iters, device)
            if rank == 0:
                alg_bw = nbytes / t / 1e9
                bus_bw = 2 * (world - 1) / world * alg_bw if world > 1 else alg_bw
                print(f"{algo},{channels},{nbytes},{t:.6f},{alg_bw:.2f},"
                      f"{bus_bw:.2f}", flush=True)
                if sampling:
                    from ..utils import xgmi

                    d = xgmi.delta(xgmi_before, xgmi.sample_xgmi())
                    if d:
                        top = sorted(d.items(), key=lambda kv: -kv[1])[:16]
                        print("# xgmi_delta "
                              + " ".join(f"{k}={v:.0f}" for k, v in top),
                              flush=True)
                    else:
                        # 1-GPU boxes report read/write as N/A; the links
                        # only count on a multi-GPU node
                        print("# xgmi_delta unavailable (no numeric link "
                              "counters on this node)", flush=True)
            nbytes *= 2
    dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
