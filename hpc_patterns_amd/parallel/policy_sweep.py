"""Placement-policy measured sweep — the reference's policy x engine matrix
(reference p2p/run.sh:9-21: {compact,spread,compact_plan} x {isend,win} x
ranks, driven through tile_mapping.sh) on MI355X: the xGMI topology tool
demonstrably DRIVES placement, and the choice is measured, not asserted.

One torchrun launch measures ONE policy (device binding happens at
process-group init); the outer policy loop lives in scripts/run_p2p.sh:

    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 -m hpc_patterns_amd.parallel.policy_sweep \
        --policy topo --csv policy_sweep.csv

CSV columns: policy,world,engine,bytes,uni_GBps,bi_GBps,pingpong_us,
device_order — the device_order column shows which GPU each rank actually
landed on, so a reader can verify the policy did something.
"""

from __future__ import annotations

import argparse
import os
import sys

import torch
import torch.distributed as dist

from .init import init_distributed
from .p2p import pairwise_bandwidth, pingpong

CSV_HEADER = ("policy,world,engine,bytes,uni_GBps,bi_GBps,pingpong_us,"
              "device_order")


def measure_policy_row(policy: str, nbytes: int, iters: int,
                       device: torch.device) -> str:
    """One CSV row for the already-initialized process group (pure of any
    argument parsing — CPU-testable on gloo)."""
    world = dist.get_world_size()
    # which device did each rank land on under this policy?
    dev_idx = device.index if device.type == "cuda" else -1
    order: list = [None] * world
    dist.all_gather_object(order, dev_idx)

    uni = pairwise_bandwidth(nbytes, iters=iters, bidirectional=False,
                             device=device)
    bi = pairwise_bandwidth(nbytes, iters=iters, bidirectional=True,
                            device=device)
    pp = pingpong(nbytes=8, iters=30, device=device)
    engine = "rccl" if dist.get_backend() == "nccl" else dist.get_backend()
    order_s = "+".join(str(d) for d in order)
    return (f"{policy},{world},{engine},{nbytes},{uni['gbps']:.2f},"
            f"{bi['gbps']:.2f},{pp['oneway_us']:.2f},{order_s}")


def main(argv=None) -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--policy", default="compact",
                    help="compact|spread|topo (parallel/placement.py)")
    ap.add_argument("--floats", type=int, default=47_185_920)
    ap.add_argument("--iters", type=int, default=5)
    ap.add_argument("--csv", default=None,
                    help="append the row to this file (header added once)")
    ap.add_argument("--backend", default=None,
                    help="override backend (gloo for CPU rehearsals)")
    args = ap.parse_args(argv)

    rank, _, world = init_distributed(backend=args.backend,
                                      policy=args.policy)
    device = (torch.device("cuda", torch.cuda.current_device())
              if torch.cuda.is_available() and args.backend != "gloo"
              else torch.device("cpu"))
    row = measure_policy_row(args.policy, args.floats * 4, args.iters, device)
    if rank == 0:
        print(row, flush=True)
        if args.csv:
            fresh = not os.path.exists(args.csv)
            with open(args.csv, "a") as f:
                if fresh:
                    f.write(CSV_HEADER + "\n")
                f.write(row + "\n")
    dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
