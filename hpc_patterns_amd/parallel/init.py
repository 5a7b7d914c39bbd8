"""Process-group bootstrap: one process per GPU, RCCL over xGMI.

Replaces the reference's mpirun/PALS launcher + ZE_AFFINITY_MASK tile binding
(reference p2p/tile_mapping.sh) with torch.distributed env:// rendezvous and
HIP device binding by LOCAL_RANK (policy-aware via placement.py).
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist

from .placement import map_rank_to_gpu


def rank_world() -> tuple[int, int, int]:
    """(rank, local_rank, world_size) from the torchrun/env contract."""
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    world = int(os.environ.get("WORLD_SIZE", 1))
    return rank, local_rank, world


def init_distributed(backend: str | None = None,
                     policy: str = "compact",
                     timeout_s: int = 300) -> tuple[int, int, int]:
    """Initialize torch.distributed and bind this process to its GPU.

    backend: None -> "nccl" (RCCL) when CUDA/HIP devices exist, else "gloo".
    policy:  rank->GPU mapping policy (compact|spread|topo), placement.py.
    """
    rank, local_rank, world = rank_world()
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"

    if torch.cuda.is_available():
        ndev = torch.cuda.device_count()
        dev = map_rank_to_gpu(local_rank, world, ndev, policy)
        torch.cuda.set_device(dev)

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    return rank, local_rank, world
