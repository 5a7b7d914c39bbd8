"""Timing discipline of the suite (reference SURVEY.md §4 "Timing discipline").

- min over N repetitions everywhere (reference bench_sycl.cpp:84-121,
  peer2pear.cpp:23-52);
- distributed intervals = MAX(end)-MIN(start) over ranks (reference
  peer2pear.cpp:49-51) — realized here as an all_reduce MAX of per-rank
  durations after a barrier;
- device-side times via hipEvents (torch.cuda.Event on ROCm).
"""

from __future__ import annotations

import time
from contextlib import contextmanager
from dataclasses import dataclass, field
from typing import Callable


@dataclass
class MinOverReps:
    """Run a callable `reps` times, track min/all wall durations (seconds)."""

    reps: int = 10
    warmup: int = 1
    times: list = field(default_factory=list)

    def run(self, fn: Callable[[], None]) -> float:
        for _ in range(self.warmup):
            fn()
        self.times = []
        for _ in range(self.reps):
            t0 = time.perf_counter()
            fn()
            self.times.append(time.perf_counter() - t0)
        return self.best

    @property
    def best(self) -> float:
        return min(self.times) if self.times else float("inf")


@contextmanager
def gpu_timer(stream=None):
    """hipEvent-based device interval timer. Yields a dict; after the
    context exits and the stream is synchronized, d['ms'] holds the device
    time."""
    import torch

    start = torch.cuda.Event(enable_timing=True)
    stop = torch.cuda.Event(enable_timing=True)
    d = {"ms": None}
    s = stream if stream is not None else torch.cuda.current_stream()
    start.record(s)
    try:
        yield d
    finally:
        stop.record(s)
        stop.synchronize()
        d["ms"] = start.elapsed_time(stop)


def dist_interval_max(local_seconds: float, group=None) -> float:
    """MAX over ranks of a local duration (the reference's MIN(start)/MAX(end)
    clock union, peer2pear.cpp:49-51, done the RCCL way)."""
    import torch
    import torch.distributed as dist

    if not (dist.is_available() and dist.is_initialized()):
        return local_seconds
    t = torch.tensor([local_seconds], dtype=torch.float64)
    backend = dist.get_backend(group)
    if backend == "nccl":
        t = t.cuda()
    dist.all_reduce(t, op=dist.ReduceOp.MAX, group=group)
    return float(t.item())
