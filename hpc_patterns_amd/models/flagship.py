"""FlagshipPatternStep — the benchmark step of the suite.

One "step" per GPU, all submitted concurrently on separate hipStreams
(reference concurrency pattern), then joined:

  stream 0:      K1 busy-wait compute kernel (calibrated to the copy time)
  stream 1:      K2 shader D2D copy, nontemporal (1 GiB payload by default)
  SDMA engine 0: H2D from pinned host memory (explicit
                 hsa_amd_memory_async_copy_on_engine — see _h2d)
  SDMA engine 1: D2H to pinned host memory
  world>1 :      all-reduce of 2^25 floats (RCCL) + pairwise 188.7 MB P2P
                 exchange (ncclSend/Recv over xGMI) — the reference's
                 C4/C5/C1 communication patterns

The benchmark value is whole-job aggregate bandwidth: payload bytes moved by
all ranks divided by step time (max over ranks). Compute contributes no
bytes — it is there to prove copies and collectives overlap compute, which
is the reference's headline criterion (speedup within 30% of theoretical,
main.cpp:12).
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field

import torch
import torch.distributed as dist

# Reference workload sizes: 1 GB copy commands (concurency/main.cpp:100-105),
# 47,185,920-float P2P pairs (peer2pear.cpp:115), 2^25-float allreduce
# (allreduce-mpi-sycl.cpp:99).
DEFAULT_CONFIG = dict(
    d2d_floats=1 << 28,        # 1 GiB shader-copy payload
    h2d_bytes=256 << 20,       # 256 MiB pinned H2D
    d2h_bytes=256 << 20,       # 256 MiB pinned D2H
    p2p_floats=47_185_920,     # 188.7 MB pairwise exchange
    allreduce_floats=1 << 25,  # 128 MiB all-reduce
    tripcount=-1,              # autotuned to match the slowest copy
    compute_globalsize=1 << 20,
)

SMOKE_CONFIG = dict(
    d2d_floats=1 << 20,
    h2d_bytes=1 << 20,
    d2h_bytes=1 << 20,
    p2p_floats=1 << 18,
    allreduce_floats=1 << 18,
    tripcount=64,
    compute_globalsize=1 << 14,
)


@dataclass
class FlagshipPatternStep:
    device: torch.device
    rank: int = 0
    world_size: int = 1
    config: dict = field(default_factory=lambda: dict(DEFAULT_CONFIG))
    use_distributed: bool = True

    def __post_init__(self):
        from .. import ops
        from .._native import native

        self.ops = ops
        self.hpk = native()
        self._copy_handles = []
        cfg = self.config
        dev = self.device
        torch.cuda.set_device(dev)
        self.streams = [torch.cuda.Stream(device=dev) for _ in range(4)]

        # buffers
        self.d2d_src = torch.empty(cfg["d2d_floats"], dtype=torch.float32, device=dev)
        self.d2d_dst = torch.empty_like(self.d2d_src)
        ops.fill(self.d2d_src, 1.0)
        # H2D/D2H through hipHostMalloc + hipMemcpyAsync (SDMA): measured
        # 57 GB/s per direction on MI355X vs 34 GB/s for torch pinned
        # copies (profiles/h2d_d2h_matrix_r4 / membench r2).
        self.h2d_host = self.hpk.host_malloc(cfg["h2d_bytes"])
        self.h2d_dev = self.hpk.hip_malloc(cfg["h2d_bytes"])
        self.d2h_dev = self.hpk.hip_malloc(cfg["d2h_bytes"])
        self.d2h_host = self.hpk.host_malloc(cfg["d2h_bytes"])
        self.hpk.fill_f32(self.d2h_dev, 2.0, cfg["d2h_bytes"] // 4,
                          torch.cuda.current_stream().cuda_stream)
        self.compute_out = torch.empty(cfg["compute_globalsize"],
                                       dtype=torch.float32, device=dev)
        try:
            self._sdma_engines = self.hpk.sdma_num_engines(dev.index or 0)
        except Exception:
            self._sdma_engines = 0

        self.distributed = (self.use_distributed and self.world_size > 1
                            and dist.is_initialized())
        if self.distributed:
            self.ar_buf = torch.ones(cfg["allreduce_floats"], dtype=torch.float32,
                                     device=dev)
            self.p2p_send = torch.empty(cfg["p2p_floats"], dtype=torch.float32,
                                        device=dev)
            ops.iota(self.p2p_send)
            self.p2p_recv = torch.empty_like(self.p2p_send)
            self.peer = self._pair_peer()

        torch.cuda.synchronize()
        if cfg["tripcount"] == -1:
            cfg["tripcount"] = self._calibrate_tripcount()

    def close(self):
        """Release the raw native buffers (torch tensors free themselves)."""
        if getattr(self, "hpk", None) is None:
            return
        torch.cuda.synchronize()
        for attr, free in (("h2d_host", self.hpk.host_free),
                           ("h2d_dev", self.hpk.hip_free),
                           ("d2h_dev", self.hpk.hip_free),
                           ("d2h_host", self.hpk.host_free)):
            ptr = getattr(self, attr, None)
            if isinstance(ptr, int) and ptr:
                free(ptr)
                setattr(self, attr, 0)

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def _pair_peer(self):
        peer = self.rank + 1 if self.rank % 2 == 0 else self.rank - 1
        return peer if peer < self.world_size else None

    # H2D/D2H go through EXPLICIT SDMA engines (engine 0 / engine 1): in a
    # torch process rocclr routes pinned D2H hipMemcpyAsync through the
    # shader-blit kernel, which steals CUs from the compute command and
    # serializes against H2D (measured: profiles/copypath_r9). The explicit
    # path returns a completion handle; _wait_copies() joins them.
    def _h2d(self, stream=None) -> None:
        if self._sdma_engines >= 1:
            self._copy_handles.append(self.hpk.sdma_copy_begin(
                self.h2d_dev, self.h2d_host, self.config["h2d_bytes"],
                self.device.index or 0, 0))
        else:
            s = stream if stream is not None else torch.cuda.current_stream()
            self.hpk.memcpy_async(self.h2d_dev, self.h2d_host,
                                  self.config["h2d_bytes"], s.cuda_stream)

    def _d2h(self, stream=None) -> None:
        if self._sdma_engines >= 2:
            self._copy_handles.append(self.hpk.sdma_copy_begin(
                self.d2h_host, self.d2h_dev, self.config["d2h_bytes"],
                self.device.index or 0, 1))
        else:
            s = stream if stream is not None else torch.cuda.current_stream()
            self.hpk.memcpy_async(self.d2h_host, self.d2h_dev,
                                  self.config["d2h_bytes"], s.cuda_stream)

    def _wait_copies(self) -> None:
        for h in self._copy_handles:
            self.hpk.sdma_wait(h)
        self._copy_handles.clear()

    # ---- calibration: make the compute command last about as long as the
    # slowest copy (linear model, reference autotuner main.cpp:226-258) ----
    def _calibrate_tripcount(self, probe: int = 200) -> int:
        dev_sync = torch.cuda.synchronize
        cur = torch.cuda.current_stream()
        self._h2d(cur)  # first-touch warmup
        self._wait_copies()
        dev_sync()
        t0 = time.perf_counter()
        self.ops.copy_kernel(self.d2d_dst, self.d2d_src)
        dev_sync()
        t_d2d = time.perf_counter() - t0
        t0 = time.perf_counter()
        self._h2d(cur)
        self._wait_copies()
        dev_sync()
        t_h2d = time.perf_counter() - t0
        target = max(t_d2d, t_h2d)

        t0 = time.perf_counter()
        self.ops.busy_wait(self.compute_out, probe,
                           self.config["compute_globalsize"])
        dev_sync()
        t_probe = max(time.perf_counter() - t0, 1e-7)
        return max(int(probe * target / t_probe), 1)

    # ---- the step ----
    def step(self) -> None:
        cfg = self.config
        s0, s1, s2, s3 = self.streams
        with torch.cuda.stream(s0):
            self.ops.busy_wait(self.compute_out, cfg["tripcount"],
                               cfg["compute_globalsize"], stream=s0)
        with torch.cuda.stream(s1):
            self.ops.copy_kernel(self.d2d_dst, self.d2d_src, stream=s1)
        self._h2d(s2)
        self._d2h(s3)

        if self.distributed:
            # collectives ride torch's comm stream, overlapping the local work
            ar_work = dist.all_reduce(self.ar_buf, async_op=True)
            if self.peer is not None:
                reqs = dist.batch_isend_irecv([
                    dist.P2POp(dist.isend, self.p2p_send, self.peer),
                    dist.P2POp(dist.irecv, self.p2p_recv, self.peer),
                ])
                for r in reqs:
                    r.wait()
            ar_work.wait()

        self._wait_copies()
        # one device-wide sync instead of four per-stream syncs: measurably
        # less host overhead per step, identical semantics at step boundary
        torch.cuda.synchronize()

    # ---- accounting ----
    def bytes_per_step_per_rank(self) -> int:
        cfg = self.config
        b = cfg["d2d_floats"] * 4 + cfg["h2d_bytes"] + cfg["d2h_bytes"]
        if self.distributed:
            if self.peer is not None:
                b += 2 * cfg["p2p_floats"] * 4  # send + recv payload
            # ring-equivalent bus bytes per rank for an allreduce
            n = self.world_size
            b += int(2 * (n - 1) / n * cfg["allreduce_floats"] * 4)
        return b

    # ---- overlap diagnostic (the reference speedup criterion, local only) --
    def measure_overlap(self, reps: int = 5) -> dict:
        """Serial vs concurrent submission of the 4 local commands."""
        cfg = self.config

        def serial_once():
            cur = torch.cuda.current_stream()
            times = []
            for fn in (
                lambda: self.ops.busy_wait(self.compute_out, cfg["tripcount"],
                                           cfg["compute_globalsize"]),
                lambda: self.ops.copy_kernel(self.d2d_dst, self.d2d_src),
                lambda: self._h2d(cur),
                lambda: self._d2h(cur),
            ):
                t0 = time.perf_counter()
                fn()
                self._wait_copies()
                torch.cuda.synchronize()
                times.append(time.perf_counter() - t0)
            return times

        def concurrent_once():
            s0, s1, s2, s3 = self.streams
            t0 = time.perf_counter()
            with torch.cuda.stream(s0):
                self.ops.busy_wait(self.compute_out, cfg["tripcount"],
                                   cfg["compute_globalsize"], stream=s0)
            with torch.cuda.stream(s1):
                self.ops.copy_kernel(self.d2d_dst, self.d2d_src, stream=s1)
            self._h2d(s2)
            self._d2h(s3)
            self._wait_copies()
            torch.cuda.synchronize()
            return time.perf_counter() - t0

        serial_best, per_cmd_best = float("inf"), None
        conc_best = float("inf")
        for _ in range(reps):
            times = serial_once()
            if sum(times) < serial_best:
                serial_best, per_cmd_best = sum(times), times
            conc_best = min(conc_best, concurrent_once())
        theoretical = serial_best / max(per_cmd_best)
        speedup = serial_best / conc_best
        return {
            "serial_s": serial_best,
            "concurrent_s": conc_best,
            "per_command_s": per_cmd_best,
            "theoretical_speedup": theoretical,
            "speedup": speedup,
            "overlap_efficiency": speedup / theoretical,
        }
