#!/usr/bin/env python3
"""bench.py — flagship benchmark of the MI355X pattern suite.

Driver contract:
    python bench.py --gpus N --steps K --warmup W
N>1 is launched as `python -m torch.distributed.run --nnodes=1
--nproc-per-node N ... bench.py` (one rank per GPU over RCCL); rank/world
come from the environment. W untimed warmup steps, then exactly K timed
steps bracketed by barrier + torch.cuda.synchronize on both sides, MAX over
ranks, one JSON line from rank 0.

The step (models/flagship.py) bundles the suite's headline patterns —
compute/copy stream overlap, pairwise xGMI P2P exchange, RCCL all-reduce —
and the value is whole-job aggregate payload bandwidth in GB/s
(higher-is-better; BASELINE.json metric components reported alongside).
The reference publishes no absolute numbers, so vs_baseline
is null.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time


def _component(components: dict, key: str, fn):
    """Run one diagnostic component; a failure degrades to a null field and
    a stderr note instead of killing the whole bench JSON line (the 8-GPU
    scale run must survive any single component failing — VERDICT r1 #1).

    HPK_BENCH_FAULT=<key> injects a failure into that component (CI
    rehearsal of the degradation path; symmetric across ranks, so no rank
    is stranded in a half-entered collective)."""
    try:
        if os.environ.get("HPK_BENCH_FAULT") == key:
            raise RuntimeError("injected fault (HPK_BENCH_FAULT)")
        val = fn()
    except Exception as e:  # noqa: BLE001 — any component error is non-fatal
        print(f"# component {key} failed: {type(e).__name__}: {e}",
              file=sys.stderr, flush=True)
        components[key] = None
        return None
    if val is not None:
        components[key] = val
    return val


def _measure_xgmi_peer_copy(nbytes: int, iters: int = 5):
    """Direct hipMemcpyPeerAsync GB/s between GPU 0 and 1 (single process
    sees the whole node under torchrun) — the xGMI SDMA path, complementing
    the RCCL pt2pt number. Returns None when <2 GPUs are visible."""
    import time

    import torch

    from hpc_patterns_amd._native import native

    if torch.cuda.device_count() < 2:
        return None
    hpk = native()
    cur = torch.cuda.current_device()
    allocs = []  # (device, ptr) — freed in finally (ADVICE r1 leak fix)
    try:
        hpk.set_device(0)
        hpk.enable_peer_access(1)
        src = hpk.hip_malloc(nbytes)
        allocs.append((0, src))
        hpk.set_device(1)
        hpk.enable_peer_access(0)
        dst = hpk.hip_malloc(nbytes)
        allocs.append((1, dst))
        hpk.set_device(0)
        best = float("inf")
        for _ in range(iters + 1):  # first iter = warmup
            t0 = time.perf_counter()
            hpk.memcpy_peer_async(dst, 1, src, 0, nbytes, 0)
            hpk.stream_synchronize(0)
            dt = time.perf_counter() - t0
            best = min(best, dt)
        return nbytes / best / 1e9
    finally:
        for dev, ptr in allocs:
            try:
                hpk.set_device(dev)
                hpk.hip_free(ptr)
            except Exception:
                pass
        torch.cuda.set_device(cur)


def _measure_xgmi_multiengine_copy(nbytes: int, iters: int = 5):
    """Chunked GPU0->GPU1 copy spread across the xGMI SDMA engines
    (MI355X: 14 per GPU) via hsa_amd_memory_async_copy_on_engine — the
    multi-engine sibling of the single hipMemcpyPeerAsync path. Returns
    None when <2 GPUs or no peer engines are visible."""
    import time

    import torch

    from hpc_patterns_amd._native import native

    if torch.cuda.device_count() < 2:
        return None
    hpk = native()
    cur = torch.cuda.current_device()
    allocs = []
    try:
        n_eng = hpk.sdma_num_engines_pair(1, 0)
        if n_eng < 1:
            return None
        n_chunks = min(n_eng, 7)
        hpk.set_device(0)
        hpk.enable_peer_access(1)
        src = hpk.hip_malloc(nbytes)
        allocs.append((0, src))
        hpk.set_device(1)
        hpk.enable_peer_access(0)
        dst = hpk.hip_malloc(nbytes)
        allocs.append((1, dst))
        hpk.set_device(0)
        chunk = nbytes // n_chunks
        best = float("inf")
        for it in range(iters + 1):
            t0 = time.perf_counter()
            handles = []
            for c in range(n_chunks):
                off = c * chunk
                ln = chunk if c < n_chunks - 1 else nbytes - off
                handles.append(hpk.sdma_copy_begin(dst + off, src + off, ln,
                                                   0, c))
            for h in handles:
                hpk.sdma_wait(h)
            dt = time.perf_counter() - t0
            if it > 0:  # first is warmup
                best = min(best, dt)
        return nbytes / best / 1e9
    finally:
        for dev, ptr in allocs:
            try:
                hpk.set_device(dev)
                hpk.hip_free(ptr)
            except Exception:
                pass
        torch.cuda.set_device(cur)


def _measure_d2d_copy_tbps(step, iters: int = 5) -> float:
    """Shader-copy payload TB/s on the step's resident 1 GiB buffers — the
    hand-written K2 kernel's quality, surfaced in the driver-visible record
    instead of only in profiles/ (VERDICT r1 weak#6 / next#8)."""
    import time

    import torch

    nbytes = step.config["d2d_floats"] * 4
    best = float("inf")
    for _ in range(iters + 1):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        step.ops.copy_kernel(step.d2d_dst, step.d2d_src)
        torch.cuda.synchronize()
        best = min(best, time.perf_counter() - t0)
    return nbytes / best / 1e12


def _measure_gemm_tf(step, size: int = 4096, iters: int = 3) -> float:
    """K7 LDS-tiled bf16 MFMA GEMM TFLOP/s at size^3 on random operands
    (the r2 showcase kernel; 2*size^3 FLOP per call)."""
    import time

    import torch

    from hpc_patterns_amd import ops

    a = (torch.rand(size, size, device=step.device) * 2 - 1).to(torch.bfloat16)
    b = (torch.rand(size, size, device=step.device) * 2 - 1).to(torch.bfloat16)
    c = torch.empty(size, size, dtype=torch.float32, device=step.device)
    ops.gemm_bf16(c, a, b)  # warmup
    torch.cuda.synchronize()
    best = float("inf")
    for _ in range(iters):
        t0 = time.perf_counter()
        ops.gemm_bf16(c, a, b)
        torch.cuda.synchronize()
        best = min(best, time.perf_counter() - t0)
    return 2.0 * size ** 3 / best / 1e12


def _measure_gemm_mx4_tf(step, size: int = 4096, iters: int = 3) -> float:
    """K7-mx4 OCP MX-fp4 GEMM TFLOP/s at size^3 (random packed nibbles,
    unit scales) — the family's fastest member (256^2 32x32x64 kernel)."""
    import time

    import torch

    from hpc_patterns_amd import ops

    p4a = torch.randint(0, 256, (size, size // 2), dtype=torch.uint8,
                        device=step.device)
    p4b = torch.randint(0, 256, (size, size // 2), dtype=torch.uint8,
                        device=step.device)
    s1 = torch.full((size, size // 32), 127, dtype=torch.uint8,
                    device=step.device)
    c = torch.empty(size, size, dtype=torch.float32, device=step.device)
    ops.gemm_mxfp4(c, p4a, p4b, s1, s1)  # warmup
    torch.cuda.synchronize()
    best = float("inf")
    for _ in range(iters):
        t0 = time.perf_counter()
        ops.gemm_mxfp4(c, p4a, p4b, s1, s1)
        torch.cuda.synchronize()
        best = min(best, time.perf_counter() - t0)
    return 2.0 * size ** 3 / best / 1e12


def _measure_mfma_tf(step, n_waves: int = 2048, tripcount: int = 20000,
                     iters: int = 3) -> float:
    """bf16 MFMA busy-loop TFLOP/s (v_mfma_f32_16x16x32_bf16 chains):
    tripcount MFMA instructions per wave, 2*16*16*32 = 16384 FLOP each."""
    import time

    import torch

    out = torch.empty(((n_waves * 64 + 255) // 256) * 256,
                      dtype=torch.float32, device=step.device)
    step.ops.busy_wait_mfma(out, tripcount, n_waves)  # warmup
    torch.cuda.synchronize()
    best = float("inf")
    for _ in range(iters):
        t0 = time.perf_counter()
        step.ops.busy_wait_mfma(out, tripcount, n_waves)
        torch.cuda.synchronize()
        best = min(best, time.perf_counter() - t0)
    return n_waves * tripcount * 16384 / best / 1e12


def _preflight(rank: int, world: int) -> None:
    """Log the node shape before any collective: link matrix + partition
    modes at rank 0, so an RCCL init hang on the scale day leaves evidence
    of WHAT the node looked like."""
    try:
        from hpc_patterns_amd._native import native

        hpk = native()
        if rank == 0:
            m = hpk.link_matrix()
            n = len(m)
            print(f"# preflight: {n} HIP device(s), world={world}",
                  file=sys.stderr)
            for i in range(n):
                row = " ".join(
                    "-" if i == j else
                    f"{m[i][j]['p2p']}/{m[i][j]['link_type']}"
                    for j in range(n))
                print(f"# preflight: gpu{i} p2p/type: {row}", file=sys.stderr)
            try:
                parts = hpk.partition_info()
                print(f"# preflight: partitions "
                      f"{[(p['compute'], p['memory']) for p in parts]}",
                      file=sys.stderr)
            except Exception:
                pass
        print(f"# preflight: rank {rank} ok", file=sys.stderr, flush=True)
    except Exception as e:
        print(f"# preflight failed (non-fatal): {e}", file=sys.stderr,
              flush=True)


class _CpuPlumbingStep:
    """CI-only stub: exercises bench.py's exact distributed control flow
    (collectives, pt2pt pairing, byte accounting) on gloo/CPU so the
    multi-GPU path is correct by construction before it ever sees 8 GPUs."""

    def __init__(self, rank, world, cfg):
        import torch
        import torch.distributed as dist

        self.config = cfg
        self.rank, self.world_size = rank, world
        self.distributed = world > 1 and dist.is_initialized()
        self.peer = (rank + 1 if rank % 2 == 0 else rank - 1)
        if self.peer >= world:
            self.peer = None
        n = cfg["p2p_floats"]
        self.ar_buf = torch.ones(cfg["allreduce_floats"])
        self.p2p_send = torch.arange(n, dtype=torch.float32)
        self.p2p_recv = torch.empty(n)
        self.local = torch.empty(cfg["d2d_floats"])

    def step(self):
        import torch.distributed as dist

        self.local.fill_(1.0)  # stand-in for the local stream bundle
        if self.distributed:
            work = dist.all_reduce(self.ar_buf, async_op=True)
            if self.peer is not None:
                reqs = dist.batch_isend_irecv([
                    dist.P2POp(dist.isend, self.p2p_send, self.peer),
                    dist.P2POp(dist.irecv, self.p2p_recv, self.peer),
                ])
                for r in reqs:
                    r.wait()
            work.wait()

    def bytes_per_step_per_rank(self):
        from hpc_patterns_amd.models.flagship import FlagshipPatternStep

        return FlagshipPatternStep.bytes_per_step_per_rank(self)


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--smoke", action="store_true",
                    help="tiny config (seconds, used by __graft_entry__)")
    ap.add_argument("--cpu", action="store_true",
                    help="CPU/gloo plumbing check of the distributed bench "
                         "path (CI only — NOT a performance measurement)")
    ap.add_argument("--policy", default=os.environ.get("HPK_PLACEMENT_POLICY",
                                                       "compact"),
                    help="rank->GPU placement policy: compact|spread|topo "
                         "(parallel/placement.py; reference tile_mapping.sh)")
    args = ap.parse_args()

    import torch

    if not torch.cuda.is_available() and not args.cpu:
        print("bench.py requires a GPU (MI355X)", file=sys.stderr)
        return 1

    import torch.distributed as dist

    from hpc_patterns_amd.models import SMOKE_CONFIG, FlagshipPatternStep
    from hpc_patterns_amd.models.flagship import DEFAULT_CONFIG
    from hpc_patterns_amd.parallel import init_distributed
    from hpc_patterns_amd.parallel.p2p import pairwise_bandwidth, pingpong

    world = int(os.environ.get("WORLD_SIZE", 1))
    if world > 1 and not args.cpu:
        # scale-day hardening: leave RCCL warnings in stderr, and never let a
        # wedged collective hang past the watchdog
        os.environ.setdefault("NCCL_DEBUG", "WARN")
        _preflight(int(os.environ.get("RANK", 0)), world)
    if world > 1:
        rank, local_rank, world = init_distributed(
            backend="gloo" if args.cpu else None, policy=args.policy,
            timeout_s=int(os.environ.get("HPK_NCCL_TIMEOUT_S", "300")))
    else:
        rank, local_rank = 0, 0
        if not args.cpu:
            torch.cuda.set_device(0)

    if args.cpu:
        device = torch.device("cpu")
        cfg = dict(SMOKE_CONFIG)
        cfg["tripcount"] = 1
        step = _CpuPlumbingStep(rank, world, cfg)
    else:
        device = torch.device("cuda", torch.cuda.current_device())
        cfg = dict(SMOKE_CONFIG if args.smoke else DEFAULT_CONFIG)
        step = FlagshipPatternStep(device=device, rank=rank, world_size=world,
                                   config=cfg)

    def barrier_sync():
        if world > 1:
            dist.barrier()
        if not args.cpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        step.step()

    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step.step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    bytes_per_step = step.bytes_per_step_per_rank() * world
    agg_gbps = bytes_per_step * args.steps / elapsed / 1e9
    ms_per_step = elapsed / args.steps * 1e3

    # ---- component diagnostics (outside the timed region) ----
    # Every component is individually try/except-wrapped: on the 8-GPU scale
    # run a single failing diagnostic must degrade to a null field, not kill
    # the JSON line. Collective components are NOT wrapped
    # per-rank-divergently: if they raise they raise on all ranks (same code
    # path), so no rank is left hanging in a half-entered collective.
    components = {}
    if not args.cpu:
        def _overlap():
            overlap = step.measure_overlap(reps=3)
            if world > 1:
                # report the WORST rank's overlap efficiency, not rank 0's
                t = torch.tensor([overlap["overlap_efficiency"]],
                                 dtype=torch.float64, device=device)
                dist.all_reduce(t, op=dist.ReduceOp.MIN)
                overlap["overlap_efficiency"] = float(t.item())
            components["stream_overlap_pct"] = round(
                100.0 * overlap["overlap_efficiency"], 1)
            components["stream_overlap_speedup"] = round(overlap["speedup"], 3)
            components["theoretical_speedup"] = round(
                overlap["theoretical_speedup"], 3)
            return [round(t * 1e3, 3) for t in overlap["per_command_s"]]

        _component(components, "per_command_ms", _overlap)
        # hand-written-kernel quality in the driver-visible record
        # (VERDICT r1 next#8): shader-copy TB/s + MFMA TFLOP/s at every rank's
        # own GPU, reported from rank 0
        _component(components, "d2d_copy_TBps",
                   lambda: round(_measure_d2d_copy_tbps(step), 2))
        _component(components, "mfma_TFbf16",
                   lambda: round(_measure_mfma_tf(step), 1))
        _component(components, "gemm_TFbf16",
                   lambda: round(_measure_gemm_tf(step), 1))
        _component(components, "gemm_TFmx4",
                   lambda: round(_measure_gemm_mx4_tf(step), 1))
    if world > 1:
        def _p2p():
            bw = pairwise_bandwidth(cfg["p2p_floats"] * 4, iters=5,
                                    bidirectional=False, device=device)
            components["p2p_uni_GBps"] = round(bw["gbps"], 2)
            return bool(bw["checksum_ok"])

        _component(components, "p2p_checksum_ok", _p2p)

        def _p2p_bidir():
            bw = pairwise_bandwidth(cfg["p2p_floats"] * 4, iters=5,
                                    bidirectional=True, device=device)
            return round(bw["gbps"], 2)

        _component(components, "p2p_bi_GBps", _p2p_bidir)
        _component(components, "pingpong_us", lambda: round(
            pingpong(nbytes=8, iters=50, device=device)["oneway_us"], 2))
        # latency->bandwidth ladder (device-buffer ping-pong, one-way µs)
        for nb, key in ((8 << 10, "pingpong_8k_us"), (8 << 20, "pingpong_8m_us")):
            _component(components, key, lambda nb=nb: round(
                pingpong(nbytes=nb, iters=20, device=device)["oneway_us"], 2))
        if rank == 0 and not args.cpu:
            _component(components, "xgmi_peer_copy_GBps", lambda: (
                lambda v: round(v, 2) if v is not None else None)(
                    _measure_xgmi_peer_copy(cfg["p2p_floats"] * 4)))
            _component(components, "xgmi_multiengine_GBps", lambda: (
                lambda v: round(v, 2) if v is not None else None)(
                    _measure_xgmi_multiengine_copy(cfg["p2p_floats"] * 4)))
        dist.barrier()

    if rank == 0:
        result = {
            "metric": ("cpu_plumbing_check" if args.cpu
                       else "pattern_aggregate_GBps"),
            "value": round(agg_gbps, 2),
            "unit": "GB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic (random-init/iota payloads, reference sizes)",
            "config": {
                "model": "flagship-pattern-step",
                "d2d_bytes": cfg["d2d_floats"] * 4,
                "h2d_bytes": cfg["h2d_bytes"],
                "d2h_bytes": cfg["d2h_bytes"],
                "p2p_bytes": cfg["p2p_floats"] * 4,
                "allreduce_bytes": cfg["allreduce_floats"] * 4,
                "tripcount_C": cfg["tripcount"],
                "parallelism": f"dp{world}",
                "placement_policy": args.policy,
            },
            "components": components,
        }
        print(json.dumps(result))

    if world > 1:
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
