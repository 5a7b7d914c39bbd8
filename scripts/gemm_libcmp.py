#!/usr/bin/env python3
"""gemm_libcmp.py — drift-controlled comparison of the K7 GEMM family
against the vendor libraries on the same box and tensors.

Methodology (findings.md #21): MI355X clocks ramp ~20% within a run, so
every variant is measured in INTERLEAVED rounds (best-of kept per
variant) and the whole sweep is bracketed by a fixed hipBLASLt control
whose start/end delta exposes residual drift.

Usage (GPU box):  python scripts/gemm_libcmp.py [--size 8192] [--rounds 3]
"""

from __future__ import annotations

import argparse
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def time_gpu(fn, reps=6, warm=2):
    import torch

    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    best = float("inf")
    for _ in range(reps):
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        best = min(best, time.perf_counter() - t0)
    return best


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", type=int, default=8192)
    ap.add_argument("--rounds", type=int, default=3)
    ap.add_argument("--groups", default="0",
                    help="comma list of HPK_GEMM_GROUP values to sweep "
                         "(0 = the shape-adaptive default)")
    args = ap.parse_args()

    import torch

    from hpc_patterns_amd import ops

    dev = torch.device("cuda", 0)
    sz = args.size
    fl = 2.0 * sz**3
    a = (torch.rand(sz, sz, device=dev) * 2 - 1).to(torch.bfloat16)
    b = (torch.rand(sz, sz, device=dev) * 2 - 1).to(torch.bfloat16)
    c = torch.empty(sz, sz, dtype=torch.float32, device=dev)
    a8, b8 = a.to(torch.float8_e4m3fn), b.to(torch.float8_e4m3fn)
    s1 = torch.full((sz, sz // 32), 127, dtype=torch.uint8, device=dev)
    p4a = torch.randint(0, 256, (sz, sz // 2), dtype=torch.uint8, device=dev)
    p4b = torch.randint(0, 256, (sz, sz // 2), dtype=torch.uint8, device=dev)
    ai = torch.randint(-128, 128, (sz, sz), dtype=torch.int8, device=dev)
    bi = torch.randint(-128, 128, (sz, sz), dtype=torch.int8, device=dev)
    ci = torch.empty(sz, sz, dtype=torch.int32, device=dev)

    def control():
        return fl / time_gpu(lambda: torch.matmul(a, b.t())) / 1e12

    print(f"hipBLASLt bf16 control (start): {control():7.1f} TF", flush=True)
    groups = [g for g in args.groups.split(",")]
    variants = {}
    for g in groups:
        tag = f"g{g}" if g != "0" else "auto"
        variants[f"bf16 {tag}"] = (g, lambda: ops.gemm_bf16(c, a, b))
        variants[f"fp8 {tag}"] = (g, lambda: ops.gemm_fp8(c, a8, b8))
        variants[f"mx8 {tag}"] = (g, lambda: ops.gemm_mxfp8(c, a8, b8, s1, s1))
        variants[f"i8 {tag}"] = (g, lambda: ops.gemm_i8(ci, ai, bi))
        variants[f"mx4 {tag}"] = (g, lambda: ops.gemm_mxfp4(
            c, p4a, p4b, s1, s1))
    try:
        sa = torch.ones(sz, 1, device=dev)
        sb = torch.ones(1, sz, device=dev)
        variants["torch._scaled_mm fp8"] = ("0", lambda: torch._scaled_mm(
            a8, b8.t().contiguous().t(), scale_a=sa, scale_b=sb,
            out_dtype=torch.bfloat16))
    except Exception:
        pass
    best = {k: 0.0 for k in variants}
    for rnd in range(args.rounds):
        for name, (g, fn) in variants.items():
            if g == "0":
                os.environ.pop("HPK_GEMM_GROUP", None)
            else:
                os.environ["HPK_GEMM_GROUP"] = g
            best[name] = max(best[name], fl / time_gpu(fn) / 1e12)
        print(f"after round {rnd}: " +
              "  ".join(f"{k}={v:.0f}" for k, v in best.items()), flush=True)
    print(f"hipBLASLt bf16 control (end)  : {control():7.1f} TF")


if __name__ == "__main__":
    main()
