#!/usr/bin/env python3
"""membench.py — on-device micro-benchmarks for the hand-written kernels.

Sweeps the shader-copy kernel (unroll x grid-cap x size) against
hipMemcpyAsync (SDMA/blit), and times fill/accumulate/reduce — the data
that picks kernel launch shapes (committed under profiles/).

Usage (on a GPU box):  python scripts/membench.py [--quick]
"""

from __future__ import annotations

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def time_gpu(fn, reps=5, warmup=2):
    import torch

    for _ in range(warmup):
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
    ap.add_argument("--quick", action="store_true")
    args = ap.parse_args()

    import torch

    from hpc_patterns_amd._native import native

    hpk = native()
    torch.cuda.set_device(0)
    dev = torch.device("cuda", 0)

    sizes = [64 << 20, 256 << 20, 1 << 30] if not args.quick else [64 << 20]
    print("# shader-copy sweep: payload GB/s (bytes moved = 2x payload)")
    print(f"{'bytes':>12} {'engine':>22} {'time_ms':>9} {'GB/s':>9}")
    for nbytes in sizes:
        src = torch.empty(nbytes // 4, dtype=torch.float32, device=dev)
        dst = torch.empty_like(src)
        src.fill_(1.0)

        variants = [("hipMemcpyAsync", lambda: hpk.memcpy_async(
            dst.data_ptr(), src.data_ptr(), nbytes,
            torch.cuda.current_stream().cuda_stream))]
        for unroll in (1, 4, 5):  # 5 = nontemporal streaming hints
            for cap in (16384, 65536, 131072):
                variants.append((
                    f"kernel u{unroll} cap{cap}",
                    lambda u=unroll, c=cap: hpk.copy_kernel_tuned(
                        dst.data_ptr(), src.data_ptr(), nbytes,
                        torch.cuda.current_stream().cuda_stream, u, c)))
        for name, fn in variants:
            t = time_gpu(fn)
            print(f"{nbytes:12d} {name:>22} {t*1e3:9.3f} {nbytes/t/1e9:9.1f}",
                  flush=True)
        del src, dst
        torch.cuda.empty_cache()

    print("\n# fill / accumulate / reduce (1 GiB)")
    n = (1 << 30) // 4
    a = torch.empty(n, dtype=torch.float32, device=dev)
    b = torch.empty_like(a)
    s = torch.cuda.current_stream().cuda_stream
    t = time_gpu(lambda: hpk.fill_f32(a.data_ptr(), 1.0, n, s))
    print(f"fill        {t*1e3:9.3f} ms  {n*4/t/1e9:9.1f} GB/s (write)")
    t = time_gpu(lambda: hpk.acc_f32(a.data_ptr(), b.data_ptr(), n, s))
    print(f"accumulate  {t*1e3:9.3f} ms  {3*n*4/t/1e9:9.1f} GB/s (2r+1w)")
    t = time_gpu(lambda: hpk.acc_f32_nt(a.data_ptr(), b.data_ptr(), n, s))
    print(f"acc (NT)    {t*1e3:9.3f} ms  {3*n*4/t/1e9:9.1f} GB/s (2r+1w)")
    t = time_gpu(lambda: hpk.reduce_sum_f32(a.data_ptr(), n, s))
    print(f"reduce_sum  {t*1e3:9.3f} ms  {n*4/t/1e9:9.1f} GB/s (read)")

    print("\n# busy_wait FMA rate (globalsize 1M, tripcount 2000)")
    out = torch.empty(1 << 20, dtype=torch.float32, device=dev)
    t = time_gpu(lambda: hpk.busy_wait(out.data_ptr(), 2000, 1 << 20, s))
    flops = (1 << 20) * 64 * 2000 * 2
    print(f"busy_wait   {t*1e3:9.3f} ms  {flops/t/1e12:9.2f} TFLOP/s fp32")

    print("\n# busy_wait_mfma rate (2048 waves, tripcount 20000)")
    out2 = torch.empty(2048 * 64, dtype=torch.float32, device=dev)
    t = time_gpu(lambda: hpk.busy_wait_mfma(out2.data_ptr(), 20000, 2048, s))
    # each mfma_f32_16x16x32_bf16: 2*16*16*32 = 16384 FLOP per wave-instr
    flops = 2048 * 20000 * 16384
    print(f"mfma        {t*1e3:9.3f} ms  {flops/t/1e12:9.2f} TFLOP/s bf16")

    print("\n# K7 LDS-tiled bf16 GEMM (C = A x B^T, fp32 accumulate, "
          "random [-1,1) operands)")
    from hpc_patterns_amd import ops

    for sz in (2048, 4096, 8192):
        a = (torch.rand(sz, sz, device=dev) * 2 - 1).to(torch.bfloat16)
        b = (torch.rand(sz, sz, device=dev) * 2 - 1).to(torch.bfloat16)
        c = torch.empty(sz, sz, dtype=torch.float32, device=dev)
        fl = 2.0 * sz * sz * sz
        t = time_gpu(lambda: ops.gemm_bf16(c, a, b))
        print(f"gemm {sz:5d} bf16 {t*1e3:9.3f} ms  {fl/t/1e12:9.1f} TFLOP/s")
        a8 = a.to(torch.float8_e4m3fn)
        b8 = b.to(torch.float8_e4m3fn)
        t = time_gpu(lambda: ops.gemm_fp8(c, a8, b8))
        print(f"gemm {sz:5d} fp8  {t*1e3:9.3f} ms  {fl/t/1e12:9.1f} TFLOP/s")
        s1 = torch.full((sz, sz // 32), 127, dtype=torch.uint8, device=dev)
        t = time_gpu(lambda: ops.gemm_mxfp8(c, a8, b8, s1, s1))
        print(f"gemm {sz:5d} mx8  {t*1e3:9.3f} ms  {fl/t/1e12:9.1f} TFLOP/s")
        p4 = torch.randint(0, 256, (sz, sz // 2), dtype=torch.uint8,
                           device=dev)
        t = time_gpu(lambda: ops.gemm_mxfp4(c, p4, p4, s1, s1))
        print(f"gemm {sz:5d} mx4  {t*1e3:9.3f} ms  {fl/t/1e12:9.1f} TFLOP/s")
        ai = torch.randint(-128, 128, (sz, sz), dtype=torch.int8, device=dev)
        bi = torch.randint(-128, 128, (sz, sz), dtype=torch.int8, device=dev)
        ci = torch.empty(sz, sz, dtype=torch.int32, device=dev)
        t = time_gpu(lambda: ops.gemm_i8(ci, ai, bi))
        print(f"gemm {sz:5d} i8   {t*1e3:9.3f} ms  {fl/t/1e12:9.1f} TOP/s")
        del a, b, a8, b8, s1, p4, ai, bi, ci, c


if __name__ == "__main__":
    main()
