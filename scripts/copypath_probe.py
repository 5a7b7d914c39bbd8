#!/usr/bin/env python3
"""copypath_probe.py — pin down ROCm's H2D/D2H engine selection.

Matrix: {Default kind, explicit kind} x {fresh stream, kernel-tainted
stream} x {H2D, D2H} at 256 MiB pinned. Run it twice: plain (rates) and
under `rocprofv3 --kernel-trace --stats` (blit-kernel counts — SDMA copies
are invisible to the kernel trace, __amd_rocclr_copyBuffer calls are the
blit path).
"""

from __future__ import annotations

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    import torch

    from hpc_patterns_amd._native import native

    hpk = native()
    torch.cuda.set_device(0)
    nbytes = 256 << 20

    host = hpk.host_malloc(nbytes)
    dev = hpk.hip_malloc(nbytes)

    fresh_h2d = torch.cuda.Stream()
    fresh_d2h = torch.cuda.Stream()
    tainted = torch.cuda.Stream()
    out = torch.empty(256, dtype=torch.float32, device="cuda")
    with torch.cuda.stream(tainted):
        hpk.busy_wait(out.data_ptr(), 100, 256, tainted.cuda_stream)
    tainted.synchronize()

    H2D, D2H, DEFAULT = 1, 2, 4

    def timed(fn, stream):
        fn()
        stream.synchronize()
        best = float("inf")
        for _ in range(5):
            t0 = time.perf_counter()
            fn()
            stream.synchronize()
            best = min(best, time.perf_counter() - t0)
        return nbytes / best / 1e9

    cases = [
        ("h2d default fresh", fresh_h2d,
         lambda: hpk.memcpy_async_kind(dev, host, nbytes, DEFAULT,
                                       fresh_h2d.cuda_stream)),
        ("h2d explicit fresh", fresh_h2d,
         lambda: hpk.memcpy_async_kind(dev, host, nbytes, H2D,
                                       fresh_h2d.cuda_stream)),
        ("h2d default tainted", tainted,
         lambda: hpk.memcpy_async_kind(dev, host, nbytes, DEFAULT,
                                       tainted.cuda_stream)),
        ("h2d explicit tainted", tainted,
         lambda: hpk.memcpy_async_kind(dev, host, nbytes, H2D,
                                       tainted.cuda_stream)),
        ("d2h default fresh", fresh_d2h,
         lambda: hpk.memcpy_async_kind(host, dev, nbytes, DEFAULT,
                                       fresh_d2h.cuda_stream)),
        ("d2h explicit fresh", fresh_d2h,
         lambda: hpk.memcpy_async_kind(host, dev, nbytes, D2H,
                                       fresh_d2h.cuda_stream)),
        ("d2h default tainted", tainted,
         lambda: hpk.memcpy_async_kind(host, dev, nbytes, DEFAULT,
                                       tainted.cuda_stream)),
        ("d2h explicit tainted", tainted,
         lambda: hpk.memcpy_async_kind(host, dev, nbytes, D2H,
                                       tainted.cuda_stream)),
    ]
    for name, stream, fn in cases:
        print(f"{name:24s} {timed(fn, stream):8.1f} GB/s", flush=True)

    # concurrent duplex on fresh streams, both kinds
    def duplex(kind_h2d, kind_d2h, s1, s2):
        host2 = hpk.host_malloc(nbytes)
        dev2 = hpk.hip_malloc(nbytes)
        best = float("inf")
        for _ in range(6):
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            hpk.memcpy_async_kind(dev, host, nbytes, kind_h2d, s1.cuda_stream)
            hpk.memcpy_async_kind(host2, dev2, nbytes, kind_d2h, s2.cuda_stream)
            s1.synchronize()
            s2.synchronize()
            best = min(best, time.perf_counter() - t0)
        hpk.host_free(host2)
        hpk.hip_free(dev2)
        return 2 * nbytes / best / 1e9

    print(f"{'duplex default fresh':24s} "
          f"{duplex(DEFAULT, DEFAULT, fresh_h2d, fresh_d2h):8.1f} GB/s")
    print(f"{'duplex explicit fresh':24s} "
          f"{duplex(H2D, D2H, fresh_h2d, fresh_d2h):8.1f} GB/s")

    # ---- allocation-history sequence (reproduces the conc-bench pattern:
    # the first H buffer in a process copies at ~57 GB/s, later small H
    # buffers allocated after big alloc/free cycles measured ~29 GB/s) ----
    print("# allocation-history sequence (22.5 MB pinned buffers)")
    small = 22_500_000

    def rate_on(hbuf, dbuf, n, stream):
        best = float("inf")
        hpk.memcpy_async_kind(dbuf, hbuf, n, DEFAULT, stream.cuda_stream)
        stream.synchronize()
        for _ in range(5):
            t0 = time.perf_counter()
            hpk.memcpy_async_kind(dbuf, hbuf, n, DEFAULT, stream.cuda_stream)
            stream.synchronize()
            best = min(best, time.perf_counter() - t0)
        return n / best / 1e9

    s = torch.cuda.Stream()
    h_a = hpk.host_malloc(small)
    d_a = hpk.hip_malloc(small)
    print(f"{'seq: first small H':24s} {rate_on(h_a, d_a, small, s):8.1f} GB/s")
    # big alloc/free churn (what the earlier command lists did)
    for _ in range(3):
        hb = hpk.host_malloc(1 << 30)
        db = hpk.hip_malloc(1 << 30)
        hpk.memcpy_async_kind(db, hb, 1 << 30, DEFAULT, s.cuda_stream)
        s.synchronize()
        hpk.host_free(hb)
        hpk.hip_free(db)
    h_b = hpk.host_malloc(small)
    d_b = hpk.hip_malloc(small)
    print(f"{'seq: post-churn new H':24s} {rate_on(h_b, d_b, small, s):8.1f} GB/s")
    print(f"{'seq: original H again':24s} {rate_on(h_a, d_a, small, s):8.1f} GB/s")


if __name__ == "__main__":
    main()
