"""GPU numerics tests: every hand-written HIP kernel vs a plain PyTorch fp32
reference (SURVEY.md §4 oracle style — exact where the math is exact)."""

import pytest
import torch

pytestmark = [pytest.mark.gpu, pytest.mark.timeout(600)]


@pytest.fixture(scope="module")
def dev():
    torch.cuda.set_device(0)
    return torch.device("cuda", 0)


@pytest.fixture(scope="module")
def ops():
    from hpc_patterns_amd import ops as _ops
    from hpc_patterns_amd._native import native

    native()  # fail loudly, not skip: on a GPU box the extension must exist
    return _ops


@pytest.mark.parametrize("n", [1, 3, 4, 255, 1 << 20, (1 << 20) + 7])
def test_fill(ops, dev, n):
    t = torch.empty(n, dtype=torch.float32, device=dev)
    ops.fill(t, 3.5)
    torch.cuda.synchronize()
    assert torch.equal(t, torch.full((n,), 3.5, device=dev))


@pytest.mark.parametrize("n", [1, 17, 1 << 20])
def test_iota(ops, dev, n):
    t = torch.empty(n, dtype=torch.float32, device=dev)
    ops.iota(t)
    torch.cuda.synchronize()
    assert torch.equal(t, torch.arange(n, dtype=torch.float32, device=dev))


@pytest.mark.parametrize("n", [4, 1023, 1 << 22, (1 << 24) + 3])
def test_accumulate(ops, dev, n):
    a = torch.rand(n, device=dev)
    b = torch.rand(n, device=dev)
    ref = a + b
    ops.accumulate(a, b)
    torch.cuda.synchronize()
    assert torch.equal(a, ref)  # fp32 add is exact vs torch's fp32 add


@pytest.mark.parametrize("nbytes", [16, 4096, (1 << 22) + 36])
def test_copy_kernel(ops, dev, nbytes):
    n = nbytes // 4
    src = torch.rand(n, device=dev)
    dst = torch.empty_like(src)
    ops.copy_kernel(dst, src)
    torch.cuda.synchronize()
    assert torch.equal(dst, src)


def test_copy_kernel_unaligned_tail(ops, dev):
    base = torch.rand(1029, device=dev)  # odd length -> 16B body + tail
    dst = torch.empty_like(base)
    ops.copy_kernel(dst, base)
    torch.cuda.synchronize()
    assert torch.equal(dst, base)


@pytest.mark.parametrize("n", [1, 1000, 1 << 24])
def test_reduce_sum_exact_iota(ops, dev, n):
    t = torch.empty(n, dtype=torch.float32, device=dev)
    ops.iota(t)
    got = ops.reduce_sum(t)
    expected = ops.iota_checksum(n)
    assert got == expected  # exact: integer-valued addends in double


def test_reduce_sum_vs_torch(ops, dev):
    t = torch.rand(1 << 20, device=dev)
    got = ops.reduce_sum(t)
    ref = float(t.to(torch.float64).sum())
    assert abs(got - ref) < 1e-6 * max(abs(ref), 1.0)


def test_busy_wait_runs_and_scales(ops, dev):
    import time

    out = torch.empty(256, dtype=torch.float32, device=dev)
    # correctness: writes a finite value everywhere in [0, globalsize)
    ops.busy_wait(out, tripcount=10, globalsize=256)
    torch.cuda.synchronize()
    assert torch.isfinite(out).all()

    def timed(trip):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        ops.busy_wait(out, tripcount=trip, globalsize=256)
        torch.cuda.synchronize()
        return time.perf_counter() - t0

    timed(1000)  # warm
    t1 = min(timed(2000) for _ in range(3))
    t4 = min(timed(8000) for _ in range(3))
    # linearity within 2x slack (the autotuner's model)
    assert 2.0 < t4 / t1 < 8.0, (t1, t4)


def test_busy_wait_mfma_runs(ops, dev):
    out = torch.empty(256, dtype=torch.float32, device=dev)
    ops.busy_wait_mfma(out, tripcount=1000, n_waves=4)
    torch.cuda.synchronize()
    assert torch.isfinite(out).all()


def test_ops_reject_wrong_dtype(ops, dev):
    t = torch.empty(8, dtype=torch.float64, device=dev)
    with pytest.raises(TypeError):
        ops.fill(t, 0.0)


def test_interop_torch_allocator_shared_with_hip_kernels(ops, dev):
    """The suite's runtime-interop capability (reference
    sycl_omp_ze_interopt): torch's caching allocator, torch streams and the
    raw HIP kernels drive the same memory with no copies."""
    s = torch.cuda.Stream()
    t = torch.zeros(1 << 16, dtype=torch.float32, device=dev)
    with torch.cuda.stream(s):
        ops.fill(t, 7.0, stream=s)        # native kernel on torch stream
        u = t * 2.0                        # torch op, same stream, same memory
    s.synchronize()
    assert torch.equal(u, torch.full_like(t, 14.0))


@pytest.mark.parametrize("unroll,cap", [(1, 4096), (4, 65536)])
def test_copy_kernel_tuned(ops, dev, unroll, cap):
    from hpc_patterns_amd._native import native

    hpk = native()
    n = (1 << 20) + 5
    src = torch.rand(n, device=dev)
    dst = torch.zeros_like(src)
    torch.cuda.synchronize()
    hpk.copy_kernel_tuned(dst.data_ptr(), src.data_ptr(), n * 4,
                          torch.cuda.current_stream().cuda_stream, unroll, cap)
    torch.cuda.synchronize()
    assert torch.equal(dst, src)


def test_roctx_markers_noop_safe(ops, dev):
    from hpc_patterns_amd._native import native

    hpk = native()
    hpk.trace_push("test_range")
    hpk.trace_mark("test_mark")
    hpk.trace_pop()


def test_sdma_explicit_engine_copy(ops, dev):
    """Explicit SDMA-engine copies: correctness + handle lifecycle."""
    from hpc_patterns_amd._native import native

    hpk = native()
    n_engines = hpk.sdma_num_engines(0)
    assert n_engines >= 1, "MI355X should expose SDMA engines"
    nbytes = 32 << 20
    host = hpk.host_malloc(nbytes)
    dev_buf = hpk.hip_malloc(nbytes)
    host2 = hpk.host_malloc(nbytes)
    # pattern into host via a device round-trip
    t = torch.arange(nbytes // 4, dtype=torch.float32, device=dev)
    torch.cuda.synchronize()
    hpk.memcpy_async(host, t.data_ptr(), nbytes, 0)
    hpk.stream_synchronize(0)
    # H2D on engine 0, D2H on engine 1 (or 0 if single-engine)
    h1 = hpk.sdma_copy_begin(dev_buf, host, nbytes, 0, 0)
    hpk.sdma_wait(h1)
    h2 = hpk.sdma_copy_begin(host2, dev_buf, nbytes, 0,
                             1 if n_engines >= 2 else 0)
    hpk.sdma_wait(h2)
    out = torch.empty_like(t)
    hpk.memcpy_async(out.data_ptr(), host2, nbytes, 0)
    hpk.stream_synchronize(0)
    assert torch.equal(out, t)
    for p in (host, host2):
        hpk.host_free(p)
    hpk.hip_free(dev_buf)


def test_fill_large_nontemporal_path(ops, dev):
    # >32 MB exercises the nontemporal-store fill variant
    n = (64 << 20) // 4 + 3
    t = torch.empty(n, dtype=torch.float32, device=dev)
    ops.fill(t, -2.25)
    torch.cuda.synchronize()
    assert torch.all(t == -2.25)


def test_copy_large_nontemporal_path(ops, dev):
    n = (64 << 20) // 4 + 5
    src = torch.rand(n, device=dev)
    dst = torch.zeros_like(src)
    ops.copy_kernel(dst, src)
    torch.cuda.synchronize()
    assert torch.equal(dst, src)


def test_sdma_engine_pair_query(ops, dev):
    from hpc_patterns_amd._native import native

    hpk = native()
    # same-device pair: must not crash; >=0 engines
    n = hpk.sdma_num_engines_pair(0, 0)
    assert n >= 0


def test_accumulate_nt_variant(ops, dev):
    from hpc_patterns_amd._native import native

    hpk = native()
    n = (48 << 20) // 4  # beyond-L3 class size, 16B aligned
    a = torch.rand(n, device=dev)
    b = torch.rand(n, device=dev)
    ref = a + b
    torch.cuda.synchronize()
    hpk.acc_f32_nt(a.data_ptr(), b.data_ptr(), n,
                   torch.cuda.current_stream().cuda_stream)
    torch.cuda.synchronize()
    assert torch.equal(a, ref)


@pytest.mark.parametrize("mb", [40, 64])
def test_nt_kernels_coherence_stress(ops, dev, mb):
    """Hunt cache-hint staleness in the NT performance kernels: write fresh
    data with NORMAL stores (dirty L2), immediately run the NT copy /
    accumulate on it, verify exactly. Repeated at L2-window sizes — if NT
    loads/stores bypassed coherence this flakes (cf. the reverted NT
    reduction, profiles/README r24 note)."""
    from hpc_patterns_amd._native import native

    hpk = native()
    n = (mb << 20) // 4
    s = torch.cuda.current_stream().cuda_stream
    for trial in range(4):
        src = torch.rand(n, device=dev)          # normal stores
        dst = torch.zeros(n, device=dev)         # normal stores (dirty zeros)
        torch.cuda.synchronize()
        ops.copy_kernel(dst, src)                # NT path at these sizes
        torch.cuda.synchronize()
        assert torch.equal(dst, src), f"NT copy stale data (trial {trial})"

        a = torch.rand(n, device=dev)
        b = torch.rand(n, device=dev)
        ref = a + b
        torch.cuda.synchronize()
        hpk.acc_f32_nt(a.data_ptr(), b.data_ptr(), n, s)
        torch.cuda.synchronize()
        assert torch.equal(a, ref), f"NT accumulate stale data (trial {trial})"
        del src, dst, a, b, ref


def test_staged_pageable_copy_roundtrip(ops, dev):
    """Pipelined pinned-staging copies for pageable memory (staged.hip):
    pageable->device on engine 0, device->pageable on engine 1, exact."""
    import numpy as np

    from hpc_patterns_amd._native import native

    hpk = native()
    n = (40 << 20) // 4 + 13  # 40MB + odd tail, several 8MB chunks
    src = np.random.rand(n).astype(np.float32)  # pageable numpy memory
    out = np.zeros_like(src)
    dbuf = hpk.hip_malloc(n * 4)
    hpk.staged_copy(dbuf, src.ctypes.data, n * 4, 0, 0, True)
    hpk.staged_copy(out.ctypes.data, dbuf, n * 4, 0,
                    1 if hpk.sdma_num_engines(0) >= 2 else 0, False)
    assert np.array_equal(out, src)
    hpk.hip_free(dbuf)


@pytest.mark.parametrize("nbytes", [1024, 8 << 20, (8 << 20) + 4,
                                    (16 << 20) - 4, 65 << 20])
def test_staged_copy_chunk_boundaries(ops, dev, nbytes):
    """staged.hip chunk/thread-split logic at boundary sizes (8 MiB chunks,
    4-way threading beyond 64 MiB)."""
    import numpy as np

    from hpc_patterns_amd._native import native

    hpk = native()
    n = nbytes // 4
    src = np.random.rand(n).astype(np.float32)
    out = np.zeros_like(src)
    dbuf = hpk.hip_malloc(nbytes)
    hpk.staged_copy(dbuf, src.ctypes.data, nbytes, 0, 0, True)
    hpk.staged_copy(out.ctypes.data, dbuf, nbytes, 0, 0, False)
    assert np.array_equal(out, src), nbytes
    hpk.hip_free(dbuf)


# ---------------------------------------------------------------------------
# K7 (r2): LDS-tiled bf16 MFMA GEMM
# ---------------------------------------------------------------------------

def test_gemm_bf16_identity_asymmetric():
    """A = I with an ASYMMETRIC B catches any row/col-swapped fragment or
    epilogue mapping (the guide's A=I-check): C = I @ B^T = B^T exactly."""
    from hpc_patterns_amd import ops

    torch.manual_seed(7)
    m = n = k = 128
    a = torch.eye(m, k, device="cuda").to(torch.bfloat16)
    b = torch.randn(n, k, device="cuda").to(torch.bfloat16)
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_bf16(c, a, b)
    torch.cuda.synchronize()
    assert torch.equal(c, b.t().float()), (c - b.t().float()).abs().max()


def test_gemm_bf16_exact_integers():
    """Small-integer payloads make every product and partial sum exactly
    representable in fp32, so the MFMA result must EQUAL the torch fp32
    reference bit for bit regardless of summation order."""
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(11)
    m, n, k = 256, 384, 512
    a = torch.randint(-4, 5, (m, k), generator=g).to(torch.bfloat16).cuda()
    b = torch.randint(-4, 5, (n, k), generator=g).to(torch.bfloat16).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_bf16(c, a, b)
    ref = torch.matmul(a.float(), b.float().t())
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()


@pytest.mark.parametrize("swizzle", [True, False])
def test_gemm_bf16_large_close(swizzle):
    """1024^3 random normal: products of bf16 values are exact in fp32, so
    the only divergence from the torch fp32 reference is summation order —
    tight allclose."""
    from hpc_patterns_amd import ops

    torch.manual_seed(3)
    m = n = k = 1024
    a = torch.randn(m, k, device="cuda").to(torch.bfloat16)
    b = torch.randn(n, k, device="cuda").to(torch.bfloat16)
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_bf16(c, a, b, xcd_swizzle=swizzle)
    ref = torch.matmul(a.float(), b.float().t())
    torch.cuda.synchronize()
    assert torch.allclose(c, ref, rtol=1e-4, atol=1e-3), \
        (c - ref).abs().max()


def test_gemm_bf16_shape_guards():
    from hpc_patterns_amd import ops

    a = torch.zeros(128, 64, dtype=torch.bfloat16, device="cuda")
    b = torch.zeros(128, 64, dtype=torch.bfloat16, device="cuda")
    c = torch.zeros(128, 128, dtype=torch.float32, device="cuda")
    ops.gemm_bf16(c, a, b)  # minimal legal shape
    with pytest.raises((RuntimeError, ValueError)):
        ops.gemm_bf16(c, a[:, :32].contiguous(), b[:, :32].contiguous())
    with pytest.raises(TypeError):
        ops.gemm_bf16(c, a.float(), b)


def test_gemm_bf16_db_variant_exact():
    """HPK_GEMM_VARIANT=db (double-buffered LDS + raw barriers + counted
    vmcnt) must be bitwise-identical to the reference on integer payloads
    — race-screened 30x at two shapes in profiles/gemm_db_r2.log; this
    keeps one exact check in CI."""
    import os

    from hpc_patterns_amd import ops

    os.environ["HPK_GEMM_VARIANT"] = "db"
    try:
        g = torch.Generator(device="cpu").manual_seed(23)
        m, n, k = 256, 256, 640
        a = torch.randint(-4, 5, (m, k), generator=g).to(torch.bfloat16).cuda()
        b = torch.randint(-4, 5, (n, k), generator=g).to(torch.bfloat16).cuda()
        c = torch.empty(m, n, dtype=torch.float32, device="cuda")
        ops.gemm_bf16(c, a, b)
        ref = torch.matmul(a.float(), b.float().t())
        torch.cuda.synchronize()
        assert torch.equal(c, ref)
    finally:
        del os.environ["HPK_GEMM_VARIANT"]


def test_gemm_fp8_exact_integers():
    """fp8 e4m3 twin: {-4..4} are exactly representable in e4m3, so the
    result must EQUAL the torch fp32 reference bitwise."""
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(17)
    m, n, k = 256, 384, 512
    a = torch.randint(-4, 5, (m, k), generator=g).to(torch.float8_e4m3fn).cuda()
    b = torch.randint(-4, 5, (n, k), generator=g).to(torch.float8_e4m3fn).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_fp8(c, a, b)
    ref = torch.matmul(a.float(), b.float().t())
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()


def test_gemm_fp8_identity_asymmetric():
    from hpc_patterns_amd import ops

    torch.manual_seed(5)
    m = n = k = 128
    a = torch.eye(m, k, device="cuda").to(torch.float8_e4m3fn)
    b = (torch.randint(-4, 5, (n, k)).float()).to(torch.float8_e4m3fn).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_fp8(c, a, b)
    torch.cuda.synchronize()
    assert torch.equal(c, b.float().t())


def test_gemm_bf16_8phase_exact_integers():
    """The deep-pipelined 256^2 8-phase kernel (default for shapes with
    M,N % 256 == 0 and K % 128 == 0): bitwise equality on integer payloads
    at an eligible shape — plus the identity check at 256^2."""
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(29)
    m, n, k = 512, 256, 640
    a = torch.randint(-3, 4, (m, k), generator=g).to(torch.bfloat16).cuda()
    b = torch.randint(-3, 4, (n, k), generator=g).to(torch.bfloat16).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_bf16(c, a, b)
    ref = torch.matmul(a.float(), b.float().t())
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()
    ai = torch.eye(256, 256, device="cuda").to(torch.bfloat16)
    bi = torch.randn(256, 256, device="cuda").to(torch.bfloat16)
    ci = torch.empty(256, 256, dtype=torch.float32, device="cuda")
    ops.gemm_bf16(ci, ai, bi)
    torch.cuda.synchronize()
    assert torch.equal(ci, bi.t().float())


def test_gemm_fp8_8phase_exact_integers(monkeypatch):
    """fp8 8-phase pipeline (pinned via HPK_GEMM_VARIANT — 256-divisible
    shapes now default to the 32x32x64 kernel): bitwise equality."""
    from hpc_patterns_amd import ops

    monkeypatch.setenv("HPK_GEMM_VARIANT", "8ph")

    g = torch.Generator(device="cpu").manual_seed(37)
    m, n, k = 256, 512, 640
    a = torch.randint(-3, 4, (m, k), generator=g).to(torch.float8_e4m3fn).cuda()
    b = torch.randint(-3, 4, (n, k), generator=g).to(torch.float8_e4m3fn).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_fp8(c, a, b)
    ref = torch.matmul(a.float(), b.float().t())
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()


def _mx_ref(a, b, a_scale, b_scale):
    """fp32 reference of the MX semantics: dequant per 32-block, matmul."""
    af = a.float() * (2.0 ** (a_scale.float() - 127)).repeat_interleave(32, 1)
    bf = b.float() * (2.0 ** (b_scale.float() - 127)).repeat_interleave(32, 1)
    return torch.matmul(af, bf.t())


def test_gemm_mxfp8_unit_scales_exact():
    """All scales 127 (=1.0): must equal the plain-fp8 exact result."""
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(41)
    m, n, k = 128, 256, 384
    a = torch.randint(-4, 5, (m, k), generator=g).to(torch.float8_e4m3fn).cuda()
    b = torch.randint(-4, 5, (n, k), generator=g).to(torch.float8_e4m3fn).cuda()
    sa = torch.full((m, k // 32), 127, dtype=torch.uint8, device="cuda")
    sb = torch.full((n, k // 32), 127, dtype=torch.uint8, device="cuda")
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_mxfp8(c, a, b, sa, sb)
    ref = torch.matmul(a.float(), b.float().t())
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()


def test_gemm_mxfp8_power_of_two_scales_exact():
    """Random per-block power-of-2 scales: products stay exact in fp32, so
    the HW-dequant result must EQUAL the fp32 reference bitwise."""
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(43)
    m, n, k = 256, 128, 512
    a = torch.randint(-3, 4, (m, k), generator=g).to(torch.float8_e4m3fn).cuda()
    b = torch.randint(-3, 4, (n, k), generator=g).to(torch.float8_e4m3fn).cuda()
    sa = torch.randint(124, 131, (m, k // 32), generator=g,
                       dtype=torch.uint8).cuda()
    sb = torch.randint(124, 131, (n, k // 32), generator=g,
                       dtype=torch.uint8).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_mxfp8(c, a, b, sa, sb)
    ref = _mx_ref(a, b, sa, sb)
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()


def test_gemm_mxfp8_identity():
    from hpc_patterns_amd import ops

    m = n = k = 128
    a = torch.eye(m, k, device="cuda").to(torch.float8_e4m3fn)
    b = torch.randint(-4, 5, (n, k)).float().to(torch.float8_e4m3fn).cuda()
    sa = torch.full((m, k // 32), 127, dtype=torch.uint8, device="cuda")
    sb = torch.full((n, k // 32), 127, dtype=torch.uint8, device="cuda")
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_mxfp8(c, a, b, sa, sb)
    torch.cuda.synchronize()
    assert torch.equal(c, b.float().t())


def test_gemm_mxfp8_large_square_exact():
    """256-divisible shape with random power-of-2 scales (covers the tile
    shapes the removed deep-pipelined variant used to take)."""
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(47)
    m, n, k = 256, 256, 512
    a = torch.randint(-3, 4, (m, k), generator=g).to(torch.float8_e4m3fn).cuda()
    b = torch.randint(-3, 4, (n, k), generator=g).to(torch.float8_e4m3fn).cuda()
    sa = torch.randint(124, 131, (m, k // 32), generator=g,
                       dtype=torch.uint8).cuda()
    sb = torch.randint(124, 131, (n, k // 32), generator=g,
                       dtype=torch.uint8).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_mxfp8(c, a, b, sa, sb)
    ref = _mx_ref(a, b, sa, sb)
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()


def test_gemm_i8_exact():
    """int8 GEMM: int32 accumulate is exact by construction — full-range
    payloads must EQUAL the int64 torch reference."""
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(53)
    m, n, k = 256, 384, 512
    a = torch.randint(-128, 128, (m, k), generator=g, dtype=torch.int8).cuda()
    b = torch.randint(-128, 128, (n, k), generator=g, dtype=torch.int8).cuda()
    c = torch.empty(m, n, dtype=torch.int32, device="cuda")
    ops.gemm_i8(c, a, b)
    # torch has no int CUDA matmul — exact reference on host
    ref = torch.matmul(a.cpu().long(), b.cpu().long().t()).to(torch.int32).cuda()
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()


def test_gemm_i8_identity():
    from hpc_patterns_amd import ops

    m = n = k = 128
    a = torch.eye(m, k).to(torch.int8).cuda()
    b = torch.randint(-5, 6, (n, k), dtype=torch.int8).cuda()
    c = torch.empty(m, n, dtype=torch.int32, device="cuda")
    ops.gemm_i8(c, a, b)
    torch.cuda.synchronize()
    assert torch.equal(c, b.int().t())


def test_gemm_i8_8phase_exact(monkeypatch):
    """The 8-phase 16x16x64 i8 pipeline (pinned via HPK_GEMM_VARIANT —
    256-divisible shapes now default to the 32x32x32 kernel)."""
    from hpc_patterns_amd import ops

    monkeypatch.setenv("HPK_GEMM_VARIANT", "8ph")

    g = torch.Generator(device="cpu").manual_seed(61)
    m, n, k = 256, 512, 512
    ah = torch.randint(-128, 128, (m, k), generator=g, dtype=torch.int8)
    bh = torch.randint(-128, 128, (n, k), generator=g, dtype=torch.int8)
    c = torch.empty(m, n, dtype=torch.int32, device="cuda")
    ops.gemm_i8(c, ah.cuda(), bh.cuda())
    ref = torch.matmul(ah.long(), bh.long().t()).to(torch.int32).cuda()
    torch.cuda.synchronize()
    assert torch.equal(c, ref)


# ---------------------------------------------------------------------------
# matmul_nt — arbitrary-shape front door (zero-padding dispatch)
# ---------------------------------------------------------------------------

def test_matmul_nt_bf16_odd_shapes_exact():
    """Odd (non-tile-multiple) shapes through the padding path: small
    integers keep everything exact, so the sliced result must EQUAL the
    torch fp32 reference."""
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(71)
    for m, n, k in ((300, 520, 736), (1, 1, 64), (257, 129, 192)):
        a = torch.randint(-4, 5, (m, k), generator=g).to(torch.bfloat16).cuda()
        b = torch.randint(-4, 5, (n, k), generator=g).to(torch.bfloat16).cuda()
        c = ops.matmul_nt(a, b)
        ref = torch.matmul(a.float(), b.float().t())
        torch.cuda.synchronize()
        assert c.shape == (m, n) and c.dtype == torch.float32
        assert torch.equal(c, ref), (m, n, k)


def test_matmul_nt_i8_odd_shapes_exact():
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(73)
    m, n, k = 100, 300, 700
    ah = torch.randint(-128, 128, (m, k), generator=g, dtype=torch.int8)
    bh = torch.randint(-128, 128, (n, k), generator=g, dtype=torch.int8)
    c = ops.matmul_nt(ah.cuda(), bh.cuda())
    ref = torch.matmul(ah.long(), bh.long().t()).to(torch.int32).cuda()
    torch.cuda.synchronize()
    assert c.shape == (m, n) and c.dtype == torch.int32
    assert torch.equal(c, ref)


def test_matmul_nt_fp8_and_mx_odd_shapes():
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(79)
    m, n, k = 200, 136, 160
    a = torch.randint(-4, 5, (m, k), generator=g).float() \
        .to(torch.float8_e4m3fn).cuda()
    b = torch.randint(-4, 5, (n, k), generator=g).float() \
        .to(torch.float8_e4m3fn).cuda()
    ref = torch.matmul(a.float(), b.float().t())
    c = ops.matmul_nt(a, b)
    torch.cuda.synchronize()
    assert torch.equal(c, ref)
    # mx path: unit scales (127 = 2^0) must reproduce the plain result
    s_a = torch.full((m, k // 32), 127, dtype=torch.uint8, device="cuda")
    s_b = torch.full((n, k // 32), 127, dtype=torch.uint8, device="cuda")
    cmx = ops.matmul_nt(a, b, a_scale=s_a, b_scale=s_b)
    torch.cuda.synchronize()
    assert torch.equal(cmx, ref)


def test_matmul_nt_aligned_no_copy():
    """Tile-multiple shapes skip padding entirely: result tensor IS the
    kernel's output buffer (no slice copy)."""
    from hpc_patterns_amd import ops

    torch.manual_seed(83)
    a = torch.randn(256, 128, device="cuda").to(torch.bfloat16)
    b = torch.randn(256, 128, device="cuda").to(torch.bfloat16)
    c = ops.matmul_nt(a, b)
    torch.cuda.synchronize()
    ref = torch.matmul(a.float(), b.float().t())
    assert c.is_contiguous() and c.shape == (256, 256)
    assert torch.allclose(c, ref, rtol=1e-3, atol=1e-3)


def test_gemm_grouped_order_exact(monkeypatch):
    """The L2-aware grouped tile order (HPK_GEMM_GROUP) is a pure launch
    reordering: results must stay bitwise identical, including grids whose
    tiles_n is not a multiple of the band width (partial last band)."""
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(89)
    m, n, k = 768, 1792, 512  # 8ph grid: 3 x 7 tiles
    a = torch.randint(-4, 5, (m, k), generator=g).to(torch.bfloat16).cuda()
    b = torch.randint(-4, 5, (n, k), generator=g).to(torch.bfloat16).cuda()
    ref = torch.matmul(a.float(), b.float().t())
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    for grp in ("2", "4", "8", "16", "64"):
        monkeypatch.setenv("HPK_GEMM_GROUP", grp)
        ops.gemm_bf16(c, a, b)
        torch.cuda.synchronize()
        assert torch.equal(c, ref), grp


def test_gemm_mxfp8_grouped_order_exact(monkeypatch):
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(97)
    m, n, k = 384, 1152, 256  # 128-tile grid: 3 x 9
    a = torch.randint(-4, 5, (m, k), generator=g).float() \
        .to(torch.float8_e4m3fn).cuda()
    b = torch.randint(-4, 5, (n, k), generator=g).float() \
        .to(torch.float8_e4m3fn).cuda()
    sa = torch.full((m, k // 32), 127, dtype=torch.uint8, device="cuda")
    sb = torch.full((n, k // 32), 127, dtype=torch.uint8, device="cuda")
    ref = torch.matmul(a.float(), b.float().t())
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    for grp in ("4", "16", "32"):
        monkeypatch.setenv("HPK_GEMM_GROUP", grp)
        ops.gemm_mxfp8(c, a, b, sa, sb)
        torch.cuda.synchronize()
        assert torch.equal(c, ref), grp


# ---------------------------------------------------------------------------
# K7-mx4: block-scaled OCP MX-fp4 GEMM (4x rate class)
# ---------------------------------------------------------------------------

def _mx4_operands(m, n, k, seed):
    """Random exactly-representable e2m1 payloads + packed forms."""
    g = torch.Generator(device="cpu").manual_seed(seed)
    from hpc_patterns_amd import ops

    vals = torch.tensor([0., 0.5, -0.5, 1., -1., 1.5, -1.5, 2., -2.])
    fa = vals[torch.randint(0, 9, (m, k), generator=g)]
    fb = vals[torch.randint(0, 9, (n, k), generator=g)]
    return fa, fb, ops.e2m1_pack(fa).cuda(), ops.e2m1_pack(fb).cuda()


def test_gemm_mxfp4_unit_scales_exact():
    """Products/sums of e2m1 values at these magnitudes are exact in fp32:
    the result must EQUAL the fp32 reference bitwise."""
    from hpc_patterns_amd import ops

    m, n, k = 128, 256, 384
    fa, fb, pa, pb = _mx4_operands(m, n, k, 113)
    sa = torch.full((m, k // 32), 127, dtype=torch.uint8, device="cuda")
    sb = torch.full((n, k // 32), 127, dtype=torch.uint8, device="cuda")
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_mxfp4(c, pa, pb, sa, sb)
    ref = torch.matmul(fa, fb.t()).cuda()
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()


def test_gemm_mxfp4_identity():
    """A = packed identity with an ASYMMETRIC B: C = B^T exactly."""
    from hpc_patterns_amd import ops

    m = n = k = 128
    fa = torch.eye(m, k)
    g = torch.Generator(device="cpu").manual_seed(127)
    vals = torch.tensor([0., 0.5, -0.5, 1., -1., 1.5, -1.5, 2., -2.])
    fb = vals[torch.randint(0, 9, (n, k), generator=g)]
    pa, pb = ops.e2m1_pack(fa).cuda(), ops.e2m1_pack(fb).cuda()
    sa = torch.full((m, k // 32), 127, dtype=torch.uint8, device="cuda")
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_mxfp4(c, pa, pb, sa, sa.clone())
    torch.cuda.synchronize()
    assert torch.equal(c, fb.t().cuda()), (c - fb.t().cuda()).abs().max()


def test_gemm_mxfp4_power_of_two_scales_exact():
    """Random e8m0 scales near 127: dequant is exact power-of-2 scaling,
    still bitwise-equal to the fp32 reference."""
    from hpc_patterns_amd import ops

    m, n, k = 256, 128, 256
    fa, fb, pa, pb = _mx4_operands(m, n, k, 131)
    g = torch.Generator(device="cpu").manual_seed(137)
    sa = torch.randint(124, 131, (m, k // 32), generator=g,
                       dtype=torch.int16).to(torch.uint8)
    sb = torch.randint(124, 131, (n, k // 32), generator=g,
                       dtype=torch.int16).to(torch.uint8)
    da = fa * torch.pow(2.0, sa.float() - 127).repeat_interleave(32, dim=1)
    db = fb * torch.pow(2.0, sb.float() - 127).repeat_interleave(32, dim=1)
    ref = torch.matmul(da, db.t()).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_mxfp4(c, pa, pb, sa.cuda(), sb.cuda())
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()


def test_gemm_mxfp4_large_square_exact():
    from hpc_patterns_amd import ops

    m = n = k = 1024
    fa, fb, pa, pb = _mx4_operands(m, n, k, 139)
    sa = torch.full((m, k // 32), 127, dtype=torch.uint8, device="cuda")
    sb = torch.full((n, k // 32), 127, dtype=torch.uint8, device="cuda")
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_mxfp4(c, pa, pb, sa, sb)
    ref = torch.matmul(fa, fb.t()).cuda()
    torch.cuda.synchronize()
    assert torch.equal(c, ref)


def test_gemm_mxfp4_grouped_order_exact(monkeypatch):
    from hpc_patterns_amd import ops

    m, n, k = 384, 1152, 256  # 3 x 9 tile grid (partial bands)
    fa, fb, pa, pb = _mx4_operands(m, n, k, 149)
    sa = torch.full((m, k // 32), 127, dtype=torch.uint8, device="cuda")
    sb = torch.full((n, k // 32), 127, dtype=torch.uint8, device="cuda")
    ref = torch.matmul(fa, fb.t()).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    for grp in ("4", "16", "32"):
        monkeypatch.setenv("HPK_GEMM_GROUP", grp)
        ops.gemm_mxfp4(c, pa, pb, sa, sb)
        torch.cuda.synchronize()
        assert torch.equal(c, ref), grp


def test_gemm_mxfp4_256tile_power_of_two_scales_exact():
    """256-divisible shapes take the 256^2-tile 32x32x64 kernel (the
    default fast path): full exactness incl. its size-4 glds scale
    staging with non-uniform scales."""
    from hpc_patterns_amd import ops

    m, n, k = 256, 512, 384
    fa, fb, pa, pb = _mx4_operands(m, n, k, 151)
    g = torch.Generator(device="cpu").manual_seed(157)
    sa = torch.randint(124, 131, (m, k // 32), generator=g,
                       dtype=torch.int16).to(torch.uint8)
    sb = torch.randint(124, 131, (n, k // 32), generator=g,
                       dtype=torch.int16).to(torch.uint8)
    da = fa * torch.pow(2.0, sa.float() - 127).repeat_interleave(32, dim=1)
    db = fb * torch.pow(2.0, sb.float() - 127).repeat_interleave(32, dim=1)
    ref = torch.matmul(da, db.t()).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_mxfp4(c, pa, pb, sa.cuda(), sb.cuda())
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()


def test_matmul_nt_mxfp4_odd_shapes_exact():
    """Packed-fp4 through the matmul_nt padding path: zero nibbles pad
    exactly, scales pad with 127."""
    from hpc_patterns_amd import ops

    m, n, k = 200, 136, 192  # odd vs the 256/128 fast-path multiples
    fa, fb, pa, pb = _mx4_operands(m, n, k, 167)
    sa = torch.full((m, k // 32), 127, dtype=torch.uint8, device="cuda")
    sb = torch.full((n, k // 32), 127, dtype=torch.uint8, device="cuda")
    c = ops.matmul_nt(pa, pb, a_scale=sa, b_scale=sb)
    ref = torch.matmul(fa, fb.t()).cuda()
    torch.cuda.synchronize()
    assert c.shape == (m, n) and torch.equal(c, ref)


def test_gemm_mxfp8_256tile_power_of_two_scales_exact():
    """256-divisible shapes take the 256^2-tile 32x32x64 mx8 kernel (the
    r2-follow-up default): full exactness incl. non-uniform scales."""
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(173)
    m, n, k = 256, 512, 384
    fa = torch.randint(-4, 5, (m, k), generator=g).float()
    fb = torch.randint(-4, 5, (n, k), generator=g).float()
    a = fa.to(torch.float8_e4m3fn).cuda()
    b = fb.to(torch.float8_e4m3fn).cuda()
    sa = torch.randint(124, 131, (m, k // 32), generator=g,
                       dtype=torch.int16).to(torch.uint8)
    sb = torch.randint(124, 131, (n, k // 32), generator=g,
                       dtype=torch.int16).to(torch.uint8)
    da = fa * torch.pow(2.0, sa.float() - 127).repeat_interleave(32, dim=1)
    db = fb * torch.pow(2.0, sb.float() - 127).repeat_interleave(32, dim=1)
    ref = torch.matmul(da, db.t()).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_mxfp8(c, a, b, sa.cuda(), sb.cuda())
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()


def test_gemm_fp8_32x32_default_exact():
    """Plain fp8 at 256-divisible shapes defaults to the 256^2 32x32x64
    scaled-MFMA kernel with hardcoded x1.0 scales: bitwise equality."""
    from hpc_patterns_amd import ops

    g = torch.Generator(device="cpu").manual_seed(179)
    m, n, k = 256, 512, 640
    a = torch.randint(-3, 4, (m, k), generator=g).to(torch.float8_e4m3fn).cuda()
    b = torch.randint(-3, 4, (n, k), generator=g).to(torch.float8_e4m3fn).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_fp8(c, a, b)
    ref = torch.matmul(a.float(), b.float().t())
    torch.cuda.synchronize()
    assert torch.equal(c, ref), (c - ref).abs().max()


def test_gemm_i8_32x32_variant_exact(monkeypatch):
    """The selectable 256^2 32x32x32 i8 kernel (HPK_GEMM_VARIANT=32 — a
    measured perf negative, kept for the design-space record): full-range
    int8, exact int32, bitwise vs the int64 host reference."""
    from hpc_patterns_amd import ops

    monkeypatch.setenv("HPK_GEMM_VARIANT", "32")

    g = torch.Generator(device="cpu").manual_seed(181)
    m, n, k = 256, 512, 384
    ah = torch.randint(-128, 128, (m, k), generator=g, dtype=torch.int8)
    bh = torch.randint(-128, 128, (n, k), generator=g, dtype=torch.int8)
    c = torch.empty(m, n, dtype=torch.int32, device="cuda")
    ops.gemm_i8(c, ah.cuda(), bh.cuda())
    ref = torch.matmul(ah.long(), bh.long().t()).to(torch.int32).cuda()
    torch.cuda.synchronize()
    assert torch.equal(c, ref)


def test_gemm_mxfp4_32h_variant_exact(monkeypatch):
    """The 256x128 occupancy-experiment kernel (HPK_MX4_WAVES=32h, a
    measured perf negative kept as the design-space record): exactness
    with random power-of-2 scales."""
    from hpc_patterns_amd import ops

    monkeypatch.setenv("HPK_MX4_WAVES", "32h")
    m, n, k = 512, 384, 256
    fa, fb, pa, pb = _mx4_operands(m, n, k, 191)
    g = torch.Generator(device="cpu").manual_seed(193)
    sa = torch.randint(124, 131, (m, k // 32), generator=g,
                       dtype=torch.int16).to(torch.uint8)
    sb = torch.randint(124, 131, (n, k // 32), generator=g,
                       dtype=torch.int16).to(torch.uint8)
    da = fa * torch.pow(2.0, sa.float() - 127).repeat_interleave(32, dim=1)
    db = fb * torch.pow(2.0, sb.float() - 127).repeat_interleave(32, dim=1)
    ref = torch.matmul(da, db.t()).cuda()
    c = torch.empty(m, n, dtype=torch.float32, device="cuda")
    ops.gemm_mxfp4(c, pa, pb, sa.cuda(), sb.cuda())
    torch.cuda.synchronize()
    assert torch.equal(c, ref)
