"""ops — hand-written HIP/CDNA4 kernels exposed on torch tensors.

Python face of the native kernels in native/kernels.hip (the MI355X
equivalents of the reference's device-code sites). All
functions launch on the CURRENT torch CUDA stream unless given one, and all
REQUIRE the native extension when a GPU is present — no silent eager
fallback.
"""

from __future__ import annotations

import torch

from .._native import native


def _stream_handle(stream=None) -> int:
    if stream is not None:
        return stream.cuda_stream if hasattr(stream, "cuda_stream") else int(stream)
    return torch.cuda.current_stream().cuda_stream


def _check_f32(t: torch.Tensor, name: str):
    if t.dtype != torch.float32 or not t.is_cuda or not t.is_contiguous():
        raise TypeError(f"{name} must be a contiguous float32 CUDA tensor")


def busy_wait(out: torch.Tensor, tripcount: int, globalsize: int | None = None,
              stream=None) -> None:
    """K1: every work-item runs 64*tripcount dependent FMAs, stores 1 float.

    Reference payload: concurency/bench.hpp:23-31 (MAD_64) submitted at
    bench_sycl.cpp:92-97.
    """
    _check_f32(out, "out")
    gs = out.numel() if globalsize is None else globalsize
    if gs > out.numel():
        raise ValueError("globalsize exceeds out buffer")
    native().busy_wait(out.data_ptr(), int(tripcount), int(gs),
                       _stream_handle(stream))


def busy_wait_mfma(out: torch.Tensor, tripcount: int, n_waves: int = 1,
                   stream=None) -> None:
    """K1-MFMA: chained v_mfma_f32_16x16x32_bf16 busy loop (matrix cores)."""
    _check_f32(out, "out")
    if out.numel() < ((n_waves * 64 + 255) // 256) * 256:
        raise ValueError("out too small: need >= ceil(n_waves*64/256)*256 floats")
    native().busy_wait_mfma(out.data_ptr(), int(tripcount), int(n_waves),
                            _stream_handle(stream))


def copy_kernel(dst: torch.Tensor, src: torch.Tensor, stream=None) -> None:
    """K2: vectorized shader copy (the blit-engine sibling of hipMemcpyAsync)."""
    if dst.numel() * dst.element_size() != src.numel() * src.element_size():
        raise ValueError("size mismatch")
    if not (dst.is_contiguous() and src.is_contiguous()):
        raise TypeError("contiguous tensors required")
    if not (dst.is_cuda and src.is_cuda):
        raise TypeError("dst and src must be CUDA tensors (a host pointer "
                        "would crash the GPU copy kernel)")
    native().copy_kernel(dst.data_ptr(), src.data_ptr(),
                         dst.numel() * dst.element_size(),
                         _stream_handle(stream))


def fill(dst: torch.Tensor, value: float, stream=None) -> None:
    """K4: dst[:] = value (reference Initialize, allreduce-mpi-sycl.cpp:34-41)."""
    _check_f32(dst, "dst")
    native().fill_f32(dst.data_ptr(), float(value), dst.numel(),
                      _stream_handle(stream))


def iota(dst: torch.Tensor, stream=None) -> None:
    """dst[i] = float(i) — device-side checksum payload generator."""
    _check_f32(dst, "dst")
    native().iota_f32(dst.data_ptr(), dst.numel(), _stream_handle(stream))


def accumulate(dst: torch.Tensor, src: torch.Tensor, stream=None) -> None:
    """K3: dst += src (reference Accumulate, allreduce-mpi-sycl.cpp:27-31)."""
    _check_f32(dst, "dst")
    _check_f32(src, "src")
    if dst.numel() != src.numel():
        raise ValueError("size mismatch")
    native().acc_f32(dst.data_ptr(), src.data_ptr(), dst.numel(),
                     _stream_handle(stream))


def reduce_sum(src: torch.Tensor, stream=None) -> float:
    """Exact float64 sum of a float32 tensor (device reduction; synchronizes).

    Order-independent when all addends are integer-valued (every partial sum
    stays below 2^53), which is what the checksum payloads guarantee —
    replaces the reference's host sort+sum (peer2pear.cpp:56-63).
    """
    _check_f32(src, "src")
    return native().reduce_sum_f32(src.data_ptr(), src.numel(),
                                   _stream_handle(stream))


def iota_checksum(n: int) -> float:
    """Expected exact double sum of [float(i) for i in range(n)]."""
    import numpy as np

    # float32 rounding of the iota values, summed exactly in float64
    return float(np.arange(n, dtype=np.float32).astype(np.float64).sum())


def gemm_bf16(c: torch.Tensor, a: torch.Tensor, b: torch.Tensor,
              stream=None, xcd_swizzle: bool = False) -> None:
    """K7 (r2): C[M,N] fp32 = A[M,K] @ B[N,K]^T, A/B bf16 K-contiguous.

    Hand-written LDS-tiled v_mfma_f32_16x16x32_bf16 kernel (native/gemm.hip):
    zero-bank-conflict cyclic-skew LDS layout, 16-byte global_load_lds
    staging, and (for M,N % 256, K % 128 shapes) the 256^2-tile 8-phase
    counted-vmcnt pipeline selected by default — 1085 TF bf16 at 8192^3 on
    random operands (43% of the 2.5 PF dense peak;
    profiles/gemm_showcase_r2.log). HPK_GEMM_VARIANT=plain|db|8ph forces a
    variant. xcd_swizzle enables the bijective XCD workgroup remap
    (measured slower here, off by default). Requires M,N multiples of 128
    and K a multiple of 64 (use matmul_nt for arbitrary shapes).
    """
    if c.dtype != torch.float32 or a.dtype != torch.bfloat16 \
            or b.dtype != torch.bfloat16:
        raise TypeError("c must be fp32; a, b must be bf16")
    for t, name in ((c, "c"), (a, "a"), (b, "b")):
        if not t.is_cuda or not t.is_contiguous() or t.dim() != 2:
            raise TypeError(f"{name} must be a contiguous 2-D CUDA tensor")
    m, k = a.shape
    n, kb = b.shape
    if kb != k or c.shape != (m, n):
        raise ValueError(f"shape mismatch: A{tuple(a.shape)} B{tuple(b.shape)}"
                         f" C{tuple(c.shape)}")
    native().gemm_bf16_nt(c.data_ptr(), a.data_ptr(), b.data_ptr(),
                          m, n, k, _stream_handle(stream), int(xcd_swizzle))


def gemm_fp8(c: torch.Tensor, a: torch.Tensor, b: torch.Tensor,
             stream=None, xcd_swizzle: bool = False) -> None:
    """K7-fp8: C[M,N] fp32 = A[M,K] @ B[N,K]^T, A/B torch.float8_e4m3fn.

    gfx950 fp8 is OCP e4m3 — exactly torch's float8_e4m3fn. Default for
    256-divisible shapes: the 256^2-tile 32x32x64 scaled-MFMA kernel with
    hardcoded x1.0 scales (2154-2194 TF — no non-scaled 32x32x64 fp8 MFMA
    exists); otherwise the mfma_f32_16x16x32_fp8_fp8 family
    (HPK_GEMM_VARIANT=8ph|plain|db selects). Same shape constraints as
    gemm_bf16.
    """
    if c.dtype != torch.float32 or a.dtype != torch.float8_e4m3fn \
            or b.dtype != torch.float8_e4m3fn:
        raise TypeError("c must be fp32; a, b must be float8_e4m3fn")
    for t, name in ((c, "c"), (a, "a"), (b, "b")):
        if not t.is_cuda or not t.is_contiguous() or t.dim() != 2:
            raise TypeError(f"{name} must be a contiguous 2-D CUDA tensor")
    m, k = a.shape
    n, kb = b.shape
    if kb != k or c.shape != (m, n):
        raise ValueError(f"shape mismatch: A{tuple(a.shape)} B{tuple(b.shape)}"
                         f" C{tuple(c.shape)}")
    native().gemm_fp8_nt(c.data_ptr(), a.data_ptr(), b.data_ptr(),
                         m, n, k, _stream_handle(stream), int(xcd_swizzle))


def gemm_mxfp8(c: torch.Tensor, a: torch.Tensor, b: torch.Tensor,
               a_scale: torch.Tensor, b_scale: torch.Tensor,
               stream=None, xcd_swizzle: bool = False) -> None:
    """K7-mx: block-scaled MX-fp8 GEMM (the 2x-bf16 rate class;
    1824-1863 TF via the default 256^2-tile 32x32x64 kernel for
    256-divisible shapes, HPK_MX8_VARIANT=plain selects the 128^2 one).

    C[M,N] fp32 = (A * 2^(As-127)) @ (B * 2^(Bs-127))^T — A/B are
    float8_e4m3fn [M,K]/[N,K]; a_scale/b_scale are uint8 e8m0 exponents
    with ONE scale per 32-element K-block ([M,K//32] / [N,K//32]; 127 =
    scale 1.0) — the OCP MX-FP8 format, dequantized in hardware by
    mfma_scale_f32_16x16x128_f8f6f4. Requires M,N,K multiples of 128.
    """
    if c.dtype != torch.float32 or a.dtype != torch.float8_e4m3fn \
            or b.dtype != torch.float8_e4m3fn:
        raise TypeError("c must be fp32; a, b must be float8_e4m3fn")
    if a_scale.dtype != torch.uint8 or b_scale.dtype != torch.uint8:
        raise TypeError("scales must be uint8 (e8m0 exponents)")
    for t, name in ((c, "c"), (a, "a"), (b, "b"),
                    (a_scale, "a_scale"), (b_scale, "b_scale")):
        if not t.is_cuda or not t.is_contiguous() or t.dim() != 2:
            raise TypeError(f"{name} must be a contiguous 2-D CUDA tensor")
    m, k = a.shape
    n, kb = b.shape
    if kb != k or c.shape != (m, n):
        raise ValueError(f"shape mismatch: A{tuple(a.shape)} B{tuple(b.shape)}"
                         f" C{tuple(c.shape)}")
    if a_scale.shape != (m, k // 32) or b_scale.shape != (n, k // 32):
        raise ValueError("scales must be [rows, K//32]")
    native().gemm_mxfp8_nt(c.data_ptr(), a.data_ptr(), b.data_ptr(),
                           a_scale.data_ptr(), b_scale.data_ptr(),
                           m, n, k, _stream_handle(stream), int(xcd_swizzle))


# the 16 OCP e2m1 (fp4) values, indexed by nibble
_E2M1_VALUES = [0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0,
                -0.0, -0.5, -1.0, -1.5, -2.0, -3.0, -4.0, -6.0]


def e2m1_pack(t: torch.Tensor) -> torch.Tensor:
    """Pack a float tensor of exactly-representable e2m1 values
    (0, ±0.5, ±1, ±1.5, ±2, ±3, ±4, ±6) into uint8 nibbles, two per byte
    (low nibble = even index along the last dim). Raises on values
    outside the e2m1 set — this is a test/packing utility, not a
    quantizer. Last dim must be even."""
    if t.shape[-1] % 2 != 0:
        raise ValueError("last dim must be even")
    table = torch.tensor(_E2M1_VALUES[:8], dtype=torch.float32,
                         device=t.device)
    mag = t.abs().float()
    idx = (mag.unsqueeze(-1) == table).to(torch.uint8).argmax(-1)
    ok = table[idx] == mag
    if not bool(ok.all()):
        raise ValueError("tensor contains values not representable in e2m1")
    nib = (idx + 8 * (t < 0).to(idx.dtype)).to(torch.uint8)
    lo = nib[..., 0::2]
    hi = nib[..., 1::2]
    return (lo | (hi << 4)).contiguous()


def e2m1_decode(p: torch.Tensor) -> torch.Tensor:
    """Inverse of e2m1_pack: uint8 nibble-packed -> float32 (last dim
    doubles)."""
    table = torch.tensor(_E2M1_VALUES, dtype=torch.float32, device=p.device)
    lo = table[(p & 0xF).long()]
    hi = table[(p >> 4).long()]
    out = torch.stack([lo, hi], dim=-1)
    return out.view(*p.shape[:-1], p.shape[-1] * 2)


def gemm_mxfp4(c: torch.Tensor, a: torch.Tensor, b: torch.Tensor,
               a_scale: torch.Tensor, b_scale: torch.Tensor,
               stream=None, xcd_swizzle: bool = False) -> None:
    """K7-mx4: block-scaled OCP MX-fp4 GEMM — the 4x-bf16 MFMA rate class.

    C[M,N] fp32 = (A * 2^(As-127)) @ (B * 2^(Bs-127))^T where A/B are
    nibble-PACKED e2m1 uint8 tensors [M,K//2]/[N,K//2] (low nibble = even
    k; e2m1_pack produces this layout) and a_scale/b_scale are e8m0
    exponents [M,K//32]/[N,K//32] as for gemm_mxfp8. The fp4 mode of the
    scaled MFMAs; operand/scale layout measured on hardware (diagonal —
    scripts/probes/fp4_probe*). Default kernel: the 256^2-tile
    double-buffered 32x32x64 design, 2903-3197 TF at 8192^3-16384^3
    (3.2 PFLOP/s; HPK_MX4_WAVES=db|8|4 selects the 128^2 variants, used
    automatically when M or N is not a multiple of 256). K =
    2*a.shape[1]; requires M,N,K % 128 == 0.
    """
    if c.dtype != torch.float32 or a.dtype != torch.uint8 \
            or b.dtype != torch.uint8:
        raise TypeError("c must be fp32; a, b must be nibble-packed uint8")
    if a_scale.dtype != torch.uint8 or b_scale.dtype != torch.uint8:
        raise TypeError("scales must be uint8 (e8m0 exponents)")
    for t, name in ((c, "c"), (a, "a"), (b, "b"),
                    (a_scale, "a_scale"), (b_scale, "b_scale")):
        if not t.is_cuda or not t.is_contiguous() or t.dim() != 2:
            raise TypeError(f"{name} must be a contiguous 2-D CUDA tensor")
    m, kb = a.shape
    n, kb2 = b.shape
    k = kb * 2
    if kb2 != kb or c.shape != (m, n):
        raise ValueError(f"shape mismatch: A{tuple(a.shape)} B{tuple(b.shape)}"
                         f" C{tuple(c.shape)}")
    if a_scale.shape != (m, k // 32) or b_scale.shape != (n, k // 32):
        raise ValueError("scales must be [rows, K//32]")
    native().gemm_mxfp4_nt(c.data_ptr(), a.data_ptr(), b.data_ptr(),
                           a_scale.data_ptr(), b_scale.data_ptr(),
                           m, n, k, _stream_handle(stream), int(xcd_swizzle))


def gemm_i8(c: torch.Tensor, a: torch.Tensor, b: torch.Tensor,
            stream=None, xcd_swizzle: bool = False) -> None:
    """K7-i8: C[M,N] int32 = A[M,K] @ B[N,K]^T, int8 operands.

    mfma_i32_16x16x64_i8 at ~2x the bf16 MFMA rate with EXACT int32
    accumulation — the quantized-inference dtype. Same NT layout and
    shape constraints as gemm_bf16 (M,N % 128, K % 64).
    """
    if c.dtype != torch.int32 or a.dtype != torch.int8 or b.dtype != torch.int8:
        raise TypeError("c must be int32; a, b must be int8")
    for t, name in ((c, "c"), (a, "a"), (b, "b")):
        if not t.is_cuda or not t.is_contiguous() or t.dim() != 2:
            raise TypeError(f"{name} must be a contiguous 2-D CUDA tensor")
    m, k = a.shape
    n, kb = b.shape
    if kb != k or c.shape != (m, n):
        raise ValueError(f"shape mismatch: A{tuple(a.shape)} B{tuple(b.shape)}"
                         f" C{tuple(c.shape)}")
    native().gemm_i8_nt(c.data_ptr(), a.data_ptr(), b.data_ptr(),
                        m, n, k, _stream_handle(stream), int(xcd_swizzle))


def gemm_pad_shapes(kind: str, m: int, n: int, k: int):
    """Padded (M, N, K) that puts a (m, n, k) problem on the fast path.

    kind is "bf16" / "fp8" (8-phase needs M,N % 256, K % 128), "i8"
    (M,N % 256, K % 256), "mxfp8" (plain kernel only: M,N,K % 128) or
    "mxfp4" (the 256^2 32x32x64 kernel: M,N % 256, K % 128).
    Pure shape math — unit-tested on CPU (tests/test_gemm_skew_logic.py).
    """
    if kind not in ("bf16", "fp8", "i8", "mxfp8", "mxfp4"):
        raise ValueError(f"unknown gemm kind {kind!r}")
    def up(x, q):
        return -(-x // q) * q
    if kind == "mxfp8":
        return up(m, 128), up(n, 128), up(k, 128)
    kq = 256 if kind == "i8" else 128
    return up(m, 256), up(n, 256), up(k, kq)


def _pad2d(t: torch.Tensor, rows: int, cols: int, fill=0):
    if t.shape == (rows, cols):
        return t
    out = torch.zeros(rows, cols, dtype=t.dtype, device=t.device) if fill == 0 \
        else torch.full((rows, cols), fill, dtype=t.dtype, device=t.device)
    out[: t.shape[0], : t.shape[1]] = t
    return out


def matmul_nt(a: torch.Tensor, b: torch.Tensor,
              a_scale: torch.Tensor = None, b_scale: torch.Tensor = None,
              stream=None, xcd_swizzle: bool = False) -> torch.Tensor:
    """Arbitrary-shape front door to the K7 GEMM family: returns A @ B^T.

    Dispatches on dtype — bf16 -> gemm_bf16, float8_e4m3fn -> gemm_fp8
    (or gemm_mxfp8 when e8m0 scales are given), int8 -> gemm_i8, uint8 +
    scales -> gemm_mxfp4 (nibble-packed e2m1; shapes in elements) — after
    zero-padding the operands up to the fast-path tile multiples
    (gemm_pad_shapes). Zero rows/columns contribute nothing, so the
    result is bit-identical to the unpadded kernel output; the padded
    region of C is sliced away (result is a fresh contiguous tensor when
    padding occurred). Scale padding uses 127 (= 2^0) — irrelevant since
    the padded data is zero.
    """
    if a.dim() != 2 or b.dim() != 2 or a.shape[1] != b.shape[1]:
        raise ValueError(f"need A[M,K], B[N,K]; got A{tuple(a.shape)} "
                         f"B{tuple(b.shape)}")
    m, k = a.shape
    n = b.shape[0]
    if a.dtype == torch.bfloat16:
        kind, fn, out_dtype = "bf16", gemm_bf16, torch.float32
    elif a.dtype == torch.int8:
        kind, fn, out_dtype = "i8", gemm_i8, torch.int32
    elif a.dtype == torch.float8_e4m3fn:
        if a_scale is not None:
            kind, fn, out_dtype = "mxfp8", None, torch.float32
        else:
            kind, fn, out_dtype = "fp8", gemm_fp8, torch.float32
    elif a.dtype == torch.uint8:
        # nibble-packed e2m1 (e2m1_pack layout): shapes are in ELEMENTS,
        # K = 2 * packed columns; zero nibbles pad exactly (+0.0)
        if a_scale is None:
            raise TypeError("packed-fp4 operands need e8m0 scales")
        kind, fn, out_dtype = "mxfp4", None, torch.float32
        k = 2 * k
    else:
        raise TypeError(f"unsupported operand dtype {a.dtype}")
    mp, np_, kp = gemm_pad_shapes(kind, m, n, k)
    if kind == "mxfp4":
        if b_scale is None or a_scale.shape != (m, k // 32) \
                or b_scale.shape != (n, k // 32):
            raise ValueError("mx4 path needs a_scale [M,K//32] and "
                             "b_scale [N,K//32] (K = 2*packed cols, "
                             "K % 32 == 0)")
        ap = _pad2d(a.contiguous(), mp, kp // 2)
        bp = _pad2d(b.contiguous(), np_, kp // 2)
        c = torch.empty(mp, np_, dtype=out_dtype, device=a.device)
        asp = _pad2d(a_scale.contiguous(), mp, kp // 32, fill=127)
        bsp = _pad2d(b_scale.contiguous(), np_, kp // 32, fill=127)
        gemm_mxfp4(c, ap, bp, asp, bsp, stream=stream,
                   xcd_swizzle=xcd_swizzle)
        if (mp, np_) == (m, n):
            return c
        return c[:m, :n].contiguous()
    ap = _pad2d(a.contiguous(), mp, kp)
    bp = _pad2d(b.contiguous(), np_, kp)
    c = torch.empty(mp, np_, dtype=out_dtype, device=a.device)
    if kind == "mxfp8":
        if b_scale is None or a_scale.shape != (m, k // 32) \
                or b_scale.shape != (n, k // 32):
            raise ValueError("mx path needs a_scale [M,K//32] and "
                             "b_scale [N,K//32] (K % 32 == 0)")
        asp = _pad2d(a_scale.contiguous(), mp, kp // 32, fill=127)
        bsp = _pad2d(b_scale.contiguous(), np_, kp // 32, fill=127)
        gemm_mxfp8(c, ap, bp, asp, bsp, stream=stream,
                   xcd_swizzle=xcd_swizzle)
    else:
        fn(c, ap, bp, stream=stream, xcd_swizzle=xcd_swizzle)
    if (mp, np_) == (m, n):
        return c
    return c[:m, :n].contiguous()
