// kernels.hip — hand-written gfx950 (CDNA4) kernels for the pattern suite.
//
// These are the MI355X-native equivalents of the reference's device-code
// sites: the SYCL busy_wait parallel_for
// (reference concurency/bench_sycl.cpp:92-97 + bench.hpp:23-31), the A2B copy
// (bench_sycl.cpp:100), Accumulate / Initialize
// (allreduce-mpi-sycl.cpp:27-41), and the checksum payload
// (peer2pear.cpp:8-17). All kernels use 256-thread blocks (4 wave64) and
// grid-stride loops sized well past the 256 CUs / 8 XCDs.

#include "include/hpk.h"

#include <cstdio>
#include <stdexcept>
#include <string>

namespace hpk {

void check_hip(hipError_t e, const char* what) {
  if (e != hipSuccess) {
    throw std::runtime_error(std::string("HIP error in ") + what + ": " +
                             hipGetErrorString(e));
  }
}

namespace {

constexpr int kBlock = 256; // 4 wave64 per workgroup
// Streaming grid cap: 256 CUs want >> 256 workgroups in flight. Measured on
// MI355X (profiles/membench_1gpu_r3.log): 1 GiB copy at cap 65536 runs
// 2.60 TB/s payload vs 2.32 at 16384 — fewer grid-stride iterations per
// thread wins for pure streaming.
constexpr size_t kMaxGrid = 65536;

inline size_t stream_grid(size_t n_items) {
  size_t blocks = (n_items + kBlock - 1) / kBlock;
  if (blocks == 0) blocks = 1;
  if (blocks > kMaxGrid) blocks = kMaxGrid;
  return blocks;
}

// --------------------------------------------------------------------------
// K1: busy-wait FMA chain. Each work-item performs 64*tripcount dependent
// v_fma_f32 ops on registers and stores one float. The dependency chain makes
// the kernel's duration linear in tripcount (the property the autotuner's
// linear regression relies on) and independent of memory traffic.
// --------------------------------------------------------------------------
__global__ void k_busy_wait(float* __restrict__ out, long tripcount,
                            long globalsize) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < globalsize; i += stride) {
    float x = 1.0f + (float)i * 1e-9f;
    const float y = 1.000001f;
    for (long t = 0; t < tripcount; ++t) {
#pragma unroll
      for (int k = 0; k < 64; ++k) {
        x = __builtin_fmaf(y, x, y); // dependent chain: not foldable
      }
    }
    out[i] = x;
  }
}

// --------------------------------------------------------------------------
// K1-MFMA: matrix-core busy loop. Each wave chains `tripcount`
// v_mfma_f32_16x16x32_bf16 instructions through one accumulator, so the
// kernel occupies the MFMA pipe (visible as MfmaUtil in rocprof PMC runs)
// while doing no memory traffic.
// --------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

__global__ void k_busy_wait_mfma(float* __restrict__ out, long tripcount) {
  // Arbitrary nonzero fragments; value is irrelevant, occupancy of the MFMA
  // pipe is the payload.
  short seed = (short)(threadIdx.x + 1);
  bf16x8_t a = {seed, seed, seed, seed, seed, seed, seed, seed};
  bf16x8_t b = {1, 2, 3, 4, 5, 6, 7, 8};
  // Two independent accumulators: the 16x16x32 MFMA's dependent-accumulator
  // latency exceeds its issue interval, so a single chain leaves the pipe
  // under-issued (measured 1.63 PF with one chain at 2 waves/SIMD).
  f32x4_t acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4_t acc1 = {0.f, 0.f, 0.f, 0.f};
  for (long t = 0; t < tripcount; t += 2) {
    acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc1, 0, 0, 0);
  }
  // One store per thread keeps the accumulators alive without noise.
  out[(size_t)blockIdx.x * blockDim.x + threadIdx.x] =
      (acc0[0] + acc1[0]) * 1e-30f; // scaled to avoid overflow for readers
}

// --------------------------------------------------------------------------
// K2: shader copy, 16 B per lane per iteration (1 KiB per wave-instruction).
// The SDMA sibling is plain hipMemcpyAsync; this kernel measures the
// shader-blit path and doubles as a D2D bandwidth workload.
// --------------------------------------------------------------------------
__global__ void k_copy_b16(const uint4* __restrict__ src, uint4* __restrict__ dst,
                           size_t n16) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n16; i += stride) dst[i] = src[i];
}

// Nontemporal variant: streaming load/store hints (slc) — tells the cache
// hierarchy not to retain lines; candidate win for copies far beyond L3.
typedef __attribute__((ext_vector_type(4))) unsigned int uint4_ev_t;

__global__ void k_copy_b16_nt(const uint4* __restrict__ src,
                              uint4* __restrict__ dst, size_t n16) {
  const uint4_ev_t* s = (const uint4_ev_t*)src;
  uint4_ev_t* d = (uint4_ev_t*)dst;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n16; i += stride) {
    uint4_ev_t v = __builtin_nontemporal_load(s + i);
    __builtin_nontemporal_store(v, d + i);
  }
}

// 4x-unrolled variant: 64 B per thread per iteration, 4 loads in flight
// before the first store — deeper MLP for the HBM path at large sizes.
__global__ void k_copy_b16x4(const uint4* __restrict__ src,
                             uint4* __restrict__ dst, size_t n16) {
  size_t stride = (size_t)gridDim.x * blockDim.x;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i + 3 * stride < n16; i += 4 * stride) {
    uint4 a = src[i];
    uint4 b = src[i + stride];
    uint4 c = src[i + 2 * stride];
    uint4 d = src[i + 3 * stride];
    dst[i] = a;
    dst[i + stride] = b;
    dst[i + 2 * stride] = c;
    dst[i + 3 * stride] = d;
  }
  for (; i < n16; i += stride) dst[i] = src[i];
}

__global__ void k_copy_b1(const unsigned char* __restrict__ src,
                          unsigned char* __restrict__ dst, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = src[i];
}

// --------------------------------------------------------------------------
// K4: fills (reference Initialize) and iota payload.
// --------------------------------------------------------------------------
__global__ void k_fill_f4(float4* __restrict__ dst, float v, size_t n4) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  float4 val = make_float4(v, v, v, v);
  for (; i < n4; i += stride) dst[i] = val;
}

typedef __attribute__((ext_vector_type(4))) float f4_ev_t;

__global__ void k_fill_f4_nt(float4* __restrict__ dst, float v, size_t n4) {
  f4_ev_t* d = (f4_ev_t*)dst;
  f4_ev_t val = {v, v, v, v};
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) __builtin_nontemporal_store(val, d + i);
}

__global__ void k_fill_f1(float* __restrict__ dst, float v, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = v;
}

__global__ void k_iota_f32(float* __restrict__ dst, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = (float)i;
}

// --------------------------------------------------------------------------
// K3: accumulate dst[i] += src[i] (the ring-allreduce reduction kernel).
// float4 vectorized body + scalar tail.
// --------------------------------------------------------------------------
__global__ void k_acc_f4(float4* __restrict__ dst, const float4* __restrict__ src,
                         size_t n4) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) {
    float4 d = dst[i];
    float4 s = src[i];
    d.x += s.x;
    d.y += s.y;
    d.z += s.z;
    d.w += s.w;
    dst[i] = d;
  }
}

__global__ void k_acc_f1(float* __restrict__ dst, const float* __restrict__ src,
                         size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] += src[i];
}

// NT variant: src is read once (never re-read) -> nontemporal load; dst is
// read-modify-write with an NT store (the line won't be revisited either).
__global__ void k_acc_f4_nt(float4* __restrict__ dst,
                            const float4* __restrict__ src, size_t n4) {
  const f4_ev_t* s = (const f4_ev_t*)src;
  f4_ev_t* d = (f4_ev_t*)dst;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) {
    f4_ev_t a = __builtin_nontemporal_load(d + i);
    f4_ev_t b = __builtin_nontemporal_load(s + i);
    __builtin_nontemporal_store(a + b, d + i);
  }
}

// --------------------------------------------------------------------------
// int32 twins of fill/accumulate (the reference instantiates its miniapps
// for float AND int via -DAPP_DATA_TYPE, mpi-sycl/CMakeLists.txt:4-5).
// --------------------------------------------------------------------------
__global__ void k_fill_i4(int4* __restrict__ dst, int v, size_t n4) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  int4 val = make_int4(v, v, v, v);
  for (; i < n4; i += stride) dst[i] = val;
}

__global__ void k_fill_i1(int* __restrict__ dst, int v, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = v;
}

__global__ void k_acc_i4(int4* __restrict__ dst, const int4* __restrict__ src,
                         size_t n4) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) {
    int4 d = dst[i];
    int4 s = src[i];
    d.x += s.x;
    d.y += s.y;
    d.z += s.z;
    d.w += s.w;
    dst[i] = d;
  }
}

__global__ void k_acc_i1(int* __restrict__ dst, const int* __restrict__ src,
                         size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] += src[i];
}

__global__ void k_reduce_partial_i32(const int* __restrict__ src, size_t n,
                                     long long* __restrict__ partial) {
  long long s = 0;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    s += src[i];
  for (int off = 32; off > 0; off >>= 1) s += __shfl_down(s, off, 64);
  __shared__ long long wsum_i[kBlock / 64];
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  if (lane == 0) wsum_i[wave] = s;
  __syncthreads();
  if (wave == 0) {
    long long b = (lane < kBlock / 64) ? wsum_i[lane] : 0;
    for (int off = 32; off > 0; off >>= 1) b += __shfl_down(b, off, 64);
    if (lane == 0) partial[blockIdx.x] = b;
  }
}

// --------------------------------------------------------------------------
// Exact checksum: double partial sums per block (wave64 shuffle reduction,
// then LDS across the block's 4 waves), host-side final sum of <=1024
// partials.
// --------------------------------------------------------------------------
__global__ void k_reduce_partial_f32(const float* __restrict__ src, size_t n,
                                     double* __restrict__ partial) {
  // float4 loads + 4 independent f64 accumulators: breaks the serial f64-add
  // dependency chain that capped the first version at 2.2 TB/s read.
  double s0 = 0.0, s1 = 0.0, s2 = 0.0, s3 = 0.0;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  size_t n4 = n / 4;
  if (((uintptr_t)src % 16) == 0) {
    // PLAIN loads on purpose: this kernel is the suite's checksum oracle.
    // An NT-load variant read a stale/partial view of freshly-copied data
    // once on gfx950 (p2p checksum exactly half, r24) — verification must
    // not depend on cache-hint semantics.
    const float4* src4 = (const float4*)src;
    for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
         i += stride) {
      float4 v = src4[i];
      s0 += (double)v.x;
      s1 += (double)v.y;
      s2 += (double)v.z;
      s3 += (double)v.w;
    }
    // tail elements by block 0 / thread 0..3
    if (blockIdx.x == 0 && threadIdx.x < n - n4 * 4)
      s0 += (double)src[n4 * 4 + threadIdx.x];
  } else {
    for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride)
      s0 += (double)src[i];
  }
  double s = (s0 + s1) + (s2 + s3);

  // wave64 reduction
  for (int off = 32; off > 0; off >>= 1) s += __shfl_down(s, off, 64);

  __shared__ double wsum[kBlock / 64];
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  if (lane == 0) wsum[wave] = s;
  __syncthreads();
  if (wave == 0) {
    double b = (lane < kBlock / 64) ? wsum[lane] : 0.0;
    for (int off = 32; off > 0; off >>= 1) b += __shfl_down(b, off, 64);
    if (lane == 0) partial[blockIdx.x] = b;
  }
}

} // namespace

void launch_busy_wait(float* out, long tripcount, long globalsize,
                      hipStream_t stream) {
  size_t grid = stream_grid((size_t)globalsize);
  hipLaunchKernelGGL(k_busy_wait, dim3(grid), dim3(kBlock), 0, stream, out,
                     tripcount, globalsize);
  check_hip(hipGetLastError(), "launch_busy_wait");
}

void launch_busy_wait_mfma(float* out, long tripcount, long n_waves,
                           hipStream_t stream) {
  if (n_waves < 1) n_waves = 1;
  size_t blocks = ((size_t)n_waves * 64 + kBlock - 1) / kBlock;
  hipLaunchKernelGGL(k_busy_wait_mfma, dim3(blocks), dim3(kBlock), 0, stream,
                     out, tripcount);
  check_hip(hipGetLastError(), "launch_busy_wait_mfma");
}

void launch_copy_kernel_tuned(void* dst, const void* src, size_t nbytes,
                              hipStream_t stream, int unroll,
                              size_t grid_cap) {
  uintptr_t d = (uintptr_t)dst, s = (uintptr_t)src;
  if ((d % 16) || (s % 16)) {
    launch_copy_kernel(dst, src, nbytes, stream);
    return;
  }
  size_t n16 = nbytes / 16;
  size_t tail = nbytes - n16 * 16;
  size_t blocks = (n16 + kBlock - 1) / kBlock;
  if (blocks == 0) blocks = 1;
  if (blocks > grid_cap) blocks = grid_cap;
  if (unroll == 5) { // nontemporal streaming variant
    hipLaunchKernelGGL(k_copy_b16_nt, dim3(blocks), dim3(kBlock), 0, stream,
                       (const uint4*)src, (uint4*)dst, n16);
  } else if (unroll >= 4) {
    hipLaunchKernelGGL(k_copy_b16x4, dim3(blocks), dim3(kBlock), 0, stream,
                       (const uint4*)src, (uint4*)dst, n16);
  } else {
    hipLaunchKernelGGL(k_copy_b16, dim3(blocks), dim3(kBlock), 0, stream,
                       (const uint4*)src, (uint4*)dst, n16);
  }
  if (tail) {
    hipLaunchKernelGGL(k_copy_b1, dim3(1), dim3(kBlock), 0, stream,
                       (const unsigned char*)src + n16 * 16,
                       (unsigned char*)dst + n16 * 16, tail);
  }
  check_hip(hipGetLastError(), "launch_copy_kernel_tuned");
}

void launch_copy_kernel(void* dst, const void* src, size_t nbytes,
                        hipStream_t stream) {
  uintptr_t d = (uintptr_t)dst, s = (uintptr_t)src;
  if ((d % 16 == 0) && (s % 16 == 0)) {
    size_t n16 = nbytes / 16;
    size_t tail = nbytes - n16 * 16;
    if (n16) {
      // Large copies: nontemporal streaming variant at a 131072-block grid —
      // measured 3.02 TB/s payload vs 2.60 for the plain kernel at 1 GiB
      // (96% of the 6.3 TB/s achievable HBM rate; membench r12). Small
      // copies keep the plain kernel (NT hints only pay beyond cache scale).
      if (nbytes >= (32u << 20)) {
        size_t blocks = (n16 + kBlock - 1) / kBlock;
        if (blocks > 131072) blocks = 131072;
        hipLaunchKernelGGL(k_copy_b16_nt, dim3(blocks), dim3(kBlock), 0,
                           stream, (const uint4*)src, (uint4*)dst, n16);
      } else {
        hipLaunchKernelGGL(k_copy_b16, dim3(stream_grid(n16)), dim3(kBlock), 0,
                           stream, (const uint4*)src, (uint4*)dst, n16);
      }
    }
    if (tail) {
      hipLaunchKernelGGL(k_copy_b1, dim3(1), dim3(kBlock), 0, stream,
                         (const unsigned char*)src + n16 * 16,
                         (unsigned char*)dst + n16 * 16, tail);
    }
  } else {
    hipLaunchKernelGGL(k_copy_b1, dim3(stream_grid(nbytes)), dim3(kBlock), 0,
                       stream, (const unsigned char*)src, (unsigned char*)dst,
                       nbytes);
  }
  check_hip(hipGetLastError(), "launch_copy_kernel");
}

void launch_fill_f32(float* dst, float value, size_t n, hipStream_t stream) {
  if (((uintptr_t)dst % 16 == 0) && n >= 4) {
    size_t n4 = n / 4;
    size_t tail = n - n4 * 4;
    if (n * 4 >= (32u << 20)) {
      size_t blocks = (n4 + kBlock - 1) / kBlock;
      if (blocks > 131072) blocks = 131072;
      hipLaunchKernelGGL(k_fill_f4_nt, dim3(blocks), dim3(kBlock), 0, stream,
                         (float4*)dst, value, n4);
    } else
    hipLaunchKernelGGL(k_fill_f4, dim3(stream_grid(n4)), dim3(kBlock), 0,
                       stream, (float4*)dst, value, n4);
    if (tail) {
      hipLaunchKernelGGL(k_fill_f1, dim3(1), dim3(64), 0, stream, dst + n4 * 4,
                         value, tail);
    }
  } else {
    hipLaunchKernelGGL(k_fill_f1, dim3(stream_grid(n)), dim3(kBlock), 0, stream,
                       dst, value, n);
  }
  check_hip(hipGetLastError(), "launch_fill_f32");
}

void launch_iota_f32(float* dst, size_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_iota_f32, dim3(stream_grid(n)), dim3(kBlock), 0, stream,
                     dst, n);
  check_hip(hipGetLastError(), "launch_iota_f32");
}

void launch_acc_f32(float* dst, const float* src, size_t n, hipStream_t stream) {
  if (((uintptr_t)dst % 16 == 0) && ((uintptr_t)src % 16 == 0) && n >= 4) {
    // beyond-L3 sizes: nontemporal variant (5.87 vs 5.11 TB/s at 1 GiB,
    // membench r22) — the ring-allreduce accumulate hot path
    if (n * 4 >= (32u << 20)) {
      launch_acc_f32_nt(dst, src, n, stream);
      return;
    }
    size_t n4 = n / 4;
    size_t tail = n - n4 * 4;
    hipLaunchKernelGGL(k_acc_f4, dim3(stream_grid(n4)), dim3(kBlock), 0, stream,
                       (float4*)dst, (const float4*)src, n4);
    if (tail) {
      hipLaunchKernelGGL(k_acc_f1, dim3(1), dim3(64), 0, stream, dst + n4 * 4,
                         src + n4 * 4, tail);
    }
  } else {
    hipLaunchKernelGGL(k_acc_f1, dim3(stream_grid(n)), dim3(kBlock), 0, stream,
                       dst, src, n);
  }
  check_hip(hipGetLastError(), "launch_acc_f32");
}

void launch_fill_i32(int* dst, int value, size_t n, hipStream_t stream) {
  if (((uintptr_t)dst % 16 == 0) && n >= 4) {
    size_t n4 = n / 4;
    size_t tail = n - n4 * 4;
    hipLaunchKernelGGL(k_fill_i4, dim3(stream_grid(n4)), dim3(kBlock), 0,
                       stream, (int4*)dst, value, n4);
    if (tail)
      hipLaunchKernelGGL(k_fill_i1, dim3(1), dim3(64), 0, stream, dst + n4 * 4,
                         value, tail);
  } else {
    hipLaunchKernelGGL(k_fill_i1, dim3(stream_grid(n)), dim3(kBlock), 0,
                       stream, dst, value, n);
  }
  check_hip(hipGetLastError(), "launch_fill_i32");
}

void launch_acc_i32(int* dst, const int* src, size_t n, hipStream_t stream) {
  if (((uintptr_t)dst % 16 == 0) && ((uintptr_t)src % 16 == 0) && n >= 4) {
    size_t n4 = n / 4;
    size_t tail = n - n4 * 4;
    hipLaunchKernelGGL(k_acc_i4, dim3(stream_grid(n4)), dim3(kBlock), 0,
                       stream, (int4*)dst, (const int4*)src, n4);
    if (tail)
      hipLaunchKernelGGL(k_acc_i1, dim3(1), dim3(64), 0, stream, dst + n4 * 4,
                         src + n4 * 4, tail);
  } else {
    hipLaunchKernelGGL(k_acc_i1, dim3(stream_grid(n)), dim3(kBlock), 0, stream,
                       dst, src, n);
  }
  check_hip(hipGetLastError(), "launch_acc_i32");
}

long long reduce_sum_i32(const int* src, size_t n, hipStream_t stream) {
  check_hip(hipDeviceSynchronize(), "reduce_sum_i32 pre-sync"); // see f32
  constexpr size_t kRedBlocks = 4096;
  size_t blocks = (n + (size_t)kBlock * 8 - 1) / ((size_t)kBlock * 8);
  if (blocks == 0) blocks = 1;
  if (blocks > kRedBlocks) blocks = kRedBlocks;
  long long* d_partial = nullptr;
  check_hip(hipMallocAsync((void**)&d_partial, blocks * sizeof(long long),
                           stream),
            "reduce_sum_i32 hipMallocAsync");
  hipLaunchKernelGGL(k_reduce_partial_i32, dim3(blocks), dim3(kBlock), 0,
                     stream, src, n, d_partial);
  check_hip(hipGetLastError(), "reduce_sum_i32 kernel");
  std::vector<long long> h(blocks);
  check_hip(hipMemcpyAsync(h.data(), d_partial, blocks * sizeof(long long),
                           hipMemcpyDeviceToHost, stream),
            "reduce_sum_i32 D2H");
  check_hip(hipFreeAsync(d_partial, stream), "reduce_sum_i32 hipFreeAsync");
  check_hip(hipStreamSynchronize(stream), "reduce_sum_i32 sync");
  long long s = 0;
  for (long long v : h) s += v;
  return s;
}

void launch_acc_f32_nt(float* dst, const float* src, size_t n,
                       hipStream_t stream) {
  if (((uintptr_t)dst % 16 == 0) && ((uintptr_t)src % 16 == 0) && n >= 4) {
    size_t n4 = n / 4;
    size_t tail = n - n4 * 4;
    size_t blocks = (n4 + kBlock - 1) / kBlock;
    if (blocks > 131072) blocks = 131072;
    hipLaunchKernelGGL(k_acc_f4_nt, dim3(blocks), dim3(kBlock), 0, stream,
                       (float4*)dst, (const float4*)src, n4);
    if (tail)
      hipLaunchKernelGGL(k_acc_f1, dim3(1), dim3(64), 0, stream, dst + n4 * 4,
                         src + n4 * 4, tail);
    check_hip(hipGetLastError(), "launch_acc_f32_nt");
  } else {
    launch_acc_f32(dst, src, n, stream);
  }
}

double reduce_sum_f32(const float* src, size_t n, hipStream_t stream) {
  // This is the suite's checksum oracle: device-wide sync first. Measured
  // on ROCm 7.2 (profiles/p2p_fail_r37.log): a kernel launched on another
  // stream immediately after hipStreamSynchronize() of the producing
  // stream intermittently (~1/25) read a partially-visible buffer; the
  // data was correct after hipDeviceSynchronize(). Verification must not
  // depend on cross-stream visibility timing.
  check_hip(hipDeviceSynchronize(), "reduce_sum_f32 pre-sync");
  constexpr size_t kRedBlocks = 4096;
  size_t blocks = (n + (size_t)kBlock * 8 - 1) / ((size_t)kBlock * 8);
  if (blocks == 0) blocks = 1;
  if (blocks > kRedBlocks) blocks = kRedBlocks;

  double* d_partial = nullptr;
  check_hip(hipMallocAsync((void**)&d_partial, blocks * sizeof(double), stream),
            "reduce_sum_f32 hipMallocAsync");
  hipLaunchKernelGGL(k_reduce_partial_f32, dim3(blocks), dim3(kBlock), 0,
                     stream, src, n, d_partial);
  check_hip(hipGetLastError(), "reduce_sum_f32 kernel");

  std::vector<double> h(blocks);
  check_hip(hipMemcpyAsync(h.data(), d_partial, blocks * sizeof(double),
                           hipMemcpyDeviceToHost, stream),
            "reduce_sum_f32 D2H");
  check_hip(hipFreeAsync(d_partial, stream), "reduce_sum_f32 hipFreeAsync");
  check_hip(hipStreamSynchronize(stream), "reduce_sum_f32 sync");

  double s = 0.0;
  for (double v : h) s += v;
  return s;
}

} // namespace hpk
