// gemm.hip — K7: the LDS-tiled MFMA GEMM family (beyond-parity showcase).
//
// The reference suite's device kernels (K1-K6) are streaming/elementwise;
// these kernels demonstrate the OTHER half of the CDNA4 execution model
// the suite is designed around: matrix cores fed through LDS. All compute
//   C[M,N] (fp32) = A[M,K] x B[N,K]^T  ("NT": both operands K-contiguous,
// the natural MFMA feeding order), fp32 accumulate, verified BITWISE
// against torch fp32 matmul on exactly-representable payloads
// (tests/test_gpu_kernels.py) and race-screened.
//
// Members (measured on MI355X, random operands, profiles/gemm_showcase_r2):
//   k_gemm_bf16_nt<2,2|2,4>   plain two-barrier 128^2 tile; the 8-wave
//                             64x32 decomposition wins (76 VGPR -> 6
//                             waves/SIMD): 840/935 TF at 4096^3/8192^3
//   k_gemm_bf16_nt_db         + double-buffered LDS, raw barriers,
//                             counted vmcnt (917/879 TF)
//   k_gemm_bf16_8ph           256^2 deep pipeline, 8 phases, two K-tiles
//                             in 8 rotating LDS half-slots, vmcnt(4) only
//                             at tile switches — DEFAULT for eligible
//                             shapes: 968/1085 TF, zero LDS bank
//                             conflicts (true-lane-group cyclic skew)
//   k_gemm_fp8_nt / _8ph      OCP fp8 e4m3 at the 16x16x32 MFMA (half
//                             the bf16 staging bytes): up to 1441 TF;
//                             256-divisible shapes default to the
//                             mxfp8_nt_32<false> route below (2.2 PF)
//   k_gemm_mxfp8_nt           block-scaled OCP MX-fp8 via
//                             mfma_scale_f32_16x16x128_f8f6f4 (HW-fused
//                             e8m0 dequant; scale lane layout
//                             reverse-engineered on hardware —
//                             scripts/probes/): 1216-1482 TF
//   k_gemm_mxfp8_nt_32<S>     256^2-tile 32x32x64 version — DEFAULT for
//                             256-divisible shapes: 1863 TF scaled (S=
//                             true), 2194 TF as the plain-fp8 route
//                             (S=false, hardcoded x1.0 scales); a THIRD
//                             operand/scale association, half-interleaved
//                             (findings #24)
//   k_gemm_mxfp4_nt*          block-scaled OCP MX-fp4 (e2m1, the 4x rate
//                             class): hardware-probed DIAGONAL layout;
//                             default _32 kernel (256^2 tile, 32x32x64
//                             MFMA, chunk-rotation bank fix):
//                             2903/3044-3197 TF at 8192^3/16384^3
//   k_gemm_i8_nt / _8ph / _32 int8 with EXACT int32 accumulation
//                             (mfma_i32_16x16x64_i8, ~2x bf16 rate,
//                             4-VGPR fragments): 1522 / 2214-2393 TOPS
//                             (8ph stays default; the _32 port measured
//                             negative, findings #25)
//
// Fragment mappings (16x16x32 bf16/fp8): A/B lane L holds 8 contiguous K
// elements at k = 8*(L>>4), row/col = L&15; C/D lane L reg r holds
// row = 4*(L>>4)+r, col = L&15 (identical across shapes/dtypes). The MX
// instruction's operand/scale layout is documented at its kernel.

#include "include/hpk.h"

#include <hip/hip_bf16.h>

#include <cstdlib>
#include <string>
#include <stdexcept>

namespace hpk {
namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int BM = 128, BN = 128, BK = 64;

// one K-step's staging: A[128][64] + B[128][64] bf16 = 32 KiB
constexpr int TILE_HALF = BM * BK;            // elements per operand tile

// Cyclic-skew LDS layout: linear [128][64]-bf16 rows are 128 B, so
// ds_read_b128 fragment reads pile multiple lanes onto the same 4-dword
// bank window. gfx950's b128 lane groups are NOT contiguous 16-lane
// blocks — they mix rows AND kfrag halves (e.g. {0-3,12-15,20-27}:
// rows {0-3,12-15} at kfrag q and rows {4-11} at kfrag q+1,
// MI355X_MICROARCH.md LDS table) — so the shift must solve the mixed
// sets: rotating each row's K range by 16 elements per (row>>1)&3 class,
//   LDS(row, k) = (row, (k + 16*((row>>1)&3)) & 63)
// gives every row class an EVEN 4-dword window slot and the interleaved
// kfrag half the odd slots: all 16 lanes of each true lane group land on
// 16 distinct windows (verified by enumeration; a naive per-row-pair
// 8-element shift measured SQ_LDS_BANK_CONFLICT = 0.5x IDX_ACTIVE —
// 2-way residual — precisely because of the group mixing).
// The 16-element granule keeps every 16-B glds chunk and every 8-element
// fragment read contiguous, so the staging pre-applies the inverse on the
// GLOBAL source chunk address while the LDS write stays lane-linear (the
// global_load_lds requirement).
__device__ __forceinline__ long lds_skew(long e) { // tile elem -> LDS slot
  long row = e >> 6, k = e & 63;
  return (row << 6) | ((k + 16 * ((row >> 1) & 3)) & 63);
}
__device__ __forceinline__ long lds_unskew(long y) { // LDS slot -> tile elem
  long row = y >> 6, k = y & 63;
  return (row << 6) | ((k - 16 * ((row >> 1) & 3)) & 63);
}

// L2/MALL-aware grouped tile order (the CUTLASS threadblock-swizzle idea,
// re-derived for the 256 MiB MALL): remap the linear workgroup id so the
// ~256 CONCURRENTLY-resident workgroups cover a near-square super-block of
// output tiles — their A row-panels and B column-panels then stay resident
// in the LLC and each panel is streamed from HBM once per super-block
// instead of once per tile row. `group` = band width in N-tiles
// (HPK_GEMM_GROUP; <=1 keeps the row-major order). Bijective for any
// grid: the last band is simply narrower (gw < group).
__device__ __forceinline__ int hpk_group_remap(int wg, int tiles_n, int nwg,
                                               int group) {
  if (group <= 1) return wg;
  const int tiles_m = nwg / tiles_n;
  const int band = group * tiles_m;
  const int b = wg / band;
  const int within = wg - b * band;
  int gw = tiles_n - b * group;
  if (gw > group) gw = group;
  const int tm = within / gw;
  const int tn = b * group + within % gw;
  return tm * tiles_n + tn;
}

// Wave-grid decomposition is a template knob: <2,2> = 4 waves of 64x64
// (4x4 fragments, 190 VGPR+AGPR, 2 waves/SIMD), <2,4> = 8 waves of
// 64x32 (4x2 fragments, 76 VGPR, 6 waves/SIMD). Measured on MI355X with
// random operands: 8 waves wins everywhere (935 vs 765 TF at 8192^3
// under the final skew) — occupancy-driven latency hiding beats the
// bigger per-wave MFMA batch. HPK_GEMM_WAVES=4|8 overrides.
template <int WAVES_M, int WAVES_N>
__global__ __launch_bounds__(WAVES_M* WAVES_N * 64) void k_gemm_bf16_nt(
    float* __restrict__ C, const __hip_bfloat16* __restrict__ A,
    const __hip_bfloat16* __restrict__ B, int M, int N, int K,
    int tiles_n, int nwg, int xcd_swizzle, int group) {
  constexpr int THREADS = WAVES_M * WAVES_N * 64;
  constexpr int MREP = BM / (WAVES_M * 16);
  constexpr int NREP = BN / (WAVES_N * 16);
  constexpr int WTM = BM / WAVES_M; // wave sub-tile rows
  constexpr int WTN = BN / WAVES_N; // wave sub-tile cols
  __shared__ __hip_bfloat16 lds[2 * TILE_HALF]; // [A tile][B tile]

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    // bijective 8-XCD round-robin -> tile-linear remap
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const int tile_m = wg / tiles_n;
  const int tile_n = wg % tiles_n;
  const long brow = (long)tile_m * BM;
  const long bcol = (long)tile_n * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WAVES_N; // wave's 64x64 sub-tile position
  const int wc = wid % WAVES_N;

  // staging plan: each thread DMAs 16 B (8 bf16) per glds; one operand
  // tile is 16 KiB = 256 threads x 16 B x 4 issues. global_load_lds
  // writes to (wave-uniform LDS base) + lane*16, so the LDS pointer we
  // pass is the WAVE chunk base and only the GLOBAL address is per-lane;
  // the row-major [128][64] tile image is lane-linear by construction.
  constexpr long elems_per_issue = (long)THREADS * 8;
  f32x4 acc[MREP][NREP] = {};

  constexpr int ISSUES = TILE_HALF / (THREADS * 8);
  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads(); // previous K-step's reads done before overwrite
    for (int issue = 0; issue < ISSUES; ++issue) {
      long o_base = (long)issue * elems_per_issue + (long)wid * (64 * 8);
      // lane's LDS slot is o_base + lane*8 (lane-linear); fetch the global
      // chunk that belongs at that slot under the skewed image
      long o = lds_unskew(o_base + (long)lane * 8);
      int row = (int)(o / BK);
      int kk = (int)(o % BK);
      const __hip_bfloat16* ga = A + (brow + row) * (long)K + k0 + kk;
      const __hip_bfloat16* gb = B + (bcol + row) * (long)K + k0 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(lds + o_base), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)(lds + TILE_HALF + o_base),
          16, 0, 0);
    }
    __syncthreads(); // carries vmcnt(0): glds queue drained

    const __hip_bfloat16* la = lds;
    const __hip_bfloat16* lb = lds + TILE_HALF;
    for (int kk = 0; kk < BK; kk += 32) {
      // fragment k-base for this lane: 8 contiguous bf16
      const int kfrag = kk + 8 * (lane >> 4);
      bf16x8 afrag[MREP], bfrag[NREP];
      for (int m = 0; m < MREP; ++m) {
        int row = wr * WTM + m * 16 + (lane & 15);
        afrag[m] = *(const bf16x8*)(la + lds_skew(row * BK + kfrag));
      }
      for (int n = 0; n < NREP; ++n) {
        int col = wc * WTN + n * 16 + (lane & 15);
        bfrag[n] = *(const bf16x8*)(lb + lds_skew(col * BK + kfrag));
      }
      for (int m = 0; m < MREP; ++m)
        for (int n = 0; n < NREP; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[m], bfrag[n], acc[m][n], 0, 0, 0);
    }
  }

  // epilogue: C/D mapping row = 4*(lane>>4)+r, col = lane&15
  for (int m = 0; m < MREP; ++m)
    for (int n = 0; n < NREP; ++n) {
      long row0 = brow + wr * WTM + m * 16 + 4 * (lane >> 4);
      long col = bcol + wc * WTN + n * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r)
        C[(row0 + r) * (long)N + col] = acc[m][n][r];
    }
}

// Double-buffered variant: the plain kernel's __syncthreads() after the
// glds issues carries an implicit s_waitcnt vmcnt(0) that drains the DMA
// queue before ANY wave crosses — the documented ~20% stall of the
// two-barrier structure. Here K-tile t+1's DMA is issued BEFORE waiting
// for tile t (FIFO per wave, so `s_waitcnt vmcnt(8)` retires exactly
// tile t's 8 DMAs while t+1's stay in flight), and the barriers are raw
// s_barrier + lgkmcnt(0) so nothing re-drains the queue. 64 KiB LDS.
template <int WAVES_M, int WAVES_N>
__global__ __launch_bounds__(WAVES_M* WAVES_N * 64) void k_gemm_bf16_nt_db(
    float* __restrict__ C, const __hip_bfloat16* __restrict__ A,
    const __hip_bfloat16* __restrict__ B, int M, int N, int K,
    int tiles_n, int nwg, int xcd_swizzle, int group) {
  constexpr int THREADS = WAVES_M * WAVES_N * 64;
  constexpr int MREP = BM / (WAVES_M * 16);
  constexpr int NREP = BN / (WAVES_N * 16);
  constexpr int WTM = BM / WAVES_M;
  constexpr int WTN = BN / WAVES_N;
  __shared__ __hip_bfloat16 lds[2 * 2 * TILE_HALF]; // 2 buffers x (A|B)

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * BM;
  const long bcol = (long)(wg % tiles_n) * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WAVES_N;
  const int wc = wid % WAVES_N;

  constexpr long elems_per_issue = (long)THREADS * 8;
  constexpr int ISSUES = TILE_HALF / (THREADS * 8);
  f32x4 acc[MREP][NREP] = {};

  auto stage = [&](int buf, int k0) {
    __hip_bfloat16* dst = lds + (long)buf * 2 * TILE_HALF;
    for (int issue = 0; issue < ISSUES; ++issue) {
      long o_base = (long)issue * elems_per_issue + (long)wid * (64 * 8);
      long o = lds_unskew(o_base + (long)lane * 8);
      int row = (int)(o / BK);
      int kk = (int)(o % BK);
      const __hip_bfloat16* ga = A + (brow + row) * (long)K + k0 + kk;
      const __hip_bfloat16* gb = B + (bcol + row) * (long)K + k0 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(dst + o_base), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)(dst + TILE_HALF + o_base),
          16, 0, 0);
    }
  };

  stage(0, 0);
  for (int k0 = 0; k0 < K; k0 += BK) {
    const int cur = (k0 / BK) & 1;
    const bool more = (k0 + BK) < K;
    if (more) stage(cur ^ 1, k0 + BK); // next tile's DMA, other buffer
    // retire exactly the CURRENT tile's DMAs (2*ISSUES per thread issued
    // first; FIFO), leaving the prefetch in flight across the barrier
    if (more)
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2 * ISSUES) : "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier(); // current buffer complete for all waves

    const __hip_bfloat16* la = lds + (long)cur * 2 * TILE_HALF;
    const __hip_bfloat16* lb = la + TILE_HALF;
    for (int kk = 0; kk < BK; kk += 32) {
      const int kfrag = kk + 8 * (lane >> 4);
      bf16x8 afrag[MREP], bfrag[NREP];
      for (int m = 0; m < MREP; ++m) {
        int row = wr * WTM + m * 16 + (lane & 15);
        afrag[m] = *(const bf16x8*)(la + lds_skew(row * BK + kfrag));
      }
      for (int n = 0; n < NREP; ++n) {
        int col = wc * WTN + n * 16 + (lane & 15);
        bfrag[n] = *(const bf16x8*)(lb + lds_skew(col * BK + kfrag));
      }
      for (int m = 0; m < MREP; ++m)
        for (int n = 0; n < NREP; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[m], bfrag[n], acc[m][n], 0, 0, 0);
    }
    // every wave done READING buf[cur] before the next iteration's
    // prefetch overwrites it
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  for (int m = 0; m < MREP; ++m)
    for (int n = 0; n < NREP; ++n) {
      long row0 = brow + wr * WTM + m * 16 + 4 * (lane >> 4);
      long col = bcol + wc * WTN + n * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r)
        C[(row0 + r) * (long)N + col] = acc[m][n][r];
    }
}

// ---------------------------------------------------------------------------
// K7-fp8: same GEMM at OCP fp8 e4m3 (__builtin_amdgcn_mfma_f32_16x16x32_
// fp8_fp8 runs at the bf16 MFMA rate but the staging traffic halves).
// A/B fragments are 8 packed fp8 in one i64 (2 VGPRs); LDS(row,k) =
// (row, (k + 16*((row>>2)&3)) & 63) spreads the fragment reads.
//
// Known 2-way LDS-conflict residual (measured SQ_LDS_BANK_CONFLICT =
// 0.5x IDX_ACTIVE, ~6% of wall): the compiler emits the 8-byte fragment
// reads as paired ds_read2st64_b64 whose banking is (a/4) mod 32 with
// 16-contiguous-lane groups; a conflict-FREE assignment there needs
// 8-byte K-rotations, but an 8-byte shift makes some 16-byte
// global_load_lds chunks wrap the 64-byte K range (source becomes
// non-contiguous), and row padding breaks glds lane-linearity — so
// 16-byte granularity (2-way) is the floor for this staging scheme.
// bf16's 16-bit elements dodge this: its 16-element rotation is one
// b128 window and reaches zero conflicts (see lds_skew above).
// ---------------------------------------------------------------------------
__device__ __forceinline__ long lds_skew8(long e) { // tile elem -> LDS slot
  long row = e >> 6, k = e & 63;
  return (row << 6) | ((k + 16 * ((row >> 2) & 3)) & 63);
}
__device__ __forceinline__ long lds_unskew8(long y) {
  long row = y >> 6, k = y & 63;
  return (row << 6) | ((k - 16 * ((row >> 2) & 3)) & 63);
}

template <int WAVES_M, int WAVES_N>
__global__ __launch_bounds__(WAVES_M* WAVES_N * 64) void k_gemm_fp8_nt(
    float* __restrict__ C, const unsigned char* __restrict__ A,
    const unsigned char* __restrict__ B, int M, int N, int K,
    int tiles_n, int nwg, int xcd_swizzle, int group) {
  constexpr int THREADS = WAVES_M * WAVES_N * 64;
  constexpr int MREP = BM / (WAVES_M * 16);
  constexpr int NREP = BN / (WAVES_N * 16);
  constexpr int WTM = BM / WAVES_M;
  constexpr int WTN = BN / WAVES_N;
  __shared__ unsigned char lds[2 * TILE_HALF]; // bytes: [A tile][B tile]

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * BM;
  const long bcol = (long)(wg % tiles_n) * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WAVES_N;
  const int wc = wid % WAVES_N;

  constexpr long bytes_per_issue = (long)THREADS * 16;
  constexpr int ISSUES = TILE_HALF / (THREADS * 16); // 1 byte/elem
  static_assert(ISSUES >= 1, "tile too small for the staging plan");
  f32x4 acc[MREP][NREP] = {};

  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();
    for (int issue = 0; issue < ISSUES; ++issue) {
      long o_base = (long)issue * bytes_per_issue + (long)wid * (64 * 16);
      // one 16-B chunk = 16 fp8 elements spanning k-bits 0-3; the skew
      // shift granule is 16 elements, so chunks stay contiguous
      long o = lds_unskew8(o_base + (long)lane * 16);
      int row = (int)(o / BK);
      int kk = (int)(o % BK);
      const unsigned char* ga = A + (brow + row) * (long)K + k0 + kk;
      const unsigned char* gb = B + (bcol + row) * (long)K + k0 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(lds + o_base), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)(lds + TILE_HALF + o_base),
          16, 0, 0);
    }
    __syncthreads();

    const unsigned char* la = lds;
    const unsigned char* lb = lds + TILE_HALF;
    for (int kk = 0; kk < BK; kk += 32) {
      const int kfrag = kk + 8 * (lane >> 4);
      long afrag[MREP], bfrag[NREP];
      for (int m = 0; m < MREP; ++m) {
        int row = wr * WTM + m * 16 + (lane & 15);
        afrag[m] = *(const long*)__builtin_assume_aligned(
            la + lds_skew8(row * BK + kfrag), 8);
      }
      for (int n = 0; n < NREP; ++n) {
        int col = wc * WTN + n * 16 + (lane & 15);
        bfrag[n] = *(const long*)__builtin_assume_aligned(
            lb + lds_skew8(col * BK + kfrag), 8);
      }
      for (int m = 0; m < MREP; ++m)
        for (int n = 0; n < NREP; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              afrag[m], bfrag[n], acc[m][n], 0, 0, 0);
    }
  }

  for (int m = 0; m < MREP; ++m)
    for (int n = 0; n < NREP; ++n) {
      long row0 = brow + wr * WTM + m * 16 + 4 * (lane >> 4);
      long col = bcol + wc * WTN + n * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r)
        C[(row0 + r) * (long)N + col] = acc[m][n][r];
    }
}

// ---------------------------------------------------------------------------
// K7-8ph: 256x256-tile deep-pipelined bf16 GEMM ("8-phase" schedule).
//
// The plain kernels pay an implicit vmcnt(0) drain at every K-step barrier;
// this variant never drains: staging runs a FIXED global schedule of one
// 16-KiB half-tile DMA per phase, two K-tiles stay LDS-resident in 8
// rotating half-slots, and the only VM waits are counted vmcnt(4) at the
// two tile switches per iteration (retiring exactly the tile about to be
// read; per-wave FIFO order makes the count exact).
//
// Geometry: 512 threads = 8 waves as 2(M)x4(N); per-wave output 128x64 =
// 8x4 fragments (128 acc VGPRs); BK=64; LDS = 8 x 16 KiB half-slots
// (A/B halves of 2 K-tiles) = 128 KiB -> 1 block/CU, the regime where
// manual pipelining pays.
//
// Phase schedule per iteration (tiles t even, t+1; parity p = t&1):
//   ph0 vmcnt(4) | read A(t) m0-3 + B(t) n-low | glds A(t+1)h0 | 16 MFMA
//   ph1          | read B(t) n-high            | glds A(t+1)h1 | 16 MFMA
//   ph2          | read A(t) m4-7              | glds B(t+2)h0 | 16 MFMA
//   ph3          |                             | glds B(t+2)h1 | 16 MFMA
//   ph4 vmcnt(4) | ... same for tile t+1 ...   | glds A(t+2)h0 ...
//   ph7          |                             | glds B(t+3)h1 | 16 MFMA
// Every phase: reads+glds, s_barrier, lgkmcnt(0), setprio(1), 16 MFMA,
// setprio(0), s_barrier. A slot is overwritten >=1 full phase after its
// last read (the reader's lgkmcnt(0) precedes its second barrier, which
// precedes any wave's next-phase glds issue). Prefetches past the last
// K-tile re-fetch the final tile into never-read slots, keeping the
// vmcnt counts uniform with no tail branches.
// ---------------------------------------------------------------------------
constexpr int PHALF = 128 * 64; // elements per half-tile image (16 KiB)

__global__ __launch_bounds__(512) void k_gemm_bf16_8ph(
    float* __restrict__ C, const __hip_bfloat16* __restrict__ A,
    const __hip_bfloat16* __restrict__ B, int M, int N, int K,
    int tiles_n, int nwg, int xcd_swizzle, int group) {
  __shared__ __hip_bfloat16 lds[8 * PHALF]; // 128 KiB

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * 256;
  const long bcol = (long)(wg % tiles_n) * 256;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 2;       // 2 wave-rows of 128
  const int wc = wid & 3;        // 4 wave-cols of 64

  const int ntiles = K / 64;

  // half ids within a tile: 0 = B cols 0-127, 1 = B cols 128-255,
  // 2 = A rows 0-127, 3 = A rows 128-255 (also the per-tile issue order)
  auto stage_half = [&](int tile, int half) {
    int k0 = (tile < ntiles ? tile : ntiles - 1) * 64; // clamp = pad refetch
    int slot = (tile & 1) * 4 + half;
    const __hip_bfloat16* G;
    long rbase;
    if (half < 2) {
      G = B;
      rbase = bcol + half * 128;
    } else {
      G = A;
      rbase = brow + (half - 2) * 128;
    }
    for (int issue = 0; issue < 2; ++issue) {
      long o_base = (long)issue * 4096 + (long)wid * 512;
      long o = lds_unskew(o_base + (long)lane * 8);
      int row = (int)(o >> 6);
      int kk = (int)(o & 63);
      const __hip_bfloat16* g = G + (rbase + row) * (long)K + k0 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)g,
          (__attribute__((address_space(3))) void*)(lds + (long)slot * PHALF +
                                                    o_base),
          16, 0, 0);
    }
  };

  f32x4 acc[8][4] = {};
  bf16x8 af[4][2];  // current m-group (4 m) x kk
  bf16x8 bl[2][2];  // n0-1 x kk
  bf16x8 bh[2][2];  // n2-3 x kk

  const int a_half = 2 + wr;       // this wave's A half id
  const int b_half = wc >> 1;      // this wave's B half id
  const int bcol_in_half = (wc & 1) * 64;

  auto read_a = [&](int parity, int mg) {
    const __hip_bfloat16* sa = lds + (long)(parity * 4 + a_half) * PHALF;
    for (int m = 0; m < 4; ++m)
      for (int k2 = 0; k2 < 2; ++k2) {
        int row = mg * 64 + m * 16 + (lane & 15);
        int kf = k2 * 32 + 8 * (lane >> 4);
        af[m][k2] = *(const bf16x8*)(sa + lds_skew(row * 64 + kf));
      }
  };
  auto read_b = [&](int parity, int ng, bf16x8 (*dst)[2]) {
    const __hip_bfloat16* sb = lds + (long)(parity * 4 + b_half) * PHALF;
    for (int n = 0; n < 2; ++n)
      for (int k2 = 0; k2 < 2; ++k2) {
        int col = bcol_in_half + (ng * 2 + n) * 16 + (lane & 15);
        int kf = k2 * 32 + 8 * (lane >> 4);
        dst[n][k2] = *(const bf16x8*)(sb + lds_skew(col * 64 + kf));
      }
  };
  auto mfma16 = [&](int mg, int ng, bf16x8 (*bfr)[2]) {
    for (int m = 0; m < 4; ++m)
      for (int n = 0; n < 2; ++n)
        for (int k2 = 0; k2 < 2; ++k2)
          acc[mg * 4 + m][ng * 2 + n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[m][k2], bfr[n][k2], acc[mg * 4 + m][ng * 2 + n], 0, 0, 0);
  };
  auto phase_sync_pre = [&] {
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
  };
  auto phase_sync_post = [&] {
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  };

  // prologue: B(0)h0 h1, A(0)h0 h1, B(1)h0 h1 — the steady-state tail.
  // The vmcnt+barrier pair publishes tile 0: every wave retires its own
  // share of the tile's DMAs BEFORE the barrier, so after it the whole
  // tile is LDS-visible to every reader (vmcnt is per-wave; the barrier
  // is what turns "my DMAs done" into "all DMAs done").
  stage_half(0, 0);
  stage_half(0, 1);
  stage_half(0, 2);
  stage_half(0, 3);
  stage_half(1, 0);
  stage_half(1, 1);
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); // my tile-0 DMAs done
  __builtin_amdgcn_s_barrier();                    // ...everyone's done

  for (int t = 0; t < ntiles; t += 2) {
    const int p0 = 0, p1 = 1; // even tile -> parity 0 slots
    // ph0
    stage_half(t + 1, 2);
    read_a(p0, 0);
    read_b(p0, 0, bl);
    phase_sync_pre();
    mfma16(0, 0, bl);
    phase_sync_post();
    // ph1
    stage_half(t + 1, 3);
    read_b(p0, 1, bh);
    phase_sync_pre();
    mfma16(0, 1, bh);
    phase_sync_post();
    // ph2
    stage_half(t + 2, 0);
    read_a(p0, 1);
    phase_sync_pre();
    mfma16(1, 0, bl);
    phase_sync_post();
    // ph3 — closing barrier also publishes tile t+1 (vmcnt before it)
    stage_half(t + 2, 1);
    phase_sync_pre();
    mfma16(1, 1, bh);
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); // my t+1 DMAs done
    __builtin_amdgcn_s_barrier();                    // all t+1 DMAs done
    // ph4 — tile t+1 (parity 1)
    stage_half(t + 2, 2);
    read_a(p1, 0);
    read_b(p1, 0, bl);
    phase_sync_pre();
    mfma16(0, 0, bl);
    phase_sync_post();
    // ph5
    stage_half(t + 2, 3);
    read_b(p1, 1, bh);
    phase_sync_pre();
    mfma16(0, 1, bh);
    phase_sync_post();
    // ph6
    stage_half(t + 3, 0);
    read_a(p1, 1);
    phase_sync_pre();
    mfma16(1, 0, bl);
    phase_sync_post();
    // ph7 — closing barrier also publishes tile t+2 for the next ph0
    stage_half(t + 3, 1);
    phase_sync_pre();
    mfma16(1, 1, bh);
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); // my t+2 DMAs done
    __builtin_amdgcn_s_barrier();                    // all t+2 DMAs done
  }

  for (int m = 0; m < 8; ++m)
    for (int n = 0; n < 4; ++n) {
      long row0 = brow + wr * 128 + m * 16 + 4 * (lane >> 4);
      long col = bcol + wc * 64 + n * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r)
        C[(row0 + r) * (long)N + col] = acc[m][n][r];
    }
}

// ---------------------------------------------------------------------------
// K7-8ph-fp8: the deep-pipelined 256^2 schedule at OCP fp8 e4m3 —
// half-tile images are 8 KiB (64 KiB LDS total -> 2 blocks/CU), one glds
// per half per thread (tile-switch waits become vmcnt(2)), fragments are
// 8 packed fp8 in an i64 read by ds_read_b64 through the fp8 skew.
// ---------------------------------------------------------------------------
constexpr int PHALF8 = 128 * 64; // fp8 half-tile image (8 KiB)

__global__ __launch_bounds__(512) void k_gemm_fp8_8ph(
    float* __restrict__ C, const unsigned char* __restrict__ A,
    const unsigned char* __restrict__ B, int M, int N, int K,
    int tiles_n, int nwg, int xcd_swizzle, int group) {
  __shared__ unsigned char lds[8 * PHALF8]; // 64 KiB -> 2 blocks/CU

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * 256;
  const long bcol = (long)(wg % tiles_n) * 256;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 2;       // 2 wave-rows of 128
  const int wc = wid & 3;        // 4 wave-cols of 64

  const int ntiles = K / 64;

  // half ids within a tile: 0 = B cols 0-127, 1 = B cols 128-255,
  // 2 = A rows 0-127, 3 = A rows 128-255 (also the per-tile issue order)
  auto stage_half = [&](int tile, int half) {
    int k0 = (tile < ntiles ? tile : ntiles - 1) * 64; // clamp = pad refetch
    int slot = (tile & 1) * 4 + half;
    const unsigned char* G;
    long rbase;
    if (half < 2) {
      G = B;
      rbase = bcol + half * 128;
    } else {
      G = A;
      rbase = brow + (half - 2) * 128;
    }
    {
      // one issue: 512 threads x 16 B = the whole 8-KiB half image
      long o_base = (long)wid * 1024;
      long o = lds_unskew8(o_base + (long)lane * 16);
      int row = (int)(o >> 6);
      int kk = (int)(o & 63);
      const unsigned char* g = G + (rbase + row) * (long)K + k0 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)g,
          (__attribute__((address_space(3))) void*)(lds + (long)slot * PHALF8 +
                                                    o_base),
          16, 0, 0);
    }
  };

  f32x4 acc[8][4] = {};
  long af[4][2];  // current m-group (4 m) x kk (8 packed fp8)
  long bl[2][2];  // n0-1 x kk
  long bh[2][2];  // n2-3 x kk

  const int a_half = 2 + wr;       // this wave's A half id
  const int b_half = wc >> 1;      // this wave's B half id
  const int bcol_in_half = (wc & 1) * 64;

  auto read_a = [&](int parity, int mg) {
    const unsigned char* sa = lds + (long)(parity * 4 + a_half) * PHALF8;
    for (int m = 0; m < 4; ++m)
      for (int k2 = 0; k2 < 2; ++k2) {
        int row = mg * 64 + m * 16 + (lane & 15);
        int kf = k2 * 32 + 8 * (lane >> 4);
        af[m][k2] = *(const long*)__builtin_assume_aligned(
            sa + lds_skew8(row * 64 + kf), 8);
      }
  };
  auto read_b = [&](int parity, int ng, long (*dst)[2]) {
    const unsigned char* sb = lds + (long)(parity * 4 + b_half) * PHALF8;
    for (int n = 0; n < 2; ++n)
      for (int k2 = 0; k2 < 2; ++k2) {
        int col = bcol_in_half + (ng * 2 + n) * 16 + (lane & 15);
        int kf = k2 * 32 + 8 * (lane >> 4);
        dst[n][k2] = *(const long*)__builtin_assume_aligned(
            sb + lds_skew8(col * 64 + kf), 8);
      }
  };
  auto mfma16 = [&](int mg, int ng, long (*bfr)[2]) {
    for (int m = 0; m < 4; ++m)
      for (int n = 0; n < 2; ++n)
        for (int k2 = 0; k2 < 2; ++k2)
          acc[mg * 4 + m][ng * 2 + n] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
              af[m][k2], bfr[n][k2], acc[mg * 4 + m][ng * 2 + n], 0, 0, 0);
  };
  auto phase_sync_pre = [&] {
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
  };
  auto phase_sync_post = [&] {
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  };

  // prologue: B(0)h0 h1, A(0)h0 h1, B(1)h0 h1 — the steady-state tail.
  // The vmcnt+barrier pair publishes tile 0: every wave retires its own
  // share of the tile's DMAs BEFORE the barrier, so after it the whole
  // tile is LDS-visible to every reader (vmcnt is per-wave; the barrier
  // is what turns "my DMAs done" into "all DMAs done").
  stage_half(0, 0);
  stage_half(0, 1);
  stage_half(0, 2);
  stage_half(0, 3);
  stage_half(1, 0);
  stage_half(1, 1);
  asm volatile("s_waitcnt vmcnt(2)" ::: "memory"); // my tile-0 DMAs done
  __builtin_amdgcn_s_barrier();                    // ...everyone's done

  for (int t = 0; t < ntiles; t += 2) {
    const int p0 = 0, p1 = 1; // even tile -> parity 0 slots
    // ph0
    stage_half(t + 1, 2);
    read_a(p0, 0);
    read_b(p0, 0, bl);
    phase_sync_pre();
    mfma16(0, 0, bl);
    phase_sync_post();
    // ph1
    stage_half(t + 1, 3);
    read_b(p0, 1, bh);
    phase_sync_pre();
    mfma16(0, 1, bh);
    phase_sync_post();
    // ph2
    stage_half(t + 2, 0);
    read_a(p0, 1);
    phase_sync_pre();
    mfma16(1, 0, bl);
    phase_sync_post();
    // ph3 — closing barrier also publishes tile t+1 (vmcnt before it)
    stage_half(t + 2, 1);
    phase_sync_pre();
    mfma16(1, 1, bh);
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory"); // my t+1 DMAs done
    __builtin_amdgcn_s_barrier();                    // all t+1 DMAs done
    // ph4 — tile t+1 (parity 1)
    stage_half(t + 2, 2);
    read_a(p1, 0);
    read_b(p1, 0, bl);
    phase_sync_pre();
    mfma16(0, 0, bl);
    phase_sync_post();
    // ph5
    stage_half(t + 2, 3);
    read_b(p1, 1, bh);
    phase_sync_pre();
    mfma16(0, 1, bh);
    phase_sync_post();
    // ph6
    stage_half(t + 3, 0);
    read_a(p1, 1);
    phase_sync_pre();
    mfma16(1, 0, bl);
    phase_sync_post();
    // ph7 — closing barrier also publishes tile t+2 for the next ph0
    stage_half(t + 3, 1);
    phase_sync_pre();
    mfma16(1, 1, bh);
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory"); // my t+2 DMAs done
    __builtin_amdgcn_s_barrier();                    // all t+2 DMAs done
  }

  for (int m = 0; m < 8; ++m)
    for (int n = 0; n < 4; ++n) {
      long row0 = brow + wr * 128 + m * 16 + 4 * (lane >> 4);
      long col = bcol + wc * 64 + n * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r)
        C[(row0 + r) * (long)N + col] = acc[m][n][r];
    }
}

// ---------------------------------------------------------------------------
// K7-mxfp8: block-scaled MX-fp8 GEMM — __builtin_amdgcn_mfma_scale_f32_
// 16x16x128_f8f6f4 runs at TWICE the bf16/fp8 MFMA rate (the only path to
// the low-precision peaks on gfx950) with HW-fused dequant: every
// 32-element K-block of A and B carries one e8m0 scale (byte; 127 = 1.0),
// exactly the OCP MX-FP8 format.
//   C[M,N] fp32 = (A .* 2^(As-127)) x (B .* 2^(Bs-127))^T
// A/B fp8 e4m3 [M][K]/[N][K]; As/Bs uint8 [M][K/32]/[N][K/32].
// Fragment mapping verified on hardware by bin/mx_probe: lane L holds 32
// contiguous k at 32*(L>>4), row/col L&15; C/D standard; the scale VGPR's
// byte 0 (opsel 0) scales the lane's block.
// Structure: plain two-barrier K-loop, BK = 128 (one MFMA per fragment
// pair), 8 waves of 64x32; scales ride ordinary loads + ds_write into the
// same LDS block (the __syncthreads drain makes them free here — in the
// deep-pipelined schedule they would puncture the counted-vmcnt pipeline,
// the documented trap 4(b)).
// LDS skew (b128 reads at 16-B granule in a [128][128]-byte image): the
// +4-invariant window family w = {0,1,4,5,8,9,12,13}: rotation
// k' = (k + 16*((row>>1)&5)) & 127 makes ALL FOUR true ds_read_b128 lane
// groups conflict-free (enumerated like lds_skew above).
// ---------------------------------------------------------------------------
constexpr int MXK = 128; // K-step
__device__ __forceinline__ long mx_skew(long e) { // byte index in [128][128]
  long row = e >> 7, k = e & 127;
  return (row << 7) | ((k + 16 * ((row >> 1) & 5)) & 127);
}
__device__ __forceinline__ long mx_unskew(long y) {
  long row = y >> 7, k = y & 127;
  return (row << 7) | ((k - 16 * ((row >> 1) & 5)) & 127);
}

typedef __attribute__((ext_vector_type(8))) int i32x8;

__global__ __launch_bounds__(512) void k_gemm_mxfp8_nt(
    float* __restrict__ C, const unsigned char* __restrict__ A,
    const unsigned char* __restrict__ B, const unsigned char* __restrict__ As,
    const unsigned char* __restrict__ Bs, int M, int N, int K, int tiles_n,
    int nwg, int xcd_swizzle, int group) {
  constexpr int MREP = 4, NREP = 2; // 8 waves as 2x4, 64x32 per wave
  // [A data 16K][B data 16K][A scales 512][B scales 512]
  __shared__ unsigned char lds[2 * 128 * MXK + 2 * 512];

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * 128;
  const long bcol = (long)(wg % tiles_n) * 128;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 2;
  const int wc = wid & 3;
  const int ks = K / 32; // scale row stride

  unsigned char* sA = lds + 2 * 128 * MXK;
  unsigned char* sB = sA + 512;
  f32x4 acc[MREP][NREP] = {};

  for (int k0 = 0; k0 < K; k0 += MXK) {
    __syncthreads();
    // data: [128][128] bytes per operand = 512 threads x 16 B x 2 issues
    for (int issue = 0; issue < 2; ++issue) {
      long o_base = (long)issue * 8192 + (long)wid * 1024;
      long o = mx_unskew(o_base + (long)lane * 16);
      int row = (int)(o >> 7);
      int kk = (int)(o & 127);
      const unsigned char* ga = A + (brow + row) * (long)K + k0 + kk;
      const unsigned char* gb = B + (bcol + row) * (long)K + k0 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(lds + o_base), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)(lds + 128 * MXK + o_base),
          16, 0, 0);
    }
    // scales: 128 rows x 4 k-blocks per operand; thread t stages one byte
    // of each (ordinary load + ds_write — the barrier drains them anyway)
    {
      int row = tid >> 2, kb = tid & 3;
      sA[tid] = As[(brow + row) * (long)ks + k0 / 32 + kb];
      sB[tid] = Bs[(bcol + row) * (long)ks + k0 / 32 + kb];
    }
    __syncthreads();

    // Operand/scale layout of v_mfma_scale_f32_16x16x128_f8f6f4,
    // REVERSE-ENGINEERED ON HARDWARE (scripts/probes/mx_probe8.hip, and
    // the lane-mask identification probes before it): the instruction's
    // per-lane scale byte does NOT cover the 32 bytes in the same lane's
    // operand — scale lane (row, gs) covers the same 16-byte half of the
    // data-lane PAIR {2a, 2a+1} with gs = a + 2h. Feeding each data lane
    //   half h <- k [32*(2a+h) + 16*(g&1), +16)   (a = g>>1)
    // makes every scale byte cover exactly ONE contiguous OCP MX
    // 32-block, whose index for scale purposes is blk = 2*(g&1)+(g>>1).
    // (Matmul is K-permutation-invariant, so the data permutation is free
    // as long as A and B use the same one.)
    const int g = lane >> 4;
    const int kh0 = 64 * (g >> 1) + 16 * (g & 1); // half-0 K byte offset
    const int blk = 2 * (g & 1) + (g >> 1);       // lane's MX block index
    typedef __attribute__((ext_vector_type(4))) int i32x4;
    auto frag32 = [&](const unsigned char* base, long e) {
      i32x4 lo = *(const i32x4*)__builtin_assume_aligned(
          base + mx_skew(e), 16);
      i32x4 hi = *(const i32x4*)__builtin_assume_aligned(
          base + mx_skew(e + 32), 16);
      i32x8 f;
      for (int j = 0; j < 4; ++j) {
        f[j] = lo[j];
        f[4 + j] = hi[j];
      }
      return f;
    };
    i32x8 afrag[MREP];
    int asc[MREP];
    for (int m = 0; m < MREP; ++m) {
      int row = wr * 64 + m * 16 + (lane & 15);
      afrag[m] = frag32(lds, (long)row * MXK + kh0);
      asc[m] = sA[row * 4 + blk];
    }
    for (int n = 0; n < NREP; ++n) {
      int col = wc * 32 + n * 16 + (lane & 15);
      i32x8 bfrag = frag32(lds + 128 * MXK, (long)col * MXK + kh0);
      int bsc = sB[col * 4 + blk];
      for (int m = 0; m < MREP; ++m)
        acc[m][n] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            afrag[m], bfrag, acc[m][n], 0, 0, 0, asc[m], 0, bsc);
    }
  }

  for (int m = 0; m < MREP; ++m)
    for (int n = 0; n < NREP; ++n) {
      long row0 = brow + wr * 64 + m * 16 + 4 * (lane >> 4);
      long col = bcol + wc * 32 + n * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r)
        C[(row0 + r) * (long)N + col] = acc[m][n][r];
    }
}

// ---------------------------------------------------------------------------
// K7-i8: integer GEMM — mfma_i32_16x16x64_i8 runs at ~2x the bf16 rate
// with exact int32 accumulation (the quantized-inference dtype).
//   C[M,N] (int32) = A[M,K] (int8) x B[N,K]^T (int8)
// K=64 per MFMA: each lane holds 16 contiguous int8 = one ds_read_b128.
// No scales, so the lane->k assignment only needs A/B consistency
// (matmul is K-permutation-invariant) — the natural k = 16*(lane>>4)
// order is used. LDS images are [128][64] bytes; the b128 lane-group skew
// for this geometry is a 32-byte rotation per (row>>3)&1 class (same
// derivation style as lds_skew: row classes on even 4-dword windows, the
// interleaved kfrag half on odd ones).
// ---------------------------------------------------------------------------
__device__ __forceinline__ long i8_skew(long e) {
  long row = e >> 6, k = e & 63;
  return (row << 6) | ((k + 32 * ((row >> 3) & 1)) & 63);
}
__device__ __forceinline__ long i8_unskew(long y) {
  long row = y >> 6, k = y & 63;
  return (row << 6) | ((k - 32 * ((row >> 3) & 1)) & 63);
}

typedef __attribute__((ext_vector_type(4))) int i32x4v;

template <int WAVES_M, int WAVES_N>
__global__ __launch_bounds__(WAVES_M* WAVES_N * 64) void k_gemm_i8_nt(
    int* __restrict__ C, const signed char* __restrict__ A,
    const signed char* __restrict__ B, int M, int N, int K, int tiles_n,
    int nwg, int xcd_swizzle, int group) {
  constexpr int THREADS = WAVES_M * WAVES_N * 64;
  constexpr int MREP = BM / (WAVES_M * 16);
  constexpr int NREP = BN / (WAVES_N * 16);
  constexpr int WTM = BM / WAVES_M;
  constexpr int WTN = BN / WAVES_N;
  __shared__ signed char lds[2 * TILE_HALF]; // bytes: [A tile][B tile]

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * BM;
  const long bcol = (long)(wg % tiles_n) * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid / WAVES_N;
  const int wc = wid % WAVES_N;

  constexpr long bytes_per_issue = (long)THREADS * 16;
  constexpr int ISSUES = TILE_HALF / (THREADS * 16);
  static_assert(ISSUES >= 1, "tile too small");
  i32x4v acc[MREP][NREP] = {};

  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads();
    for (int issue = 0; issue < ISSUES; ++issue) {
      long o_base = (long)issue * bytes_per_issue + (long)wid * (64 * 16);
      long o = i8_unskew(o_base + (long)lane * 16);
      int row = (int)(o >> 6);
      int kk = (int)(o & 63);
      const signed char* ga = A + (brow + row) * (long)K + k0 + kk;
      const signed char* gb = B + (bcol + row) * (long)K + k0 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(lds + o_base), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)(lds + TILE_HALF + o_base),
          16, 0, 0);
    }
    __syncthreads();

    const int kfrag = 16 * (lane >> 4);
    i32x4v afrag[MREP];
    for (int m = 0; m < MREP; ++m) {
      int row = wr * WTM + m * 16 + (lane & 15);
      afrag[m] = *(const i32x4v*)__builtin_assume_aligned(
          lds + i8_skew(row * BK + kfrag), 16);
    }
    for (int n = 0; n < NREP; ++n) {
      int col = wc * WTN + n * 16 + (lane & 15);
      i32x4v bfrag = *(const i32x4v*)__builtin_assume_aligned(
          lds + TILE_HALF + i8_skew(col * BK + kfrag), 16);
      for (int m = 0; m < MREP; ++m)
        acc[m][n] = __builtin_amdgcn_mfma_i32_16x16x64_i8(afrag[m], bfrag,
                                                          acc[m][n], 0, 0, 0);
    }
  }

  for (int m = 0; m < MREP; ++m)
    for (int n = 0; n < NREP; ++n) {
      long row0 = brow + wr * WTM + m * 16 + 4 * (lane >> 4);
      long col = bcol + wc * WTN + n * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r)
        C[(row0 + r) * (long)N + col] = acc[m][n][r];
    }
}

// NOTE (r2): a deep-pipelined 8-phase MX variant was built and measured,
// then REMOVED: MX fragments are 8 VGPRs per operand (32 bytes) and the
// 256^2 schedule's 128-VGPR accumulator plus the scale plumbing exceeded
// the 256-VGPR/wave budget of the 8-wave geometry — 152 B/lane of
// in-loop spills made it SLOWER than this plain kernel (653 vs 1146 TF
// at 4096^3) and the spill-pressured build also mis-scheduled the scale
// pipeline. The plain two-barrier structure is the right home for the
// scaled instruction at this tile geometry.

} // namespace

// ---------------------------------------------------------------------------
// K7-i8-8ph: the deep-pipelined 8-phase schedule at int8 — [128][128]-byte
// half images (16 KiB, same slot geometry and vmcnt(4) schedule as the
// bf16 8-phase kernel), 16-byte b128 fragments (4 VGPRs each, the
// LIGHTEST of the family), two K=64 MFMAs per fragment pair per K-tile.
// Skew: 32-byte rotation per (row>>1)&3 class — the evens window family
// for this geometry's +-1-unit lane-group mixing.
// ---------------------------------------------------------------------------
__device__ __forceinline__ long i8s_skew(long e) {
  long row = e >> 7, k = e & 127;
  return (row << 7) | ((k + 32 * ((row >> 1) & 3)) & 127);
}
__device__ __forceinline__ long i8s_unskew(long y) {
  long row = y >> 7, k = y & 127;
  return (row << 7) | ((k - 32 * ((row >> 1) & 3)) & 127);
}

constexpr int PH8I = 128 * 128; // i8 half-tile image bytes (16 KiB)

__global__ __launch_bounds__(512) void k_gemm_i8_8ph(
    int* __restrict__ C, const signed char* __restrict__ A,
    const signed char* __restrict__ B, int M, int N, int K,
    int tiles_n, int nwg, int xcd_swizzle, int group) {
  __shared__ signed char lds[8 * PH8I]; // 128 KiB

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * 256;
  const long bcol = (long)(wg % tiles_n) * 256;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 2;       // 2 wave-rows of 128
  const int wc = wid & 3;        // 4 wave-cols of 64

  const int ntiles = K / 128;

  // half ids within a tile: 0 = B cols 0-127, 1 = B cols 128-255,
  // 2 = A rows 0-127, 3 = A rows 128-255 (also the per-tile issue order)
  auto stage_half = [&](int tile, int half) {
    int k0 = (tile < ntiles ? tile : ntiles - 1) * 128; // clamp = pad refetch
    int slot = (tile & 1) * 4 + half;
    const signed char* G;
    long rbase;
    if (half < 2) {
      G = B;
      rbase = bcol + half * 128;
    } else {
      G = A;
      rbase = brow + (half - 2) * 128;
    }
    for (int issue = 0; issue < 2; ++issue) {
      long o_base = (long)issue * 8192 + (long)wid * 1024;
      long o = i8s_unskew(o_base + (long)lane * 16);
      int row = (int)(o >> 7);
      int kk = (int)(o & 127);
      const signed char* g = G + (rbase + row) * (long)K + k0 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)g,
          (__attribute__((address_space(3))) void*)(lds + (long)slot * PH8I +
                                                    o_base),
          16, 0, 0);
    }
  };

  i32x4v acc[8][4] = {};
  i32x4v af[4][2];  // current m-group (4 m) x kk
  i32x4v bl[2][2];  // n0-1 x kk
  i32x4v bh[2][2];  // n2-3 x kk

  const int a_half = 2 + wr;       // this wave's A half id
  const int b_half = wc >> 1;      // this wave's B half id
  const int bcol_in_half = (wc & 1) * 64;

  auto read_a = [&](int parity, int mg) {
    const signed char* sa = lds + (long)(parity * 4 + a_half) * PH8I;
    for (int m = 0; m < 4; ++m)
      for (int k2 = 0; k2 < 2; ++k2) {
        int row = mg * 64 + m * 16 + (lane & 15);
        int kf = k2 * 64 + 16 * (lane >> 4);
        af[m][k2] = *(const i32x4v*)__builtin_assume_aligned(
            sa + i8s_skew(row * 128 + kf), 16);
      }
  };
  auto read_b = [&](int parity, int ng, i32x4v (*dst)[2]) {
    const signed char* sb = lds + (long)(parity * 4 + b_half) * PH8I;
    for (int n = 0; n < 2; ++n)
      for (int k2 = 0; k2 < 2; ++k2) {
        int col = bcol_in_half + (ng * 2 + n) * 16 + (lane & 15);
        int kf = k2 * 64 + 16 * (lane >> 4);
        dst[n][k2] = *(const i32x4v*)__builtin_assume_aligned(
            sb + i8s_skew(col * 128 + kf), 16);
      }
  };
  auto mfma16 = [&](int mg, int ng, i32x4v (*bfr)[2]) {
    for (int m = 0; m < 4; ++m)
      for (int n = 0; n < 2; ++n)
        for (int k2 = 0; k2 < 2; ++k2)
          acc[mg * 4 + m][ng * 2 + n] = __builtin_amdgcn_mfma_i32_16x16x64_i8(
              af[m][k2], bfr[n][k2], acc[mg * 4 + m][ng * 2 + n], 0, 0, 0);
  };
  auto phase_sync_pre = [&] {
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
  };
  auto phase_sync_post = [&] {
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  };

  // prologue: B(0)h0 h1, A(0)h0 h1, B(1)h0 h1 — the steady-state tail.
  // The vmcnt+barrier pair publishes tile 0: every wave retires its own
  // share of the tile's DMAs BEFORE the barrier, so after it the whole
  // tile is LDS-visible to every reader (vmcnt is per-wave; the barrier
  // is what turns "my DMAs done" into "all DMAs done").
  stage_half(0, 0);
  stage_half(0, 1);
  stage_half(0, 2);
  stage_half(0, 3);
  stage_half(1, 0);
  stage_half(1, 1);
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); // my tile-0 DMAs done
  __builtin_amdgcn_s_barrier();                    // ...everyone's done

  for (int t = 0; t < ntiles; t += 2) {
    const int p0 = 0, p1 = 1; // even tile -> parity 0 slots
    // ph0
    stage_half(t + 1, 2);
    read_a(p0, 0);
    read_b(p0, 0, bl);
    phase_sync_pre();
    mfma16(0, 0, bl);
    phase_sync_post();
    // ph1
    stage_half(t + 1, 3);
    read_b(p0, 1, bh);
    phase_sync_pre();
    mfma16(0, 1, bh);
    phase_sync_post();
    // ph2
    stage_half(t + 2, 0);
    read_a(p0, 1);
    phase_sync_pre();
    mfma16(1, 0, bl);
    phase_sync_post();
    // ph3 — closing barrier also publishes tile t+1 (vmcnt before it)
    stage_half(t + 2, 1);
    phase_sync_pre();
    mfma16(1, 1, bh);
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); // my t+1 DMAs done
    __builtin_amdgcn_s_barrier();                    // all t+1 DMAs done
    // ph4 — tile t+1 (parity 1)
    stage_half(t + 2, 2);
    read_a(p1, 0);
    read_b(p1, 0, bl);
    phase_sync_pre();
    mfma16(0, 0, bl);
    phase_sync_post();
    // ph5
    stage_half(t + 2, 3);
    read_b(p1, 1, bh);
    phase_sync_pre();
    mfma16(0, 1, bh);
    phase_sync_post();
    // ph6
    stage_half(t + 3, 0);
    read_a(p1, 1);
    phase_sync_pre();
    mfma16(1, 0, bl);
    phase_sync_post();
    // ph7 — closing barrier also publishes tile t+2 for the next ph0
    stage_half(t + 3, 1);
    phase_sync_pre();
    mfma16(1, 1, bh);
    __builtin_amdgcn_s_setprio(0);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); // my t+2 DMAs done
    __builtin_amdgcn_s_barrier();                    // all t+2 DMAs done
  }

  for (int m = 0; m < 8; ++m)
    for (int n = 0; n < 4; ++n) {
      long row0 = brow + wr * 128 + m * 16 + 4 * (lane >> 4);
      long col = bcol + wc * 64 + n * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r)
        C[(row0 + r) * (long)N + col] = acc[m][n][r];
    }
}

// Band width (in N-tiles) of the grouped tile order above. Defaults come
// from interleaved A/B sweeps at 8192^3 and 16384^3 (clock-drift-controlled;
// docs/gemm.md): row-major is already LLC-friendly up to ~32 N-tiles, wider
// grids gain 4-7% from 16/32-wide bands. HPK_GEMM_GROUP overrides.

// 256^2-tile 32x32x32 int8 kernel — the fp8 _32 design (see
// k_gemm_mxfp8_nt_32) at the non-scaled i8 MFMA: per-lane fragment = 16
// contiguous K elements at k = 16*(lane>>5) of each K-32 step (one
// ds_read_b128), exact int32 accumulation, no scale staging (vmcnt(8)
// prefetch count), same (row>>2)&7 anti-conflict chunk rotation on the
// 128-byte rows.
__global__ __launch_bounds__(512) void k_gemm_i8_nt_32(
    int* __restrict__ C, const signed char* __restrict__ A,
    const signed char* __restrict__ B, int M, int N, int K, int tiles_n,
    int nwg, int xcd_swizzle, int group) {
  constexpr int TILE = 256 * 128; // 32 KiB per operand per buffer (K-128)
  __shared__ unsigned char lds[2 * 2 * TILE];

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * 256;
  const long bcol = (long)(wg % tiles_n) * 256;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1;
  const int wc = wid & 1;

  typedef __attribute__((ext_vector_type(16))) int i32x16;
  typedef __attribute__((ext_vector_type(4))) int i32x4;
  i32x16 acc[2][4] = {};

  auto stage = [&](int buf, int k0) {
    unsigned char* dst = lds + (long)buf * 2 * TILE;
    for (int issue = 0; issue < 4; ++issue) {
      long o_base = (long)issue * 8192 + (long)wid * 1024;
      long o = o_base + (long)lane * 16;
      int row = (int)(o >> 7);
      int p = (int)((o & 127) >> 4);
      int kk = ((p - (row >> 2)) & 7) * 16;
      const signed char* ga = A + (brow + row) * (long)K + k0 + kk;
      const signed char* gb = B + (bcol + row) * (long)K + k0 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(dst + o_base), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)(dst + TILE + o_base), 16,
          0, 0);
    }
  };

  const int g = lane >> 5;
  const int r31 = lane & 31;
  int a_off[2], b_off[4];
  for (int mf = 0; mf < 2; ++mf) a_off[mf] = (wr * 64 + mf * 32 + r31) * 128;
  for (int nf = 0; nf < 4; ++nf) b_off[nf] = (wc * 128 + nf * 32 + r31) * 128;

  stage(0, 0);
  for (int k0 = 0; k0 < K; k0 += 128) {
    const int cur = (k0 >> 7) & 1;
    const bool more = (k0 + 128) < K;
    if (more) stage(cur ^ 1, k0 + 128);
    if (more)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    const unsigned char* la = lds + (long)cur * 2 * TILE;
    const unsigned char* lb = la + TILE;
    for (int kk = 0; kk < 4; ++kk) {
      i32x4 afrag[2];
      for (int mf = 0; mf < 2; ++mf) {
        int row = wr * 64 + mf * 32 + r31;
        int ch = (2 * kk + g + (row >> 2)) & 7;
        afrag[mf] = *(const i32x4*)__builtin_assume_aligned(
            la + a_off[mf] + 16 * ch, 16);
      }
      for (int nf = 0; nf < 4; ++nf) {
        int col = wc * 128 + nf * 32 + r31;
        int ch = (2 * kk + g + (col >> 2)) & 7;
        i32x4 bfrag = *(const i32x4*)__builtin_assume_aligned(
            lb + b_off[nf] + 16 * ch, 16);
        for (int mf = 0; mf < 2; ++mf)
          acc[mf][nf] = __builtin_amdgcn_mfma_i32_32x32x32_i8(
              afrag[mf], bfrag, acc[mf][nf], 0, 0, 0);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  for (int mf = 0; mf < 2; ++mf)
    for (int nf = 0; nf < 4; ++nf) {
      long col = bcol + wc * 128 + nf * 32 + r31;
      for (int r = 0; r < 16; ++r) {
        long row = brow + wr * 64 + mf * 32 + (r & 3) + 8 * (r >> 2) + 4 * g;
        C[row * (long)N + col] = acc[mf][nf][r];
      }
    }
}


static int gemm_group(int tiles_n) {
  if (const char* env = std::getenv("HPK_GEMM_GROUP")) return std::atoi(env);
  if (tiles_n >= 128) return 32;
  if (tiles_n >= 64) return 16;
  return 1;
}

void launch_gemm_i8_nt(int* C, const void* A, const void* B, long M,
                       long N, long K, hipStream_t stream, int xcd_swizzle) {
  if (M % BM != 0 || N % BN != 0 || K % BK != 0)
    throw std::runtime_error(
        "gemm_i8_nt requires M,N % 128 == 0 and K % 64 == 0");
  const int grp = gemm_group((int)(N / 256));
  const char* var = std::getenv("HPK_GEMM_VARIANT");
  // HPK_GEMM_VARIANT=32: the 256^2 32x32x32 kernel — measured NEGATIVE
  // for i8 (1984/2074 vs the 8-phase 16x16x64 pipeline's 2294/2393 TOPS
  // at 8192/16384^3: the 16x16x64 instruction already has the same
  // 16-byte-operand economy and the deep pipeline out-schedules the db
  // structure); kept selectable, default stays 8ph
  if (var && std::string(var) == "32" && M % 256 == 0 && N % 256 == 0 &&
      K % 128 == 0) {
    int tn32 = (int)(N / 256);
    int n32 = (int)(M / 256) * tn32;
    const char* genv = std::getenv("HPK_GEMM_GROUP");
    const int grp32 = genv ? std::atoi(genv) : 1;
    hipLaunchKernelGGL(k_gemm_i8_nt_32, dim3(n32), dim3(512), 0, stream, C,
                       (const signed char*)A, (const signed char*)B, (int)M,
                       (int)N, (int)K, tn32, n32, xcd_swizzle, grp32);
    check_hip(hipGetLastError(), "launch_gemm_i8_nt(32)");
    return;
  }
  const bool ph8 = !var || std::string(var) == "8ph";
  if (ph8 && M % 256 == 0 && N % 256 == 0 && K % 256 == 0) {
    int tn = (int)(N / 256);
    int n8 = (int)(M / 256) * tn;
    hipLaunchKernelGGL(k_gemm_i8_8ph, dim3(n8), dim3(512), 0, stream, C,
                       (const signed char*)A, (const signed char*)B, (int)M,
                       (int)N, (int)K, tn, n8, xcd_swizzle, grp);
    check_hip(hipGetLastError(), "launch_gemm_i8_nt(8ph)");
    return;
  }
  int tiles_n = (int)(N / BN);
  int nwg = (int)(M / BM) * tiles_n;
  hipLaunchKernelGGL((k_gemm_i8_nt<2, 4>), dim3(nwg), dim3(512), 0, stream,
                     C, (const signed char*)A, (const signed char*)B, (int)M,
                     (int)N, (int)K, tiles_n, nwg, xcd_swizzle, grp);
  check_hip(hipGetLastError(), "launch_gemm_i8_nt");
}


// ---------------------------------------------------------------------------
// K7-mxfp4: block-scaled OCP MX-fp4 GEMM — the same scaled MFMA in fp4
// (e2m1) mode, the 4x-bf16 rate class (~10 PF dense spec; fp4 shares the
// fp6 rate on CDNA4). fp4_probe2/fp4_probe1 (scripts/probes/) measured a
// DIAGONAL operand/scale layout — unlike fp8's cross-lane pattern: data
// lane (row, g) supplies the ONE contiguous OCP 32-block k in
// [32g, 32g+32) packed 2 elements/byte (within-block nibble order is a
// free consistent K-permutation — both orders validated), and the lane's
// OWN scale operand byte 0 (opsel 0) covers exactly that block. Each
// fragment is therefore ONE contiguous 16-byte LDS read, and the natural
// [128 rows][64 bytes] tile is bank-conflict-free for the true b128 lane
// groups (window (4*row + g) mod 32 — enumerated in
// tests/test_gemm_skew_logic.py): NO skew is needed. Operands live in
// the low 4 VGPRs of the v8i32 builtin argument (fp4 reads 16 B/lane).
constexpr int MX4B = 64; // packed bytes per row per K-tile (128 elements)
__global__ __launch_bounds__(512) void k_gemm_mxfp4_nt(
    float* __restrict__ C, const unsigned char* __restrict__ A,
    const unsigned char* __restrict__ B, const unsigned char* __restrict__ As,
    const unsigned char* __restrict__ Bs, int M, int N, int K, int tiles_n,
    int nwg, int xcd_swizzle, int group) {
  constexpr int MREP = 4, NREP = 2; // 8 waves as 2x4, 64x32 per wave
  // [A data 8K][B data 8K][A scales 512][B scales 512]
  __shared__ unsigned char lds[2 * 128 * MX4B + 2 * 512];

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * 128;
  const long bcol = (long)(wg % tiles_n) * 128;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 2;
  const int wc = wid & 3;
  const int ks = K / 32;       // scale row stride
  const long Kb = (long)K / 2; // packed data row stride (bytes)

  unsigned char* sA = lds + 2 * 128 * MX4B;
  unsigned char* sB = sA + 512;
  f32x4 acc[MREP][NREP] = {};

  for (int k0 = 0; k0 < K; k0 += 128) {
    __syncthreads();
    // data: [128][64] bytes per operand = 512 threads x 16 B, one issue
    {
      long o_base = (long)wid * 1024;
      long o = o_base + (long)lane * 16;
      int row = (int)(o >> 6);
      int kk = (int)(o & 63);
      const unsigned char* ga = A + (brow + row) * Kb + k0 / 2 + kk;
      const unsigned char* gb = B + (bcol + row) * Kb + k0 / 2 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(lds + o_base), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)(lds + 128 * MX4B + o_base),
          16, 0, 0);
    }
    // scales: 128 rows x 4 k-blocks per operand (as in the mx8 kernel)
    {
      int row = tid >> 2, kb = tid & 3;
      sA[tid] = As[(brow + row) * (long)ks + k0 / 32 + kb];
      sB[tid] = Bs[(bcol + row) * (long)ks + k0 / 32 + kb];
    }
    __syncthreads();

    const int g = lane >> 4;
    typedef __attribute__((ext_vector_type(4))) int i32x4;
    auto frag16 = [&](const unsigned char* base, long byteoff) {
      i32x4 lo = *(const i32x4*)__builtin_assume_aligned(base + byteoff, 16);
      i32x8 f = {};
      for (int j = 0; j < 4; ++j) f[j] = lo[j];
      return f;
    };
    i32x8 afrag[MREP];
    int asc[MREP];
    for (int m = 0; m < MREP; ++m) {
      int row = wr * 64 + m * 16 + (lane & 15);
      afrag[m] = frag16(lds, (long)row * MX4B + 16 * g);
      asc[m] = sA[row * 4 + g];
    }
    for (int n = 0; n < NREP; ++n) {
      int col = wc * 32 + n * 16 + (lane & 15);
      i32x8 bfrag = frag16(lds + 128 * MX4B, (long)col * MX4B + 16 * g);
      int bsc = sB[col * 4 + g];
      for (int m = 0; m < MREP; ++m)
        acc[m][n] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            afrag[m], bfrag, acc[m][n], 4, 4, 0, asc[m], 0, bsc);
    }
  }

  for (int m = 0; m < MREP; ++m)
    for (int n = 0; n < NREP; ++n) {
      long row0 = brow + wr * 64 + m * 16 + 4 * (lane >> 4);
      long col = bcol + wc * 32 + n * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r)
        C[(row0 + r) * (long)N + col] = acc[m][n][r];
    }
}


// 256^2-tile 32x32x64 MX-fp8 kernel — the mx4 _32 design (see
// k_gemm_mxfp4_nt_32) ported to fp8: same arithmetic-intensity argument
// (the 128^2 mx8 kernel's 127 FLOP/staged-byte = a ~1 PF naive HBM
// ceiling, measured 1.4-1.5 PF with LLC help; 256^2 doubles it), same
// double-buffered all-glds staging. fp8 differences, both measured on
// hardware (scripts/probes/fp8_probe*_32x32): the operand/scale layout
// at this shape is HALF-INTERLEAVED, not diagonal — scale lane (row,gs)
// covers bytes [16gs,+16) of BOTH g-lanes, so half h of lane (row,g)
// feeds from k [32h + 16g, +16) and the lane passes the scale byte for
// block g — and the 128-byte rows need a (row>>2)&7 chunk rotation
// (4 rows share each (8row mod 32) window band; their row>>2 values are
// distinct mod 8 — enumerated in tests/test_gemm_skew_logic.py).
// SCALED=false turns it into the plain-fp8 kernel: the scaled MFMA with
// a hardcoded 127 (= x1.0) scale IS the non-scaled fp8 GEMM, and no
// 32x32x64 non-scaled fp8 MFMA exists — scale staging drops out
// entirely (one fewer glds per wave per tile).
template <bool SCALED>
__global__ __launch_bounds__(512) void k_gemm_mxfp8_nt_32(
    float* __restrict__ C, const unsigned char* __restrict__ A,
    const unsigned char* __restrict__ B, const unsigned char* __restrict__ As,
    const unsigned char* __restrict__ Bs, int M, int N, int K, int tiles_n,
    int nwg, int xcd_swizzle, int group) {
  constexpr int TILE = 256 * 128; // 32 KiB per operand per buffer (K-128)
  __shared__ unsigned char lds[2 * 2 * TILE + 2 * 2048];
  unsigned char* const sbase = lds + 2 * 2 * TILE;

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * 256;
  const long bcol = (long)(wg % tiles_n) * 256;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1;
  const int wc = wid & 1;
  const int ks = K / 32;

  typedef __attribute__((ext_vector_type(16))) float f32x16;
  typedef __attribute__((ext_vector_type(4))) int i32x4;
  f32x16 acc[2][4] = {};

  auto stage = [&](int buf, int k0) {
    unsigned char* dst = lds + (long)buf * 2 * TILE;
    for (int issue = 0; issue < 4; ++issue) {
      long o_base = (long)issue * 8192 + (long)wid * 1024;
      long o = o_base + (long)lane * 16;
      int row = (int)(o >> 7);
      int p = (int)((o & 127) >> 4);
      int kk = ((p - (row >> 2)) & 7) * 16;
      const unsigned char* ga = A + (brow + row) * (long)K + k0 + kk;
      const unsigned char* gb = B + (bcol + row) * (long)K + k0 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(dst + o_base), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)(dst + TILE + o_base), 16,
          0, 0);
    }
    if (SCALED) {
      const unsigned char* S = (wid < 4) ? As : Bs;
      long rbase = (wid < 4) ? brow : bcol;
      int srow = (wid & 3) * 64 + lane;
      const unsigned char* gs = S + (rbase + srow) * (long)ks + k0 / 32;
      unsigned char* sdst =
          sbase + (long)buf * 2048 + (wid >= 4 ? 1024 : 0) + (wid & 3) * 256;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gs,
          (__attribute__((address_space(3))) void*)sdst, 4, 0, 0);
    }
  };

  const int g = lane >> 5;
  const int r31 = lane & 31;
  int a_off[2], asc_off[2], b_off[4], bsc_off[4];
  for (int mf = 0; mf < 2; ++mf) {
    int row = wr * 64 + mf * 32 + r31;
    a_off[mf] = row * 128;
    asc_off[mf] = row * 4 + g;
  }
  for (int nf = 0; nf < 4; ++nf) {
    int col = wc * 128 + nf * 32 + r31;
    b_off[nf] = col * 128;
    bsc_off[nf] = col * 4 + g;
  }
  i32x8 afrag[2];
  i32x8 bfrag[4];

  stage(0, 0);
  for (int k0 = 0; k0 < K; k0 += 128) {
    const int cur = (k0 >> 7) & 1;
    const bool more = (k0 + 128) < K;
    if (more) stage(cur ^ 1, k0 + 128);
    if (more)
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(SCALED ? 9 : 8) : "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    const unsigned char* la = lds + (long)cur * 2 * TILE;
    const unsigned char* lb = la + TILE;
    const unsigned char* sA = sbase + (long)cur * 2048;
    const unsigned char* sB = sA + 1024;
    for (int kk = 0; kk < 2; ++kk) {
      int asc[2];
      for (int mf = 0; mf < 2; ++mf) {
        int row = wr * 64 + mf * 32 + r31;
        int rot = row >> 2;
        int ch0 = (4 * kk + g + rot) & 7;      // half h=0
        int ch1 = (4 * kk + 2 + g + rot) & 7;  // half h=1
        *(i32x4*)&afrag[mf] = *(const i32x4*)__builtin_assume_aligned(
            la + a_off[mf] + 16 * ch0, 16);
        *((i32x4*)&afrag[mf] + 1) = *(const i32x4*)__builtin_assume_aligned(
            la + a_off[mf] + 16 * ch1, 16);
        asc[mf] = SCALED ? sA[asc_off[mf] + 2 * kk] : 127;
      }
      for (int nf = 0; nf < 4; ++nf) {
        int col = wc * 128 + nf * 32 + r31;
        int rot = col >> 2;
        int ch0 = (4 * kk + g + rot) & 7;
        int ch1 = (4 * kk + 2 + g + rot) & 7;
        *(i32x4*)&bfrag[nf] = *(const i32x4*)__builtin_assume_aligned(
            lb + b_off[nf] + 16 * ch0, 16);
        *((i32x4*)&bfrag[nf] + 1) = *(const i32x4*)__builtin_assume_aligned(
            lb + b_off[nf] + 16 * ch1, 16);
        int bsc = SCALED ? sB[bsc_off[nf] + 2 * kk] : 127;
        for (int mf = 0; mf < 2; ++mf)
          acc[mf][nf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              afrag[mf], bfrag[nf], acc[mf][nf], 0, 0, 0, asc[mf], 0, bsc);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  for (int mf = 0; mf < 2; ++mf)
    for (int nf = 0; nf < 4; ++nf) {
      long col = bcol + wc * 128 + nf * 32 + r31;
      for (int r = 0; r < 16; ++r) {
        long row = brow + wr * 64 + mf * 32 + (r & 3) + 8 * (r >> 2) + 4 * g;
        C[row * (long)N + col] = acc[mf][nf][r];
      }
    }
}

void launch_gemm_mxfp8_nt(float* C, const void* A, const void* B,
                          const void* As, const void* Bs, long M, long N,
                          long K, hipStream_t stream, int xcd_swizzle) {
  if (M % 128 != 0 || N % 128 != 0 || K % 128 != 0)
    throw std::runtime_error("gemm_mxfp8_nt requires M,N,K % 128 == 0");
  // default: the 256^2 32x32x64 kernel for eligible shapes
  // (HPK_MX8_VARIANT=plain forces the 128^2 16x16x128 kernel)
  const char* v8 = std::getenv("HPK_MX8_VARIANT");
  const bool m32 = (!v8 || std::string(v8) == "32");
  if (m32 && M % 256 == 0 && N % 256 == 0) {
    int tn32 = (int)(N / 256);
    int n32 = (int)(M / 256) * tn32;
    const char* genv = std::getenv("HPK_GEMM_GROUP");
    const int grp32 = genv ? std::atoi(genv) : 1; // row-major (as mx4 _32)
    hipLaunchKernelGGL((k_gemm_mxfp8_nt_32<true>), dim3(n32), dim3(512), 0,
                       stream,
                       C, (const unsigned char*)A, (const unsigned char*)B,
                       (const unsigned char*)As, (const unsigned char*)Bs,
                       (int)M, (int)N, (int)K, tn32, n32, xcd_swizzle,
                       grp32);
    check_hip(hipGetLastError(), "launch_gemm_mxfp8_nt(32)");
    return;
  }
  const int grp = gemm_group((int)(N / 128));
  int tiles_n = (int)(N / 128);
  int nwg = (int)(M / 128) * tiles_n;
  hipLaunchKernelGGL(k_gemm_mxfp8_nt, dim3(nwg), dim3(512), 0, stream, C,
                     (const unsigned char*)A, (const unsigned char*)B,
                     (const unsigned char*)As, (const unsigned char*)Bs,
                     (int)M, (int)N, (int)K, tiles_n, nwg, xcd_swizzle, grp);
  check_hip(hipGetLastError(), "launch_gemm_mxfp8_nt");
}



// 4-wave variant: same diagonal-layout staging, 64x64 per wave (MREP =
// NREP = 4) — 16 MFMAs per wave per K-tile instead of 8, halving the
// relative cost of the per-tile barrier + staging-drain stall that PMC
// shows dominating the 8-wave kernel (MfmaUtil 20.6%). HPK_MX4_WAVES
// selects; the launcher defaults to the measured winner.
__global__ __launch_bounds__(256) void k_gemm_mxfp4_nt_w4(
    float* __restrict__ C, const unsigned char* __restrict__ A,
    const unsigned char* __restrict__ B, const unsigned char* __restrict__ As,
    const unsigned char* __restrict__ Bs, int M, int N, int K, int tiles_n,
    int nwg, int xcd_swizzle, int group) {
  constexpr int MREP = 4, NREP = 4; // 4 waves as 2x2, 64x64 per wave
  __shared__ unsigned char lds[2 * 128 * MX4B + 2 * 512];

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * 128;
  const long bcol = (long)(wg % tiles_n) * 128;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1;
  const int wc = wid & 1;
  const int ks = K / 32;
  const long Kb = (long)K / 2;

  unsigned char* sA = lds + 2 * 128 * MX4B;
  unsigned char* sB = sA + 512;
  f32x4 acc[MREP][NREP] = {};

  for (int k0 = 0; k0 < K; k0 += 128) {
    __syncthreads();
    // data: [128][64] bytes per operand = 256 threads x 16 B x 2 issues
    for (int issue = 0; issue < 2; ++issue) {
      long o_base = (long)issue * 4096 + (long)wid * 1024;
      long o = o_base + (long)lane * 16;
      int row = (int)(o >> 6);
      int kk = (int)(o & 63);
      const unsigned char* ga = A + (brow + row) * Kb + k0 / 2 + kk;
      const unsigned char* gb = B + (bcol + row) * Kb + k0 / 2 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(lds + o_base), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)(lds + 128 * MX4B + o_base),
          16, 0, 0);
    }
    // scales: 512 entries per operand, 2 per thread
    {
      int e0 = tid * 2;
      int row = e0 >> 2, kb = e0 & 3;
      sA[e0] = As[(brow + row) * (long)ks + k0 / 32 + kb];
      sB[e0] = Bs[(bcol + row) * (long)ks + k0 / 32 + kb];
      int e1 = e0 + 1;
      int row1 = e1 >> 2, kb1 = e1 & 3;
      sA[e1] = As[(brow + row1) * (long)ks + k0 / 32 + kb1];
      sB[e1] = Bs[(bcol + row1) * (long)ks + k0 / 32 + kb1];
    }
    __syncthreads();

    const int g = lane >> 4;
    typedef __attribute__((ext_vector_type(4))) int i32x4;
    auto frag16 = [&](const unsigned char* base, long byteoff) {
      i32x4 lo = *(const i32x4*)__builtin_assume_aligned(base + byteoff, 16);
      i32x8 f = {};
      for (int j = 0; j < 4; ++j) f[j] = lo[j];
      return f;
    };
    i32x8 afrag[MREP];
    int asc[MREP];
    for (int m = 0; m < MREP; ++m) {
      int row = wr * 64 + m * 16 + (lane & 15);
      afrag[m] = frag16(lds, (long)row * MX4B + 16 * g);
      asc[m] = sA[row * 4 + g];
    }
    for (int n = 0; n < NREP; ++n) {
      int col = wc * 64 + n * 16 + (lane & 15);
      i32x8 bfrag = frag16(lds + 128 * MX4B, (long)col * MX4B + 16 * g);
      int bsc = sB[col * 4 + g];
      for (int m = 0; m < MREP; ++m)
        acc[m][n] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            afrag[m], bfrag, acc[m][n], 4, 4, 0, asc[m], 0, bsc);
    }
  }

  for (int m = 0; m < MREP; ++m)
    for (int n = 0; n < NREP; ++n) {
      long row0 = brow + wr * 64 + m * 16 + 4 * (lane >> 4);
      long col = bcol + wc * 64 + n * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r)
        C[(row0 + r) * (long)N + col] = acc[m][n][r];
    }
}


// Double-buffered variant: tile t+1's staging DMAs issue before tile t's
// compute and a counted s_waitcnt retires exactly tile t's (the bf16 db
// discipline, findings #18). ALL staging is global_load_lds — including
// the scales, gathered as 4-byte dwords by waves 0-3 (K % 128 makes the
// scale rows dword-aligned) — so the per-wave vmcnt count is exact
// (3 issues for waves 0-3, 2 for waves 4-7; mixing ordinary loads into
// the queue would break the count — the guide's load-kind trap). PMC
// motivation: the plain kernel's MfmaUtil is 20.6% — each tile eats a
// full memory latency inside the barrier window; with 34 KiB LDS and 85
// VGPRs two 8-wave blocks stay resident on top of the intra-block
// prefetch. HPK_MX4_WAVES=db selects... (launcher: HPK_GEMM_VARIANT
// conventions kept: default is the measured winner).
__global__ __launch_bounds__(512) void k_gemm_mxfp4_nt_db(
    float* __restrict__ C, const unsigned char* __restrict__ A,
    const unsigned char* __restrict__ B, const unsigned char* __restrict__ As,
    const unsigned char* __restrict__ Bs, int M, int N, int K, int tiles_n,
    int nwg, int xcd_swizzle, int group) {
  constexpr int MREP = 4, NREP = 2;
  constexpr int TILE = 128 * MX4B; // 8 KiB per operand per buffer
  // ONE shared array (a second __shared__ object would force vmcnt(0)
  // before every ds_read): [buf][A|B data] x2, then [buf][sA|sB] x2
  __shared__ unsigned char lds[2 * 2 * TILE + 2 * 1024];
  unsigned char* const sbase = lds + 2 * 2 * TILE;

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * 128;
  const long bcol = (long)(wg % tiles_n) * 128;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 2;
  const int wc = wid & 3;
  const int ks = K / 32;
  const long Kb = (long)K / 2;

  f32x4 acc[MREP][NREP] = {};

  auto stage = [&](int buf, int k0) {
    unsigned char* dst = lds + (long)buf * 2 * TILE;
    long o_base = (long)wid * 1024;
    long o = o_base + (long)lane * 16;
    int row = (int)(o >> 6);
    int kk = (int)(o & 63);
    const unsigned char* ga = A + (brow + row) * Kb + k0 / 2 + kk;
    const unsigned char* gb = B + (bcol + row) * Kb + k0 / 2 + kk;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)ga,
        (__attribute__((address_space(3))) void*)(dst + o_base), 16, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)gb,
        (__attribute__((address_space(3))) void*)(dst + TILE + o_base), 16, 0,
        0);
    if (wid < 4) {
      // scale dwords: wave w, lane l gathers rows (w&1)*64+l of As (w<2)
      // or Bs — dest layout stays s[row*4 + kb]
      const unsigned char* S = (wid < 2) ? As : Bs;
      long rbase = (wid < 2) ? brow : bcol;
      int srow = (wid & 1) * 64 + lane;
      const unsigned char* gs = S + (rbase + srow) * (long)ks + k0 / 32;
      unsigned char* sdst =
          sbase + (long)buf * 1024 + (wid >= 2 ? 512 : 0) + (wid & 1) * 256;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gs,
          (__attribute__((address_space(3))) void*)sdst, 4, 0, 0);
    }
  };

  // VALU-diet (PMC showed 4 VALU per MFMA on the lambda version):
  // per-lane LDS byte offsets are K-invariant — compute once; operand
  // i32x8s persist across the loop so their top halves (unused in fp4
  // mode) stay zero with no per-tile re-init/copies.
  typedef __attribute__((ext_vector_type(4))) int i32x4;
  const int g = lane >> 4;
  int a_off[MREP], asc_off[MREP], b_off[NREP], bsc_off[NREP];
  for (int m = 0; m < MREP; ++m) {
    int row = wr * 64 + m * 16 + (lane & 15);
    a_off[m] = row * MX4B + 16 * g;
    asc_off[m] = row * 4 + g;
  }
  for (int n = 0; n < NREP; ++n) {
    int col = wc * 32 + n * 16 + (lane & 15);
    b_off[n] = col * MX4B + 16 * g;
    bsc_off[n] = col * 4 + g;
  }
  i32x8 afrag[MREP] = {};
  i32x8 bfrag[NREP] = {};

  stage(0, 0);
  for (int k0 = 0; k0 < K; k0 += 128) {
    const int cur = (k0 >> 7) & 1;
    const bool more = (k0 + 128) < K;
    if (more) stage(cur ^ 1, k0 + 128);
    if (more) {
      if (wid < 4)
        asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    const unsigned char* la = lds + (long)cur * 2 * TILE;
    const unsigned char* lb = la + TILE;
    const unsigned char* sA = sbase + (long)cur * 1024;
    const unsigned char* sB = sA + 512;
    int asc[MREP];
    for (int m = 0; m < MREP; ++m) {
      *(i32x4*)&afrag[m] = *(const i32x4*)__builtin_assume_aligned(
          la + a_off[m], 16);
      asc[m] = sA[asc_off[m]];
    }
    for (int n = 0; n < NREP; ++n) {
      *(i32x4*)&bfrag[n] = *(const i32x4*)__builtin_assume_aligned(
          lb + b_off[n], 16);
      int bsc = sB[bsc_off[n]];
      for (int m = 0; m < MREP; ++m)
        acc[m][n] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            afrag[m], bfrag[n], acc[m][n], 4, 4, 0, asc[m], 0, bsc);
    }
    // all waves done reading buf[cur] before the next prefetch overwrites
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  for (int m = 0; m < MREP; ++m)
    for (int n = 0; n < NREP; ++n) {
      long row0 = brow + wr * 64 + m * 16 + 4 * (lane >> 4);
      long col = bcol + wc * 32 + n * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r)
        C[(row0 + r) * (long)N + col] = acc[m][n][r];
    }
}


// 32x32x64 variant at a 256^2 tile: the bigger scaled MFMA
// (mfma_scale_f32_32x32x64_f8f6f4, 9099 TF ubench) plus the bigger tile
// attack the two measured walls in order: an un-skewed 64-byte-stride
// tile read 8-way bank conflicts (SQ_LDS_BANK_CONFLICT = 4x MFMA count,
// fixed by the (row>>3)&3 chunk rotation below), and the 128^2 tile's
// arithmetic intensity (254 FLOP/staged-byte = a ~2.0 PF HBM ceiling at
// 8 TB/s — the measured 2.0-2.1 PF plateau of ALL 128^2 fp4 variants).
// 256^2 doubles intensity (508 FLOP/B -> ~4 PF ceiling). fp4_probe3
// measured the same DIAGONAL operand/scale layout at this shape: lane
// (row=lane&31, g=lane>>5) supplies OCP block k in [32g,32g+32),
// own-lane scale byte. 8 waves as 4x2 (64x128 per wave, 16 MFMAs per
// K-128 tile), f32x16 accumulators (128 AGPRs), double-buffered
// all-glds staging (scale rows as size-4 glds: sub-dword glds lands in
// 4-byte-per-lane LDS slots — measured, and a K-128 tile's 4 scale
// bytes fill the slot exactly).
__global__ __launch_bounds__(512) void k_gemm_mxfp4_nt_32(
    float* __restrict__ C, const unsigned char* __restrict__ A,
    const unsigned char* __restrict__ B, const unsigned char* __restrict__ As,
    const unsigned char* __restrict__ Bs, int M, int N, int K, int tiles_n,
    int nwg, int xcd_swizzle, int group) {
  constexpr int TILE = 256 * 64; // 16 KiB per operand per buffer (K-128)
  __shared__ unsigned char lds[2 * 2 * TILE + 2 * 2048];
  unsigned char* const sbase = lds + 2 * 2 * TILE;

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * 256;
  const long bcol = (long)(wg % tiles_n) * 256;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1; // 4 wave-rows of 64
  const int wc = wid & 1;  // 2 wave-cols of 128
  const int ks = K / 32;
  const long Kb = (long)K / 2;

  typedef __attribute__((ext_vector_type(16))) float f32x16;
  typedef __attribute__((ext_vector_type(4))) int i32x4;
  f32x16 acc[2][4] = {};

  auto stage = [&](int buf, int k0) {
    unsigned char* dst = lds + (long)buf * 2 * TILE;
    // anti-bank-conflict chunk rotation: LDS chunk position p of row r
    // holds global chunk (p - (r>>3)) & 3 (true-lane-group enumeration
    // in tests/test_gemm_skew_logic.py); glds keeps LDS lane-linear so
    // the rotation is applied to the SOURCE chunk address
    for (int issue = 0; issue < 2; ++issue) {
      long o_base = (long)issue * 8192 + (long)wid * 1024;
      long o = o_base + (long)lane * 16;
      int row = (int)(o >> 6);
      int p = (int)((o & 63) >> 4);
      int kk = ((p - (row >> 3)) & 3) * 16;
      const unsigned char* ga = A + (brow + row) * Kb + k0 / 2 + kk;
      const unsigned char* gb = B + (bcol + row) * Kb + k0 / 2 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(dst + o_base), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)(dst + TILE + o_base), 16,
          0, 0);
    }
    // scales: one dword per row (K-128 tile = 4 e8m0 bytes); waves 0-3
    // gather As rows wid*64+lane, waves 4-7 Bs
    const unsigned char* S = (wid < 4) ? As : Bs;
    long rbase = (wid < 4) ? brow : bcol;
    int srow = (wid & 3) * 64 + lane;
    const unsigned char* gs = S + (rbase + srow) * (long)ks + k0 / 32;
    unsigned char* sdst =
        sbase + (long)buf * 2048 + (wid >= 4 ? 1024 : 0) + (wid & 3) * 256;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)gs,
        (__attribute__((address_space(3))) void*)sdst, 4, 0, 0);
  };

  // K-invariant per-lane offsets
  const int g = lane >> 5;
  const int r31 = lane & 31;
  int a_off[2], asc_off[2], b_off[4], bsc_off[4];
  for (int mf = 0; mf < 2; ++mf) {
    int row = wr * 64 + mf * 32 + r31;
    a_off[mf] = row * 64;
    asc_off[mf] = row * 4 + g;
  }
  for (int nf = 0; nf < 4; ++nf) {
    int col = wc * 128 + nf * 32 + r31;
    b_off[nf] = col * 64;
    bsc_off[nf] = col * 4 + g;
  }
  i32x8 afrag[2] = {};
  i32x8 bfrag[4] = {};

  stage(0, 0);
  for (int k0 = 0; k0 < K; k0 += 128) {
    const int cur = (k0 >> 7) & 1;
    const bool more = (k0 + 128) < K;
    if (more) stage(cur ^ 1, k0 + 128);
    if (more)
      asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    const unsigned char* la = lds + (long)cur * 2 * TILE;
    const unsigned char* lb = la + TILE;
    const unsigned char* sA = sbase + (long)cur * 2048;
    const unsigned char* sB = sA + 1024;
    for (int kk = 0; kk < 2; ++kk) {
      int asc[2];
      for (int mf = 0; mf < 2; ++mf) {
        int row = wr * 64 + mf * 32 + r31;
        int ch = (g + 2 * kk + (row >> 3)) & 3;
        *(i32x4*)&afrag[mf] = *(const i32x4*)__builtin_assume_aligned(
            la + a_off[mf] + 16 * ch, 16);
        asc[mf] = sA[asc_off[mf] + 2 * kk];
      }
      for (int nf = 0; nf < 4; ++nf) {
        int col = wc * 128 + nf * 32 + r31;
        int ch = (g + 2 * kk + (col >> 3)) & 3;
        *(i32x4*)&bfrag[nf] = *(const i32x4*)__builtin_assume_aligned(
            lb + b_off[nf] + 16 * ch, 16);
        int bsc = sB[bsc_off[nf] + 2 * kk];
        for (int mf = 0; mf < 2; ++mf)
          acc[mf][nf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              afrag[mf], bfrag[nf], acc[mf][nf], 4, 4, 0, asc[mf], 0, bsc);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  for (int mf = 0; mf < 2; ++mf)
    for (int nf = 0; nf < 4; ++nf) {
      long col = bcol + wc * 128 + nf * 32 + r31;
      for (int r = 0; r < 16; ++r) {
        long row = brow + wr * 64 + mf * 32 + (r & 3) + 8 * (r >> 2) + 4 * g;
        C[row * (long)N + col] = acc[mf][nf][r];
      }
    }
}


// 256x128-tile variant of the 32x32x64 mx4 kernel: the occupancy
// experiment the final PMC motivates (181 VGPR x 2 waves/SIMD = one
// resident block; nothing covers the barrier windows, MfmaUtil 35%).
// Halving the N-tile drops the accumulator floor to 64 VGPRs/lane so
// TWO 8-wave blocks co-reside (4 waves/SIMD) and cover each other's
// waits — at the price of arithmetic intensity (341 vs 508
// FLOP/staged-byte). 8 waves as 4x2 of 64x64; per-wave glds counts
// differ (A 2 issues + B 1 for all waves, + 1 scale issue for waves
// 0-5), so the counted vmcnt branches per wave.
__global__ __launch_bounds__(512) void k_gemm_mxfp4_nt_32h(
    float* __restrict__ C, const unsigned char* __restrict__ A,
    const unsigned char* __restrict__ B, const unsigned char* __restrict__ As,
    const unsigned char* __restrict__ Bs, int M, int N, int K, int tiles_n,
    int nwg, int xcd_swizzle, int group) {
  constexpr int ATILE = 256 * 64; // 16 KiB
  constexpr int BTILE = 128 * 64; // 8 KiB
  __shared__ unsigned char lds[2 * (ATILE + BTILE) + 2 * 1536];
  unsigned char* const sbase = lds + 2 * (ATILE + BTILE);

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  wg = hpk_group_remap(wg, tiles_n, nwg, group);
  const long brow = (long)(wg / tiles_n) * 256;
  const long bcol = (long)(wg % tiles_n) * 128;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1; // 4 wave-rows of 64
  const int wc = wid & 1;  // 2 wave-cols of 64
  const int ks = K / 32;
  const long Kb = (long)K / 2;

  typedef __attribute__((ext_vector_type(16))) float f32x16;
  typedef __attribute__((ext_vector_type(4))) int i32x4;
  f32x16 acc[2][2] = {};

  auto stage = [&](int buf, int k0) {
    unsigned char* dst = lds + (long)buf * (ATILE + BTILE);
    for (int issue = 0; issue < 2; ++issue) { // A: 256 rows
      long o_base = (long)issue * 8192 + (long)wid * 1024;
      long o = o_base + (long)lane * 16;
      int row = (int)(o >> 6);
      int p = (int)((o & 63) >> 4);
      int kk = ((p - (row >> 3)) & 3) * 16;
      const unsigned char* ga = A + (brow + row) * Kb + k0 / 2 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)ga,
          (__attribute__((address_space(3))) void*)(dst + o_base), 16, 0, 0);
    }
    { // B: 128 rows, one issue
      long o_base = (long)wid * 1024;
      long o = o_base + (long)lane * 16;
      int row = (int)(o >> 6);
      int p = (int)((o & 63) >> 4);
      int kk = ((p - (row >> 3)) & 3) * 16;
      const unsigned char* gb = B + (bcol + row) * Kb + k0 / 2 + kk;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gb,
          (__attribute__((address_space(3))) void*)(dst + ATILE + o_base), 16,
          0, 0);
    }
    if (wid < 6) { // scales: A rows 0-255 (waves 0-3), B rows 0-127 (4-5)
      const unsigned char* S = (wid < 4) ? As : Bs;
      long rbase = (wid < 4) ? brow : bcol;
      int srow = (wid < 4 ? (wid & 3) : (wid & 1)) * 64 + lane;
      const unsigned char* gs = S + (rbase + srow) * (long)ks + k0 / 32;
      unsigned char* sdst = sbase + (long)buf * 1536 +
                            (wid < 4 ? (wid & 3) * 256 : 1024 + (wid & 1) * 256);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gs,
          (__attribute__((address_space(3))) void*)sdst, 4, 0, 0);
    }
  };

  const int g = lane >> 5;
  const int r31 = lane & 31;
  int a_off[2], asc_off[2], b_off[2], bsc_off[2];
  for (int mf = 0; mf < 2; ++mf) {
    int row = wr * 64 + mf * 32 + r31;
    a_off[mf] = row * 64;
    asc_off[mf] = row * 4 + g;
  }
  for (int nf = 0; nf < 2; ++nf) {
    int col = wc * 64 + nf * 32 + r31;
    b_off[nf] = col * 64;
    bsc_off[nf] = col * 4 + g;
  }
  i32x8 afrag[2] = {};
  i32x8 bfrag[2] = {};

  stage(0, 0);
  for (int k0 = 0; k0 < K; k0 += 128) {
    const int cur = (k0 >> 7) & 1;
    const bool more = (k0 + 128) < K;
    if (more) stage(cur ^ 1, k0 + 128);
    if (more) {
      if (wid < 6)
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    const unsigned char* la = lds + (long)cur * (ATILE + BTILE);
    const unsigned char* lb = la + ATILE;
    const unsigned char* sA = sbase + (long)cur * 1536;
    const unsigned char* sB = sA + 1024;
    for (int kk = 0; kk < 2; ++kk) {
      int asc[2];
      for (int mf = 0; mf < 2; ++mf) {
        int row = wr * 64 + mf * 32 + r31;
        int ch = (g + 2 * kk + (row >> 3)) & 3;
        *(i32x4*)&afrag[mf] = *(const i32x4*)__builtin_assume_aligned(
            la + a_off[mf] + 16 * ch, 16);
        asc[mf] = sA[asc_off[mf] + 2 * kk];
      }
      for (int nf = 0; nf < 2; ++nf) {
        int col = wc * 64 + nf * 32 + r31;
        int ch = (g + 2 * kk + (col >> 3)) & 3;
        *(i32x4*)&bfrag[nf] = *(const i32x4*)__builtin_assume_aligned(
            lb + b_off[nf] + 16 * ch, 16);
        int bsc = sB[bsc_off[nf] + 2 * kk];
        for (int mf = 0; mf < 2; ++mf)
          acc[mf][nf] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              afrag[mf], bfrag[nf], acc[mf][nf], 4, 4, 0, asc[mf], 0, bsc);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  for (int mf = 0; mf < 2; ++mf)
    for (int nf = 0; nf < 2; ++nf) {
      long col = bcol + wc * 64 + nf * 32 + r31;
      for (int r = 0; r < 16; ++r) {
        long row = brow + wr * 64 + mf * 32 + (r & 3) + 8 * (r >> 2) + 4 * g;
        C[row * (long)N + col] = acc[mf][nf][r];
      }
    }
}

void launch_gemm_mxfp4_nt(float* C, const void* A, const void* B,
                          const void* As, const void* Bs, long M, long N,
                          long K, hipStream_t stream, int xcd_swizzle) {
  if (M % 128 != 0 || N % 128 != 0 || K % 128 != 0)
    throw std::runtime_error("gemm_mxfp4_nt requires M,N,K % 128 == 0");
  const int grp = gemm_group((int)(N / 128));
  int tiles_n = (int)(N / 128);
  int nwg = (int)(M / 128) * tiles_n;
  // variants: 32 (the 256^2-tile 32x32x64 kernel — DEFAULT for
  // 256-divisible shapes: 2903/3044 TF at 8192^3/16384^3 vs db 2003),
  // db (double-buffered 16x16x128 at 128^2, the fallback), 8 (plain
  // 8-wave), 4 (plain 4-wave — measured negative)
  const char* v = std::getenv("HPK_MX4_WAVES");
  std::string waves = v ? v : "32";
  if (waves == "32" && M % 256 == 0 && N % 256 == 0) {
    int tn32 = (int)(N / 256);
    int n32 = (int)(M / 256) * tn32;
    // measured (16384^3 interleaved sweep): row-major beats 16/32-wide
    // bands here — one resident block/CU makes the concurrent footprint
    // 4 full tile rows, which the MALL already covers
    const char* genv = std::getenv("HPK_GEMM_GROUP");
    const int grp32 = genv ? std::atoi(genv) : 1;
    hipLaunchKernelGGL(k_gemm_mxfp4_nt_32, dim3(n32), dim3(512), 0, stream,
                       C, (const unsigned char*)A, (const unsigned char*)B,
                       (const unsigned char*)As, (const unsigned char*)Bs,
                       (int)M, (int)N, (int)K, tn32, n32, xcd_swizzle,
                       grp32);
  } else if (waves == "32h" && M % 256 == 0) {
    int tnh = (int)(N / 128);
    int nh = (int)(M / 256) * tnh;
    const char* genv = std::getenv("HPK_GEMM_GROUP");
    const int grph = genv ? std::atoi(genv) : 1;
    hipLaunchKernelGGL(k_gemm_mxfp4_nt_32h, dim3(nh), dim3(512), 0, stream,
                       C, (const unsigned char*)A, (const unsigned char*)B,
                       (const unsigned char*)As, (const unsigned char*)Bs,
                       (int)M, (int)N, (int)K, tnh, nh, xcd_swizzle, grph);
  } else if (waves == "db" || waves == "32" || waves == "32h") {
    hipLaunchKernelGGL(k_gemm_mxfp4_nt_db, dim3(nwg), dim3(512), 0, stream,
                       C, (const unsigned char*)A, (const unsigned char*)B,
                       (const unsigned char*)As, (const unsigned char*)Bs,
                       (int)M, (int)N, (int)K, tiles_n, nwg, xcd_swizzle,
                       grp);
  } else if (waves == "4") {
    hipLaunchKernelGGL(k_gemm_mxfp4_nt_w4, dim3(nwg), dim3(256), 0, stream,
                       C, (const unsigned char*)A, (const unsigned char*)B,
                       (const unsigned char*)As, (const unsigned char*)Bs,
                       (int)M, (int)N, (int)K, tiles_n, nwg, xcd_swizzle,
                       grp);
  } else {
    hipLaunchKernelGGL(k_gemm_mxfp4_nt, dim3(nwg), dim3(512), 0, stream, C,
                       (const unsigned char*)A, (const unsigned char*)B,
                       (const unsigned char*)As, (const unsigned char*)Bs,
                       (int)M, (int)N, (int)K, tiles_n, nwg, xcd_swizzle,
                       grp);
  }
  check_hip(hipGetLastError(), "launch_gemm_mxfp4_nt");
}

void launch_gemm_fp8_nt(float* C, const void* A, const void* B, long M,
                        long N, long K, hipStream_t stream,
                        int xcd_swizzle) {
  if (M % BM != 0 || N % BN != 0 || K % BK != 0)
    throw std::runtime_error(
        "gemm_fp8_nt requires M,N % 128 == 0 and K % 64 == 0");
  const int grp = gemm_group((int)(N / 256));
  const char* var = std::getenv("HPK_GEMM_VARIANT");
  const bool ph8 = !var || std::string(var) == "8ph";
  // default: the 256^2 32x32x64 scaled-MFMA kernel with hardcoded x1.0
  // scales (no non-scaled 32x32x64 fp8 MFMA exists) — measured above the
  // 16x16x128 8-phase pipeline; HPK_GEMM_VARIANT=8ph|plain|db forces it off
  if (!var && M % 256 == 0 && N % 256 == 0 && K % 128 == 0) {
    int tn32 = (int)(N / 256);
    int n32 = (int)(M / 256) * tn32;
    const char* genv = std::getenv("HPK_GEMM_GROUP");
    const int grp32 = genv ? std::atoi(genv) : 1;
    hipLaunchKernelGGL((k_gemm_mxfp8_nt_32<false>), dim3(n32), dim3(512), 0,
                       stream, C, (const unsigned char*)A,
                       (const unsigned char*)B, nullptr, nullptr, (int)M,
                       (int)N, (int)K, tn32, n32, xcd_swizzle, grp32);
    check_hip(hipGetLastError(), "launch_gemm_fp8_nt(32)");
    return;
  }
  if (ph8 && M % 256 == 0 && N % 256 == 0 && K % 128 == 0) {
    int tn = (int)(N / 256);
    int n8 = (int)(M / 256) * tn;
    hipLaunchKernelGGL(k_gemm_fp8_8ph, dim3(n8), dim3(512), 0, stream, C,
                       (const unsigned char*)A, (const unsigned char*)B,
                       (int)M, (int)N, (int)K, tn, n8, xcd_swizzle, grp);
    check_hip(hipGetLastError(), "launch_gemm_fp8_nt(8ph)");
    return;
  }
  int tiles_m = (int)(M / BM), tiles_n = (int)(N / BN);
  int nwg = tiles_m * tiles_n;
  hipLaunchKernelGGL((k_gemm_fp8_nt<2, 4>), dim3(nwg), dim3(512), 0, stream,
                     C, (const unsigned char*)A, (const unsigned char*)B,
                     (int)M, (int)N, (int)K, tiles_n, nwg, xcd_swizzle, grp);
  check_hip(hipGetLastError(), "launch_gemm_fp8_nt");
}

void launch_gemm_bf16_nt(float* C, const void* A, const void* B, long M,
                         long N, long K, hipStream_t stream,
                         int xcd_swizzle) {
  if (M % BM != 0 || N % BN != 0 || K % BK != 0)
    throw std::runtime_error(
        "gemm_bf16_nt requires M,N % 128 == 0 and K % 64 == 0");
  const int grp = gemm_group((int)(N / 256));
  int tiles_m = (int)(M / BM), tiles_n = (int)(N / BN);
  int nwg = tiles_m * tiles_n;
  // measured (profiles/gemm_r2 logs, random [-1,1) operands): 8 waves
  // 777/924 TF at 4096^3/8192^3 vs 4 waves 686/765 — the 64x32 sub-tile
  // variant's 76-VGPR budget lifts occupancy 2 -> 6 waves/SIMD and wins
  // everywhere; it is the default.
  int waves = 8;
  if (const char* env = std::getenv("HPK_GEMM_WAVES")) waves = std::atoi(env);
  const char* var = std::getenv("HPK_GEMM_VARIANT");
  const bool dbuf = var && std::string(var) == "db";
  // default: the deep-pipelined 256^2 8-phase kernel wherever its shape
  // constraints hold (measured 887/1026 TF at 4096^3/8192^3 vs the plain
  // 8-wave kernel's 777/924; 50 exact-equality race-screen runs clean).
  // HPK_GEMM_VARIANT=plain|db forces the simpler kernels.
  const bool ph8 = !var || std::string(var) == "8ph";
  if (ph8 && M % 256 == 0 && N % 256 == 0 && K % 128 == 0) {
    int tn = (int)(N / 256);
    int n8 = (int)(M / 256) * tn;
    hipLaunchKernelGGL(k_gemm_bf16_8ph, dim3(n8), dim3(512), 0, stream, C,
                       (const __hip_bfloat16*)A, (const __hip_bfloat16*)B,
                       (int)M, (int)N, (int)K, tn, n8, xcd_swizzle, grp);
    check_hip(hipGetLastError(), "launch_gemm_bf16_nt(8ph)");
    return;
  }
  if (dbuf && waves == 8) {
    hipLaunchKernelGGL((k_gemm_bf16_nt_db<2, 4>), dim3(nwg), dim3(512), 0,
                       stream, C, (const __hip_bfloat16*)A,
                       (const __hip_bfloat16*)B, (int)M, (int)N, (int)K,
                       tiles_n, nwg, xcd_swizzle, grp);
  } else if (dbuf) {
    hipLaunchKernelGGL((k_gemm_bf16_nt_db<2, 2>), dim3(nwg), dim3(256), 0,
                       stream, C, (const __hip_bfloat16*)A,
                       (const __hip_bfloat16*)B, (int)M, (int)N, (int)K,
                       tiles_n, nwg, xcd_swizzle, grp);
  } else if (waves == 8) {
    hipLaunchKernelGGL((k_gemm_bf16_nt<2, 4>), dim3(nwg), dim3(512), 0,
                       stream, C, (const __hip_bfloat16*)A,
                       (const __hip_bfloat16*)B, (int)M, (int)N, (int)K,
                       tiles_n, nwg, xcd_swizzle, grp);
  } else {
    hipLaunchKernelGGL((k_gemm_bf16_nt<2, 2>), dim3(nwg), dim3(256), 0,
                       stream, C, (const __hip_bfloat16*)A,
                       (const __hip_bfloat16*)B, (int)M, (int)N, (int)K,
                       tiles_n, nwg, xcd_swizzle, grp);
  }
  check_hip(hipGetLastError(), "launch_gemm_bf16_nt");
}

} // namespace hpk
