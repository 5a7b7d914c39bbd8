// gemm.hip — K7: LDS-tiled bf16 MFMA GEMM (beyond-parity showcase).
//
// The reference suite's device kernels (K1-K6) are streaming/elementwise;
// this kernel demonstrates the OTHER half of the CDNA4 execution model the
// suite is designed around: matrix cores fed through LDS.
//   C[M,N] (fp32) = A[M,K] (bf16, row-major) x B[N,K]^T (bf16, row-major)
// i.e. an "NT" GEMM — both operands K-contiguous, the natural
// v_mfma_f32_16x16x32_bf16 feeding order.
//
// Structure (the plain-HIP two-barrier K-loop):
//   - 128x128 output tile per 256-thread workgroup (4 waves, 2x2 wave
//     grid, each wave owns a 64x64 sub-tile as 4x4 MFMA fragments).
//   - K-step 64: A-tile [128][64] and B-tile [128][64] staged into ONE
//     32 KiB LDS buffer per K-step via __builtin_amdgcn_global_load_lds
//     (16-byte direct-to-LDS DMA; the LDS image is lane-linear by
//     construction, which that instruction requires).
//   - two barriers per K-step: [sync] stage [sync] 16 ds_read_b128 + 32
//     chained MFMAs per wave.
//   - bijective XCD-aware workgroup swizzle so consecutive XCDs see
//     neighbouring C tiles (L2 locality when HBM-bound).
// Fragment mappings (16x16x32 bf16): A/B lane L holds 8 contiguous K
// elements at k = 8*(L>>4), row/col = L&15; C/D lane L reg r holds
// row = 4*(L>>4)+r, col = L&15.
//
// Numerics: fp32 accumulate; verified against torch fp32 matmul with
// exactly-representable integer payloads (tests/test_gpu_kernels.py).

#include "include/hpk.h"

#include <hip/hip_bf16.h>

#include <stdexcept>

namespace hpk {
namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int WAVES_M = 2, WAVES_N = 2;       // 2x2 waves of 64x64 each
constexpr int THREADS = WAVES_M * WAVES_N * 64;
constexpr int MREP = 4, NREP = 4;             // 16x16 fragments per wave

// one K-step's staging: A[128][64] + B[128][64] bf16 = 32 KiB
constexpr int TILE_HALF = BM * BK;            // elements per operand tile

__global__ __launch_bounds__(THREADS) void k_gemm_bf16_nt(
    float* __restrict__ C, const __hip_bfloat16* __restrict__ A,
    const __hip_bfloat16* __restrict__ B, int M, int N, int K,
    int tiles_n, int nwg, int xcd_swizzle) {
  __shared__ __hip_bfloat16 lds[2 * TILE_HALF]; // [A tile][B tile]

  int wg = (int)blockIdx.x;
  if (xcd_swizzle) {
    // bijective 8-XCD round-robin -> tile-linear remap
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, i = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
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
  const long elems_per_issue = (long)THREADS * 8;
  f32x4 acc[MREP][NREP] = {};

  for (int k0 = 0; k0 < K; k0 += BK) {
    __syncthreads(); // previous K-step's reads done before overwrite
    for (int issue = 0; issue < 4; ++issue) {
      long o_base = (long)issue * elems_per_issue + (long)wid * (64 * 8);
      long o = o_base + (long)lane * 8; // this lane's element offset
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
        int row = wr * 64 + m * 16 + (lane & 15);
        afrag[m] = *(const bf16x8*)(la + row * BK + kfrag);
      }
      for (int n = 0; n < NREP; ++n) {
        int col = wc * 64 + n * 16 + (lane & 15);
        bfrag[n] = *(const bf16x8*)(lb + col * BK + kfrag);
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
      long row0 = brow + wr * 64 + m * 16 + 4 * (lane >> 4);
      long col = bcol + wc * 64 + n * 16 + (lane & 15);
      for (int r = 0; r < 4; ++r)
        C[(row0 + r) * (long)N + col] = acc[m][n][r];
    }
}

} // namespace

void launch_gemm_bf16_nt(float* C, const void* A, const void* B, long M,
                         long N, long K, hipStream_t stream,
                         int xcd_swizzle) {
  if (M % BM != 0 || N % BN != 0 || K % BK != 0)
    throw std::runtime_error(
        "gemm_bf16_nt requires M,N % 128 == 0 and K % 64 == 0");
  int tiles_m = (int)(M / BM), tiles_n = (int)(N / BN);
  int nwg = tiles_m * tiles_n;
  hipLaunchKernelGGL(k_gemm_bf16_nt, dim3(nwg), dim3(THREADS), 0, stream,
                     C, (const __hip_bfloat16*)A, (const __hip_bfloat16*)B,
                     (int)M, (int)N, (int)K, tiles_n, nwg, xcd_swizzle);
  check_hip(hipGetLastError(), "launch_gemm_bf16_nt");
}

} // namespace hpk
