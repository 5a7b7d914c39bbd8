// hpk.h — C++ API of the MI355X-native HPC-patterns core library.
//
// Brand-new CDNA4/HIP design with the capabilities of the reference suite
// (argonne-lcf/HPC-Patterns): hand-written gfx950 kernels (the reference's
// SYCL parallel_for / omp-target loops, see reference concurency/bench.hpp:7-31,
// allreduce-mpi-sycl.cpp:27-41), a multi-hipStream/hipGraph concurrency engine
// (reference bench_sycl.cpp:19-144 / bench_omp.cpp), xGMI topology discovery
// (reference p2p/topology.cpp, Level-Zero Sysman -> rocm_smi/HIP here), and
// HIP-IPC one-sided transport (reference MPI_Win/MPI_Put path,
// p2p/peer2pear.cpp:68-102).
#pragma once

#include <hip/hip_runtime.h>

#include <cstddef>
#include <cstdint>
#include <map>
#include <string>
#include <vector>

namespace hpk {

// ---------------------------------------------------------------------------
// Error handling
// ---------------------------------------------------------------------------
void check_hip(hipError_t e, const char* what);

// ---------------------------------------------------------------------------
// Kernels (kernels.hip) — all launch asynchronously on `stream` unless noted.
// ---------------------------------------------------------------------------

// K1 "C" compute command: every work-item runs 64*tripcount dependent FMAs on
// registers and stores one float (reference bench.hpp:23-31 MAD_64 payload).
void launch_busy_wait(float* out, long tripcount, long globalsize,
                      hipStream_t stream);

// K1-MFMA variant: every wave runs `tripcount` chained
// v_mfma_f32_16x16x32_bf16 ops — lights up the matrix cores so concurrency
// profiles show MFMA utilisation. n_waves waves of 64 lanes.
void launch_busy_wait_mfma(float* out, long tripcount, long n_waves,
                           hipStream_t stream);

// K2 shader-copy: vectorized 16B grid-stride copy (the "shader blit" sibling
// of hipMemcpyAsync's SDMA path). Pointers may be in any HIP-visible space.
void launch_copy_kernel(void* dst, const void* src, size_t nbytes,
                        hipStream_t stream);
// Tunable variant for micro-benchmarking: unroll in {1,4} (16 B vs 64 B per
// thread per iteration), grid_cap = max workgroups.
void launch_copy_kernel_tuned(void* dst, const void* src, size_t nbytes,
                              hipStream_t stream, int unroll,
                              size_t grid_cap);

// K4 fills (reference Initialize kernel, allreduce-mpi-sycl.cpp:34-41).
void launch_fill_f32(float* dst, float value, size_t n, hipStream_t stream);
// dst[i] = (float)i — device-side iota payload generator (reference
// fill_randomly host shuffle, peer2pear.cpp:8-17; shuffling is done on host,
// the plain iota fill runs on device).
void launch_iota_f32(float* dst, size_t n, hipStream_t stream);

// K3 accumulate: dst[i] += src[i] (reference Accumulate,
// allreduce-mpi-sycl.cpp:27-31), vectorized float4 grid-stride.
void launch_acc_f32(float* dst, const float* src, size_t n, hipStream_t stream);
// nontemporal variant (read-once src / rmw-once dst, beyond-L3 sizes)
void launch_acc_f32_nt(float* dst, const float* src, size_t n,
                       hipStream_t stream);

// K7 (r2, beyond-parity showcase): LDS-tiled bf16 MFMA GEMM —
// C[M,N] (fp32) = A[M,K] x B[N,K]^T, both operands bf16 K-contiguous.
// Default for M,N % 256 / K % 128: the 256^2-tile 8-phase deep pipeline
// (v_mfma_f32_16x16x32_bf16, counted vmcnt, zero bank conflicts);
// otherwise the plain/db 128^2 kernels (HPK_GEMM_VARIANT /
// HPK_GEMM_WAVES select). Optional bijective XCD workgroup swizzle.
// Requires M,N % 128 == 0 and K % 64 == 0 (throws otherwise).
void launch_gemm_bf16_nt(float* C, const void* A, const void* B, long M,
                         long N, long K, hipStream_t stream,
                         int xcd_swizzle = 1);
// fp8 (OCP e4m3): 256-divisible shapes default to the 256^2 32x32x64
// scaled-MFMA kernel with hardcoded x1.0 scales (2.2 PF); otherwise the
// mfma_f32_16x16x32_fp8_fp8 family.
void launch_gemm_fp8_nt(float* C, const void* A, const void* B, long M,
                        long N, long K, hipStream_t stream,
                        int xcd_swizzle = 0);
// int8 GEMM (mfma_i32_16x16x64_i8, ~2x the bf16 rate, exact int32
// accumulation): C[M,N] int32 = A[M,K] x B[N,K]^T, int8 operands.
void launch_gemm_i8_nt(int* C, const void* A, const void* B, long M,
                       long N, long K, hipStream_t stream,
                       int xcd_swizzle = 0);
// Block-scaled MX-fp8 (mfma_scale_f32_16x16x128_f8f6f4, 2x the bf16
// rate): C = (A .* 2^(As-127)) x (B .* 2^(Bs-127))^T with one e8m0 scale
// byte per 32-element K-block (As: [M][K/32], Bs: [N][K/32], uint8).
// Requires M,N,K % 128 == 0.
void launch_gemm_mxfp8_nt(float* C, const void* A, const void* B,
                          const void* As, const void* Bs, long M, long N,
                          long K, hipStream_t stream, int xcd_swizzle = 0);
// Block-scaled MX-fp4 (e2m1, the 4x-bf16 rate class): A/B are
// nibble-PACKED uint8 [rows][K/2] (low nibble = even k), scales as for
// mxfp8. Operand/scale layout measured on hardware: diagonal (one OCP
// 32-block per lane, own-lane scale — scripts/probes/fp4_probe*).
void launch_gemm_mxfp4_nt(float* C, const void* A, const void* B,
                          const void* As, const void* Bs, long M, long N,
                          long K, hipStream_t stream, int xcd_swizzle = 0);

// Exact double-precision sum of n floats. Synchronizes `stream`.
// Replaces the reference's O(N log N) host sort+sum checksum
// (peer2pear.cpp:56-63) with an order-independent exact device reduction.
double reduce_sum_f32(const float* src, size_t n, hipStream_t stream);

// int32 twins (the reference instantiates miniapps for float AND int).
void launch_fill_i32(int* dst, int value, size_t n, hipStream_t stream);
void launch_acc_i32(int* dst, const int* src, size_t n, hipStream_t stream);
long long reduce_sum_i32(const int* src, size_t n, hipStream_t stream);

// ---------------------------------------------------------------------------
// Concurrency engine (conc.hip) — reference bench<T>() ABI, bench.hpp:37-40.
// ---------------------------------------------------------------------------

// Modes:
//   serial       — one stream, synchronize after every command (baseline)
//   in_order     — N hipStreams (default: one per command), round-robin
//   host_threads — one std::thread + stream per command (reference
//                  bench_omp.cpp host_threads mode)
//   graph        — all commands as independent nodes of one hipGraph; the
//                  HIP analog of a SYCL out-of-order queue
//   out_of_order — alias of graph (HIP streams are strictly in-order; the
//                  graph scheduler is the runtime-managed concurrency path)
//   graph_explicit — same graph semantics but built with explicit node-API
//                  calls (hipGraphAddMemcpyNode1D / child-graph kernel
//                  nodes, all independent roots) instead of stream capture
//   nowait       — alias of in_order (reference bench_omp.cpp nowait mode)
extern const std::string allowed_modes;
bool mode_is_allowed(const std::string& mode);

// copy-engine selection for A2B commands (the reference's copy-engine env
// knobs as a first-class CLI switch):
//   auto   -> hipMemcpyAsync (runtime picks SDMA or blit)
//   shader -> hand-written K2 copy kernel (CU path)
//   sdma   -> explicit hsa_amd_memory_async_copy_on_engine (DMA path)
enum CopyEngine { kCopyEngineAuto = 0, kCopyEngineShader = 1,
                  kCopyEngineSdma = 2 };

struct ConcResult {
  long total_us = 0;                  // min over repetitions
  std::vector<long> per_cmd_us;       // serial mode: min per-command wall time
  std::vector<double> per_cmd_dev_ms; // hipEvent device time (profiling only)
  // unmeasured per-command entries are -1 (e.g. per_cmd_us outside serial
  // mode, per_cmd_dev_ms without --enable_profiling)
};

// commands: "C" or "A2B"/"AB" with A,B in {M,D,H,S} =
// malloc/hipMalloc/hipHostMalloc/hipMallocManaged (reference
// bench_sycl.cpp:55-72 letter DSL).
// params keys: tripcount_C, globalsize_C, globalsize_<CMD> (floats).
ConcResult conc_bench(const std::string& mode,
                      const std::vector<std::string>& commands,
                      const std::map<std::string, size_t>& params,
                      bool enable_profiling, int n_queues, int n_repetitions,
                      bool verbose, int copy_engine);

// ---------------------------------------------------------------------------
// Topology (topo.hip) — xGMI link discovery (reference p2p/topology.cpp).
// ---------------------------------------------------------------------------

struct LinkInfo {
  int p2p_accessible = 0; // hipDeviceCanAccessPeer
  int link_type = -1;     // hipExtGetLinkTypeAndHopCount type (2 == xGMI)
  int hops = -1;
  long min_bw_mbps = -1;  // rocm_smi min/max link bandwidth (if available)
  long max_bw_mbps = -1;
  long weight = -1;       // rocm_smi link weight (if available)
};

int device_count();
// NxN matrix; [i][i] is zero-initialized LinkInfo.
std::vector<std::vector<LinkInfo>> link_matrix();

// XCD compute-partition (SPX/DPX/CPX...) and memory-partition (NPS*) mode
// per device — the MI355X analog of the reference's tile-fission awareness.
struct PartitionInfo {
  std::string compute; // e.g. "SPX", "CPX"; empty if unavailable
  std::string memory;  // e.g. "NPS1"; empty if unavailable
};
std::vector<PartitionInfo> partition_info();
// Connected components under direct-P2P reachability (the reference's
// "connectivity planes", topology.cpp:76-89). MI355X nodes are fully
// connected, so this is usually one plane — the interesting data is the
// per-pair link table above.
std::vector<std::vector<int>> p2p_planes();

// ---------------------------------------------------------------------------
// IPC one-sided transport (ipc.hip) — reference MPI_Win/MPI_Put analog.
// ---------------------------------------------------------------------------

// 64-byte opaque handle for a hipMalloc'd region, exchangeable between
// processes (dmabuf IPC; requires HSA_ENABLE_IPC_MODE_LEGACY=0).
std::vector<uint8_t> ipc_get_handle(void* dptr);
void* ipc_open_handle(const std::vector<uint8_t>& handle);
void ipc_close_handle(void* dptr);

// ---------------------------------------------------------------------------
// Explicit SDMA-engine copies (sdma.hip) — hsa_amd_memory_async_copy[_on_
// engine]: copies placed on a named DMA engine, bypassing rocclr's
// blit-fallback heuristics (the reference's copy-engine selection knobs,
// done natively).
// ---------------------------------------------------------------------------
int sdma_num_engines(int device);
// Pipelined pageable<->device copy through pinned double-buffer staging on
// a named SDMA engine (blocking call; run it on its own thread to overlap
// with other commands). h2d: src pageable -> dst device; else reverse.
void staged_copy(void* dst, const void* src, size_t nbytes, int device,
                 int engine_index, bool h2d);
// engines usable for src_device -> dst_device peer copies (xGMI SDMA).
int sdma_num_engines_pair(int dst_device, int src_device);
// engine_index < 0 lets ROCr pick. Returns a handle; copy completes when
// sdma_wait(handle) returns (handle is consumed).
void* sdma_copy_begin(void* dst, const void* src, size_t nbytes, int device,
                      int engine_index);
void sdma_wait(void* handle);

// ---------------------------------------------------------------------------
// Tracing (trace.hip) — roctx ranges for rocprofv3 --marker-trace.
// ---------------------------------------------------------------------------
void trace_push(const char* name);
void trace_pop();
void trace_mark(const char* name);

void enable_peer_access(int peer_device);
void memcpy_peer_async(void* dst, int dst_dev, const void* src, int src_dev,
                       size_t nbytes, hipStream_t stream);

} // namespace hpk
