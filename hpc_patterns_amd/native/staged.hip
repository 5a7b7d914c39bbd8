// staged.hip — pipelined pageable<->device copies through pinned staging.
//
// The reference's OMP-map variant hands pageable host buffers to the
// runtime's mapping tables (reference allreduce-map-mpi-omp-offload.cpp);
// HIP's hipMemcpyAsync stages pageable copies internally through ONE shared
// staging path, which is why M2D || D2M never overlaps in the sweep matrix
// (profiles/hip_sweep_r15_tables.log). This module is the explicit version:
// each copy gets its own pinned double-buffer and its own SDMA engine, with
// the CPU memcpy of chunk i overlapping the DMA of chunk i-1 — so two
// opposite-direction pageable copies can genuinely run concurrently.

#include "include/hpk.h"

#include <cstring>
#include <mutex>
#include <thread>
#include <vector>

namespace hpk {

namespace {

constexpr size_t kChunk = 8 << 20; // 8 MiB staging chunks

// Process-lifetime pool of pinned staging chunks (hipHostMalloc costs ms;
// a staged copy borrows two and returns them).
class StagingPool {
 public:
  void* take() {
    std::lock_guard<std::mutex> lock(mu_);
    if (!free_.empty()) {
      void* p = free_.back();
      free_.pop_back();
      return p;
    }
    void* p = nullptr;
    check_hip(hipHostMalloc(&p, kChunk, hipHostMallocDefault),
              "staging hipHostMalloc");
    return p;
  }
  void give(void* p) {
    std::lock_guard<std::mutex> lock(mu_);
    free_.push_back(p);
  }

 private:
  std::mutex mu_;
  std::vector<void*> free_;
};

StagingPool& pool() {
  static StagingPool p;
  return p;
}

} // namespace

namespace {
void staged_copy_range(void* dst, const void* src, size_t nbytes, int device,
                       int engine_index, bool h2d) {
  void* stage[2] = {pool().take(), pool().take()};
  void* pending = nullptr; // in-flight DMA handle
  size_t off = 0;
  int buf = 0;

  if (h2d) {
    // pageable host -> device: CPU memcpy chunk i || DMA chunk i-1
    while (off < nbytes) {
      size_t len = nbytes - off < kChunk ? nbytes - off : kChunk;
      std::memcpy(stage[buf], (const char*)src + off, len);
      if (pending) sdma_wait(pending);
      pending = sdma_copy_begin((char*)dst + off, stage[buf], len, device,
                                engine_index);
      off += len;
      buf ^= 1;
      if (off < nbytes && pending) {
        // before reusing the other buffer two chunks later, its DMA must be
        // done — with depth 2 that is exactly the wait above next iteration
      }
    }
    if (pending) sdma_wait(pending);
  } else {
    // device -> pageable host: DMA chunk i || CPU memcpy chunk i-1
    size_t prev_off = 0, prev_len = 0;
    int prev_buf = 0;
    while (off < nbytes) {
      size_t len = nbytes - off < kChunk ? nbytes - off : kChunk;
      void* h = sdma_copy_begin(stage[buf], (const char*)src + off, len,
                                device, engine_index);
      if (pending) {
        sdma_wait(pending);
        std::memcpy((char*)dst + prev_off, stage[prev_buf], prev_len);
      }
      pending = h;
      prev_off = off;
      prev_len = len;
      prev_buf = buf;
      off += len;
      buf ^= 1;
    }
    if (pending) {
      sdma_wait(pending);
      std::memcpy((char*)dst + prev_off, stage[prev_buf], prev_len);
    }
  }

  pool().give(stage[0]);
  pool().give(stage[1]);
}
} // namespace

void staged_copy(void* dst, const void* src, size_t nbytes, int device,
                 int engine_index, bool h2d) {
  // The pipeline is CPU-memcpy-bound (~23-33 GB/s single thread for the
  // D2H drain, measured r27); split large copies across worker threads,
  // each with its own staging pair on the same engine.
  constexpr size_t kParallelCut = 64 << 20;
  const int nthreads = nbytes >= kParallelCut ? 4 : 1;
  if (nthreads == 1) {
    staged_copy_range(dst, src, nbytes, device, engine_index, h2d);
    return;
  }
  size_t part = ((nbytes / nthreads) + 15) & ~size_t(15); // 16B-aligned parts
  std::vector<std::thread> ts;
  for (int t = 0; t < nthreads; ++t) {
    size_t off = (size_t)t * part;
    if (off >= nbytes) break;
    size_t len = nbytes - off < part ? nbytes - off : part;
    ts.emplace_back([=] {
      staged_copy_range((char*)dst + off, (const char*)src + off, len, device,
                        engine_index, h2d);
    });
  }
  for (auto& th : ts) th.join();
}

} // namespace hpk
