// membench_main.cpp — hpk_membench: native kernel micro-benchmarks.
//
// C++ twin of scripts/membench.py: rates for the hand-written gfx950
// kernels (copy variants vs hipMemcpyAsync, fills, accumulate, exact
// reductions, FMA and MFMA busy loops) straight from the native library —
// the numbers that size every launch shape in kernels.hip.
//
//   hpk_membench [--quick] [--floats N]

#include "../hpc_patterns_amd/native/include/hpk.h"

#include <chrono>
#include <cstdio>
#include <cstring>
#include <functional>
#include <string>

namespace {

double now_s() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

double time_best(const std::function<void()>& fn, int reps = 5, int warm = 2) {
  for (int i = 0; i < warm; ++i) fn();
  hpk::check_hip(hipDeviceSynchronize(), "warm sync");
  double best = 1e30;
  for (int i = 0; i < reps; ++i) {
    double t0 = now_s();
    fn();
    hpk::check_hip(hipDeviceSynchronize(), "sync");
    double dt = now_s() - t0;
    if (dt < best) best = dt;
  }
  return best;
}

} // namespace

int main(int argc, char* argv[]) {
  bool quick = false;
  size_t n = (1ull << 30) / 4; // 1 GiB of floats
  for (int i = 1; i < argc; ++i) {
    std::string s = argv[i];
    if (s == "--quick") quick = true;
    else if (s == "--floats" && i + 1 < argc) n = std::strtoull(argv[++i], nullptr, 10);
    else {
      std::printf("Usage: %s [--quick] [--floats N]\n", argv[0]);
      return s == "-h" || s == "--help" ? 0 : 1;
    }
  }
  if (quick) n = (64ull << 20) / 4;
  size_t bytes = n * sizeof(float);

  hpk::check_hip(hipSetDevice(0), "set device");
  float *a = nullptr, *b = nullptr;
  hpk::check_hip(hipMalloc(&a, bytes), "a");
  hpk::check_hip(hipMalloc(&b, bytes), "b");
  hpk::launch_fill_f32(a, 1.f, n, nullptr);
  hpk::launch_fill_f32(b, 2.f, n, nullptr);
  hpk::check_hip(hipDeviceSynchronize(), "init");

  std::printf("# hpk_membench: %zu floats (%.1f MB)\n", n, bytes / 1e6);
  double t;

  t = time_best([&] {
    hpk::check_hip(hipMemcpyAsync(b, a, bytes, hipMemcpyDeviceToDevice, 0),
                   "memcpy");
  });
  std::printf("%-26s %8.3f ms %9.1f GB/s payload\n", "hipMemcpyAsync D2D",
              t * 1e3, bytes / t / 1e9);
  t = time_best([&] { hpk::launch_copy_kernel(b, a, bytes, nullptr); });
  std::printf("%-26s %8.3f ms %9.1f GB/s payload\n", "copy kernel (auto NT)",
              t * 1e3, bytes / t / 1e9);
  t = time_best([&] { hpk::launch_fill_f32(a, 3.f, n, nullptr); });
  std::printf("%-26s %8.3f ms %9.1f GB/s write\n", "fill", t * 1e3,
              bytes / t / 1e9);
  t = time_best([&] { hpk::launch_acc_f32(b, a, n, nullptr); });
  std::printf("%-26s %8.3f ms %9.1f GB/s (2r+1w)\n", "accumulate", t * 1e3,
              3.0 * bytes / t / 1e9);
  t = time_best([&] { (void)hpk::reduce_sum_f32(a, n, nullptr); });
  std::printf("%-26s %8.3f ms %9.1f GB/s read\n", "reduce_sum (exact f64)",
              t * 1e3, bytes / t / 1e9);

  // busy loops
  long gs = 1 << 20, trip = quick ? 500 : 2000;
  float* out = nullptr;
  hpk::check_hip(hipMalloc(&out, gs * sizeof(float)), "out");
  t = time_best([&] { hpk::launch_busy_wait(out, trip, gs, nullptr); });
  std::printf("%-26s %8.3f ms %9.2f TFLOP/s fp32\n", "busy_wait FMA", t * 1e3,
              (double)gs * 64 * trip * 2 / t / 1e12);
  long waves = 2048, mtrip = quick ? 5000 : 20000;
  t = time_best([&] { hpk::launch_busy_wait_mfma(out, mtrip, waves, nullptr); });
  std::printf("%-26s %8.3f ms %9.2f TFLOP/s bf16\n", "busy_wait MFMA", t * 1e3,
              (double)waves * mtrip * 16384 / t / 1e12);

  (void)hipFree(a);
  (void)hipFree(b);
  (void)hipFree(out);
  return 0;
}
