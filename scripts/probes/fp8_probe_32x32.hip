// 32x32x64 FP8 layout validation. The diagonal hypothesis FAILED
// (866/1024); fp8_probe2_32x32's scale-bump map showed scale lane
// (row, gs) covers operand bytes [16gs, 16gs+16) of BOTH g-lanes of the
// row. Feed that makes each scale byte one OCP 32-block: half h of lane
// (row, g) <- k [32h + 16g, +16); the lane passes the scale byte for
// block g of its row (HW reads half h scales from lane h*32+row).
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstring>
#include <cmath>
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(8))) int i32x8;

__global__ void k_probe(float* C, const unsigned char* A,
                        const unsigned char* B, const unsigned char* As,
                        const unsigned char* Bs) {
  int lane = threadIdx.x & 63;
  int row = lane & 31, g = lane >> 5;
  unsigned char ab[32], bb[32];
  for (int h = 0; h < 2; ++h)
    for (int i = 0; i < 16; ++i) {
      int k = 32 * h + 16 * g + i;
      ab[16 * h + i] = A[row * 64 + k];
      bb[16 * h + i] = B[row * 64 + k];
    }
  i32x8 av, bv;
  __builtin_memcpy(&av, ab, 32);
  __builtin_memcpy(&bv, bb, 32);
  int sa = As[row * 2 + g];
  int sb = Bs[row * 2 + g];
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
      av, bv, acc, 0, 0, 0, sa, 0, sb);
  for (int r = 0; r < 16; ++r) {
    int orow = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    C[orow * 32 + (lane & 31)] = acc[r];
  }
}

unsigned char enc8(float v) { // e4m3
  if (v == 0) return 0;
  unsigned char s = v < 0 ? 0x80 : 0;
  float a = v < 0 ? -v : v; int e = 0;
  while (a >= 2.f) { a /= 2.f; ++e; }
  while (a < 1.f) { a *= 2.f; --e; }
  return s | ((e + 7) << 3) | (int)((a - 1.f) * 8.f + 0.5f);
}

int main() {
  unsigned char hA[2048], hB[2048], hAs[64], hBs[64];
  float fA[2048], fB[2048];
  srand(23);
  for (int i = 0; i < 2048; ++i) { // 32 rows x 64 k
    int v = (rand() % 5) - 2;
    int w = (rand() % 5) - 2;
    fA[i] = (float)v; hA[i] = enc8(fA[i]);
    fB[i] = (float)w; hB[i] = enc8(fB[i]);
  }
  for (int i = 0; i < 64; ++i) {
    hAs[i] = 125 + (rand() % 5);
    hBs[i] = 125 + (rand() % 5);
  }
  float ref[1024];
  for (int r = 0; r < 32; ++r)
    for (int c = 0; c < 32; ++c) {
      float s = 0;
      for (int k = 0; k < 64; ++k)
        s += fA[r * 64 + k] * exp2f((float)hAs[r * 2 + k / 32] - 127.f) *
             fB[c * 64 + k] * exp2f((float)hBs[c * 2 + k / 32] - 127.f);
      ref[r * 32 + c] = s;
    }
  unsigned char *dA, *dB, *dAs, *dBs; float* dC;
  (void)hipMalloc(&dA, 2048); (void)hipMalloc(&dB, 2048);
  (void)hipMalloc(&dAs, 64); (void)hipMalloc(&dBs, 64);
  (void)hipMalloc(&dC, 4096);
  (void)hipMemcpy(dA, hA, 2048, hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hB, 2048, hipMemcpyHostToDevice);
  (void)hipMemcpy(dAs, hAs, 64, hipMemcpyHostToDevice);
  (void)hipMemcpy(dBs, hBs, 64, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(k_probe, dim3(1), dim3(64), 0, 0, dC, dA, dB, dAs, dBs);
  hipError_t le = hipGetLastError();
  if (le != hipSuccess) { printf("launch err %d\n", (int)le); return 1; }
  float out[1024];
  (void)hipMemcpy(out, dC, 4096, hipMemcpyDeviceToHost);
  int bad = 0;
  for (int i = 0; i < 1024; ++i)
    if (out[i] != ref[i]) ++bad;
  printf("fp8 32x32x64 half-interleaved layout: %s (%d/1024)\n",
         bad ? "FAIL" : "PASS", bad);
  if (bad)
    for (int i = 0; i < 6; ++i)
      printf("  C[%d]=%g ref=%g\n", i, out[i], ref[i]);
  return 0;
}
