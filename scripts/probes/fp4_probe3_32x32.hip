// 32x32x64 fp4 probe: does the diagonal layout hold for
// mfma_scale_f32_32x32x64_f8f6f4 (the bigger MFMA: 4x output per operand
// byte)? Hypothesis: lane (row=lane&31, g=lane>>5) supplies the one OCP
// 32-block k in [32g, 32g+32) nibble-packed, own-lane scale byte 0.
// C/D: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5).
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
  unsigned char ab[16] = {}, bb[16] = {};
  for (int i = 0; i < 32; ++i) {
    int k = 32 * g + i;
    unsigned an = (A[row * 32 + k / 2] >> (4 * (k & 1))) & 0xf;
    unsigned bn = (B[row * 32 + k / 2] >> (4 * (k & 1))) & 0xf;
    ab[i / 2] |= an << (4 * (i & 1));
    bb[i / 2] |= bn << (4 * (i & 1));
  }
  i32x8 av = {}, bv = {};
  __builtin_memcpy(&av, ab, 16);
  __builtin_memcpy(&bv, bb, 16);
  int sa = As[row * 2 + g];
  int sb = Bs[row * 2 + g];
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
      av, bv, acc, 4, 4, 0, sa, 0, sb);
  for (int r = 0; r < 16; ++r) {
    int orow = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    C[orow * 32 + (lane & 31)] = acc[r];
  }
}

unsigned char enc4(float v) {
  unsigned char s = v < 0 ? 0x8 : 0;
  float x = v < 0 ? -v : v;
  if (x == 0) return 0;
  if (x == 0.5f) return s | 1;
  if (x == 1.f) return s | 2;
  if (x == 1.5f) return s | 3;
  if (x == 2.f) return s | 4;
  if (x == 3.f) return s | 5;
  if (x == 4.f) return s | 6;
  return s | 7;
}

int main() {
  unsigned char hA[1024], hB[1024], hAs[64], hBs[64];
  float fA[2048], fB[2048];
  const float vals[9] = {0, 0.5f, -0.5f, 1, -1, 1.5f, -1.5f, 2, -2};
  srand(19);
  memset(hA, 0, 1024); memset(hB, 0, 1024);
  for (int i = 0; i < 2048; ++i) { // 32 rows x 64 k
    fA[i] = vals[rand() % 9];
    fB[i] = vals[rand() % 9];
    hA[i / 2] |= enc4(fA[i]) << (4 * (i & 1));
    hB[i / 2] |= enc4(fB[i]) << (4 * (i & 1));
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
  (void)hipMalloc(&dA, 1024); (void)hipMalloc(&dB, 1024);
  (void)hipMalloc(&dAs, 64); (void)hipMalloc(&dBs, 64);
  (void)hipMalloc(&dC, 4096);
  (void)hipMemcpy(dA, hA, 1024, hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hB, 1024, hipMemcpyHostToDevice);
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
  printf("fp4 32x32x64 diagonal layout: %s (%d/1024)\n",
         bad ? "FAIL" : "PASS", bad);
  if (bad)
    for (int i = 0; i < 6; ++i)
      printf("  C[%d]=%g ref=%g\n", i, out[i], ref[i]);
  return 0;
}
