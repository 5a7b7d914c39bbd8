// Validated-layout test: feed data so each scale byte covers ONE
// contiguous OCP MX 32-block.
//   data-lane g (a=g>>1? NO: a=g>>1? define a = g/2, odd = g&1):
//     half h (16B) <- k [32*(2a+h) + 16*odd, +16)
//   scale-lane g provides MX block (2*(g&1) + (g>>1)) of its row.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstring>
#include <cmath>
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) int i32x8;

__global__ void k_probe(float* C, const unsigned char* A,
                        const unsigned char* B, const unsigned char* As,
                        const unsigned char* Bs) {
  int lane = threadIdx.x & 63;
  int row = lane & 15, g = lane >> 4;
  int a = g >> 1, odd = g & 1;
  unsigned char ab[32], bb[32];
  for (int h = 0; h < 2; ++h) {
    int kst = 32 * (2 * a + h) + 16 * odd;
    for (int i = 0; i < 16; ++i) {
      ab[16 * h + i] = A[row * 128 + kst + i];
      bb[16 * h + i] = B[row * 128 + kst + i];
    }
  }
  i32x8 av, bv;
  __builtin_memcpy(&av, ab, 32);
  __builtin_memcpy(&bv, bb, 32);
  int blk = 2 * (g & 1) + (g >> 1);
  int sa = As[row * 4 + blk];
  int sb = Bs[row * 4 + blk];
  f32x4 acc = {};
  acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      av, bv, acc, 0, 0, 0, sa, 0, sb);
  for (int r = 0; r < 4; ++r)
    C[(4 * g + r) * 16 + row] = acc[r];
}

unsigned char enc(float v) {
  if (v == 0) return 0;
  unsigned char s = v < 0 ? 0x80 : 0;
  float x = v < 0 ? -v : v; int e = 0;
  while (x >= 2.f) { x /= 2.f; ++e; }
  while (x < 1.f) { x *= 2.f; --e; }
  return s | ((e + 7) << 3) | (int)((x - 1.f) * 8.f + 0.5f);
}

int main() {
  unsigned char hA[2048], hB[2048], hAs[64], hBs[64];
  float fA[2048], fB[2048];
  srand(13);
  for (int i = 0; i < 2048; ++i) {
    int v = (rand() % 5) - 2; fA[i] = (float)v; hA[i] = enc((float)v);
    int w = (rand() % 5) - 2; fB[i] = (float)w; hB[i] = enc((float)w);
  }
  for (int i = 0; i < 64; ++i) {
    hAs[i] = 125 + (rand() % 5);
    hBs[i] = 125 + (rand() % 5);
  }
  float ref[256];
  for (int r = 0; r < 16; ++r)
    for (int c = 0; c < 16; ++c) {
      float s = 0;
      for (int k = 0; k < 128; ++k)
        s += fA[r * 128 + k] * exp2f((float)hAs[r * 4 + k / 32] - 127.f) *
             fB[c * 128 + k] * exp2f((float)hBs[c * 4 + k / 32] - 127.f);
      ref[r * 16 + c] = s;
    }
  unsigned char *dA, *dB, *dAs, *dBs; float* dC;
  (void)hipMalloc(&dA, 2048); (void)hipMalloc(&dB, 2048);
  (void)hipMalloc(&dAs, 64); (void)hipMalloc(&dBs, 64);
  (void)hipMalloc(&dC, 1024);
  (void)hipMemcpy(dA, hA, 2048, hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hB, 2048, hipMemcpyHostToDevice);
  (void)hipMemcpy(dAs, hAs, 64, hipMemcpyHostToDevice);
  (void)hipMemcpy(dBs, hBs, 64, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(k_probe, dim3(1), dim3(64), 0, 0, dC, dA, dB, dAs, dBs);
  float out[256];
  (void)hipMemcpy(out, dC, 1024, hipMemcpyDeviceToHost);
  int bad = 0;
  for (int i = 0; i < 256; ++i)
    if (out[i] != ref[i]) ++bad;
  printf("validated layout: %s (%d/256)\n", bad ? "FAIL" : "PASS", bad);
  if (bad)
    for (int i = 0; i < 6; ++i)
      printf("  C[%d]=%g ref=%g\n", i, out[i], ref[i]);
  return 0;
}
