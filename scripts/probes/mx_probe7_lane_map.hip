// Decisive map: A row k-blocks of 16 elems weighted {1,2,4,...,128};
// singleton (lane, byte)=128 runs; print doubled 16-blocks per row.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstring>
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) int i32x8;

__global__ void k_probe(float* C, const unsigned char* A,
                        const unsigned char* B, int which, int byteidx) {
  int lane = threadIdx.x & 63;
  int row = lane & 15, g = lane >> 4;
  i32x8 av, bv;
  unsigned char ab[32], bb[32];
  for (int i = 0; i < 32; ++i) {
    ab[i] = A[row * 128 + 32 * g + i];
    bb[i] = B[row * 128 + 32 * g + i];
  }
  __builtin_memcpy(&av, ab, 32);
  __builtin_memcpy(&bv, bb, 32);
  unsigned sa = 0x7f7f7f7f;
  if (lane == which)
    sa = (sa & ~(0xffu << (8 * byteidx))) | (0x80u << (8 * byteidx));
  f32x4 acc = {};
  acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      av, bv, acc, 0, 0, 0, (int)sa, 0, 0x7f7f7f7f);
  for (int r = 0; r < 4; ++r)
    C[(4 * g + r) * 16 + row] = acc[r];
}

unsigned char enc(float v) {
  if (v == 0) return 0;
  unsigned char s = v < 0 ? 0x80 : 0;
  float a = v < 0 ? -v : v; int e = 0;
  while (a >= 2.f) { a /= 2.f; ++e; }
  while (a < 1.f) { a *= 2.f; --e; }
  return s | ((e + 7) << 3) | (int)((a - 1.f) * 8.f + 0.5f);
}

int main() {
  unsigned char hA[2048], hB[2048];
  for (int r = 0; r < 16; ++r)
    for (int k = 0; k < 128; ++k) {
      hA[r * 128 + k] = enc((float)(1 << (k / 16)));
      hB[r * 128 + k] = enc(1.f);
    }
  unsigned char *dA, *dB; float* dC;
  (void)hipMalloc(&dA, 2048); (void)hipMalloc(&dB, 2048);
  (void)hipMalloc(&dC, 1024);
  (void)hipMemcpy(dA, hA, 2048, hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hB, 2048, hipMemcpyHostToDevice);
  float base = 16 * 255;
  for (int byteidx = 0; byteidx < 4; ++byteidx) {
    for (int L = 0; L < 64; ++L) {
      hipLaunchKernelGGL(k_probe, dim3(1), dim3(64), 0, 0, dC, dA, dB, L,
                         byteidx);
      float out[256];
      (void)hipMemcpy(out, dC, 1024, hipMemcpyDeviceToHost);
      char line[256] = "";
      int any = 0;
      for (int r = 0; r < 16; ++r) {
        int e = (int)((out[r * 16] - base) / 16.f + 0.5f);
        if (!e) continue;
        any = 1;
        char buf[128];
        // e's set bits = doubled 16-blocks of row r
        snprintf(buf, sizeof buf, " r%d:blocks[", r);
        strcat(line, buf);
        for (int q = 0; q < 8; ++q)
          if (e & (1 << q)) { snprintf(buf, sizeof buf, "%d,", q); strcat(line, buf); }
        strcat(line, "]");
      }
      if (any) printf("byte%d lane %2d ->%s\n", byteidx, L, line);
    }
  }
  return 0;
}
