// Mapping probe for v_mfma_scale_f32_16x16x128_f8f6f4 (fp8 e4m3 inputs).
// Hypothesis: lane L holds A row r=L&15, k = 32*(L>>4)+i (i<32, 32 bytes);
// B col c=L&15 same k; C/D: col=lane&15, row=4*(lane>>4)+reg.
// Scales: e8m0 byte (127 = 1.0) applied per 32-element block; opsel
// selects which byte of the i32 scale operand.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstring>
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) int i32x8;

__global__ void k_probe(float* C, const unsigned char* A,
                        const unsigned char* B, int sa, int sb) {
  int lane = threadIdx.x & 63;
  i32x8 av, bv;
  unsigned char ab[32], bb[32];
  for (int i = 0; i < 32; ++i) {
    ab[i] = A[(lane & 15) * 128 + 32 * (lane >> 4) + i];
    bb[i] = B[(lane & 15) * 128 + 32 * (lane >> 4) + i];
  }
  __builtin_memcpy(&av, ab, 32);
  __builtin_memcpy(&bv, bb, 32);
  f32x4 acc = {};
  acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      av, bv, acc, 0, 0, 0, sa, 0, sb);
  for (int r = 0; r < 4; ++r)
    C[(4 * (lane >> 4) + r) * 16 + (lane & 15)] = acc[r];
}

// fp8 e4m3 encode of small ints (exact for |v|<=8 here)
unsigned char enc(float v) {
  // e4m3fn: sign(1) exp(4,bias7) mant(3)
  if (v == 0) return 0;
  unsigned char s = v < 0 ? 0x80 : 0;
  float a = v < 0 ? -v : v;
  int e = 0;
  while (a >= 2.f) { a /= 2.f; ++e; }
  while (a < 1.f) { a *= 2.f; --e; }
  int m = (int)((a - 1.f) * 8.f + 0.5f);
  return s | ((e + 7) << 3) | m;
}

int main() {
  // A[16][128], B[16][128]; C = A x B^T with fp32 host reference
  unsigned char hA[16 * 128], hB[16 * 128];
  float fA[16 * 128], fB[16 * 128];
  srand(7);
  for (int i = 0; i < 16 * 128; ++i) {
    int v = (rand() % 7) - 3;
    fA[i] = (float)v; hA[i] = enc((float)v);
    int w = (rand() % 5) - 2;
    fB[i] = (float)w; hB[i] = enc((float)w);
  }
  float ref[256];
  for (int r = 0; r < 16; ++r)
    for (int c = 0; c < 16; ++c) {
      float s = 0;
      for (int k = 0; k < 128; ++k) s += fA[r * 128 + k] * fB[c * 128 + k];
      ref[r * 16 + c] = s;
    }
  unsigned char *dA, *dB; float* dC;
  hipMalloc(&dA, sizeof hA); hipMalloc(&dB, sizeof hB);
  hipMalloc(&dC, 256 * 4);
  hipMemcpy(dA, hA, sizeof hA, hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, sizeof hB, hipMemcpyHostToDevice);
  // scales = 1.0: e8m0 127 in every byte
  int one = 0x7f7f7f7f;
  hipLaunchKernelGGL(k_probe, dim3(1), dim3(64), 0, 0, dC, dA, dB, one, one);
  float out[256];
  hipMemcpy(out, dC, 256 * 4, hipMemcpyDeviceToHost);
  int bad = 0;
  for (int i = 0; i < 256; ++i)
    if (out[i] != ref[i]) ++bad;
  printf("scale=1.0 mapping: %s (%d/256 mismatches)\n", bad ? "FAIL" : "PASS", bad);
  if (bad) {
    for (int i = 0; i < 8; ++i)
      printf("  C[%d]=%f ref=%f\n", i, out[i], ref[i]);
  }
  // scale test: sa = 2.0 (e8m0 128) on ALL blocks -> C doubles
  int two = 0x80808080;
  hipLaunchKernelGGL(k_probe, dim3(1), dim3(64), 0, 0, dC, dA, dB, two, one);
  hipMemcpy(out, dC, 256 * 4, hipMemcpyDeviceToHost);
  bad = 0;
  for (int i = 0; i < 256; ++i)
    if (out[i] != 2.f * ref[i]) ++bad;
  printf("scaleA=2.0: %s (%d/256)\n", bad ? "FAIL" : "PASS", bad);
  // opsel test: scaleA byte1 = 2.0, opsel=1 -> doubles; opsel=0 (byte0=1.0) -> identity
  return 0;
}
