// Standalone probe: verify v_mfma_f32_32x32x16_bf16 operand/D layout.
// Hypothesis (by analogy with the verified f32 form and CDNA3 bf16 forms):
//   A[i][k]: lane l holds i = l&31, k = (l>>5)*8 + e  (e = 0..7)
//   B[k][j]: lane l holds j = l&31, k = (l>>5)*8 + e
//   C/D:     col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
// Values are small integers (exact in bf16) so equality is exact.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

__global__ void probe(const float* A, const float* B, float* D) {
  int l = threadIdx.x;
  bf16x8 a, b;
  for (int e = 0; e < 8; ++e) {
    int ka = (l >> 5) * 8 + e;
    a[e] = (__bf16)A[(l & 31) * 16 + ka];   // A row-major [32][16]
    b[e] = (__bf16)B[ka * 32 + (l & 31)];   // B row-major [16][32]
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  for (int g = 0; g < 16; ++g) {
    int row = (g & 3) + 8 * (g >> 2) + 4 * (l >> 5);
    D[row * 32 + (l & 31)] = acc[g];
  }
}

int main() {
  float hA[32 * 16], hB[16 * 32], hD[1024];
  for (int i = 0; i < 32; ++i)
    for (int k = 0; k < 16; ++k) hA[i * 16 + k] = (float)((i % 7) + k);
  for (int k = 0; k < 16; ++k)
    for (int j = 0; j < 32; ++j) hB[k * 32 + j] = (float)((j % 5) + 2 * k);
  float *dA, *dB, *dD;
  hipMalloc(&dA, sizeof(hA)); hipMalloc(&dB, sizeof(hB)); hipMalloc(&dD, sizeof(hD));
  hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dA, dB, dD);
  hipMemcpy(hD, dD, sizeof(hD), hipMemcpyDeviceToHost);
  int bad = 0;
  for (int i = 0; i < 32; ++i)
    for (int j = 0; j < 32; ++j) {
      float want = 0;
      for (int k = 0; k < 16; ++k) want += ((i % 7) + k) * ((j % 5) + 2 * k);
      if (hD[i * 32 + j] != want) {
        if (bad < 6)
          printf("MISMATCH i=%d j=%d got=%g want=%g\n", i, j, hD[i * 32 + j], want);
        ++bad;
      }
    }
  printf(bad ? "BF16 LAYOUT BAD (%d)\n" : "BF16 LAYOUT OK\n", bad);
  return bad != 0;
}
