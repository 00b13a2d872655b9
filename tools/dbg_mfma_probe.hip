// Standalone probe: verify v_mfma_f32_32x32x2_f32 operand/D layout claims.
#include <hip/hip_runtime.h>
#include <cstdio>
typedef float f32x16 __attribute__((ext_vector_type(16)));
__global__ void probe(const float* A, const float* B, float* D) {
  int l = threadIdx.x;
  f32x16 acc = {};
  for (int s = 0; s < 1; ++s) {
    float a = A[(l & 31) * 2 + (l >> 5)];      // A[i][k] row-major 32x2
    float b = B[(l >> 5) * 32 + (l & 31)];     // B[k][j] row-major 2x32
    acc = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc, 0, 0, 0);
  }
  for (int g = 0; g < 16; ++g) {
    int row = (g & 3) + 8 * (g >> 2) + 4 * (l >> 5);
    int col = l & 31;
    D[row * 32 + col] = acc[g];
  }
}
int main() {
  float hA[64], hB[64], hD[1024];
  for (int i = 0; i < 32; ++i) for (int kk = 0; kk < 2; ++kk) hA[i*2+kk] = i + 100.f*kk;
  for (int kk = 0; kk < 2; ++kk) for (int j = 0; j < 32; ++j) hB[kk*32+j] = 1000.f*j + 7.f*kk;
  float *dA,*dB,*dD;
  hipMalloc(&dA, 256); hipMalloc(&dB, 256); hipMalloc(&dD, 4096);
  hipMemcpy(dA, hA, 256, hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, 256, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dA, dB, dD);
  hipMemcpy(hD, dD, 4096, hipMemcpyDeviceToHost);
  int bad = 0;
  for (int i = 0; i < 32 && bad < 5; ++i)
    for (int j = 0; j < 32; ++j) {
      float want = 0;
      for (int kk = 0; kk < 2; ++kk) want += (i + 100.f*kk) * (1000.f*j + 7.f*kk);
      if (hD[i*32+j] != want) { printf("MISMATCH i=%d j=%d got=%f want=%f\n", i, j, hD[i*32+j], want); if (++bad >= 5) break; }
    }
  printf(bad ? "LAYOUT BAD\n" : "LAYOUT OK\n");
  return 0;
}
