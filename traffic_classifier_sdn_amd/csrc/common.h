// Common device helpers for the CDNA4 (gfx950) kernels.
// Target: MI355X only — wave64, 256 CUs / 8 XCDs, 160 KiB LDS per CU.
#pragma once

#include <hip/hip_runtime.h>

#define WAVE 64
#define DEV __device__ __forceinline__

// Grid sizing for memory-bound grid-stride kernels (guide G11):
// cap blocks, grid-stride the rest.
static inline int ts_grid(long long work, int block, int cap = 2048) {
  long long b = (work + block - 1) / block;
  if (b > cap) b = cap;
  if (b < 1) b = 1;
  return (int)b;
}

// 12-float feature row load: rows are 48 B, so every row is 16-B aligned
// when the base pointer is. 3x float4 per row.
struct Row12 {
  float v[12];
};

DEV Row12 load_row12(const float* __restrict__ X, long long row) {
  Row12 r;
  const float4* p = reinterpret_cast<const float4*>(X + row * 12);
  float4 a = p[0], b = p[1], c = p[2];
  r.v[0] = a.x; r.v[1] = a.y; r.v[2] = a.z; r.v[3] = a.w;
  r.v[4] = b.x; r.v[5] = b.y; r.v[6] = b.z; r.v[7] = b.w;
  r.v[8] = c.x; r.v[9] = c.y; r.v[10] = c.z; r.v[11] = c.w;
  return r;
}

// wave-level f32/f64 sum over all 64 lanes
DEV float wave_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_down(x, off, 64);
  return x;
}

DEV double wave_sum(double x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_down(x, off, 64);
  return x;
}

DEV unsigned long long wave_sum(unsigned long long x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_down(x, off, 64);
  return x;
}
