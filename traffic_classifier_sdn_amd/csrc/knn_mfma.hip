// MFMA brute-force k-NN for MI355X (gfx950) — the distance GEMM on the
// matrix cores ("brute-force distance GEMM (MFMA) +
// per-row top-k selection").
//
// Shape: queries Q[nq,12] vs reference R[nr,12], k <= 8.  The dot-product
// part of ||q-r||^2 runs as v_mfma_f32_32x32x2_f32 tiles (exact f32 at the
// 157 TF f32 rate); selection keys are formed in expanded form
//     key = ||r-c||^2 - 2 (q-c)·(r-c)      (= d^2 - ||q-c||^2)
// around the reference column means c, which removes the catastrophic
// cancellation raw byte/packet-count features would cause (translation
// leaves Euclidean distance invariant; the per-query constant drops out of
// the ordering).  Selection is a per-row threshold filter: producing lanes
// compare each key against a broadcast (stale-tolerant, conservative)
// per-row threshold and only SURVIVORS spill into a per-wave LDS key
// matrix + survivor bitmask; the owning lane pair then merges them into a
// register-resident exact top-k list.  Stale thresholds only admit extra
// survivors (the owner re-checks), never drop one, so the selection is
// exact; in steady state a candidate costs one compare beyond the MFMA.
// The final k winners per query are REFINED with the exact
// direct-difference f32 distance (same numerics as the scalar kernel).
//
// Grid: (ceil(nq/QB), S) — query blocks x reference shards; knn_merge_kernel
// folds the S partial lists per query (+ fused uniform vote).
//
// Per workgroup (256 threads = 4 waves):
//   QB=256 queries staged centered+transposed in LDS (12 KB)
//   R streamed in 128-candidate tiles, centered+transposed (6 KB) + norms
//   wave w owns query tiles w, w+4 (32 rows each; row r of a tile is
//   owned by lanes r and r+32, scanning key columns 0-15 / 16-31)

#include <hip/hip_runtime.h>

#include <cfloat>

#include "common.h"

#define KM_QB 256      // queries per workgroup (8 query tiles, 2 per wave)
#define KM_TB 128      // candidate tile (4 MFMA column-subtiles)
#define KM_KMAX 8      // max k supported by this path
#define KM_F 12
#define KM_PITCH 33    // keymat row pitch (bank-staggered)

typedef float f32x16 __attribute__((ext_vector_type(16)));

// v_mfma_f32_32x32x2_f32 C/D mapping: col = lane&31,
// row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)   (verified: tools/dbg_mfma_probe)
DEV int km_rowmap(int reg, int half) { return (reg & 3) + 8 * (reg >> 2) + 4 * half; }

// Exact direct-difference squared distance (refine numerics).
DEV float km_dist2(const float* __restrict__ q, const float* __restrict__ r) {
  float a = 0.f, b = 0.f;
#pragma unroll
  for (int j = 0; j < 6; ++j) {
    float t = q[j] - r[j];
    a = fmaf(t, t, a);
  }
#pragma unroll
  for (int j = 6; j < KM_F; ++j) {
    float t = q[j] - r[j];
    b = fmaf(t, t, b);
  }
  return a + b;
}

__launch_bounds__(256) __global__ void knn_mfma_kernel(
    const float* __restrict__ Q, const float* __restrict__ R,
    const float* __restrict__ cmean,  // [12] reference column means
    float* __restrict__ part_d,       // [S, nq, k] refined exact d^2
    int* __restrict__ part_i,         // [S, nq, k] global candidate index (-1 pad)
    long long nq, long long nr, int k, long long shard_rows) {
  __shared__ float s_qt[KM_F][KM_QB];            // centered Q^T
  __shared__ float s_rt[KM_F][KM_TB];            // centered R-tile^T
  __shared__ float s_rn[KM_TB];                  // ||r-c||^2 (FLT_MAX pad)
  // per-wave survivor key matrix [col][row] + per-row survivor bitmask:
  // producers only spill keys that beat the (stale-tolerant) per-row
  // threshold, so in steady state a subtile costs 16 broadcast threshold
  // reads + compares and the owner scan reads nothing at all — the LDS
  // pipe stops being the bottleneck (was: full 16-write/16-read key spill)
  __shared__ float s_km[4][32 * KM_PITCH];
  __shared__ unsigned s_mask[4][32];
  __shared__ float s_worst[KM_QB];  // per-row k-th-best key (owners update)

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wave = tid >> 6;
  const int half = lane >> 5;  // MFMA k-index; also this lane's column half
  const int l31 = lane & 31;
  const long long qb0 = (long long)blockIdx.x * KM_QB;
  const int shard = blockIdx.y;
  const long long r0 = (long long)shard * shard_rows;
  const long long r1 = min(nr, r0 + shard_rows);
  if (r0 >= r1) {  // empty shard: emit pads
    for (int row = tid; row < KM_QB; row += blockDim.x) {
      long long q = qb0 + row;
      if (q < nq)
        for (int j = 0; j < k; ++j) {
          part_d[((long long)shard * nq + q) * k + j] = FLT_MAX;
          part_i[((long long)shard * nq + q) * k + j] = -1;
        }
    }
    return;
  }

  float cm[KM_F];
#pragma unroll
  for (int j = 0; j < KM_F; ++j) cm[j] = cmean[j];

  // ---- stage the query block: centered, transposed -----------------------
  for (int i = tid; i < KM_QB; i += blockDim.x) {
    long long q = min(qb0 + i, nq - 1);  // clamp tail (outputs masked later)
    Row12 x = load_row12(Q, q);
#pragma unroll
    for (int j = 0; j < KM_F; ++j) s_qt[j][i] = x.v[j] - cm[j];
    s_worst[i] = FLT_MAX;
  }
  if (tid < 4)
#pragma unroll
    for (int r = 0; r < 32; ++r) s_mask[tid][r] = 0u;
  __syncthreads();

  // per-(lane, query-tile) top-k sub-list in registers; the lane pair
  // (r, r+32) covers key columns [0,16) / [16,32) of row r
  float lk[2][KM_KMAX];
  int li[2][KM_KMAX];
  float wkey[2];
#pragma unroll
  for (int t = 0; t < 2; ++t) {
#pragma unroll
    for (int j = 0; j < KM_KMAX; ++j) {
      lk[t][j] = FLT_MAX;
      li[t][j] = -1;
    }
    wkey[t] = FLT_MAX;
  }

  // ---- main loop over candidate tiles ------------------------------------
  // Software prefetch: each tile's global row load is issued one iteration
  // ahead (into registers), so its ~HBM round-trip overlaps the previous
  // tile's MFMA work instead of stalling the whole workgroup at staging.
  const int pf_i = tid;  // thread -> candidate slot (KM_TB <= blockDim.x)
  Row12 pf_row;
  bool pf_valid = false;
  if (pf_i < KM_TB && r0 + pf_i < r1) {
    pf_row = load_row12(R, r0 + pf_i);
    pf_valid = true;
  }
  for (long long tb = r0; tb < r1; tb += KM_TB) {
    const int cnt_t = (int)min((long long)KM_TB, r1 - tb);
    __syncthreads();  // previous tile fully consumed
    if (pf_i < KM_TB) {
      if (pf_valid) {
        float rn = 0.f;
#pragma unroll
        for (int j = 0; j < KM_F; ++j) {
          float rc = pf_row.v[j] - cm[j];
          s_rt[j][pf_i] = rc;
          rn = fmaf(rc, rc, rn);
        }
        s_rn[pf_i] = rn;
      } else {
#pragma unroll
        for (int j = 0; j < KM_F; ++j) s_rt[j][pf_i] = 0.f;
        s_rn[pf_i] = FLT_MAX;  // padded candidate never selected
      }
    }
    __syncthreads();
    // issue the NEXT tile's loads now; the waitcnt lands at the next
    // iteration's staging writes, hidden behind this tile's compute
    {
      long long nxt = tb + KM_TB + pf_i;
      pf_valid = (pf_i < KM_TB) && (nxt < r1);
      if (pf_valid) pf_row = load_row12(R, nxt);
    }

#pragma unroll
    for (int qti = 0; qti < 2; ++qti) {
      const int qt = wave + 4 * qti;
      const int rowbase = qt * 32;
      // A fragments: lane l -> Qc[rowbase + (l&31)][2s + (l>>5)]
      float afrag[6];
#pragma unroll
      for (int s = 0; s < 6; ++s) afrag[s] = s_qt[2 * s + half][rowbase + l31];
      // per-row threshold cache (broadcast reads); stale values only admit
      // extra survivors — the owner re-checks against its exact register
      // worst — never miss one
      float tau[16];
#pragma unroll
      for (int g = 0; g < 16; ++g) tau[g] = s_worst[rowbase + km_rowmap(g, half)];

      for (int ct = 0; ct < KM_TB / 32; ++ct) {
        f32x16 acc = {};
#pragma unroll
        for (int s = 0; s < 6; ++s)
          acc = __builtin_amdgcn_mfma_f32_32x32x2f32(
              afrag[s], s_rt[2 * s + half][ct * 32 + l31], acc, 0, 0, 0);
        const float rncol = s_rn[ct * 32 + l31];
        // survivors spill (key -> own slot, col bit -> row mask); padded
        // columns carry key = FLT_MAX and never pass
#pragma unroll
        for (int g = 0; g < 16; ++g) {
          float key = fmaf(-2.f, acc[g], rncol);
          if (key < tau[g]) {
            int row = km_rowmap(g, half);
            s_km[wave][l31 * KM_PITCH + row] = key;
            atomicOr(&s_mask[wave][row], 1u << l31);
          }
        }
        __builtin_amdgcn_wave_barrier();
        // owner scan: lane pair (l31, l31+32) splits the survivor bits of
        // row l31 (low/high 16 columns)
        {
          const int row = l31;
          unsigned m = s_mask[wave][row];
          unsigned mh = half ? (m >> 16) : (m & 0xffffu);
          const long long colbase = tb + ct * 32 + half * 16;
          while (mh) {
            int cc = __ffs(mh) - 1;
            mh &= mh - 1;
            float key = s_km[wave][(half * 16 + cc) * KM_PITCH + row];
            if (key < wkey[qti]) {
              int ws = 0;
              float wv = -FLT_MAX;
#pragma unroll
              for (int j = 0; j < KM_KMAX; ++j)
                if (j < k && lk[qti][j] > wv) {
                  wv = lk[qti][j];
                  ws = j;
                }
              lk[qti][ws] = key;
              li[qti][ws] = (int)(colbase + cc);
              wv = -FLT_MAX;
#pragma unroll
              for (int j = 0; j < KM_KMAX; ++j)
                if (j < k && lk[qti][j] > wv) wv = lk[qti][j];
              wkey[qti] = wv;
            }
          }
          // a produced column can land in either half's sub-list, so the
          // shared threshold must be conservative for BOTH: publish the max
          // of the lane pair's worsts (stale/loose admits extra survivors,
          // never drops one)
          float other = __shfl(wkey[qti], lane ^ 32, WAVE);
          float pub = fmaxf(wkey[qti], other);
          __builtin_amdgcn_wave_barrier();
          if (half == 0) {
            s_worst[rowbase + row] = pub;
            if (m) s_mask[wave][row] = 0u;
          }
        }
        __builtin_amdgcn_wave_barrier();
      }
    }
  }
  __syncthreads();

  // ---- merge lane pairs, refine with exact distances, emit sorted --------
#pragma unroll
  for (int qti = 0; qti < 2; ++qti) {
    const int qt = wave + 4 * qti;
    const long long q = qb0 + qt * 32 + l31;
    // pull the partner half's sub-list (lane r+32 -> lane r and vice versa;
    // only lanes < 32 emit)
    float ok[2 * KM_KMAX];
    int oi[2 * KM_KMAX];
#pragma unroll
    for (int j = 0; j < KM_KMAX; ++j) {
      ok[j] = lk[qti][j];
      oi[j] = li[qti][j];
      ok[KM_KMAX + j] = __shfl(lk[qti][j], lane ^ 32, WAVE);
      oi[KM_KMAX + j] = __shfl(li[qti][j], lane ^ 32, WAVE);
    }
    if (half == 0 && q < nq) {
      Row12 xq = load_row12(Q, q);
      // select top-k of the 2k candidates by key, refine, sort by (d, idx)
      float dk[KM_KMAX];
      int ik[KM_KMAX];
#pragma unroll
      for (int j = 0; j < KM_KMAX; ++j) {
        dk[j] = FLT_MAX;
        ik[j] = -1;
      }
      // insertion into a sorted-by-key k-list (ties: lower index first)
      for (int j = 0; j < 2 * KM_KMAX; ++j) {
        if (oi[j] < 0) continue;  // untouched slots (>= k per half) are -1
        float kv = ok[j];
        int iv = oi[j];
        if (kv > dk[k - 1] || (kv == dk[k - 1] && ik[k - 1] >= 0 && iv > ik[k - 1])) continue;
        int b = k - 1;
        while (b > 0 && (dk[b - 1] > kv ||
                         (dk[b - 1] == kv && (unsigned)ik[b - 1] > (unsigned)iv))) {
          dk[b] = dk[b - 1];
          ik[b] = ik[b - 1];
          --b;
        }
        dk[b] = kv;
        ik[b] = iv;
      }
      // refine: exact diff-form distances of the winners
      for (int j = 0; j < k; ++j) {
        if (ik[j] < 0) {
          dk[j] = FLT_MAX;
          continue;
        }
        Row12 xr = load_row12(R, ik[j]);
        dk[j] = km_dist2(xq.v, xr.v);
      }
      // final sort by (exact d, idx)
      for (int a = 1; a < k; ++a) {
        float dv = dk[a];
        int iv = ik[a];
        int b = a - 1;
        while (b >= 0 && (dk[b] > dv || (dk[b] == dv && (unsigned)ik[b] > (unsigned)iv))) {
          dk[b + 1] = dk[b];
          ik[b + 1] = ik[b];
          --b;
        }
        dk[b + 1] = dv;
        ik[b + 1] = iv;
      }
      long long o = ((long long)shard * nq + q) * k;
      for (int j = 0; j < k; ++j) {
        part_d[o + j] = dk[j];
        part_i[o + j] = ik[j];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Merge the S per-shard lists of one query into the final top-k (+ fused
// uniform vote when labels are supplied).  One thread per query.
// ---------------------------------------------------------------------------
__global__ void knn_merge_kernel(const float* __restrict__ part_d,
                                 const int* __restrict__ part_i,
                                 const unsigned char* __restrict__ ry,  // may be null
                                 float* __restrict__ out_d, int* __restrict__ out_i,
                                 int* __restrict__ out_lab, long long nq, int S,
                                 int k, int C, long long idx_base) {
  long long q = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (q >= nq) return;
  float dk[KM_KMAX];
  int ik[KM_KMAX];
#pragma unroll
  for (int j = 0; j < KM_KMAX; ++j) {
    dk[j] = FLT_MAX;
    ik[j] = -1;
  }
  for (int s = 0; s < S; ++s) {
    long long o = ((long long)s * nq + q) * k;
    for (int j = 0; j < k; ++j) {
      float d = part_d[o + j];
      int idx = part_i[o + j];
      if (idx < 0) continue;
      // shard lists are sorted: once this shard stops beating the current
      // worst, the rest of it cannot either
      if (d > dk[k - 1] || (d == dk[k - 1] && ik[k - 1] >= 0 && (unsigned)idx > (unsigned)ik[k - 1]))
        break;
      int b = k - 1;
      while (b > 0 && (dk[b - 1] > d || (dk[b - 1] == d && (unsigned)ik[b - 1] > (unsigned)idx))) {
        dk[b] = dk[b - 1];
        ik[b] = ik[b - 1];
        --b;
      }
      dk[b] = d;
      ik[b] = idx;
    }
  }
  for (int j = 0; j < k; ++j) {
    out_d[q * k + j] = dk[j];
    out_i[q * k + j] = ik[j] >= 0 ? (int)(ik[j] + idx_base) : -1;
  }
  if (out_lab && ry) {
    int votes[16];
#pragma unroll
    for (int c = 0; c < 16; ++c) votes[c] = 0;
    for (int j = 0; j < k; ++j)
      if (ik[j] >= 0) votes[ry[ik[j]] & 15] += 1;
    int best = 0, bc = 0;
    for (int c = 0; c < C; ++c)
      if (votes[c] > best) {
        best = votes[c];
        bc = c;
      }
    out_lab[q] = bc;
  }
}

extern "C" void launch_knn_mfma(const float* Q, const float* R, const float* cmean,
                                const unsigned char* ry, float* part_d, int* part_i,
                                float* out_d, int* out_i, int* out_lab,
                                long long nq, long long nr, int S, int k, int C,
                                long long idx_base, hipStream_t stream) {
  long long shard_rows = (nr + S - 1) / S;
  dim3 grid((unsigned)((nq + KM_QB - 1) / KM_QB), (unsigned)S);
  hipLaunchKernelGGL(knn_mfma_kernel, grid, dim3(256), 0, stream, Q, R, cmean,
                     part_d, part_i, nq, nr, k, shard_rows);
  int block = 256;
  int mgrid = (int)((nq + block - 1) / block);
  hipLaunchKernelGGL(knn_merge_kernel, dim3(mgrid), dim3(block), 0, stream,
                     part_d, part_i, ry, out_d, out_i, out_lab, nq, S, k, C,
                     idx_base);
}
