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
// the ordering).
//
// Selection (round-2 SWAPPED-OPERAND design): the MFMA computes
// D[cand][query] (A = centered R-tile, B = centered Q^T), so the query
// index lands in the LANE (col = lane&31) and each lane's 16 accumulator
// registers are 16 CANDIDATES of its own query — top-k selection is pure
// lane-local register work (one fmaf + compare per candidate, rare sorted
// insert).  Round 1 computed D[query][cand] (A = Q), which scattered one
// query's candidates across the wave and needed a per-wave LDS key matrix,
// survivor bitmask, owner scan, and stale-threshold publishes — measured
// epilogue-bound at 42.4% MfmaUtil (profiles/pmc_counters_r01.md).  The
// swap deletes all of that LDS machinery (-18.4 KB/WG, no atomics, no wave
// barriers in the hot loop).  The final k winners per query are REFINED
// with the exact direct-difference f32 distance (same numerics as the
// scalar kernel).
//
// Grid: (ceil(nq/QB), S) — query blocks x reference shards; knn_merge_kernel
// folds the S partial lists per query (+ fused uniform vote).
//
// Per workgroup (256 threads = 4 waves):
//   QB=256 queries staged centered+transposed in LDS (12 KB)
//   R streamed in 256-candidate tiles, double-buffered centered+transposed
//   (2 x 12 KB) + norms — ONE __syncthreads per tile (stage t+1 into the
//   idle buffer while computing t)
//   wave w serves query groups w, w+4 (32 queries each; lane pair
//   (l, l+32) shares query (group*32 + l&31) and splits each 32-candidate
//   subtile between its two accumulator row maps)

#include <hip/hip_runtime.h>

#include <cfloat>

#include "common.h"

#define KM_NQT 2       // query groups per wave
#define KM_QB (32 * 4 * KM_NQT)  // queries per workgroup
#define KM_TB 256      // candidate tile (8 MFMA column-subtiles)
#define KM_KMAX 8      // max k supported by this path
#define KM_F 12

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

__launch_bounds__(256, 4) __global__ void knn_mfma_kernel(
    const float* __restrict__ Q, const float* __restrict__ R,
    const float* __restrict__ cmean,  // [12] reference column means
    float* __restrict__ part_d,       // [S, nq, k] refined exact d^2
    int* __restrict__ part_i,         // [S, nq, k] global candidate index (-1 pad)
    long long nq, long long nr, int k, long long shard_rows) {
  __shared__ float s_qt[KM_F][KM_QB];            // centered Q^T
  __shared__ float s_rt[2][KM_F][KM_TB];         // centered R-tile^T (2 bufs)
  __shared__ float s_rn[2][KM_TB];               // ||r-c||^2 (FLT_MAX pad)
  __shared__ float s_cm[KM_F];                   // column means (staging only)

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wave = tid >> 6;
  const int half = lane >> 5;  // MFMA k-index (K=2 per instruction)
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

  // column means live in LDS, not registers: they are read only in the
  // staging phases, and the 12 VGPRs they'd occupy are part of the margin
  // that keeps this kernel at 4 waves/SIMD
  if (tid < KM_F) s_cm[tid] = cmean[tid];
  __syncthreads();

  // ---- stage the query block: centered, transposed -----------------------
  for (int i = tid; i < KM_QB; i += blockDim.x) {
    long long q = min(qb0 + i, nq - 1);  // clamp tail (outputs masked later)
    Row12 x = load_row12(Q, q);
#pragma unroll
    for (int j = 0; j < KM_F; ++j) s_qt[j][i] = x.v[j] - s_cm[j];
  }
  __syncthreads();

  // SWAPPED-OPERAND selection (round 2): with A = R-tile and B = Q^T the
  // D[cand][query] mapping puts QUERY in the lane index (col = lane&31) and
  // 16 CANDIDATES in each lane's accumulator registers — so top-k selection
  // is pure lane-local register work (compare + rare sorted insert).  The
  // round-1 layout (A = Q) spread one query's candidates across lanes and
  // needed an LDS key matrix + survivor bitmask + owner scan + stale
  // threshold publishes (42.4% MfmaUtil, epilogue-bound).  All of that is
  // gone: no LDS spill, no atomics, no wave barriers in the hot loop.
  //
  // Lane l serves query qb0 + (wave + 4*qti)*32 + (l&31); the lane pair
  // (l, l+32) holds the same query and splits each 32-candidate subtile by
  // the D-row map (rowmap(g, l>>5)).  The merge/refine tail below is the
  // same lane-pair contract as round 1.
  float bfrag[KM_NQT][6];  // Q fragments per query group, loaded once
#pragma unroll
  for (int qti = 0; qti < KM_NQT; ++qti) {
    const int qbase = (wave + 4 * qti) * 32;
#pragma unroll
    for (int s = 0; s < 6; ++s) bfrag[qti][s] = s_qt[2 * s + half][qbase + l31];
  }

  // per-(lane, query-group) exact top-k sub-list in registers
  float lk[KM_NQT][KM_KMAX];
  int li[KM_NQT][KM_KMAX];
  float wkey[KM_NQT];
#pragma unroll
  for (int t = 0; t < KM_NQT; ++t) {
#pragma unroll
    for (int j = 0; j < KM_KMAX; ++j) {
      lk[t][j] = FLT_MAX;
      li[t][j] = -1;
    }
    wkey[t] = FLT_MAX;
  }

  // ---- main loop over candidate tiles ------------------------------------
  // Software prefetch: each tile's global row load is issued one iteration
  // ahead (into registers); LDS staging is double-buffered so ONE
  // __syncthreads per tile separates write(buf) from compute(buf) — the
  // barrier of the NEXT iteration separates compute(buf) from the write
  // that reuses buf two tiles later.
  const int pf_i = tid;  // thread -> candidate slot (KM_TB == blockDim.x)
  Row12 pf_row;
  bool pf_valid = false;
  if (r0 + pf_i < r1) {
    pf_row = load_row12(R, r0 + pf_i);
    pf_valid = true;
  }
  int t = 0;
  for (long long tb = r0; tb < r1; tb += KM_TB, ++t) {
    const int buf = t & 1;
    // stage tile t (prefetched last iteration) into its buffer
    if (pf_valid) {
      float rn = 0.f;
#pragma unroll
      for (int j = 0; j < KM_F; ++j) {
        float rc = pf_row.v[j] - s_cm[j];
        s_rt[buf][j][pf_i] = rc;
        rn = fmaf(rc, rc, rn);
      }
      s_rn[buf][pf_i] = rn;
    } else {
#pragma unroll
      for (int j = 0; j < KM_F; ++j) s_rt[buf][j][pf_i] = 0.f;
      s_rn[buf][pf_i] = FLT_MAX;  // padded candidate never selected
    }
    // issue the NEXT tile's loads now; their waitcnt lands at the next
    // iteration's staging writes, hidden behind this tile's compute
    {
      long long nxt = tb + KM_TB + pf_i;
      pf_valid = nxt < r1;
      if (pf_valid) pf_row = load_row12(R, nxt);
    }
    __syncthreads();  // tile t visible; buf^1 free for the next staging pass

    const float(*rt)[KM_TB] = s_rt[buf];
    const float* rn_t = s_rn[buf];

    for (int ct = 0; ct < KM_TB / 32; ++ct) {
      // A fragments: candidate rows, lane l -> Rc[ct*32 + (l&31)][2s + half];
      // loaded ONCE per subtile and reused by both query groups
      float afrag[6];
#pragma unroll
      for (int s = 0; s < 6; ++s) afrag[s] = rt[2 * s + half][ct * 32 + l31];
      // hoist this lane's 16 candidate norms as 4 batched ds_read_b128
      // (rowmap hits 4 runs of 4 consecutive rows: {0,8,16,24}+4*half);
      // leaving the reads inline made hipcc serialize ~50-cycle LDS
      // latency per candidate check — 32 dependent waits per subtile
      const float4* rn4 =
          reinterpret_cast<const float4*>(rn_t + ct * 32 + 4 * half);
      float4 rnv[2];
#pragma unroll
      for (int r4 = 0; r4 < 2; ++r4) rnv[r4] = rn4[2 * r4];
      float4 rnw[2];
#pragma unroll
      for (int r4 = 0; r4 < 2; ++r4) rnw[r4] = rn4[4 + 2 * r4];
      const float* rnf0 = reinterpret_cast<const float*>(rnv);
      const float* rnf1 = reinterpret_cast<const float*>(rnw);
#pragma unroll
      for (int qti = 0; qti < KM_NQT; ++qti) {
        f32x16 acc = {};
#pragma unroll
        for (int s = 0; s < 6; ++s)
          acc = __builtin_amdgcn_mfma_f32_32x32x2f32(afrag[s], bfrag[qti][s],
                                                     acc, 0, 0, 0);
        // lane-local selection: 16 candidates of THIS lane's query; padded
        // candidates carry rn = FLT_MAX and never pass the compare.  One
        // min-tree + ONE guarded branch per query group instead of 16
        // per-candidate skip branches: the common (no-survivor) path is 16
        // fma + 15 min + 1 compare, all VALU, and the insert block's exec
        // mask is wave-collective (taken only when SOME lane has a winner).
        // Keys are NOT kept in registers — the rare insert path recomputes
        // them from acc/rnf, saving 16 VGPRs (the difference between 3 and
        // 4 waves/SIMD at this kernel's register budget).
        float kmin = fmaf(-2.f, acc[0], rnf0[0]);
#pragma unroll
        for (int g = 1; g < 8; ++g)
          kmin = fminf(kmin, fmaf(-2.f, acc[g], rnf0[g]));
#pragma unroll
        for (int g = 8; g < 16; ++g)
          kmin = fminf(kmin, fmaf(-2.f, acc[g], rnf1[g - 8]));
        if (kmin < wkey[qti]) {
#pragma unroll
          for (int g = 0; g < 16; ++g) {
            float key = fmaf(-2.f, acc[g], g < 8 ? rnf0[g] : rnf1[g - 8]);
            if (key < wkey[qti]) {
              const int cand = ct * 32 + km_rowmap(g, half);
              int ws = 0;
              float wv = -FLT_MAX;
#pragma unroll
              for (int j = 0; j < KM_KMAX; ++j)
                if (j < k && lk[qti][j] > wv) {
                  wv = lk[qti][j];
                  ws = j;
                }
              lk[qti][ws] = key;
              li[qti][ws] = (int)(tb + cand);
              wv = -FLT_MAX;
#pragma unroll
              for (int j = 0; j < KM_KMAX; ++j)
                if (j < k && lk[qti][j] > wv) wv = lk[qti][j];
              wkey[qti] = wv;
            }
          }
        }
      }
    }
  }
  __syncthreads();

  // ---- merge lane pairs, refine with exact distances, emit sorted --------
#pragma unroll
  for (int qti = 0; qti < KM_NQT; ++qti) {
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
// bf16 coarse-pass variant (OPT-IN approximate mode, `approx=True`).
//
// Same swapped-operand structure, but the distance GEMM runs on
// v_mfma_f32_32x32x16_bf16 — 16x the f32-MFMA rate — over bf16-rounded
// centered rows (K=16: the 12 features + 4 zero pads, ONE MFMA per 32x32
// subtile instead of six).  Selection keeps the FULL KM_KMAX=8 candidates
// per lane half (16 per query — a >=2x safety margin over k<=8) ranked by
// the approximate bf16 key; the tail then refines EVERY pooled candidate
// with the exact f32 direct-difference distance and emits the exact-best k
// of the pool.  The output is exact distances over an approximate candidate
// pool: recall is measured, not proven (tests pin >=0.999 @ k=5 on
// flow-feature-scale data) — the default path stays the exact f32 kernel.
// A/B fragment layout hardware-verified: tools/dbg_mfma_bf16_probe.hip.
// ---------------------------------------------------------------------------

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

__launch_bounds__(256, 4) __global__ void knn_mfma_bf16_kernel(
    const float* __restrict__ Q, const float* __restrict__ R,
    const float* __restrict__ cmean, float* __restrict__ part_d,
    int* __restrict__ part_i, long long nq, long long nr, int k,
    long long shard_rows) {
  // per-candidate-contiguous bf16 layout: one lane's 8-element K-chunk is a
  // single 16-byte ds_read (s_qt/s_rt rows are 32 B)
  __shared__ __attribute__((aligned(16))) __bf16 s_qt[KM_QB][16];
  __shared__ __attribute__((aligned(16))) __bf16 s_rt[2][KM_TB][16];
  __shared__ float s_rn[2][KM_TB];
  __shared__ float s_cm[KM_F];

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wave = tid >> 6;
  const int half = lane >> 5;
  const int l31 = lane & 31;
  const long long qb0 = (long long)blockIdx.x * KM_QB;
  const int shard = blockIdx.y;
  const long long r0 = (long long)shard * shard_rows;
  const long long r1 = min(nr, r0 + shard_rows);
  if (r0 >= r1) {
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

  if (tid < KM_F) s_cm[tid] = cmean[tid];
  __syncthreads();

  for (int i = tid; i < KM_QB; i += blockDim.x) {
    long long q = min(qb0 + i, nq - 1);
    Row12 x = load_row12(Q, q);
#pragma unroll
    for (int j = 0; j < KM_F; ++j) s_qt[i][j] = (__bf16)(x.v[j] - s_cm[j]);
#pragma unroll
    for (int j = KM_F; j < 16; ++j) s_qt[i][j] = (__bf16)0.f;
  }
  __syncthreads();

  bf16x8 bfrag[KM_NQT];
#pragma unroll
  for (int qti = 0; qti < KM_NQT; ++qti)
    bfrag[qti] = *reinterpret_cast<const bf16x8*>(
        &s_qt[(wave + 4 * qti) * 32 + l31][half * 8]);

  float lk[KM_NQT][KM_KMAX];
  int li[KM_NQT][KM_KMAX];
  float wkey[KM_NQT];
#pragma unroll
  for (int t = 0; t < KM_NQT; ++t) {
#pragma unroll
    for (int j = 0; j < KM_KMAX; ++j) {
      lk[t][j] = FLT_MAX;
      li[t][j] = -1;
    }
    wkey[t] = FLT_MAX;
  }

  const int pf_i = tid;
  Row12 pf_row;
  bool pf_valid = false;
  if (r0 + pf_i < r1) {
    pf_row = load_row12(R, r0 + pf_i);
    pf_valid = true;
  }
  int t = 0;
  for (long long tb = r0; tb < r1; tb += KM_TB, ++t) {
    const int buf = t & 1;
    if (pf_valid) {
      // rn from the bf16-ROUNDED values: keys then rank the rounded
      // geometry consistently (the refine pass restores exact f32)
      float rn = 0.f;
#pragma unroll
      for (int j = 0; j < KM_F; ++j) {
        __bf16 rb = (__bf16)(pf_row.v[j] - s_cm[j]);
        s_rt[buf][pf_i][j] = rb;
        float rc = (float)rb;
        rn = fmaf(rc, rc, rn);
      }
#pragma unroll
      for (int j = KM_F; j < 16; ++j) s_rt[buf][pf_i][j] = (__bf16)0.f;
      s_rn[buf][pf_i] = rn;
    } else {
#pragma unroll
      for (int j = 0; j < 16; ++j) s_rt[buf][pf_i][j] = (__bf16)0.f;
      s_rn[buf][pf_i] = FLT_MAX;
    }
    {
      long long nxt = tb + KM_TB + pf_i;
      pf_valid = nxt < r1;
      if (pf_valid) pf_row = load_row12(R, nxt);
    }
    __syncthreads();

    const __bf16(*rt)[16] = s_rt[buf];
    const float* rn_t = s_rn[buf];

    for (int ct = 0; ct < KM_TB / 32; ++ct) {
      bf16x8 afrag =
          *reinterpret_cast<const bf16x8*>(&rt[ct * 32 + l31][half * 8]);
      const float4* rn4 =
          reinterpret_cast<const float4*>(rn_t + ct * 32 + 4 * half);
      float4 rnv[2];
#pragma unroll
      for (int r4 = 0; r4 < 2; ++r4) rnv[r4] = rn4[2 * r4];
      float4 rnw[2];
#pragma unroll
      for (int r4 = 0; r4 < 2; ++r4) rnw[r4] = rn4[4 + 2 * r4];
      const float* rnf0 = reinterpret_cast<const float*>(rnv);
      const float* rnf1 = reinterpret_cast<const float*>(rnw);
#pragma unroll
      for (int qti = 0; qti < KM_NQT; ++qti) {
        f32x16 acc = {};
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(afrag, bfrag[qti], acc,
                                                      0, 0, 0);
        float kmin = fmaf(-2.f, acc[0], rnf0[0]);
#pragma unroll
        for (int g = 1; g < 8; ++g)
          kmin = fminf(kmin, fmaf(-2.f, acc[g], rnf0[g]));
#pragma unroll
        for (int g = 8; g < 16; ++g)
          kmin = fminf(kmin, fmaf(-2.f, acc[g], rnf1[g - 8]));
        if (kmin < wkey[qti]) {
#pragma unroll
          for (int g = 0; g < 16; ++g) {
            float key = fmaf(-2.f, acc[g], g < 8 ? rnf0[g] : rnf1[g - 8]);
            if (key < wkey[qti]) {
              const int cand = ct * 32 + km_rowmap(g, half);
              // pool is the FULL KM_KMAX regardless of runtime k: the
              // surplus is the refine stage's safety margin
              int ws = 0;
              float wv = -FLT_MAX;
#pragma unroll
              for (int j = 0; j < KM_KMAX; ++j)
                if (lk[qti][j] > wv) {
                  wv = lk[qti][j];
                  ws = j;
                }
              lk[qti][ws] = key;
              li[qti][ws] = (int)(tb + cand);
              wv = -FLT_MAX;
#pragma unroll
              for (int j = 0; j < KM_KMAX; ++j)
                if (lk[qti][j] > wv) wv = lk[qti][j];
              wkey[qti] = wv;
            }
          }
        }
      }
    }
  }
  __syncthreads();

  // tail: refine EVERY pooled candidate exactly, emit the exact-best k
#pragma unroll
  for (int qti = 0; qti < KM_NQT; ++qti) {
    const int qt = wave + 4 * qti;
    const long long q = qb0 + qt * 32 + l31;
    int oi[2 * KM_KMAX];
#pragma unroll
    for (int j = 0; j < KM_KMAX; ++j) {
      oi[j] = li[qti][j];
      oi[KM_KMAX + j] = __shfl(li[qti][j], lane ^ 32, WAVE);
    }
    if (half == 0 && q < nq) {
      Row12 xq = load_row12(Q, q);
      float dk[KM_KMAX];
      int ik[KM_KMAX];
#pragma unroll
      for (int j = 0; j < KM_KMAX; ++j) {
        dk[j] = FLT_MAX;
        ik[j] = -1;
      }
      for (int j = 0; j < 2 * KM_KMAX; ++j) {
        if (oi[j] < 0) continue;
        Row12 xr = load_row12(R, oi[j]);
        float dv = km_dist2(xq.v, xr.v);
        int iv = oi[j];
        if (dv > dk[k - 1] ||
            (dv == dk[k - 1] && ik[k - 1] >= 0 && (unsigned)iv > (unsigned)ik[k - 1]))
          continue;
        int b = k - 1;
        while (b > 0 && (dk[b - 1] > dv ||
                         (dk[b - 1] == dv && (unsigned)ik[b - 1] > (unsigned)iv))) {
          dk[b] = dk[b - 1];
          ik[b] = ik[b - 1];
          --b;
        }
        dk[b] = dv;
        ik[b] = iv;
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
                                long long idx_base, int approx, hipStream_t stream) {
  long long shard_rows = (nr + S - 1) / S;
  dim3 grid((unsigned)((nq + KM_QB - 1) / KM_QB), (unsigned)S);
  if (approx)
    hipLaunchKernelGGL(knn_mfma_bf16_kernel, grid, dim3(256), 0, stream, Q, R,
                       cmean, part_d, part_i, nq, nr, k, shard_rows);
  else
    hipLaunchKernelGGL(knn_mfma_kernel, grid, dim3(256), 0, stream, Q, R, cmean,
                       part_d, part_i, nq, nr, k, shard_rows);
  int block = 256;
  int mgrid = (int)((nq + block - 1) / block);
  hipLaunchKernelGGL(knn_merge_kernel, dim3(mgrid), dim3(block), 0, stream,
                     part_d, part_i, ry, out_d, out_i, out_lab, nq, S, k, C,
                     idx_base);
}
