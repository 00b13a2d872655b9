// CDNA4 predict kernels for the six estimators.
// Hand-written for gfx950: wave64 blocks, params staged in LDS, feature rows
// as float4 vector loads, grid-stride over rows.  All label outputs use
// first-maximum tie-breaking to match the sklearn semantics of the CPU
// oracles (ops/cpu.py).
//
// Register-pressure rule observed throughout: any per-lane array indexed by
// a RUNTIME value would be spilled to scratch by hipcc, so every such array
// (RF class accumulators, SVC OVO accumulators, KNN k-best lists) is sized
// by a template parameter and only indexed inside fully-unrolled loops; the
// per-row feature vector, whose index IS runtime (tree node feature ids),
// lives in LDS instead.

#include <cstdlib>

#include "common.h"

// ---------------------------------------------------------------------------
// Gaussian NB: fused joint-log-likelihood + argmax (reference N5).
//   score[c] = const[c] - 0.5 * sum_j (x_j - theta[c,j])^2 * inv_var[c,j]
// Params (C*F*2 + C floats) are broadcast via LDS.
// ---------------------------------------------------------------------------
__global__ void gnb_predict_kernel(const float* __restrict__ X,
                                   const float* __restrict__ theta,
                                   const float* __restrict__ inv_var,
                                   const float* __restrict__ cconst,
                                   int* __restrict__ out,
                                   long long n, int C) {
  constexpr int F = 12;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* s_theta = reinterpret_cast<float*>(smem);
  float* s_ivar = s_theta + C * F;
  float* s_const = s_ivar + C * F;
  for (int i = threadIdx.x; i < C * F; i += blockDim.x) {
    s_theta[i] = theta[i];
    s_ivar[i] = inv_var[i];
  }
  for (int i = threadIdx.x; i < C; i += blockDim.x) s_const[i] = cconst[i];
  __syncthreads();

  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long row = (long long)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += stride) {
    Row12 x = load_row12(X, row);
    float best = -INFINITY;
    int bi = 0;
    for (int c = 0; c < C; ++c) {
      float s = 0.f;
#pragma unroll
      for (int j = 0; j < F; ++j) {
        float d = x.v[j] - s_theta[c * F + j];
        s = fmaf(d * d, s_ivar[c * F + j], s);
      }
      s = s_const[c] - 0.5f * s;
      if (s > best) { best = s; bi = c; }
    }
    out[row] = bi;
  }
}

extern "C" void launch_gnb_predict(const float* X, const float* theta,
                                   const float* inv_var, const float* cconst,
                                   int* out, long long n, int C,
                                   hipStream_t stream) {
  const int block = 256;
  size_t lds = (size_t)(2 * C * 12 + C) * sizeof(float);
  hipLaunchKernelGGL(gnb_predict_kernel, dim3(ts_grid(n, block)), dim3(block),
                     lds, stream, X, theta, inv_var, cconst, out, n, C);
}

// ---------------------------------------------------------------------------
// Linear (logistic) argmax: out = argmax_c (W[c,:] . x + b[c])   (N1 serve)
// ---------------------------------------------------------------------------
__global__ void linear_argmax_kernel(const float* __restrict__ X,
                                     const float* __restrict__ W,
                                     const float* __restrict__ b,
                                     int* __restrict__ out,
                                     long long n, int C) {
  constexpr int F = 12;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* s_w = reinterpret_cast<float*>(smem);
  float* s_b = s_w + C * F;
  for (int i = threadIdx.x; i < C * F; i += blockDim.x) s_w[i] = W[i];
  for (int i = threadIdx.x; i < C; i += blockDim.x) s_b[i] = b[i];
  __syncthreads();

  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long row = (long long)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += stride) {
    Row12 x = load_row12(X, row);
    float best = -INFINITY;
    int bi = 0;
    for (int c = 0; c < C; ++c) {
      float s = s_b[c];
#pragma unroll
      for (int j = 0; j < F; ++j) s = fmaf(x.v[j], s_w[c * F + j], s);
      if (s > best) { best = s; bi = c; }
    }
    out[row] = bi;
  }
}

extern "C" void launch_linear_argmax(const float* X, const float* W,
                                     const float* b, int* out, long long n,
                                     int C, hipStream_t stream) {
  const int block = 256;
  size_t lds = (size_t)(C * 12 + C) * sizeof(float);
  hipLaunchKernelGGL(linear_argmax_kernel, dim3(ts_grid(n, block)), dim3(block),
                     lds, stream, X, W, b, out, n, C);
}

// ---------------------------------------------------------------------------
// KMeans assignment + partial update (N6): label = argmin_k ||x - c_k||^2,
// with per-block LDS partial (count, sum) per cluster flushed by f64 atomics
// (one atomic set per block, guide G12), plus inertia.
// ---------------------------------------------------------------------------
__global__ void kmeans_assign_kernel(const float* __restrict__ X,
                                     const float* __restrict__ centers,
                                     int* __restrict__ labels,
                                     double* __restrict__ counts,   // [K]
                                     double* __restrict__ sums,     // [K,F]
                                     double* __restrict__ inertia,  // [1]
                                     long long n, int K, int want_update) {
  constexpr int F = 12;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  double* s_sum = reinterpret_cast<double*>(smem);        // [K,F]
  double* s_cnt = s_sum + K * F;                          // [K]
  float* s_c = reinterpret_cast<float*>(s_cnt + K);       // [K,F]
  for (int i = threadIdx.x; i < K * F; i += blockDim.x) {
    s_c[i] = centers[i];
    if (want_update) s_sum[i] = 0.0;
  }
  if (want_update)
    for (int i = threadIdx.x; i < K; i += blockDim.x) s_cnt[i] = 0.0;
  __syncthreads();

  double local_inertia = 0.0;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long row = (long long)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += stride) {
    Row12 x = load_row12(X, row);
    float best = INFINITY;
    int bi = 0;
    for (int k = 0; k < K; ++k) {
      float d = 0.f;
#pragma unroll
      for (int j = 0; j < F; ++j) {
        float t = x.v[j] - s_c[k * F + j];
        d = fmaf(t, t, d);
      }
      if (d < best) { best = d; bi = k; }
    }
    labels[row] = bi;
    local_inertia += (double)best;
    if (want_update) {
      // LDS atomics: contention limited to the block's 256 lanes
      atomicAdd(&s_cnt[bi], 1.0);
#pragma unroll
      for (int j = 0; j < F; ++j) atomicAdd(&s_sum[bi * F + j], (double)x.v[j]);
    }
  }
  // inertia: wave-reduce then one atomic per wave
  local_inertia = wave_sum(local_inertia);
  if ((threadIdx.x & (WAVE - 1)) == 0) atomicAdd(inertia, local_inertia);
  if (want_update) {
    __syncthreads();
    for (int i = threadIdx.x; i < K * F; i += blockDim.x)
      atomicAdd(&sums[i], s_sum[i]);
    for (int i = threadIdx.x; i < K; i += blockDim.x)
      atomicAdd(&counts[i], s_cnt[i]);
  }
}

extern "C" void launch_kmeans_assign(const float* X, const float* centers,
                                     int* labels, double* counts, double* sums,
                                     double* inertia, long long n, int K,
                                     int want_update, hipStream_t stream) {
  const int block = 256;
  size_t lds = (size_t)(K * 12 + K) * sizeof(double) +
               (size_t)(K * 12) * sizeof(float);
  hipLaunchKernelGGL(kmeans_assign_kernel, dim3(ts_grid(n, block)), dim3(block),
                     lds, stream, X, centers, labels, counts, sums, inertia, n,
                     K, want_update);
}

// ---------------------------------------------------------------------------
// Random forest traversal + vote (N4) — the flagship predict kernel.
// Packed node = uint2 { x: f32 threshold bits | leaf-prob row index,
//                       y: (right_child_global << 8) | feature (0xff = leaf) }
// Design (profiled on MI355X): the traversal is a dependent chain of node
// fetches, so occupancy is the lever.  Only the packed forest lives in LDS
// (shared by the whole block); the per-lane feature row stays in REGISTERS
// and the runtime feature index is resolved with an 11-op cndmask select
// tree (compile-time indices, ~8 cycles — far shorter than an LDS
// round-trip and no scratch spill).  512-thread blocks with nodes-only LDS
// give 3 blocks = 24 waves per CU.
// ---------------------------------------------------------------------------

// select v[f] for runtime f in [0,12) with compile-time register indices:
// two levels — pick r = f&3 within each quad, then the quad q = f>>2.
DEV float sel12(const Row12& x, unsigned f) {
  unsigned r = f & 3u, q = f >> 2;
  float a = (r & 1u) ? x.v[1] : x.v[0];
  float b = (r & 2u) ? ((r & 1u) ? x.v[3] : x.v[2]) : a;
  float q0 = b;
  float c = (r & 1u) ? x.v[5] : x.v[4];
  float d = (r & 2u) ? ((r & 1u) ? x.v[7] : x.v[6]) : c;
  float q1 = d;
  float e = (r & 1u) ? x.v[9] : x.v[8];
  float g = (r & 2u) ? ((r & 1u) ? x.v[11] : x.v[10]) : e;
  float q2 = g;
  float lo = (q & 1u) ? q1 : q0;
  return (q & 2u) ? q2 : lo;
}

// Traversal body shared by the four table-placement variants.  LDS_NODES /
// LDS_PROBS are compile-time so the node fetch lowers to ds_read_b64 (LDS)
// or global_load_dwordx2 — a runtime ternary over the two pointers would
// force generic/flat addressing (measured: LDS instr count collapses and
// VALU doubles on 64-bit flat address math).
//
// Leaf encoding: most leaves of a fully-grown forest are PURE (one-hot
// distribution).  A pure leaf carries its class in the feature byte
// (0xf0|class) and is counted with one shift+add into a packed u64 vote
// register (10 bits per class, forests up to 1023 trees) — no memory read
// at all.  Only mixed leaves (feature byte 0xff) touch the probability
// table.
template <int C, bool LDS_NODES, bool LDS_PROBS, bool LDS_FEAT>
__launch_bounds__(512, 1)
__global__ void rf_predict_kernel(const float* __restrict__ X,
                                  const uint2* __restrict__ nodes,
                                  const int* __restrict__ roots,  // [T]
                                  const float* __restrict__ leaf_proba,
                                  int* __restrict__ out,
                                  long long n, int n_nodes, int n_leaves,
                                  int T) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint2* s_nodes = reinterpret_cast<uint2*>(smem);
  float* s_probs = reinterpret_cast<float*>(s_nodes + (LDS_NODES ? n_nodes : 0));
  // LDS_FEAT: each thread's 12 features live in LDS at pitch 13 (gcd(13,32)
  // = 1 -> the 32 lanes of a group land on distinct banks) and the per-node
  // runtime feature pick is ONE ds_read_b32 instead of sel12's ~10-op
  // register select tree — trading idle LDS-pipe cycles for the VALU issue
  // slots this kernel is bound on (VALUBusy 99.2%).
  float* s_feat = s_probs + (LDS_PROBS ? n_leaves * C : 0);

  if (LDS_NODES)
    for (int i = threadIdx.x; i < n_nodes; i += blockDim.x) s_nodes[i] = nodes[i];
  if (LDS_PROBS)
    for (int i = threadIdx.x; i < n_leaves * C; i += blockDim.x)
      s_probs[i] = leaf_proba[i];
  if (LDS_NODES || LDS_PROBS) __syncthreads();
  float* my_feat = s_feat + (int)threadIdx.x * 13;

  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long row = (long long)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += stride) {
    Row12 x = load_row12(X, row);
    if (LDS_FEAT) {
#pragma unroll
      for (int j = 0; j < 12; ++j) my_feat[j] = x.v[j];
    }
    float acc[C];
#pragma unroll
    for (int c = 0; c < C; ++c) acc[c] = 0.f;
    unsigned long long votes = 0ull;  // 10-bit packed per-class pure counts
    int decided = 0;
    for (int t = 0; t < T; ++t) {
      if (!decided) {
        int idx = roots[t];
        while (true) {
          uint2 node = LDS_NODES ? s_nodes[idx] : nodes[idx];
          unsigned feat = node.y & 0xffu;
          if (feat >= 0xf0u) {
            if (feat == 0xffu) {  // mixed leaf: read the distribution
              int pr = (int)node.x * C;
#pragma unroll
              for (int c = 0; c < C; ++c)
                acc[c] += LDS_PROBS ? s_probs[pr + c] : leaf_proba[pr + c];
            } else {  // pure leaf: class in the low nibble
              votes += 1ull << ((feat & 0xfu) * 10);
            }
            break;
          }
          float thr = __uint_as_float(node.x);
          float fv = LDS_FEAT ? my_feat[feat] : sel12(x, feat);
          idx = (fv <= thr) ? idx + 1 : (int)(node.y >> 8);
        }
      }
      // EXACT early majority exit: each remaining tree adds at most 1.0 to
      // any class score, so once leader - runner-up > trees left the argmax
      // is decided.  The kernel is VALU-bound and most rows of an accurate
      // forest are near-unanimous, so whole waves retire around tree T/2.
      if ((t & 7) == 7) {
        if (!decided) {
          float m1 = -INFINITY, m2 = -INFINITY;
#pragma unroll
          for (int c = 0; c < C; ++c) {
            float sc = acc[c] + (float)((votes >> (c * 10)) & 1023ull);
            if (sc > m1) { m2 = m1; m1 = sc; }
            else if (sc > m2) m2 = sc;
          }
          if (m1 - m2 > (float)(T - 1 - t)) decided = 1;
        }
        if (__all(decided)) break;  // work only stops wave-uniformly
      }
    }
    float best = -INFINITY;
    int bi = 0;
#pragma unroll
    for (int c = 0; c < C; ++c) {
      float sc = acc[c] + (float)((votes >> (c * 10)) & 1023ull);
      if (sc > best) { best = sc; bi = c; }
    }
    out[row] = bi;
  }
}

// ILP-2 variant (mode 3): two independent traversal chains per lane.  At
// mode 2 the kernel is latency-bound (VALUBusy 57.5%, SQ_WAIT_ANY 68% of
// wave cycles parked on the L2 node fetch + LDS feature read chains) and
// the 32-waves/CU cap is already reached — more memory-level parallelism
// has to come from within the wave.  256-thread blocks keep the two
// pitch-13 feature slabs at the same 26.6 KB the 512-thread single-row
// variant used, so occupancy is unchanged and each CU runs 2x the chains.
template <int C>
__launch_bounds__(256, 1)
__global__ void rf_predict_ilp2_kernel(const float* __restrict__ X,
                                       const uint2* __restrict__ nodes,
                                       const int* __restrict__ roots,
                                       const float* __restrict__ leaf_proba,
                                       int* __restrict__ out, long long n,
                                       int T) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* s_feat = reinterpret_cast<float*>(smem);
  float* fA = s_feat + (int)threadIdx.x * 13;
  float* fB = s_feat + (int)(blockDim.x + threadIdx.x) * 13;

  long long stride = (long long)gridDim.x * blockDim.x * 2;
  for (long long base = (long long)blockIdx.x * blockDim.x * 2; base < n;
       base += stride) {
    const long long rowA = base + threadIdx.x;
    const long long rowB = base + blockDim.x + threadIdx.x;
    if (rowA < n) {
      Row12 x = load_row12(X, rowA);
#pragma unroll
      for (int j = 0; j < 12; ++j) fA[j] = x.v[j];
    }
    if (rowB < n) {
      Row12 x = load_row12(X, rowB);
#pragma unroll
      for (int j = 0; j < 12; ++j) fB[j] = x.v[j];
    }
    float accA[C], accB[C];
#pragma unroll
    for (int c = 0; c < C; ++c) accA[c] = accB[c] = 0.f;
    unsigned long long votesA = 0ull, votesB = 0ull;
    int decidedA = rowA >= n, decidedB = rowB >= n;
    for (int t = 0; t < T; ++t) {
      int idxA = roots[t], idxB = idxA;
      bool runA = !decidedA, runB = !decidedB;
      while (runA || runB) {
        uint2 nodeA, nodeB;
        if (runA) nodeA = nodes[idxA];
        if (runB) nodeB = nodes[idxB];
        if (runA) {
          unsigned feat = nodeA.y & 0xffu;
          if (feat >= 0xf0u) {
            if (feat == 0xffu) {
              int pr = (int)nodeA.x * C;
#pragma unroll
              for (int c = 0; c < C; ++c) accA[c] += leaf_proba[pr + c];
            } else {
              votesA += 1ull << ((feat & 0xfu) * 10);
            }
            runA = false;
          } else {
            float thr = __uint_as_float(nodeA.x);
            idxA = (fA[feat] <= thr) ? idxA + 1 : (int)(nodeA.y >> 8);
          }
        }
        if (runB) {
          unsigned feat = nodeB.y & 0xffu;
          if (feat >= 0xf0u) {
            if (feat == 0xffu) {
              int pr = (int)nodeB.x * C;
#pragma unroll
              for (int c = 0; c < C; ++c) accB[c] += leaf_proba[pr + c];
            } else {
              votesB += 1ull << ((feat & 0xfu) * 10);
            }
            runB = false;
          } else {
            float thr = __uint_as_float(nodeB.x);
            idxB = (fB[feat] <= thr) ? idxB + 1 : (int)(nodeB.y >> 8);
          }
        }
      }
      if ((t & 7) == 7) {
        if (!decidedA) {
          float m1 = -INFINITY, m2 = -INFINITY;
#pragma unroll
          for (int c = 0; c < C; ++c) {
            float sc = accA[c] + (float)((votesA >> (c * 10)) & 1023ull);
            if (sc > m1) { m2 = m1; m1 = sc; }
            else if (sc > m2) m2 = sc;
          }
          if (m1 - m2 > (float)(T - 1 - t)) decidedA = 1;
        }
        if (!decidedB) {
          float m1 = -INFINITY, m2 = -INFINITY;
#pragma unroll
          for (int c = 0; c < C; ++c) {
            float sc = accB[c] + (float)((votesB >> (c * 10)) & 1023ull);
            if (sc > m1) { m2 = m1; m1 = sc; }
            else if (sc > m2) m2 = sc;
          }
          if (m1 - m2 > (float)(T - 1 - t)) decidedB = 1;
        }
        if (__all(decidedA && decidedB)) break;
      }
    }
    if (rowA < n) {
      float best = -INFINITY;
      int bi = 0;
#pragma unroll
      for (int c = 0; c < C; ++c) {
        float sc = accA[c] + (float)((votesA >> (c * 10)) & 1023ull);
        if (sc > best) { best = sc; bi = c; }
      }
      out[rowA] = bi;
    }
    if (rowB < n) {
      float best = -INFINITY;
      int bi = 0;
#pragma unroll
      for (int c = 0; c < C; ++c) {
        float sc = accB[c] + (float)((votesB >> (c * 10)) & 1023ull);
        if (sc > best) { best = sc; bi = c; }
      }
      out[rowB] = bi;
    }
  }
}

// Small-batch variant: ONE WAVE PER ROW, trees split across lanes.  At
// serve batch sizes the row-per-lane kernel runs a ~900-step serial
// dependent-load chain per row on a mostly idle chip; here each lane walks
// only ceil(T/64) trees and n waves fill the SIMDs.  Nodes are read from
// global memory — the packed forest (tens of KB) stays L2-resident.
template <int C>
__launch_bounds__(256) __global__ void rf_predict_wave_kernel(
    const float* __restrict__ X, const uint2* __restrict__ nodes,
    const int* __restrict__ roots, const float* __restrict__ leaf_proba,
    int* __restrict__ out, long long n, int T) {
  const int lane = threadIdx.x & (WAVE - 1);
  const long long row = (long long)blockIdx.x * (blockDim.x / WAVE) +
                        (threadIdx.x >> 6);
  if (row >= n) return;
  Row12 x = load_row12(X, row);  // broadcast load
  float acc[C];
#pragma unroll
  for (int c = 0; c < C; ++c) acc[c] = 0.f;
  unsigned long long votes = 0ull;  // 10-bit packed per-class pure counts
  for (int t = lane; t < T; t += WAVE) {
    int idx = roots[t];
    while (true) {
      uint2 node = nodes[idx];
      unsigned feat = node.y & 0xffu;
      if (feat >= 0xf0u) {
        if (feat == 0xffu) {
          int pr = (int)node.x * C;
#pragma unroll
          for (int c = 0; c < C; ++c) acc[c] += leaf_proba[pr + c];
        } else {
          votes += 1ull << ((feat & 0xfu) * 10);
        }
        break;
      }
      float thr = __uint_as_float(node.x);
      idx = (sel12(x, feat) <= thr) ? idx + 1 : (int)(node.y >> 8);
    }
  }
  votes = wave_sum(votes);  // per-field sums stay < 1024: no cross-field carry
#pragma unroll
  for (int c = 0; c < C; ++c) acc[c] = wave_sum(acc[c]);
  if (lane == 0) {
    float best = -INFINITY;
    int bi = 0;
#pragma unroll
    for (int c = 0; c < C; ++c) {
      float sc = acc[c] + (float)((votes >> (c * 10)) & 1023ull);
      if (sc > best) { best = sc; bi = c; }
    }
    out[row] = bi;
  }
}

extern "C" void launch_rf_predict(const float* X, const unsigned* nodes,
                                  const int* roots, const float* leaf_proba,
                                  int* out, long long n, int n_nodes,
                                  int n_leaves, int T, int C,
                                  hipStream_t stream) {
  if (n <= 131072) {  // wave-per-row fills the chip at serve batch sizes
    dim3 wgrid((unsigned)((n + 3) / 4));
#define RFW_CASE(CV)                                                        \
  case CV:                                                                  \
    hipLaunchKernelGGL((rf_predict_wave_kernel<CV>), wgrid, dim3(256), 0,   \
                       stream, X, reinterpret_cast<const uint2*>(nodes),    \
                       roots, leaf_proba, out, n, T);                       \
    return;
    switch (C) {
      RFW_CASE(2) RFW_CASE(3) RFW_CASE(4) RFW_CASE(5) RFW_CASE(6) RFW_CASE(7)
      RFW_CASE(8)
      default: break;
    }
#undef RFW_CASE
  }
  const int block = 512;
  // Occupancy first: with nodes-only LDS a 43 KB forest admits 3 blocks
  // (24 waves) per CU.  Leaf probabilities go to LDS only when everything
  // still fits the 3-block budget; big forests fall back to the
  // (L2-resident) global tables.
  //
  // Feature placement (TCSDN_RF_MODE, A/B'd on hardware — 100-tree
  // reference forest, 10M rows: mode 0 = 2.09 Gflows/s, mode 1 = 2.12,
  // mode 2 = 2.60): mode 0 keeps features in registers (sel12 select tree,
  // nodes LDS when they fit); mode 1 also stages features in LDS (pitch 13,
  // one ds_read per node) but loses a block of occupancy next to a big
  // forest; mode 2 — THE DEFAULT — stages features in LDS and leaves the
  // node table in L2 (42 KB forest ≪ 4 MB per-XCD L2; the ~200-cycle node
  // fetch hides behind 32 waves/CU while the VALU sheds the ~10-op select
  // tree this kernel was issue-bound on).
  int mode = 2;
  if (const char* e = getenv("TCSDN_RF_MODE")) mode = atoi(e);
  if (mode == 3) {  // ILP-2: two chains per lane, 256-thread blocks
    const int b2 = 256;
    size_t lds2 = (size_t)b2 * 2 * 13 * sizeof(float);
    dim3 g2(ts_grid((n + 1) / 2, b2, 4096));
#define RFI_CASE(CV)                                                         \
  case CV:                                                                   \
    hipLaunchKernelGGL((rf_predict_ilp2_kernel<CV>), g2, dim3(b2), lds2,     \
                       stream, X, reinterpret_cast<const uint2*>(nodes),     \
                       roots, leaf_proba, out, n, T);                        \
    return;
    switch (C) {
      RFI_CASE(2) RFI_CASE(3) RFI_CASE(4) RFI_CASE(5) RFI_CASE(6) RFI_CASE(7)
      RFI_CASE(8) RFI_CASE(12) RFI_CASE(16)
      default: break;
    }
#undef RFI_CASE
  }
  size_t node_bytes = (size_t)n_nodes * sizeof(uint2);
  size_t prob_bytes = (size_t)n_leaves * C * sizeof(float);
  size_t feat_bytes = (size_t)block * 13 * sizeof(float);
  size_t budget = 52 * 1024;  // 3 blocks/CU floor
  bool lds_feat = mode != 0;
  bool lds_nodes = mode != 2 && node_bytes + (lds_feat ? feat_bytes : 0) <=
                                    (mode == 1 ? (size_t)76 * 1024 : budget);
  bool lds_probs = lds_nodes && !lds_feat && (node_bytes + prob_bytes) <= budget;
  size_t lds = (lds_nodes ? node_bytes : 0) + (lds_probs ? prob_bytes : 0) +
               (lds_feat ? feat_bytes : 0);
  dim3 grid(ts_grid(n, block, 4096));
#define RF_LAUNCH(CV, LN, LP, LF)                                            \
  hipLaunchKernelGGL((rf_predict_kernel<CV, LN, LP, LF>), grid, dim3(block), \
                     lds, stream, X, reinterpret_cast<const uint2*>(nodes),  \
                     roots, leaf_proba, out, n, n_nodes, n_leaves, T)
#define RF_CASE(CV)                                                          \
  case CV:                                                                   \
    if (lds_feat) {                                                          \
      if (lds_nodes) RF_LAUNCH(CV, true, false, true);                       \
      else RF_LAUNCH(CV, false, false, true);                                \
    } else if (lds_nodes && lds_probs) RF_LAUNCH(CV, true, true, false);     \
    else if (lds_nodes) RF_LAUNCH(CV, true, false, false);                   \
    else RF_LAUNCH(CV, false, false, false);                                 \
    return;
  switch (C) {
    RF_CASE(2) RF_CASE(3) RF_CASE(4) RF_CASE(5) RF_CASE(6) RF_CASE(7)
    RF_CASE(8) RF_CASE(12) RF_CASE(16)
    default: break;
  }
#undef RF_CASE
#undef RF_LAUNCH
}

// ---------------------------------------------------------------------------
// SVC RBF + one-vs-one vote (N2).  SV tiles (features, libsvm dual rows,
// class ids) stream through LDS; each lane keeps C*(C-1) accumulators
//   acc[c][r] = sum over SVs of class c of dual[r][sv] * exp(-gamma*d2(x,sv))
// then dec(i<j) = acc[i][j-1] + acc[j][i] + b[pair], majority vote.
// The per-SV class branch is wave-UNIFORM (the sv index is uniform across
// the block), so only the matching unrolled branch executes and all
// accumulator indices stay compile-time.
// ---------------------------------------------------------------------------
#define SVC_TILE 128

template <int C>
__global__ void svc_predict_kernel(const float* __restrict__ X,
                                   const float* __restrict__ SV,    // [nsv,F]
                                   const float* __restrict__ dual,  // [C-1,nsv]
                                   const unsigned char* __restrict__ svclass,
                                   const float* __restrict__ intercept,
                                   int* __restrict__ out,
                                   long long n, int nsv, float gamma) {
  constexpr int F = 12;
  constexpr int CR = C - 1;
  constexpr int NPAIR = C * (C - 1) / 2;
  __shared__ __attribute__((aligned(16))) float s_sv[SVC_TILE * F];
  __shared__ float s_dual[CR * SVC_TILE];
  __shared__ unsigned char s_cls[SVC_TILE];
  __shared__ float s_b[NPAIR];

  for (int i = threadIdx.x; i < NPAIR; i += blockDim.x) s_b[i] = intercept[i];

  long long stride = (long long)gridDim.x * blockDim.x;
  // lock-step row tiles so the SV staging loop stays uniform per block
  for (long long base = (long long)blockIdx.x * blockDim.x; base < n;
       base += stride) {
    long long row = base + threadIdx.x;
    Row12 x;
    if (row < n) x = load_row12(X, row);
    // f64 running accumulators + per-tile f32 partials: a single f32 chain
    // over tens of thousands of SIGNED near-unit dual terms (thousands of
    // alphas at the C bound in a non-separable fit) loses the O(1) decision
    // value to rounding (measured: 24% wrong votes at 35K SVs); a 128-term
    // f32 partial folded into f64 per tile is exact to ~1e-5 at any nsv
    double acc[C * CR];
#pragma unroll
    for (int i = 0; i < C * CR; ++i) acc[i] = 0.0;

    for (int tile = 0; tile < nsv; tile += SVC_TILE) {
      int cnt = min(SVC_TILE, nsv - tile);
      __syncthreads();
      for (int i = threadIdx.x; i < cnt * F; i += blockDim.x)
        s_sv[i] = SV[(long long)tile * F + i];
#pragma unroll
      for (int r = 0; r < CR; ++r)
        for (int i = threadIdx.x; i < cnt; i += blockDim.x)
          s_dual[r * SVC_TILE + i] = dual[(long long)r * nsv + tile + i];
      for (int i = threadIdx.x; i < cnt; i += blockDim.x)
        s_cls[i] = svclass[tile + i];
      __syncthreads();
      if (row < n) {
        float accT[C * CR];
#pragma unroll
        for (int i = 0; i < C * CR; ++i) accT[i] = 0.f;
        for (int s = 0; s < cnt; ++s) {
          float d = 0.f;
#pragma unroll
          for (int j = 0; j < F; ++j) {
            float t = x.v[j] - s_sv[s * F + j];
            d = fmaf(t, t, d);
          }
          float kv = __expf(-gamma * d);
          int c = s_cls[s];  // wave-uniform
#pragma unroll
          for (int cc = 0; cc < C; ++cc) {
            if (c == cc) {
#pragma unroll
              for (int r = 0; r < CR; ++r)
                accT[cc * CR + r] =
                    fmaf(s_dual[r * SVC_TILE + s], kv, accT[cc * CR + r]);
            }
          }
        }
#pragma unroll
        for (int i = 0; i < C * CR; ++i) acc[i] += (double)accT[i];
      }
    }
    if (row >= n) continue;
    // OVO vote (first max wins ties, libsvm semantics)
    int votes[C];
#pragma unroll
    for (int c = 0; c < C; ++c) votes[c] = 0;
    int p = 0;
#pragma unroll
    for (int i = 0; i < C; ++i)
#pragma unroll
      for (int j = i + 1; j < C; ++j, ++p) {
        double dec = acc[i * CR + (j - 1)] + acc[j * CR + i] + (double)s_b[p];
        if (dec > 0.0) votes[i]++; else votes[j]++;
      }
    int best = -1, bi = 0;
#pragma unroll
    for (int c = 0; c < C; ++c)
      if (votes[c] > best) { best = votes[c]; bi = c; }
    out[row] = bi;
  }
}

// Small-batch variant: ONE WAVE PER ROW, SVs split across lanes.  At serve
// batch sizes (thousands of rows) the row-per-lane kernel leaves most of
// the chip idle and each lane walks all nsv SVs serially; here n waves fill
// the 1024 SIMDs and the per-row latency drops ~100x (serve-path profile:
// profiles/serve_kernel_trace_r01.md).  SV rows stream from global memory —
// the working set (nsv * 69 B) stays L2-resident and is shared by every
// wave.  The per-SV class is NOT wave-uniform here, so the OVO accumulator
// update is an unrolled compile-time switch on the class id.
template <int C>
__launch_bounds__(256) __global__ void svc_predict_wave_kernel(
    const float* __restrict__ X, const float* __restrict__ SV,
    const float* __restrict__ dual, const unsigned char* __restrict__ svclass,
    const float* __restrict__ intercept, int* __restrict__ out, long long n,
    int nsv, float gamma) {
  constexpr int F = 12;
  constexpr int CR = C - 1;
  constexpr int NPAIR = C * (C - 1) / 2;
  const int lane = threadIdx.x & (WAVE - 1);
  const long long row = (long long)blockIdx.x * (blockDim.x / WAVE) +
                        (threadIdx.x >> 6);
  if (row >= n) return;
  Row12 x = load_row12(X, row);  // broadcast load, all lanes same row
  // f64 per-lane accumulators: the 64-way split already conditions the sum,
  // but signed bounded-alpha duals at large nsv still overwhelm f32 (same
  // hazard as the tiled kernel — see comment there)
  double acc[C * CR];
#pragma unroll
  for (int i = 0; i < C * CR; ++i) acc[i] = 0.0;
  for (int s = lane; s < nsv; s += WAVE) {
    Row12 sv = load_row12(SV, s);
    float d = 0.f;
#pragma unroll
    for (int j = 0; j < F; ++j) {
      float t = x.v[j] - sv.v[j];
      d = fmaf(t, t, d);
    }
    float kv = __expf(-gamma * d);
    int c = svclass[s];
#pragma unroll
    for (int cc = 0; cc < C; ++cc) {
      if (c == cc) {
#pragma unroll
        for (int r = 0; r < CR; ++r)
          acc[cc * CR + r] += (double)(dual[(long long)r * nsv + s] * kv);
      }
    }
  }
  // cross-lane reduction of the C*(C-1) partial sums
#pragma unroll
  for (int i = 0; i < C * CR; ++i) acc[i] = wave_sum(acc[i]);
  if (lane == 0) {
    int votes[C];
#pragma unroll
    for (int c = 0; c < C; ++c) votes[c] = 0;
    int p = 0;
#pragma unroll
    for (int i = 0; i < C; ++i)
#pragma unroll
      for (int j = i + 1; j < C; ++j, ++p) {
        double dec = acc[i * CR + (j - 1)] + acc[j * CR + i] + (double)intercept[p];
        if (dec > 0.0) votes[i]++; else votes[j]++;
      }
    int best = -1, bi = 0;
#pragma unroll
    for (int c = 0; c < C; ++c)
      if (votes[c] > best) { best = votes[c]; bi = c; }
    out[row] = bi;
    (void)NPAIR;
  }
}

extern "C" void launch_svc_predict(const float* X, const float* SV,
                                   const float* dual,
                                   const unsigned char* svclass,
                                   const float* intercept, int* out,
                                   long long n, int nsv, int C, float gamma,
                                   hipStream_t stream) {
  const int block = 256;
  // crossover re-measured in round 2 after the f64-accumulator fix: at 65K
  // rows the wave-per-row kernel beats the tiled one (serve 34.5 -> 39.6M
  // flows/s), so the cutover moved 32768 -> 131072 (same for RF above)
  if (n <= 131072) {  // wave-per-row fills the chip at serve batch sizes
    dim3 wgrid((unsigned)((n + 3) / 4));
#define SVC_WCASE(CV)                                                       \
  case CV:                                                                  \
    hipLaunchKernelGGL((svc_predict_wave_kernel<CV>), wgrid, dim3(block),   \
                       0, stream, X, SV, dual, svclass, intercept, out, n,  \
                       nsv, gamma);                                         \
    return;
    switch (C) {
      SVC_WCASE(2) SVC_WCASE(3) SVC_WCASE(4) SVC_WCASE(5) SVC_WCASE(6)
      default: break;
    }
#undef SVC_WCASE
  }
  dim3 grid(ts_grid(n, block));
#define SVC_CASE(CV)                                                        \
  case CV:                                                                  \
    hipLaunchKernelGGL((svc_predict_kernel<CV>), grid, dim3(block), 0,      \
                       stream, X, SV, dual, svclass, intercept, out, n,     \
                       nsv, gamma);                                         \
    return;
  switch (C) {
    SVC_CASE(2) SVC_CASE(3) SVC_CASE(4) SVC_CASE(5) SVC_CASE(6)
    default: break;
  }
#undef SVC_CASE
}

// ---------------------------------------------------------------------------
// KNN brute-force top-k + fused vote (N3).  One lane per query; reference
// rows stream through LDS in tiles shared by the block; each lane keeps a
// sorted k-best (dist, idx) list in registers (K compile-time; ties keep the
// lower reference index, matching sklearn ordering).  idx_base offsets
// emitted indices for sharded reference sets (all-gather merge keys stay
// global).
// ---------------------------------------------------------------------------
#define KNN_TILE 256
#define KNN_MAXC 8

template <int K>
__global__ void knn_topk_kernel(const float* __restrict__ Q,
                                const float* __restrict__ R,
                                const unsigned char* __restrict__ ry,  // may be null
                                float* __restrict__ out_d,  // [nq,K]
                                int* __restrict__ out_i,    // [nq,K]
                                int* __restrict__ out_lab,  // [nq] or null
                                long long nq, long long nr, int C,
                                long long idx_base) {
  constexpr int F = 12;
  __shared__ __attribute__((aligned(16))) float s_r[KNN_TILE * F];
  __shared__ unsigned char s_y[KNN_TILE];

  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long base = (long long)blockIdx.x * blockDim.x; base < nq;
       base += stride) {
    long long q = base + threadIdx.x;
    Row12 x;
    if (q < nq) x = load_row12(Q, q);
    float bd[K];
    int bi_[K];
    int bl[K];
#pragma unroll
    for (int k = 0; k < K; ++k) { bd[k] = INFINITY; bi_[k] = -1; bl[k] = 0; }

    for (long long tile = 0; tile < nr; tile += KNN_TILE) {
      int cnt = (int)min((long long)KNN_TILE, nr - tile);
      __syncthreads();
      for (int i = threadIdx.x; i < cnt * F; i += blockDim.x)
        s_r[i] = R[tile * F + i];
      if (ry)
        for (int i = threadIdx.x; i < cnt; i += blockDim.x) s_y[i] = ry[tile + i];
      __syncthreads();
      if (q >= nq) continue;
      // branchless sorted insert: slot k takes bd[k-1] when the new element
      // lands above it, or the new element when it lands here; strict
      // compares keep earlier (lower) reference indices on ties.
      auto insert = [&](float d, int ins, int lab) {
#pragma unroll
        for (int k = K - 1; k > 0; --k) {
          bool above = bd[k - 1] > d;  // new element goes before slot k-1
          if (bd[k] > d) {
            bd[k] = above ? bd[k - 1] : d;
            bi_[k] = above ? bi_[k - 1] : ins;
            bl[k] = above ? bl[k - 1] : lab;
          }
        }
        if (bd[0] > d) { bd[0] = d; bi_[0] = ins; bl[0] = lab; }
      };
      // two candidates per iteration, two split accumulators each: four
      // independent FMA chains in flight hide VALU and LDS latency (a
      // single fmaf chain leaves the SIMD idle between dependent ops)
      auto dist2 = [&](const float* __restrict__ r) {
        float a = 0.f, b = 0.f;
#pragma unroll
        for (int j = 0; j < 6; ++j) {
          float t = x.v[j] - r[j];
          a = fmaf(t, t, a);
        }
#pragma unroll
        for (int j = 6; j < F; ++j) {
          float t = x.v[j] - r[j];
          b = fmaf(t, t, b);
        }
        return a + b;
      };
      int s = 0;
      for (; s + 2 <= cnt; s += 2) {
        float d0 = dist2(&s_r[s * F]);
        float d1 = dist2(&s_r[(s + 1) * F]);
        if (d0 < bd[K - 1]) insert(d0, (int)(tile + s), ry ? (int)s_y[s] : 0);
        if (d1 < bd[K - 1]) insert(d1, (int)(tile + s + 1), ry ? (int)s_y[s + 1] : 0);
      }
      if (s < cnt) {
        float d0 = dist2(&s_r[s * F]);
        if (d0 < bd[K - 1]) insert(d0, (int)(tile + s), ry ? (int)s_y[s] : 0);
      }
    }
    if (q >= nq) continue;
#pragma unroll
    for (int k = 0; k < K; ++k) {
      out_d[q * K + k] = bd[k];
      out_i[q * K + k] = bi_[k] >= 0 ? (int)(bi_[k] + idx_base) : -1;
    }
    if (out_lab) {
      int best = 0, bc = 0;
      for (int c = 0; c < C; ++c) {
        int v = 0;
#pragma unroll
        for (int k = 0; k < K; ++k) v += (bl[k] == c && bi_[k] >= 0) ? 1 : 0;
        if (v > best) { best = v; bc = c; }
      }
      out_lab[q] = bc;
    }
  }
}

extern "C" void launch_knn_topk(const float* Q, const float* R,
                                const unsigned char* ry, float* out_d,
                                int* out_i, int* out_lab, long long nq,
                                long long nr, int k, int C, long long idx_base,
                                hipStream_t stream) {
  const int block = 256;
  dim3 grid(ts_grid(nq, block));
#define KNN_CASE(KV)                                                        \
  case KV:                                                                  \
    hipLaunchKernelGGL((knn_topk_kernel<KV>), grid, dim3(block), 0, stream, \
                       Q, R, ry, out_d, out_i, out_lab, nq, nr, C, idx_base); \
    return;
  switch (k) {
    KNN_CASE(1) KNN_CASE(2) KNN_CASE(3) KNN_CASE(4) KNN_CASE(5) KNN_CASE(6)
    KNN_CASE(7) KNN_CASE(8) KNN_CASE(16) KNN_CASE(32)
    default: break;
  }
#undef KNN_CASE
}
