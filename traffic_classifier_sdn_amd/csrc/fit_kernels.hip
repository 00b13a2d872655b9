// CDNA4 fit-path kernels: sufficient statistics and fused loss/gradient
// (SURVEY.md §2.2 N1/N5/N6 fit columns).  f64 accumulation throughout — the
// host-side L-BFGS / closed-form updates match the f64 CPU oracles, and the
// resulting buffers are exactly what gets RCCL-all-reduced per step.

#include <cfloat>

#include "common.h"

// ---------------------------------------------------------------------------
// GaussianNB sufficient stats: per-class (count, sum, sum-of-squares).
// Per-block LDS accumulation (f64), one global atomic sweep per block.
// ---------------------------------------------------------------------------
__global__ void gnb_fit_stats_kernel(const double* __restrict__ X,
                                     const long long* __restrict__ y,
                                     double* __restrict__ count,  // [C]
                                     double* __restrict__ sum,    // [C,F]
                                     double* __restrict__ sumsq,  // [C,F]
                                     long long n, int C) {
  constexpr int F = 12;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  double* s_sum = reinterpret_cast<double*>(smem);  // [C,F]
  double* s_sq = s_sum + C * F;                     // [C,F]
  double* s_cnt = s_sq + C * F;                     // [C]
  for (int i = threadIdx.x; i < C * F; i += blockDim.x) {
    s_sum[i] = 0.0;
    s_sq[i] = 0.0;
  }
  for (int i = threadIdx.x; i < C; i += blockDim.x) s_cnt[i] = 0.0;
  __syncthreads();

  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long row = (long long)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += stride) {
    int c = (int)y[row];
    atomicAdd(&s_cnt[c], 1.0);
#pragma unroll
    for (int j = 0; j < F; ++j) {
      double v = X[row * F + j];
      atomicAdd(&s_sum[c * F + j], v);
      atomicAdd(&s_sq[c * F + j], v * v);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < C * F; i += blockDim.x) {
    atomicAdd(&sum[i], s_sum[i]);
    atomicAdd(&sumsq[i], s_sq[i]);
  }
  for (int i = threadIdx.x; i < C; i += blockDim.x) atomicAdd(&count[i], s_cnt[i]);
}

extern "C" void launch_gnb_fit_stats(const double* X, const long long* y,
                                     double* count, double* sum, double* sumsq,
                                     long long n, int C, hipStream_t stream) {
  const int block = 256;
  size_t lds = (size_t)(2 * C * 12 + C) * sizeof(double);
  hipLaunchKernelGGL(gnb_fit_stats_kernel, dim3(ts_grid(n, block)), dim3(block),
                     lds, stream, X, y, count, sum, sumsq, n, C);
}

// ---------------------------------------------------------------------------
// Fused multinomial-logistic loss + gradient (the lbfgs objective):
//   p = softmax(X W^T + b);  loss += -log p[y];  G_w += (p - onehot_y)^T X
// Per-row softmax in registers (C <= 16), per-block LDS gradient [C, F+1]
// f64, one global atomic sweep per block.  The l2 term is applied on host.
// ---------------------------------------------------------------------------
// Accumulation strategy (round-2 rewrite): the first version did C*(F+1)
// (=78 at C=6) LDS f64 atomicAdds PER ROW, and within a wave all 64 lanes
// hit the SAME 78 addresses — a 64-way serialized conflict chain that
// measured 65% VALUBusy / ~1 ms per 1M-row pass (profiles/
// pmc_counters_r02.md).  Now each entry of the (p - onehot) ⊗ [x,1] outer
// product is wave-reduced in registers (__shfl_down tree, guide G12:
// per-wave partial reduction first) and lane 0 adds it to a PER-WAVE LDS
// slice — zero atomics in the hot loop, one global atomic sweep per block.
template <int C>
__global__ void logistic_grad_kernel(const double* __restrict__ X,
                                     const long long* __restrict__ y,
                                     const double* __restrict__ W,  // [C,F]
                                     const double* __restrict__ b,  // [C]
                                     double* __restrict__ grad,  // [C,F+1]
                                     double* __restrict__ loss,  // [1]
                                     long long n) {
  constexpr int F = 12;
  constexpr int NG = C * (F + 1);
  extern __shared__ __attribute__((aligned(16))) char smem[];
  double* s_w = reinterpret_cast<double*>(smem);  // [C,F]
  double* s_b = s_w + C * F;                      // [C]
  double* s_g = s_b + C;                          // [nwaves][NG]
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int nwaves = blockDim.x / WAVE;
  for (int i = threadIdx.x; i < C * F; i += blockDim.x) s_w[i] = W[i];
  for (int i = threadIdx.x; i < C; i += blockDim.x) s_b[i] = b[i];
  for (int i = threadIdx.x; i < nwaves * NG; i += blockDim.x) s_g[i] = 0.0;
  __syncthreads();

  double local_loss = 0.0;
  double* gw = s_g + wid * NG;
  long long stride = (long long)gridDim.x * blockDim.x;
  // uniform trip count per block: every lane reaches the wave_sum calls
  for (long long base = (long long)blockIdx.x * blockDim.x; base < n;
       base += stride) {
    long long row = base + threadIdx.x;
    bool valid = row < n;
    double xv[F];
    double pv[C];
#pragma unroll
    for (int j = 0; j < F; ++j) xv[j] = 0.0;
#pragma unroll
    for (int c = 0; c < C; ++c) pv[c] = 0.0;
    if (valid) {
#pragma unroll
      for (int j = 0; j < F; ++j) xv[j] = X[row * F + j];
      double logits[C];
      double m = -INFINITY;
#pragma unroll
      for (int c = 0; c < C; ++c) {
        double s = s_b[c];
#pragma unroll
        for (int j = 0; j < F; ++j) s += xv[j] * s_w[c * F + j];
        logits[c] = s;
        m = fmax(m, s);
      }
      int yc = (int)y[row];
      double z = 0.0;
      // keep logit_y - m BEFORE the exp: log(exp(t)) underflows to -inf for
      // t < -745, which raw-scale flow features reach easily once the
      // weights grow — the CPU oracle's log_softmax never exponentiates the
      // margin
      double ly = 0.0;
#pragma unroll
      for (int c = 0; c < C; ++c) {
        double t = logits[c] - m;
        if (c == yc) ly = t;
        logits[c] = exp(t);
        z += logits[c];
      }
      local_loss += -(ly - log(z));
      double inv_z = 1.0 / z;
#pragma unroll
      for (int c = 0; c < C; ++c)
        pv[c] = logits[c] * inv_z - (c == yc ? 1.0 : 0.0);
    }
#pragma unroll
    for (int c = 0; c < C; ++c) {
#pragma unroll
      for (int j = 0; j <= F; ++j) {
        double v = wave_sum(j < F ? pv[c] * xv[j] : pv[c]);
        if (lane == 0) gw[c * (F + 1) + j] += v;
      }
    }
  }
  local_loss = wave_sum(local_loss);
  if (lane == 0) atomicAdd(loss, local_loss);
  __syncthreads();
  for (int i = threadIdx.x; i < NG; i += blockDim.x) {
    double s = 0.0;
    for (int w = 0; w < nwaves; ++w) s += s_g[w * NG + i];
    atomicAdd(&grad[i], s);
  }
}

extern "C" void launch_logistic_grad(const double* X, const long long* y,
                                     const double* W, const double* b,
                                     double* grad, double* loss, long long n,
                                     int C, hipStream_t stream) {
  const int block = 256;
  size_t bytes = (size_t)(C * 12 + C + (block / 64) * C * 13) * sizeof(double);
  dim3 grid(ts_grid(n, block));
#define LG_CASE(CV)                                                         \
  case CV:                                                                  \
    hipLaunchKernelGGL((logistic_grad_kernel<CV>), grid, dim3(block),       \
                       bytes, stream, X, y, W, b, grad, loss, n);           \
    return;
  switch (C) {
    LG_CASE(2) LG_CASE(3) LG_CASE(4) LG_CASE(5) LG_CASE(6) LG_CASE(7)
    LG_CASE(8) LG_CASE(12) LG_CASE(16)
    default: break;
  }
#undef LG_CASE
}

// ---------------------------------------------------------------------------
// Serve-path feature extraction (reference Flow.updateforward/-reverse math,
// traffic_classifier.py:63-96, vectorised over flows): given per-flow
// counter snapshots and timestamps, emit the 12-feature row.
// Layout per flow: fwd/rev (packets, bytes) current + previous cumulative,
// t_now, t_prev_f, t_prev_r, t_start.
// ---------------------------------------------------------------------------
__global__ void flow_features_kernel(const double* __restrict__ cur,   // [n,4] fp,fb,rp,rb
                                     const double* __restrict__ prev,  // [n,4]
                                     const double* __restrict__ times, // [n,6] tfc,tfp,trc,trp,t0,_
                                     float* __restrict__ out,          // [n,12]
                                     long long n) {
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double fp = cur[i * 4 + 0], fb = cur[i * 4 + 1];
    double rp = cur[i * 4 + 2], rb = cur[i * 4 + 3];
    double dfp = fp - prev[i * 4 + 0], dfb = fb - prev[i * 4 + 1];
    double drp = rp - prev[i * 4 + 2], drb = rb - prev[i * 4 + 3];
    double tfc = times[i * 6 + 0], tfp = times[i * 6 + 1];
    double trc = times[i * 6 + 2], trp = times[i * 6 + 3];
    double t0 = times[i * 6 + 4];
    double df = tfc - tfp, dr = trc - trp;
    double lf = tfc - t0, lr = trc - t0;
    float* o = out + i * 12;
    o[0] = (float)dfp;
    o[1] = (float)dfb;
    o[2] = (float)(df != 0.0 ? dfp / df : 0.0);
    o[3] = (float)(lf != 0.0 ? fp / lf : 0.0);
    o[4] = (float)(df != 0.0 ? dfb / df : 0.0);
    o[5] = (float)(lf != 0.0 ? fb / lf : 0.0);
    o[6] = (float)drp;
    o[7] = (float)drb;
    o[8] = (float)(dr != 0.0 ? drp / dr : 0.0);
    o[9] = (float)(lr != 0.0 ? rp / lr : 0.0);
    o[10] = (float)(dr != 0.0 ? drb / dr : 0.0);
    o[11] = (float)(lr != 0.0 ? rb / lr : 0.0);
  }
}

extern "C" void launch_flow_features(const double* cur, const double* prev,
                                     const double* times, float* out,
                                     long long n, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(flow_features_kernel, dim3(ts_grid(n, block)), dim3(block),
                     0, stream, cur, prev, times, out, n);
}

// ---------------------------------------------------------------------------
// SMO working-pair kernels for the scalable RBF-SVC dual fit (N2 fit path,
// BASELINE config #3).  The host (models/svc_fit.py) drives libsvm WSS-1
// iterations; rows are sharded across ranks and the pair candidates are
// merged with one all-gather per iteration.
// ---------------------------------------------------------------------------

// order-preserving f64->u32 key (monotone: a<b  =>  enc(a)<enc(b))
DEV unsigned enc_f32(float f) {
  unsigned u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

// out[0] = packed argmax over I_up of (-y*grad); out[1] = packed argmax
// over I_low of (+y*grad)  (i.e. argmin of -y*grad).  pack = key<<32 | idx.
__global__ void smo_select_kernel(const float* __restrict__ y,
                                  const double* __restrict__ alpha,
                                  const double* __restrict__ grad,
                                  double C, long long n,
                                  unsigned long long* __restrict__ out) {
  __shared__ unsigned long long s_up, s_low;
  if (threadIdx.x == 0) { s_up = 0; s_low = 0; }
  __syncthreads();
  unsigned long long up = 0, low = 0;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    float yt = y[t];
    double a = alpha[t];
    float myg = (float)(-(double)yt * grad[t]);
    bool in_up = (yt > 0.f && a < C) || (yt < 0.f && a > 0.0);
    bool in_low = (yt > 0.f && a > 0.0) || (yt < 0.f && a < C);
    if (in_up) {
      unsigned long long p = ((unsigned long long)enc_f32(myg) << 32) | (unsigned)t;
      up = p > up ? p : up;
    }
    if (in_low) {
      unsigned long long p = ((unsigned long long)enc_f32(-myg) << 32) | (unsigned)t;
      low = p > low ? p : low;
    }
  }
  atomicMax(&s_up, up);
  atomicMax(&s_low, low);
  __syncthreads();
  if (threadIdx.x == 0) {
    atomicMax(&out[0], s_up);
    atomicMax(&out[1], s_low);
  }
}

extern "C" void launch_smo_select(const float* y, const double* alpha,
                                  const double* grad, double C, long long n,
                                  unsigned long long* out, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(smo_select_kernel, dim3(ts_grid(n, block)), dim3(block), 0,
                     stream, y, alpha, grad, C, n, out);
}

// ---------------------------------------------------------------------------
// RF split search (round 2): fused gini scan over the level histogram.
//
// The torch formulation of the split search materialised a cumsum over the
// whole [L,12,256,C] histogram plus ~8 further GB-scale intermediates per
// level; this kernel assigns one THREAD per (node, feature), keeps the
// running class counts in registers, and resolves the per-node best
// (impurity, feature, bin) with ONE packed u64 atomicMin per thread —
// the histogram is read exactly twice (class totals, then the scan).
//   hist:   int32 [L, F=12, 256, C]
//   fsel:   uint8 [L, 12]   1 = feature is an mtry candidate for the node
//   best:   u64   [L]       init ~0; (enc_f32(imp) << 32) | (f << 16) | bin
//   cnt:    int32 [L, C]    per-node class counts (written by feature 0)
// Ties break toward the smaller (f, bin) — the same order torch's flat
// argmin used, so tree construction stays deterministic.
// ---------------------------------------------------------------------------
template <int C>
__global__ void rf_split_kernel(const int* __restrict__ hist,
                                const unsigned char* __restrict__ fsel,
                                unsigned long long* __restrict__ best,
                                int* __restrict__ cnt, int L) {
  constexpr int F = 12;
  long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long long)L * F) return;
  int node = (int)(idx / F);
  int f = (int)(idx % F);
  // the scatter may have filled only the node's SELECTED feature planes
  // (launch_rf_hist with fsel), so unselected threads exit before touching
  // totals, and the node's class counts are written by its FIRST selected
  // feature
  if (!fsel[node * F + f]) return;
  int first_sel = 0;
  while (first_sel < F && !fsel[node * F + first_sel]) ++first_sel;
  const int* h = hist + ((long long)node * F + f) * 256 * C;
  float total[C];
#pragma unroll
  for (int c = 0; c < C; ++c) total[c] = 0.f;
  for (int b = 0; b < 256; ++b)
#pragma unroll
    for (int c = 0; c < C; ++c) total[c] += (float)h[b * C + c];
  float n_node = 0.f;
#pragma unroll
  for (int c = 0; c < C; ++c) n_node += total[c];
  if (f == first_sel) {
#pragma unroll
    for (int c = 0; c < C; ++c) cnt[node * C + c] = (int)total[c];
  }
  float cum[C];
#pragma unroll
  for (int c = 0; c < C; ++c) cum[c] = 0.f;
  float best_imp = FLT_MAX;
  int best_b = -1;
  const float inv_n = 1.f / fmaxf(n_node, 1.f);
  for (int b = 0; b < 255; ++b) {  // bin 255: empty right side, never valid
    float nl = 0.f;
#pragma unroll
    for (int c = 0; c < C; ++c) {
      cum[c] += (float)h[b * C + c];
      nl += cum[c];
    }
    float nr = n_node - nl;
    if (nl < 1.f || nr < 1.f) continue;
    float sl = 0.f, sr = 0.f;
#pragma unroll
    for (int c = 0; c < C; ++c) {
      float pl = cum[c] / nl;
      float pr = (total[c] - cum[c]) / nr;
      sl = fmaf(pl, pl, sl);
      sr = fmaf(pr, pr, sr);
    }
    float imp = (nl * (1.f - sl) + nr * (1.f - sr)) * inv_n;
    if (imp < best_imp) {
      best_imp = imp;
      best_b = b;
    }
  }
  if (best_b < 0) return;
  unsigned long long p =
      ((unsigned long long)enc_f32(best_imp) << 32) |
      ((unsigned long long)f << 16) | (unsigned)best_b;
  atomicMin(&best[node], p);
}

// Fused frontier partition: one pass re-assigns every row to its child
// node (or -1 once it reaches a leaf), replacing the ~8 separate torch
// index/gather/where passes the builder ran per level.
//   lmap[id]   new local id of the LEFT child (-1: node became a leaf)
//   feat[id]   split feature of node id (valid where lmap >= 0)
//   binthr[id] split bin (go left when bin(x) <= binthr)
__global__ void rf_partition_kernel(const unsigned char* __restrict__ B,
                                    int* __restrict__ nid,
                                    const int* __restrict__ lmap,
                                    const int* __restrict__ feat,
                                    const int* __restrict__ binthr,
                                    long long n) {
  constexpr int F = 12;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long row = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       row < n; row += stride) {
    int id = nid[row];
    if (id < 0) continue;
    int nl = lmap[id];
    if (nl < 0) {
      nid[row] = -1;
      continue;
    }
    int b = B[row * F + feat[id]];
    nid[row] = nl + (b <= binthr[id] ? 0 : 1);
  }
}

extern "C" void launch_rf_partition(const unsigned char* B, int* nid,
                                    const int* lmap, const int* feat,
                                    const int* binthr, long long n,
                                    hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(rf_partition_kernel, dim3(ts_grid(n, block)), dim3(block),
                     0, stream, B, nid, lmap, feat, binthr, n);
}

extern "C" void launch_rf_split(const int* hist, const unsigned char* fsel,
                                unsigned long long* best, int* cnt, int L,
                                int C, hipStream_t stream) {
  const int block = 256;
  long long work = (long long)L * 12;
  dim3 grid((unsigned)((work + block - 1) / block));
#define RFS_CASE(CV)                                                        \
  case CV:                                                                  \
    hipLaunchKernelGGL((rf_split_kernel<CV>), grid, dim3(block), 0, stream, \
                       hist, fsel, best, cnt, L);                           \
    return;
  switch (C) {
    RFS_CASE(2) RFS_CASE(3) RFS_CASE(4) RFS_CASE(5) RFS_CASE(6) RFS_CASE(7)
    RFS_CASE(8) RFS_CASE(12) RFS_CASE(16)
    default: break;
  }
#undef RFS_CASE
}

// ---------------------------------------------------------------------------
// WSS-2 (libsvm second-order working-set selection) additions — round 2.
//
// i is still the WSS-1 argmax over I_up; j is then chosen to maximise the
// second-order objective gain  (Gmax − (−y_t G_t))² / a_t  with
// a_t = K_ii + K_tt − 2 y_i y_t K_it = 2 − 2 y_i y_t K(i,t) for RBF — which
// needs the full kernel row K(i,·).  That row is computed ONCE per
// iteration (smo_row_kernel) and REUSED by the gradient update for the
// i-half, so WSS-2 costs the same two fused row evaluations per iteration
// as WSS-1 while typically needing far fewer iterations (measured:
// profiles/).  Stopping stays libsvm's Gmax + Gmax2 < eps via the WSS-1
// low candidate (sel[1]).  sel is u64[3]: [up(i), low1(Gmax2), low2(j)].
// ---------------------------------------------------------------------------

// krow[t] = K(x_i, x_t); i decoded from the live select buffer
__global__ void smo_row_kernel(const float* __restrict__ X,
                               const unsigned long long* __restrict__ sel,
                               const double* __restrict__ sol,
                               float* __restrict__ krow, float gamma,
                               long long n) {
  __shared__ float s_xi[12];
  __shared__ int s_run;
  if (threadIdx.x == 0)
    s_run = (sol[2] == 0.0 && sel[0] != 0ull) ? 1 : 0;
  __syncthreads();
  if (!s_run) return;
  if (threadIdx.x < 12) {
    int i = (int)(sel[0] & 0xffffffffull);
    s_xi[threadIdx.x] = X[(long long)i * 12 + threadIdx.x];
  }
  __syncthreads();
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    Row12 x = load_row12(X, t);
    float d = 0.f;
#pragma unroll
    for (int k = 0; k < 12; ++k) {
      float a = x.v[k] - s_xi[k];
      d = fmaf(a, a, d);
    }
    krow[t] = __expf(-gamma * d);
  }
}

extern "C" void launch_smo_row(const float* X, const unsigned long long* sel,
                               const double* sol, float* krow, float gamma,
                               long long n, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(smo_row_kernel, dim3(ts_grid(n, block)), dim3(block), 0,
                     stream, X, sel, sol, krow, gamma, n);
}

// sel[2] = packed argmax over eligible I_low of (Gmax − (−y_t G_t))² / a_t
__global__ void smo_select2_kernel(const float* __restrict__ y,
                                   const double* __restrict__ alpha,
                                   const double* __restrict__ grad,
                                   const float* __restrict__ krow,
                                   unsigned long long* __restrict__ sel,
                                   const double* __restrict__ sol,
                                   double C, long long n) {
  __shared__ unsigned long long s_best;
  __shared__ float s_yi;
  __shared__ double s_gmax;
  __shared__ int s_run;
  if (threadIdx.x == 0) {
    s_best = 0;
    unsigned long long pu = sel[0];
    s_run = (sol[2] == 0.0 && pu != 0ull) ? 1 : 0;
    if (s_run) {
      int i = (int)(pu & 0xffffffffull);
      s_yi = y[i];
      s_gmax = -(double)s_yi * grad[i];
    }
  }
  __syncthreads();
  if (!s_run) return;
  const float yi = s_yi;
  const double gmax = s_gmax;
  unsigned long long best = 0;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    float yt = y[t];
    double a = alpha[t];
    bool in_low = (yt > 0.f && a > 0.0) || (yt < 0.f && a < C);
    if (!in_low) continue;
    double gd = gmax + (double)yt * grad[t];  // Gmax − (−y_t G_t)
    if (gd <= 0.0) continue;
    double aq = 2.0 - 2.0 * (double)yi * (double)yt * (double)krow[t];
    if (aq <= 0.0) aq = 1e-12;
    float obj = (float)(gd * gd / aq);
    unsigned long long p =
        ((unsigned long long)enc_f32(obj) << 32) | (unsigned)t;
    best = p > best ? p : best;
  }
  atomicMax(&s_best, best);
  __syncthreads();
  if (threadIdx.x == 0) atomicMax(&sel[2], s_best);
}

extern "C" void launch_smo_select2(const float* y, const double* alpha,
                                   const double* grad, const float* krow,
                                   unsigned long long* sel, const double* sol,
                                   double C, long long n, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(smo_select2_kernel, dim3(ts_grid(n, block)), dim3(block),
                     0, stream, y, alpha, grad, krow, sel, sol, C, n);
}

// Device-side analytic 2-variable solve (single-GPU fast path): decodes the
// select result, solves the pair subproblem, updates alpha IN PLACE, stages
// the two rows + deltas for the gradient kernel, and re-zeroes the select
// buffer — so a whole SMO iteration runs select→solve→update with NO host
// round-trip; the host polls the status word once per iteration chunk.
// sol layout (f64[4]): [yidai, yjdaj, status(0 run/1 converged), gap]
__global__ void smo_solve_kernel(const float* __restrict__ X,
                                 const float* __restrict__ y,
                                 double* __restrict__ alpha,
                                 const double* __restrict__ grad,
                                 unsigned long long* __restrict__ sel,
                                 float* __restrict__ rows,  // [24]
                                 double* __restrict__ sol,  // [4]
                                 double C, double tol, float gamma) {
  if (threadIdx.x != 0) return;  // tiny scalar epilogue: one lane suffices
  if (sol[2] != 0.0) return;     // already converged: stay converged
  unsigned long long pu = sel[0], pl = sel[1];
  sel[0] = 0;
  sel[1] = 0;  // re-arm the atomicMax select for the next iteration
  if (pu == 0 || pl == 0) {
    sol[0] = sol[1] = 0.0;
    sol[2] = 1.0;
    return;
  }
  int i = (int)(pu & 0xffffffffull);
  int j = (int)(pl & 0xffffffffull);
  double yi = (double)y[i], yj = (double)y[j];
  double ai = alpha[i], aj = alpha[j];
  double gi = grad[i], gj = grad[j];
  double up_val = -yi * gi;
  double low_val = -yj * gj;
  double gap = up_val - low_val;
  sol[3] = gap;
  if (gap < tol) {
    sol[0] = sol[1] = 0.0;
    sol[2] = 1.0;
    return;
  }
  double d2 = 0.0;
#pragma unroll
  for (int k = 0; k < 12; ++k) {
    float xi = X[(long long)i * 12 + k];
    float xj = X[(long long)j * 12 + k];
    rows[k] = xi;
    rows[12 + k] = xj;
    double t = (double)xi - (double)xj;
    d2 += t * t;
  }
  double kij = exp(-(double)gamma * d2);
  double a = 2.0 - 2.0 * yi * yj * kij;
  if (a <= 0.0) a = 1e-12;
  double d = gap / a;
  double ai_new = ai + yi * d;
  double s = yi * ai + yj * aj;
  ai_new = fmin(fmax(ai_new, 0.0), C);
  double aj_new = yj * (s - yi * ai_new);
  aj_new = fmin(fmax(aj_new, 0.0), C);
  ai_new = yi * (s - yj * aj_new);
  ai_new = fmin(fmax(ai_new, 0.0), C);
  double dai = ai_new - ai, daj = aj_new - aj;
  if (fabs(dai) < 1e-16 && fabs(daj) < 1e-16) {
    sol[0] = sol[1] = 0.0;
    sol[2] = 1.0;
    return;
  }
  alpha[i] = ai_new;
  alpha[j] = aj_new;
  sol[0] = yi * dai;
  sol[1] = yj * daj;
}

extern "C" void launch_smo_solve(const float* X, const float* y, double* alpha,
                                 const double* grad, unsigned long long* sel,
                                 float* rows, double* sol, double C, double tol,
                                 float gamma, hipStream_t stream) {
  hipLaunchKernelGGL(smo_solve_kernel, dim3(1), dim3(64), 0, stream, X, y,
                     alpha, grad, sel, rows, sol, C, tol, gamma);
}

// WSS-2 solve: the update pair is (i, sel[2]'s j) — falling back to the
// WSS-1 j when no second-order candidate qualified — while the STOPPING
// criterion stays libsvm's Gmax + Gmax2 < eps via the WSS-1 candidate.
__global__ void smo_solve2_kernel(const float* __restrict__ X,
                                  const float* __restrict__ y,
                                  double* __restrict__ alpha,
                                  const double* __restrict__ grad,
                                  unsigned long long* __restrict__ sel,
                                  float* __restrict__ rows,  // [24]
                                  double* __restrict__ sol,  // [4]
                                  double C, double tol, float gamma) {
  if (threadIdx.x != 0) return;
  if (sol[2] != 0.0) return;
  unsigned long long pu = sel[0], pl1 = sel[1], pl2 = sel[2];
  sel[0] = 0;
  sel[1] = 0;
  sel[2] = 0;
  if (pu == 0 || pl1 == 0) {
    sol[0] = sol[1] = 0.0;
    sol[2] = 1.0;
    return;
  }
  int i = (int)(pu & 0xffffffffull);
  int j1 = (int)(pl1 & 0xffffffffull);
  int j = pl2 != 0 ? (int)(pl2 & 0xffffffffull) : j1;
  double yi = (double)y[i];
  double gi = grad[i];
  double up_val = -yi * gi;
  // stopping gap: Gmax + Gmax2 (WSS-1 low candidate)
  double gap = up_val + (double)y[j1] * grad[j1];
  sol[3] = gap;
  if (gap < tol) {
    sol[0] = sol[1] = 0.0;
    sol[2] = 1.0;
    return;
  }
  double yj = (double)y[j];
  double ai = alpha[i], aj = alpha[j];
  double pair_gap = up_val + yj * grad[j];  // > 0 by selection / fallback
  double d2 = 0.0;
#pragma unroll
  for (int k = 0; k < 12; ++k) {
    float xi = X[(long long)i * 12 + k];
    float xj = X[(long long)j * 12 + k];
    rows[k] = xi;
    rows[12 + k] = xj;
    double t = (double)xi - (double)xj;
    d2 += t * t;
  }
  double kij = exp(-(double)gamma * d2);
  double a = 2.0 - 2.0 * yi * yj * kij;
  if (a <= 0.0) a = 1e-12;
  double d = pair_gap / a;
  double ai_new = ai + yi * d;
  double s = yi * ai + yj * aj;
  ai_new = fmin(fmax(ai_new, 0.0), C);
  double aj_new = yj * (s - yi * ai_new);
  aj_new = fmin(fmax(aj_new, 0.0), C);
  ai_new = yi * (s - yj * aj_new);
  ai_new = fmin(fmax(ai_new, 0.0), C);
  double dai = ai_new - ai, daj = aj_new - aj;
  if (fabs(dai) < 1e-16 && fabs(daj) < 1e-16) {
    sol[0] = sol[1] = 0.0;
    sol[2] = 1.0;
    return;
  }
  alpha[i] = ai_new;
  alpha[j] = aj_new;
  sol[0] = yi * dai;
  sol[1] = yj * daj;
}

extern "C" void launch_smo_solve2(const float* X, const float* y, double* alpha,
                                  const double* grad, unsigned long long* sel,
                                  float* rows, double* sol, double C,
                                  double tol, float gamma, hipStream_t stream) {
  hipLaunchKernelGGL(smo_solve2_kernel, dim3(1), dim3(64), 0, stream, X, y,
                     alpha, grad, sel, rows, sol, C, tol, gamma);
}

// grad[t] += y_t * (yi*dai*K(xi,x_t) + yj*daj*K(xj,x_t)); the two RBF rows
// are computed on the fly (fused — no kernel matrix is ever materialised).
// xi/xj come from the 24-float staging buffer `rows` (filled by the host
// each iteration; works for remote rows in the sharded fit).
__global__ void smo_update_kernel(const float* __restrict__ X,
                                  const float* __restrict__ y,
                                  double* __restrict__ grad,
                                  const float* __restrict__ rows,  // [24]
                                  double yidai, double yjdaj, float gamma,
                                  long long n) {
  __shared__ float s_rows[24];
  if (threadIdx.x < 24) s_rows[threadIdx.x] = rows[threadIdx.x];
  __syncthreads();
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    Row12 x = load_row12(X, t);
    float di = 0.f, dj = 0.f;
#pragma unroll
    for (int k = 0; k < 12; ++k) {
      float a = x.v[k] - s_rows[k];
      float b = x.v[k] - s_rows[12 + k];
      di = fmaf(a, a, di);
      dj = fmaf(b, b, dj);
    }
    double ki = (double)__expf(-gamma * di);
    double kj = (double)__expf(-gamma * dj);
    grad[t] += (double)y[t] * (yidai * ki + yjdaj * kj);
  }
}

// Variant driven by the device-resident sol buffer (smo_solve output):
// deltas come from sol[0..1]; a converged status (sol[2] != 0) no-ops.
__global__ void smo_update_dev_kernel(const float* __restrict__ X,
                                      const float* __restrict__ y,
                                      double* __restrict__ grad,
                                      const float* __restrict__ rows,  // [24]
                                      const double* __restrict__ sol,  // [4]
                                      float gamma, long long n) {
  __shared__ float s_rows[24];
  __shared__ double s_d[2];
  if (threadIdx.x < 24) s_rows[threadIdx.x] = rows[threadIdx.x];
  if (threadIdx.x < 2) s_d[threadIdx.x] = sol[threadIdx.x];
  __syncthreads();
  if (sol[2] != 0.0) return;
  const double yidai = s_d[0], yjdaj = s_d[1];
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    Row12 x = load_row12(X, t);
    float di = 0.f, dj = 0.f;
#pragma unroll
    for (int k = 0; k < 12; ++k) {
      float a = x.v[k] - s_rows[k];
      float b = x.v[k] - s_rows[12 + k];
      di = fmaf(a, a, di);
      dj = fmaf(b, b, dj);
    }
    double ki = (double)__expf(-gamma * di);
    double kj = (double)__expf(-gamma * dj);
    grad[t] += (double)y[t] * (yidai * ki + yjdaj * kj);
  }
}

extern "C" void launch_smo_update_dev(const float* X, const float* y,
                                      double* grad, const float* rows,
                                      const double* sol, float gamma,
                                      long long n, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(smo_update_dev_kernel, dim3(ts_grid(n, block)),
                     dim3(block), 0, stream, X, y, grad, rows, sol, gamma, n);
}

// WSS-2 variant: the i-half of the gradient update reads the kernel row
// smo_row_kernel already computed (no recomputation); only the j-row is
// evaluated on the fly — half the exp work of smo_update_dev.
__global__ void smo_update_dev2_kernel(const float* __restrict__ X,
                                       const float* __restrict__ y,
                                       double* __restrict__ grad,
                                       const float* __restrict__ rows,  // [24]
                                       const double* __restrict__ sol,  // [4]
                                       const float* __restrict__ krow,
                                       float gamma, long long n) {
  __shared__ float s_xj[12];
  __shared__ double s_d[2];
  if (threadIdx.x < 12) s_xj[threadIdx.x] = rows[12 + threadIdx.x];
  if (threadIdx.x < 2) s_d[threadIdx.x] = sol[threadIdx.x];
  __syncthreads();
  if (sol[2] != 0.0) return;
  const double yidai = s_d[0], yjdaj = s_d[1];
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    Row12 x = load_row12(X, t);
    float dj = 0.f;
#pragma unroll
    for (int k = 0; k < 12; ++k) {
      float b = x.v[k] - s_xj[k];
      dj = fmaf(b, b, dj);
    }
    double ki = (double)krow[t];
    double kj = (double)__expf(-gamma * dj);
    grad[t] += (double)y[t] * (yidai * ki + yjdaj * kj);
  }
}

extern "C" void launch_smo_update_dev2(const float* X, const float* y,
                                       double* grad, const float* rows,
                                       const double* sol, const float* krow,
                                       float gamma, long long n,
                                       hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(smo_update_dev2_kernel, dim3(ts_grid(n, block)),
                     dim3(block), 0, stream, X, y, grad, rows, sol, krow,
                     gamma, n);
}

extern "C" void launch_smo_update(const float* X, const float* y, double* grad,
                                  const float* rows, double yidai, double yjdaj,
                                  float gamma, long long n, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(smo_update_kernel, dim3(ts_grid(n, block)), dim3(block), 0,
                     stream, X, y, grad, rows, yidai, yjdaj, gamma, n);
}

// ---------------------------------------------------------------------------
// Random-forest histogram build (N4 fit): the hot op of the breadth-first
// level-synchronous tree builder (models/random_forest.py GPU path).  For
// every bootstrap row, scatter its class into the owning node's per-feature
// per-bin class histogram:
//     hist[nid[t]][f][bins[t,f]][y[t]] += 1   for f in 0..11
// Rows whose node is already finalised carry nid = -1 and are skipped.
// Histograms for one level can span many nodes, so the accumulation uses
// global u32 atomics; contention per (node,feat,bin,class) cell is spread
// across the whole row set and measured negligible next to the feature
// gather.  bins are u8 (256-bin quantile grid), classes <= 16.
// ---------------------------------------------------------------------------
// FSEL: when a per-node mtry mask is supplied (u8 [nodes,12], round 2) the
// scatter only touches the node's CANDIDATE features — with sklearn's
// max_features=3 that is 3 atomics per row instead of 12, and the split
// search never reads the unselected planes.  fsel == nullptr keeps the
// full-12 behaviour (the parity tests' contract).
template <bool FSEL>
__global__ void rf_hist_kernel(const unsigned char* __restrict__ bins,  // [n,12]
                               const unsigned char* __restrict__ y,    // [n]
                               const int* __restrict__ nid,            // [n]
                               const unsigned char* __restrict__ fsel, // [nodes,12] or null
                               unsigned* __restrict__ hist,  // [nodes,12,256,C]
                               long long n, int C) {
  constexpr int F = 12;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    int node = nid[t];
    if (node < 0) continue;
    int cls = y[t];
    const unsigned char* b = bins + t * F;
#pragma unroll
    for (int f = 0; f < F; ++f) {
      if (FSEL && !fsel[node * F + f]) continue;
      long long cell = (((long long)node * F + f) * 256 + b[f]) * C + cls;
      atomicAdd(&hist[cell], 1u);
    }
  }
}

// COMPACT variant (round 2): the histogram holds only each node's mf mtry
// candidate planes — [nodes, mf, 256, C] — which keeps the whole level's
// atomic working set inside the 256 MB Infinity Cache (the full 12-plane
// buffer reached 600 MB at 8192-node chunks and the scatter fell to
// HBM-latency atomics: measured 98 ms of the 112 ms level loop).
// frank[node*12+f] = slot index of f among the node's candidates (0xff:
// not a candidate).
__global__ void rf_hist_compact_kernel(
    const unsigned char* __restrict__ bins, const unsigned char* __restrict__ y,
    const int* __restrict__ nid, const unsigned char* __restrict__ frank,
    unsigned* __restrict__ hist, long long n, int C, int mf) {
  constexpr int F = 12;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long t = (long long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    int node = nid[t];
    if (node < 0) continue;
    int cls = y[t];
    const unsigned char* b = bins + t * F;
    const unsigned char* fr = frank + node * F;
#pragma unroll
    for (int f = 0; f < F; ++f) {
      int slot = fr[f];
      if (slot == 0xff) continue;
      long long cell = (((long long)node * mf + slot) * 256 + b[f]) * C + cls;
      atomicAdd(&hist[cell], 1u);
    }
  }
}

extern "C" void launch_rf_hist_compact(const unsigned char* bins,
                                       const unsigned char* y, const int* nid,
                                       const unsigned char* frank,
                                       unsigned* hist, long long n, int C,
                                       int mf, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(rf_hist_compact_kernel, dim3(ts_grid(n, block)),
                     dim3(block), 0, stream, bins, y, nid, frank, hist, n, C,
                     mf);
}

// split search over the COMPACT histogram: one thread per (node, slot);
// fidx[node*mf+slot] = the slot's real feature id (packed into the result
// so the host decode is unchanged).
template <int C>
__global__ void rf_split_compact_kernel(const int* __restrict__ hist,
                                        const unsigned char* __restrict__ fidx,
                                        unsigned long long* __restrict__ best,
                                        int* __restrict__ cnt, int L, int mf) {
  long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long long)L * mf) return;
  int node = (int)(idx / mf);
  int slot = (int)(idx % mf);
  const int* h = hist + ((long long)node * mf + slot) * 256 * C;
  float total[C];
#pragma unroll
  for (int c = 0; c < C; ++c) total[c] = 0.f;
  for (int b = 0; b < 256; ++b)
#pragma unroll
    for (int c = 0; c < C; ++c) total[c] += (float)h[b * C + c];
  float n_node = 0.f;
#pragma unroll
  for (int c = 0; c < C; ++c) n_node += total[c];
  if (slot == 0) {
#pragma unroll
    for (int c = 0; c < C; ++c) cnt[node * C + c] = (int)total[c];
  }
  float cum[C];
#pragma unroll
  for (int c = 0; c < C; ++c) cum[c] = 0.f;
  float best_imp = FLT_MAX;
  int best_b = -1;
  const float inv_n = 1.f / fmaxf(n_node, 1.f);
  for (int b = 0; b < 255; ++b) {
    float nl = 0.f;
#pragma unroll
    for (int c = 0; c < C; ++c) {
      cum[c] += (float)h[b * C + c];
      nl += cum[c];
    }
    float nr = n_node - nl;
    if (nl < 1.f || nr < 1.f) continue;
    float sl = 0.f, sr = 0.f;
#pragma unroll
    for (int c = 0; c < C; ++c) {
      float pl = cum[c] / nl;
      float pr = (total[c] - cum[c]) / nr;
      sl = fmaf(pl, pl, sl);
      sr = fmaf(pr, pr, sr);
    }
    float imp = (nl * (1.f - sl) + nr * (1.f - sr)) * inv_n;
    if (imp < best_imp) {
      best_imp = imp;
      best_b = b;
    }
  }
  if (best_b < 0) return;
  unsigned long long p =
      ((unsigned long long)enc_f32(best_imp) << 32) |
      ((unsigned long long)fidx[node * mf + slot] << 16) | (unsigned)best_b;
  atomicMin(&best[node], p);
}

extern "C" void launch_rf_split_compact(const int* hist,
                                        const unsigned char* fidx,
                                        unsigned long long* best, int* cnt,
                                        int L, int C, int mf,
                                        hipStream_t stream) {
  const int block = 256;
  long long work = (long long)L * mf;
  dim3 grid((unsigned)((work + block - 1) / block));
#define RFSC_CASE(CV)                                                       \
  case CV:                                                                  \
    hipLaunchKernelGGL((rf_split_compact_kernel<CV>), grid, dim3(block), 0, \
                       stream, hist, fidx, best, cnt, L, mf);               \
    return;
  switch (C) {
    RFSC_CASE(2) RFSC_CASE(3) RFSC_CASE(4) RFSC_CASE(5) RFSC_CASE(6)
    RFSC_CASE(7) RFSC_CASE(8) RFSC_CASE(12) RFSC_CASE(16)
    default: break;
  }
#undef RFSC_CASE
}

extern "C" void launch_rf_hist(const unsigned char* bins, const unsigned char* y,
                               const int* nid, const unsigned char* fsel,
                               unsigned* hist, long long n, int C,
                               hipStream_t stream) {
  const int block = 256;
  if (fsel)
    hipLaunchKernelGGL((rf_hist_kernel<true>), dim3(ts_grid(n, block)),
                       dim3(block), 0, stream, bins, y, nid, fsel, hist, n, C);
  else
    hipLaunchKernelGGL((rf_hist_kernel<false>), dim3(ts_grid(n, block)),
                       dim3(block), 0, stream, bins, y, nid, nullptr, hist, n,
                       C);
}
