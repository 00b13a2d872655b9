// PyTorch (ROCm) bindings for the CDNA4 kernels.  This TU is host-only;
// every kernel lives in the .hip TUs and is reached through the extern "C"
// launchers so the torch headers never touch device code.

#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include <ATen/cuda/CUDAContext.h>

namespace {

#define CHECK_IN(t, type)                                         \
  TORCH_CHECK((t).is_cuda(), #t " must be a GPU tensor");         \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous");     \
  TORCH_CHECK((t).scalar_type() == type, #t " has wrong dtype");

hipStream_t cur_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

}  // namespace

extern "C" {
void launch_gnb_predict(const float*, const float*, const float*, const float*,
                        int*, long long, int, hipStream_t);
void launch_linear_argmax(const float*, const float*, const float*, int*,
                          long long, int, hipStream_t);
void launch_kmeans_assign(const float*, const float*, int*, double*, double*,
                          double*, long long, int, int, hipStream_t);
void launch_rf_predict(const float*, const unsigned*, const int*, const float*,
                       int*, long long, int, int, int, int, hipStream_t);
void launch_svc_predict(const float*, const float*, const float*,
                        const unsigned char*, const float*, int*, long long,
                        int, int, float, hipStream_t);
void launch_knn_topk(const float*, const float*, const unsigned char*, float*,
                     int*, int*, long long, long long, int, int, long long,
                     hipStream_t);
void launch_knn_mfma(const float*, const float*, const float*,
                     const unsigned char*, float*, int*, float*, int*, int*,
                     long long, long long, int, int, int, long long, int,
                     hipStream_t);
void launch_gnb_fit_stats(const double*, const long long*, double*, double*,
                          double*, long long, int, hipStream_t);
void launch_logistic_grad(const double*, const long long*, const double*,
                          const double*, double*, double*, long long, int,
                          hipStream_t);
void launch_flow_features(const double*, const double*, const double*, float*,
                          long long, hipStream_t);
void launch_smo_select(const float*, const double*, const double*, double,
                       long long, unsigned long long*, hipStream_t);
void launch_rf_hist(const unsigned char*, const unsigned char*, const int*,
                    const unsigned char*, unsigned*, long long, int,
                    hipStream_t);
void launch_smo_solve(const float*, const float*, double*, const double*,
                      unsigned long long*, float*, double*, double, double,
                      float, hipStream_t);
void launch_smo_update_dev(const float*, const float*, double*, const float*,
                           const double*, float, long long, hipStream_t);
void launch_smo_row(const float*, const unsigned long long*, const double*,
                    float*, float, long long, hipStream_t);
void launch_rf_split(const int*, const unsigned char*, unsigned long long*,
                     int*, int, int, hipStream_t);
void launch_rf_partition(const unsigned char*, int*, const int*, const int*,
                         const int*, long long, hipStream_t);
void launch_rf_hist_compact(const unsigned char*, const unsigned char*,
                            const int*, const unsigned char*, unsigned*,
                            long long, int, int, hipStream_t);
void launch_rf_split_compact(const int*, const unsigned char*,
                             unsigned long long*, int*, int, int, int,
                             hipStream_t);
void launch_smo_select2(const float*, const double*, const double*,
                        const float*, unsigned long long*, const double*,
                        double, long long, hipStream_t);
void launch_smo_solve2(const float*, const float*, double*, const double*,
                       unsigned long long*, float*, double*, double, double,
                       float, hipStream_t);
void launch_smo_update_dev2(const float*, const float*, double*, const float*,
                            const double*, const float*, float, long long,
                            hipStream_t);
void launch_smo_update(const float*, const float*, double*, const float*,
                       double, double, float, long long, hipStream_t);
}

static torch::Tensor gnb_predict(torch::Tensor X, torch::Tensor theta,
                                 torch::Tensor inv_var, torch::Tensor cconst) {
  CHECK_IN(X, torch::kFloat32);
  CHECK_IN(theta, torch::kFloat32);
  CHECK_IN(inv_var, torch::kFloat32);
  CHECK_IN(cconst, torch::kFloat32);
  TORCH_CHECK(X.size(1) == 12, "X must be (n,12)");
  const long long n = X.size(0);
  const int C = theta.size(0);
  auto out = torch::empty({n}, X.options().dtype(torch::kInt32));
  launch_gnb_predict(X.data_ptr<float>(), theta.data_ptr<float>(),
                     inv_var.data_ptr<float>(), cconst.data_ptr<float>(),
                     out.data_ptr<int>(), n, C, cur_stream());
  return out;
}

static torch::Tensor linear_argmax(torch::Tensor X, torch::Tensor W,
                                   torch::Tensor b) {
  CHECK_IN(X, torch::kFloat32);
  CHECK_IN(W, torch::kFloat32);
  CHECK_IN(b, torch::kFloat32);
  TORCH_CHECK(X.size(1) == 12, "X must be (n,12)");
  const long long n = X.size(0);
  const int C = W.size(0);
  auto out = torch::empty({n}, X.options().dtype(torch::kInt32));
  launch_linear_argmax(X.data_ptr<float>(), W.data_ptr<float>(),
                       b.data_ptr<float>(), out.data_ptr<int>(), n, C,
                       cur_stream());
  return out;
}

static std::vector<torch::Tensor> kmeans_assign(torch::Tensor X,
                                                torch::Tensor centers,
                                                bool want_update) {
  CHECK_IN(X, torch::kFloat32);
  CHECK_IN(centers, torch::kFloat32);
  TORCH_CHECK(X.size(1) == 12, "X must be (n,12)");
  const long long n = X.size(0);
  const int K = centers.size(0);
  auto labels = torch::empty({n}, X.options().dtype(torch::kInt32));
  auto counts = torch::zeros({K}, X.options().dtype(torch::kFloat64));
  auto sums = torch::zeros({K, 12}, X.options().dtype(torch::kFloat64));
  auto inertia = torch::zeros({1}, X.options().dtype(torch::kFloat64));
  launch_kmeans_assign(X.data_ptr<float>(), centers.data_ptr<float>(),
                       labels.data_ptr<int>(), counts.data_ptr<double>(),
                       sums.data_ptr<double>(), inertia.data_ptr<double>(), n,
                       K, want_update ? 1 : 0, cur_stream());
  return {labels, counts, sums, inertia};
}

static torch::Tensor rf_predict(torch::Tensor X, torch::Tensor nodes,
                                torch::Tensor roots, torch::Tensor leaf_proba,
                                int64_t n_leaves, int64_t C) {
  CHECK_IN(X, torch::kFloat32);
  CHECK_IN(nodes, torch::kInt32);  // packed uint2 as 2x int32
  CHECK_IN(roots, torch::kInt32);
  CHECK_IN(leaf_proba, torch::kFloat32);
  TORCH_CHECK(X.size(1) == 12, "X must be (n,12)");
  TORCH_CHECK(nodes.size(1) == 2, "nodes must be (n_nodes,2) int32");
  const long long n = X.size(0);
  const int n_nodes = nodes.size(0);
  const int T = roots.size(0);
  auto out = torch::empty({n}, X.options().dtype(torch::kInt32));
  launch_rf_predict(X.data_ptr<float>(),
                    reinterpret_cast<const unsigned*>(nodes.data_ptr<int>()),
                    roots.data_ptr<int>(), leaf_proba.data_ptr<float>(),
                    out.data_ptr<int>(), n, n_nodes, (int)n_leaves, T, (int)C,
                    cur_stream());
  return out;
}

static torch::Tensor svc_predict(torch::Tensor X, torch::Tensor SV,
                                 torch::Tensor dual, torch::Tensor svclass,
                                 torch::Tensor intercept, double gamma) {
  CHECK_IN(X, torch::kFloat32);
  CHECK_IN(SV, torch::kFloat32);
  CHECK_IN(dual, torch::kFloat32);
  CHECK_IN(svclass, torch::kUInt8);
  CHECK_IN(intercept, torch::kFloat32);
  TORCH_CHECK(X.size(1) == 12, "X must be (n,12)");
  const long long n = X.size(0);
  const int nsv = SV.size(0);
  const int C = dual.size(0) + 1;
  TORCH_CHECK(C >= 2 && C <= 6, "svc kernel supports 2..6 classes");
  auto out = torch::empty({n}, X.options().dtype(torch::kInt32));
  launch_svc_predict(X.data_ptr<float>(), SV.data_ptr<float>(),
                     dual.data_ptr<float>(), svclass.data_ptr<unsigned char>(),
                     intercept.data_ptr<float>(), out.data_ptr<int>(), n, nsv,
                     C, (float)gamma, cur_stream());
  return out;
}

static std::vector<torch::Tensor> knn_topk(torch::Tensor Q, torch::Tensor R,
                                           c10::optional<torch::Tensor> ry,
                                           int64_t k, int64_t C,
                                           int64_t idx_base) {
  CHECK_IN(Q, torch::kFloat32);
  CHECK_IN(R, torch::kFloat32);
  TORCH_CHECK(Q.size(1) == 12 && R.size(1) == 12, "rows must be (n,12)");
  TORCH_CHECK(k <= 32, "k <= 32");
  const long long nq = Q.size(0);
  const long long nr = R.size(0);
  auto dist = torch::empty({nq, k}, Q.options());
  auto idx = torch::empty({nq, k}, Q.options().dtype(torch::kInt32));
  const unsigned char* ry_ptr = nullptr;
  torch::Tensor lab;
  int* lab_ptr = nullptr;
  if (ry.has_value()) {
    CHECK_IN(ry.value(), torch::kUInt8);
    ry_ptr = ry.value().data_ptr<unsigned char>();
    lab = torch::empty({nq}, Q.options().dtype(torch::kInt32));
    lab_ptr = lab.data_ptr<int>();
  }
  launch_knn_topk(Q.data_ptr<float>(), R.data_ptr<float>(), ry_ptr,
                  dist.data_ptr<float>(), idx.data_ptr<int>(), lab_ptr, nq, nr,
                  (int)k, (int)C, idx_base, cur_stream());
  if (ry.has_value()) return {dist, idx, lab};
  return {dist, idx};
}

// MFMA distance-GEMM path: R sharded across gridDim.y, per-shard partial
// top-k lists merged by a second kernel (knn_mfma.hip).
static std::vector<torch::Tensor> knn_topk_mfma(torch::Tensor Q,
                                                torch::Tensor R,
                                                torch::Tensor cmean,
                                                c10::optional<torch::Tensor> ry,
                                                int64_t k, int64_t C,
                                                int64_t idx_base,
                                                int64_t n_shards,
                                                int64_t approx) {
  CHECK_IN(Q, torch::kFloat32);
  CHECK_IN(R, torch::kFloat32);
  CHECK_IN(cmean, torch::kFloat32);
  TORCH_CHECK(Q.size(1) == 12 && R.size(1) == 12, "rows must be (n,12)");
  TORCH_CHECK(k >= 1 && k <= 8, "MFMA path supports k <= 8");
  TORCH_CHECK(cmean.numel() == 12, "cmean must be (12,)");
  const long long nq = Q.size(0);
  const long long nr = R.size(0);
  int S = (int)n_shards;
  if (S < 1) S = 1;
  if ((long long)S > (nr + 127) / 128) S = (int)((nr + 127) / 128);
  auto part_d = torch::empty({S, nq, k}, Q.options());
  auto part_i = torch::empty({S, nq, k}, Q.options().dtype(torch::kInt32));
  auto dist = torch::empty({nq, k}, Q.options());
  auto idx = torch::empty({nq, k}, Q.options().dtype(torch::kInt32));
  const unsigned char* ry_ptr = nullptr;
  torch::Tensor lab;
  int* lab_ptr = nullptr;
  if (ry.has_value()) {
    CHECK_IN(ry.value(), torch::kUInt8);
    ry_ptr = ry.value().data_ptr<unsigned char>();
    lab = torch::empty({nq}, Q.options().dtype(torch::kInt32));
    lab_ptr = lab.data_ptr<int>();
  }
  launch_knn_mfma(Q.data_ptr<float>(), R.data_ptr<float>(),
                  cmean.data_ptr<float>(), ry_ptr, part_d.data_ptr<float>(),
                  part_i.data_ptr<int>(), dist.data_ptr<float>(),
                  idx.data_ptr<int>(), lab_ptr, nq, nr, S, (int)k, (int)C,
                  idx_base, (int)approx, cur_stream());
  if (ry.has_value()) return {dist, idx, lab};
  return {dist, idx};
}

static std::vector<torch::Tensor> gnb_fit_stats(torch::Tensor X,
                                                torch::Tensor y, int64_t C) {
  CHECK_IN(X, torch::kFloat64);
  CHECK_IN(y, torch::kInt64);
  TORCH_CHECK(X.size(1) == 12, "X must be (n,12)");
  const long long n = X.size(0);
  auto count = torch::zeros({C}, X.options());
  auto sum = torch::zeros({C, 12}, X.options());
  auto sumsq = torch::zeros({C, 12}, X.options());
  launch_gnb_fit_stats(X.data_ptr<double>(), reinterpret_cast<const long long*>(y.data_ptr<int64_t>()),
                       count.data_ptr<double>(), sum.data_ptr<double>(),
                       sumsq.data_ptr<double>(), n, (int)C, cur_stream());
  return {count, sum, sumsq};
}

static std::vector<torch::Tensor> logistic_grad(torch::Tensor X,
                                                torch::Tensor y,
                                                torch::Tensor W,
                                                torch::Tensor b) {
  CHECK_IN(X, torch::kFloat64);
  CHECK_IN(y, torch::kInt64);
  CHECK_IN(W, torch::kFloat64);
  CHECK_IN(b, torch::kFloat64);
  const long long n = X.size(0);
  const int C = W.size(0);
  auto grad = torch::zeros({C, 13}, X.options());
  auto loss = torch::zeros({1}, X.options());
  launch_logistic_grad(X.data_ptr<double>(), reinterpret_cast<const long long*>(y.data_ptr<int64_t>()),
                       W.data_ptr<double>(), b.data_ptr<double>(),
                       grad.data_ptr<double>(), loss.data_ptr<double>(), n, C,
                       cur_stream());
  return {grad, loss};
}

static torch::Tensor flow_features(torch::Tensor cur, torch::Tensor prev,
                                   torch::Tensor times) {
  CHECK_IN(cur, torch::kFloat64);
  CHECK_IN(prev, torch::kFloat64);
  CHECK_IN(times, torch::kFloat64);
  const long long n = cur.size(0);
  auto out = torch::empty({n, 12}, cur.options().dtype(torch::kFloat32));
  launch_flow_features(cur.data_ptr<double>(), prev.data_ptr<double>(),
                       times.data_ptr<double>(), out.data_ptr<float>(), n,
                       cur_stream());
  return out;
}

static void smo_select(torch::Tensor y, torch::Tensor alpha,
                       torch::Tensor grad, double C, torch::Tensor out) {
  CHECK_IN(y, torch::kFloat32);
  CHECK_IN(alpha, torch::kFloat64);
  CHECK_IN(grad, torch::kFloat64);
  CHECK_IN(out, torch::kInt64);  // reinterpreted as u64 packed keys, len 2
  launch_smo_select(y.data_ptr<float>(), alpha.data_ptr<double>(),
                    grad.data_ptr<double>(), C, y.size(0),
                    reinterpret_cast<unsigned long long*>(out.data_ptr<int64_t>()),
                    cur_stream());
}

static void rf_hist(torch::Tensor bins, torch::Tensor y, torch::Tensor nid,
                    torch::Tensor hist,
                    c10::optional<torch::Tensor> fsel = c10::nullopt) {
  CHECK_IN(bins, torch::kUInt8);
  CHECK_IN(y, torch::kUInt8);
  CHECK_IN(nid, torch::kInt32);
  CHECK_IN(hist, torch::kInt32);  // u32 atomics on int32 storage
  TORCH_CHECK(bins.size(1) == 12, "bins must be (n,12)");
  TORCH_CHECK(hist.dim() == 4 && hist.size(1) == 12 && hist.size(2) == 256,
              "hist must be (nodes,12,256,C)");
  const unsigned char* fs = nullptr;
  if (fsel.has_value()) {
    CHECK_IN(fsel.value(), torch::kUInt8);
    TORCH_CHECK(fsel.value().numel() == hist.size(0) * 12,
                "fsel must be (nodes,12)");
    fs = fsel.value().data_ptr<unsigned char>();
  }
  launch_rf_hist(bins.data_ptr<unsigned char>(), y.data_ptr<unsigned char>(),
                 nid.data_ptr<int>(), fs,
                 reinterpret_cast<unsigned*>(hist.data_ptr<int>()),
                 bins.size(0), hist.size(3), cur_stream());
}

static void smo_solve(torch::Tensor X, torch::Tensor y, torch::Tensor alpha,
                      torch::Tensor grad, torch::Tensor sel, torch::Tensor rows,
                      torch::Tensor sol, double C, double tol, double gamma) {
  CHECK_IN(X, torch::kFloat32);
  CHECK_IN(y, torch::kFloat32);
  CHECK_IN(alpha, torch::kFloat64);
  CHECK_IN(grad, torch::kFloat64);
  CHECK_IN(sel, torch::kInt64);
  CHECK_IN(rows, torch::kFloat32);
  CHECK_IN(sol, torch::kFloat64);
  launch_smo_solve(X.data_ptr<float>(), y.data_ptr<float>(),
                   alpha.data_ptr<double>(), grad.data_ptr<double>(),
                   reinterpret_cast<unsigned long long*>(sel.data_ptr<int64_t>()),
                   rows.data_ptr<float>(), sol.data_ptr<double>(), C, tol,
                   (float)gamma, cur_stream());
}

static void smo_update_dev(torch::Tensor X, torch::Tensor y, torch::Tensor grad,
                           torch::Tensor rows, torch::Tensor sol, double gamma) {
  CHECK_IN(X, torch::kFloat32);
  CHECK_IN(y, torch::kFloat32);
  CHECK_IN(grad, torch::kFloat64);
  CHECK_IN(rows, torch::kFloat32);
  CHECK_IN(sol, torch::kFloat64);
  launch_smo_update_dev(X.data_ptr<float>(), y.data_ptr<float>(),
                        grad.data_ptr<double>(), rows.data_ptr<float>(),
                        sol.data_ptr<double>(), (float)gamma, X.size(0),
                        cur_stream());
}

static std::vector<torch::Tensor> rf_level_compact(torch::Tensor bins,
                                                   torch::Tensor y,
                                                   torch::Tensor nid,
                                                   torch::Tensor frank,
                                                   torch::Tensor fidx, int64_t L,
                                                   int64_t C) {
  // fused level pass: compact mtry-plane histogram scatter + split search
  CHECK_IN(bins, torch::kUInt8);
  CHECK_IN(y, torch::kUInt8);
  CHECK_IN(nid, torch::kInt32);
  CHECK_IN(frank, torch::kUInt8);
  CHECK_IN(fidx, torch::kUInt8);
  TORCH_CHECK(bins.size(1) == 12, "bins must be (n,12)");
  TORCH_CHECK(frank.numel() == L * 12, "frank must be (L,12)");
  int mf = (int)(fidx.numel() / L);
  TORCH_CHECK(mf >= 1 && (int64_t)mf * L == fidx.numel(), "fidx must be (L,mf)");
  auto hist = torch::zeros({L, mf, 256, C},
                           bins.options().dtype(torch::kInt32));
  launch_rf_hist_compact(bins.data_ptr<unsigned char>(),
                         y.data_ptr<unsigned char>(), nid.data_ptr<int>(),
                         frank.data_ptr<unsigned char>(),
                         reinterpret_cast<unsigned*>(hist.data_ptr<int>()),
                         bins.size(0), (int)C, mf, cur_stream());
  auto best = torch::full({L}, -1, bins.options().dtype(torch::kInt64));
  auto cnt = torch::zeros({L, C}, bins.options().dtype(torch::kInt32));
  launch_rf_split_compact(hist.data_ptr<int>(),
                          fidx.data_ptr<unsigned char>(),
                          reinterpret_cast<unsigned long long*>(best.data_ptr<int64_t>()),
                          cnt.data_ptr<int>(), (int)L, (int)C, mf,
                          cur_stream());
  return {best, cnt};
}

static void rf_partition(torch::Tensor B, torch::Tensor nid,
                         torch::Tensor lmap, torch::Tensor feat,
                         torch::Tensor binthr) {
  CHECK_IN(B, torch::kUInt8);
  CHECK_IN(nid, torch::kInt32);
  CHECK_IN(lmap, torch::kInt32);
  CHECK_IN(feat, torch::kInt32);
  CHECK_IN(binthr, torch::kInt32);
  TORCH_CHECK(B.size(1) == 12, "B must be (n,12)");
  launch_rf_partition(B.data_ptr<unsigned char>(), nid.data_ptr<int>(),
                      lmap.data_ptr<int>(), feat.data_ptr<int>(),
                      binthr.data_ptr<int>(), B.size(0), cur_stream());
}

static std::vector<torch::Tensor> rf_split(torch::Tensor hist,
                                           torch::Tensor fsel) {
  CHECK_IN(hist, torch::kInt32);
  CHECK_IN(fsel, torch::kUInt8);
  TORCH_CHECK(hist.dim() == 4 && hist.size(1) == 12 && hist.size(2) == 256,
              "hist must be (L,12,256,C)");
  int L = (int)hist.size(0);
  int C = (int)hist.size(3);
  auto best = torch::full({L}, -1, hist.options().dtype(torch::kInt64));
  auto cnt = torch::zeros({L, C}, hist.options());
  launch_rf_split(hist.data_ptr<int>(), fsel.data_ptr<unsigned char>(),
                  reinterpret_cast<unsigned long long*>(best.data_ptr<int64_t>()),
                  cnt.data_ptr<int>(), L, C, cur_stream());
  return {best, cnt};
}

static void smo_row(torch::Tensor X, torch::Tensor sel, torch::Tensor sol,
                    torch::Tensor krow, double gamma) {
  CHECK_IN(X, torch::kFloat32);
  CHECK_IN(sel, torch::kInt64);
  CHECK_IN(sol, torch::kFloat64);
  CHECK_IN(krow, torch::kFloat32);
  launch_smo_row(X.data_ptr<float>(),
                 reinterpret_cast<unsigned long long*>(sel.data_ptr<int64_t>()),
                 sol.data_ptr<double>(), krow.data_ptr<float>(), (float)gamma,
                 X.size(0), cur_stream());
}

static void smo_select2(torch::Tensor y, torch::Tensor alpha,
                        torch::Tensor grad, torch::Tensor krow,
                        torch::Tensor sel, torch::Tensor sol, double C) {
  CHECK_IN(y, torch::kFloat32);
  CHECK_IN(alpha, torch::kFloat64);
  CHECK_IN(grad, torch::kFloat64);
  CHECK_IN(krow, torch::kFloat32);
  CHECK_IN(sel, torch::kInt64);
  CHECK_IN(sol, torch::kFloat64);
  launch_smo_select2(y.data_ptr<float>(), alpha.data_ptr<double>(),
                     grad.data_ptr<double>(), krow.data_ptr<float>(),
                     reinterpret_cast<unsigned long long*>(sel.data_ptr<int64_t>()),
                     sol.data_ptr<double>(), C, y.size(0), cur_stream());
}

static void smo_solve2(torch::Tensor X, torch::Tensor y, torch::Tensor alpha,
                       torch::Tensor grad, torch::Tensor sel, torch::Tensor rows,
                       torch::Tensor sol, double C, double tol, double gamma) {
  CHECK_IN(X, torch::kFloat32);
  CHECK_IN(y, torch::kFloat32);
  CHECK_IN(alpha, torch::kFloat64);
  CHECK_IN(grad, torch::kFloat64);
  CHECK_IN(sel, torch::kInt64);
  CHECK_IN(rows, torch::kFloat32);
  CHECK_IN(sol, torch::kFloat64);
  TORCH_CHECK(sel.numel() >= 3, "WSS-2 sel buffer must be u64[3]");
  launch_smo_solve2(X.data_ptr<float>(), y.data_ptr<float>(),
                    alpha.data_ptr<double>(), grad.data_ptr<double>(),
                    reinterpret_cast<unsigned long long*>(sel.data_ptr<int64_t>()),
                    rows.data_ptr<float>(), sol.data_ptr<double>(), C, tol,
                    (float)gamma, cur_stream());
}

static void smo_update_dev2(torch::Tensor X, torch::Tensor y, torch::Tensor grad,
                            torch::Tensor rows, torch::Tensor sol,
                            torch::Tensor krow, double gamma) {
  CHECK_IN(X, torch::kFloat32);
  CHECK_IN(y, torch::kFloat32);
  CHECK_IN(grad, torch::kFloat64);
  CHECK_IN(rows, torch::kFloat32);
  CHECK_IN(sol, torch::kFloat64);
  CHECK_IN(krow, torch::kFloat32);
  launch_smo_update_dev2(X.data_ptr<float>(), y.data_ptr<float>(),
                         grad.data_ptr<double>(), rows.data_ptr<float>(),
                         sol.data_ptr<double>(), krow.data_ptr<float>(),
                         (float)gamma, X.size(0), cur_stream());
}

static void smo_update(torch::Tensor X, torch::Tensor y, torch::Tensor grad,
                       torch::Tensor rows, double yidai, double yjdaj,
                       double gamma) {
  CHECK_IN(X, torch::kFloat32);
  CHECK_IN(y, torch::kFloat32);
  CHECK_IN(grad, torch::kFloat64);
  CHECK_IN(rows, torch::kFloat32);
  launch_smo_update(X.data_ptr<float>(), y.data_ptr<float>(),
                    grad.data_ptr<double>(), rows.data_ptr<float>(), yidai,
                    yjdaj, (float)gamma, X.size(0), cur_stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("smo_select", &smo_select, "WSS-1 pair candidate selection");
  m.def("smo_update", &smo_update, "fused RBF-row gradient update");
  m.def("rf_hist", &rf_hist, "per-node per-feature class histograms (tree build)",
        py::arg("bins"), py::arg("y"), py::arg("nid"), py::arg("hist"),
        py::arg("fsel") = c10::nullopt);
  m.def("smo_solve", &smo_solve, "device-side SMO pair solve (fused iteration)");
  m.def("smo_update_dev", &smo_update_dev, "gradient update from device sol buffer");
  m.def("rf_split", &rf_split, "fused gini split search over a level histogram");
  m.def("rf_partition", &rf_partition, "fused frontier row partition");
  m.def("rf_level_compact", &rf_level_compact,
        "fused level pass: mtry-compact hist scatter + split search");
  m.def("smo_row", &smo_row, "K(x_i, .) kernel row for WSS-2");
  m.def("smo_select2", &smo_select2, "WSS-2 second-order j selection");
  m.def("smo_solve2", &smo_solve2, "device-side WSS-2 pair solve");
  m.def("smo_update_dev2", &smo_update_dev2, "gradient update reusing the WSS-2 kernel row");
  m.def("gnb_predict", &gnb_predict, "fused GaussianNB loglik+argmax");
  m.def("linear_argmax", &linear_argmax, "logits+argmax");
  m.def("kmeans_assign", &kmeans_assign, "Lloyd assign + partial update");
  m.def("rf_predict", &rf_predict, "packed-forest traversal + vote");
  m.def("svc_predict", &svc_predict, "RBF Gram + OVO vote");
  m.def("knn_topk", &knn_topk, "brute-force top-k (+fused vote)");
  m.def("knn_topk_mfma", &knn_topk_mfma, "MFMA distance-GEMM top-k (+fused vote)",
        py::arg("Q"), py::arg("R"), py::arg("cmean"), py::arg("ry"),
        py::arg("k"), py::arg("C"), py::arg("idx_base"), py::arg("n_shards"),
        py::arg("approx") = 0);
  m.def("gnb_fit_stats", &gnb_fit_stats, "per-class sufficient stats");
  m.def("logistic_grad", &logistic_grad, "fused CE loss+grad");
  m.def("flow_features", &flow_features, "counters -> 12 features");
}
