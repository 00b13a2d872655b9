"""Per-kernel GPU microbenchmarks: times every predict/fit op on synthetic
rows and prints one JSON line per op (rows/s).  Run on an MI355X box."""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from traffic_classifier_sdn_amd.models import load_model
from traffic_classifier_sdn_amd.ops import gpu as og
from traffic_classifier_sdn_amd.utils.datasets import load_reference_dataset, synthetic_flow_rows

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
MODELS = os.path.join(REPO, "data", "ref_models")


def timeit(fn, steps=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps


def main():
    assert torch.cuda.is_available()
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 4_000_000
    X_real, _ = load_reference_dataset()
    X = torch.from_numpy(synthetic_flow_rows(n, seed=0, reference_X=X_real)).cuda()
    out = {}

    rf = load_model(os.path.join(MODELS, "RandomForestClassifier.npz"), device="cuda")
    out["rf_predict"] = timeit(lambda: og.rf_argmax(X, rf.forest))

    gnb = load_model(os.path.join(MODELS, "GaussianNB.npz"), device="cuda")
    th = gnb.theta_.float().cuda()
    va = gnb.var_.float().cuda()
    pr = gnb.class_prior_.float().cuda()
    out["gnb_predict"] = timeit(lambda: og.gnb_argmax(X, th, va, pr))

    lr = load_model(os.path.join(MODELS, "LogisticRegression.npz"), device="cuda")
    W = lr.coef_.float().cuda()
    b = lr.intercept_.float().cuda()
    out["linear_predict"] = timeit(lambda: og.linear_argmax(X, W, b))

    km = load_model(os.path.join(MODELS, "KMeans_Clustering.npz"), device="cuda")
    cc = km.cluster_centers_.float().cuda()
    out["kmeans_labels"] = timeit(lambda: og.kmeans_labels(X, cc))
    out["kmeans_assign_update"] = timeit(lambda: og.kmeans_assign(X, cc))

    svc = load_model(os.path.join(MODELS, "SVC.npz"), device="cuda")
    SV = svc.support_vectors_.float().cuda()
    dc = svc.dual_coef_.float().cuda()
    ic = svc.intercept_.float().cuda()
    ns = svc.n_support_.cuda()
    Xs = X[: min(n, 1_000_000)]
    out["svc_predict@1M"] = timeit(lambda: og.svc_predict(Xs, SV, dc, ic, ns, svc.gamma_), steps=5)

    knn = load_model(os.path.join(MODELS, "KNeighbors.npz"), device="cuda")
    R = knn.fit_X_.float().cuda()
    y8 = knn.y_.cuda()
    Xq = X[: min(n, 1_000_000)]
    out["knn_classify@1Mq_4448r"] = timeit(lambda: og.knn_classify(Xq, R, y8, 5, 6), steps=5)

    # fit ops
    Xd = X[: min(n, 2_000_000)].double()
    y = torch.randint(0, 6, (Xd.shape[0],), device="cuda")
    out["gnb_fit_stats@2M"] = timeit(lambda: og.gnb_fit_stats(Xd, y, 6))
    W6 = torch.randn(6, 12, dtype=torch.float64, device="cuda")
    b6 = torch.randn(6, dtype=torch.float64, device="cuda")
    out["logistic_grad@2M"] = timeit(lambda: og.logistic_loss_grad(Xd, y, W6, b6))

    rows = {"n_rows": n}
    for k, v in out.items():
        base = 1_000_000 if "@1M" in k else (2_000_000 if "@2M" in k else n)
        rows[k] = {"ms": v * 1e3, "rows_per_s": base / v}
    print(json.dumps(rows, indent=1))


if __name__ == "__main__":
    main()
