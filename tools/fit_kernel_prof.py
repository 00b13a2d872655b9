"""Fit-kernel profiling driver at 1M+ rows (the fit
kernels — logistic_grad, gnb_fit_stats, rf_hist — were never profiled at the
row counts the fit story advertises).  Runs ONLY the selected kernel so a
rocprofv3 --pmc pass attributes counters cleanly:

    rocprofv3 --pmc VALUBusy LdsBankConflict -- python tools/fit_kernel_prof.py logistic 1000000
    rocprofv3 --pmc VALUBusy LdsBankConflict -- python tools/fit_kernel_prof.py gnb 4000000
    rocprofv3 --pmc VALUBusy LdsBankConflict -- python tools/fit_kernel_prof.py rf-hist 4000000

Prints a JSON line with rows/s so profiles can pair counters with rates.
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from traffic_classifier_sdn_amd.ops import gpu as og
from traffic_classifier_sdn_amd.utils.datasets import load_reference_dataset, synthetic_flow_rows


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
    which = sys.argv[1] if len(sys.argv) > 1 else "logistic"
    n = int(sys.argv[2]) if len(sys.argv) > 2 else 1_000_000
    steps = int(sys.argv[3]) if len(sys.argv) > 3 else 10
    X_real, _ = load_reference_dataset()
    Xn = synthetic_flow_rows(n, seed=0, reference_X=X_real)
    rng = np.random.default_rng(0)
    y = torch.from_numpy(rng.integers(0, 6, size=n)).cuda()

    if which == "logistic":
        X = torch.from_numpy(Xn).double().cuda()
        coef = torch.zeros(6, 12, dtype=torch.float64, device="cuda")
        b = torch.zeros(6, dtype=torch.float64, device="cuda")
        dt = timeit(lambda: og.logistic_loss_grad(X, y, coef, b), steps=steps)
    elif which == "gnb":
        X = torch.from_numpy(Xn).double().cuda()
        dt = timeit(lambda: og.gnb_fit_stats(X, y, 6), steps=steps)
    elif which == "rf-hist":
        # quantised feature bins as the hist builder produces them
        bins = torch.from_numpy(
            rng.integers(0, 256, size=(n, 12), dtype=np.uint8)
        ).cuda()
        # 64 live nodes at a mid-depth level of the tree build
        nid = torch.from_numpy(
            rng.integers(0, 64, size=n, dtype=np.int32)
        ).cuda()
        dt = timeit(lambda: og.rf_hist(bins, y.to(torch.uint8), nid, 64, 6), steps=steps)
    else:
        raise SystemExit(f"unknown kernel {which}")

    print(json.dumps({"kernel": which, "rows": n, "ms": dt * 1e3, "rows_per_s": n / dt}))


if __name__ == "__main__":
    main()
