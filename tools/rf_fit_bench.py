"""Quick RF hist-fit throughput check (GPU)."""
import sys, time
sys.path.insert(0, ".")
import numpy as np
import torch
from traffic_classifier_sdn_amd.models import RandomForestClassifier
from traffic_classifier_sdn_amd.utils.datasets import synthetic_flow_rows

n = int(sys.argv[1]) if len(sys.argv) > 1 else 1_000_000
trees = int(sys.argv[2]) if len(sys.argv) > 2 else 25
X = synthetic_flow_rows(n, seed=9)
ys = ((X[:, 1] > np.median(X[:, 1])).astype(int) * 3
      + (X[:, 4] > np.median(X[:, 4])).astype(int)
      + (X[:, 7] > np.median(X[:, 7])).astype(int)) % 6
t0 = time.time()
m = RandomForestClassifier(n_estimators=trees, seed=0, device="cuda").fit(X, ys)
torch.cuda.synchronize()
dt = time.time() - t0
nn = len(m.trees_[0]["feature"])
print(f"GPU hist fit {n} rows x {trees} trees: {dt:.2f}s = {n*trees/dt:.3g} row-trees/s, nodes/tree ~{nn}")
acc = (m.predict(X[:100000]) == ys[:100000]).mean()
print(f"train-acc (100k sample): {acc:.4f}")
