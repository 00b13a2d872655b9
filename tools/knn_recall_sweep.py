"""Recall of the bf16 coarse-pass KNN vs the exact kernel across k=1..8."""
import sys
sys.path.insert(0, ".")
import numpy as np, torch
from traffic_classifier_sdn_amd.ops.gpu import _ext, _knn_cmean
from traffic_classifier_sdn_amd.utils.datasets import synthetic_flow_rows

R = torch.from_numpy(synthetic_flow_rows(2_000_000, seed=3)).float().cuda().contiguous()
Q = torch.from_numpy(synthetic_flow_rows(8192, seed=4)).float().cuda().contiguous()
cm = _knn_cmean(R)
for k in range(1, 9):
    _, i_ex = _ext.knn_topk_mfma(Q, R, cm, None, k, 0, 0, 8, 0)
    _, i_ap = _ext.knn_topk_mfma(Q, R, cm, None, k, 0, 0, 8, 1)
    ex, ap = i_ex.cpu().numpy(), i_ap.cpu().numpy()
    rec = np.mean([len(set(ex[q]) & set(ap[q])) / k for q in range(ex.shape[0])])
    print(f"k={k}: recall={rec:.6f}")
