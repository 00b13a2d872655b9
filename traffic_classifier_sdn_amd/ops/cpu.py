"""CPU reference implementations of every compute op.

These are the numerics oracles for the CDNA4 HIP kernels (tests compare the
GPU path against these on identical inputs) and the execution path when no
GPU is present.  Everything is plain PyTorch/numpy; semantics mirror the
sklearn 1.0.1 native kernels the reference's compute ran on (SURVEY.md §2.2
N1-N6), e.g. tie-breaking is always "first maximum".
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import numpy as np
import torch


# ----------------------------------------------------------------------
# Linear / logistic (N1)
# ----------------------------------------------------------------------


def linear_logits(X: torch.Tensor, coef: torch.Tensor, intercept: torch.Tensor) -> torch.Tensor:
    """logits[n,c] = X @ coef^T + intercept  (reference sklearn
    LogisticRegression.decision_function)."""
    return X @ coef.T.to(X.dtype) + intercept.to(X.dtype)


def linear_argmax(X: torch.Tensor, coef: torch.Tensor, intercept: torch.Tensor) -> torch.Tensor:
    return torch.argmax(linear_logits(X, coef, intercept), dim=1).to(torch.int32)


def logistic_loss_grad(
    X: torch.Tensor,
    y: torch.Tensor,
    coef: torch.Tensor,
    intercept: torch.Tensor,
    l2: float = 1.0,
    sample_range: Optional[Tuple[int, int]] = None,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Full-batch multinomial logistic loss + gradients (sklearn's lbfgs
    objective: mean CE * n + 0.5*l2*||coef||^2, gradient in the same scale).

    Returns (loss_scalar, grad_coef[C,F], grad_intercept[C]).
    """
    logits = linear_logits(X, coef, intercept)
    logp = torch.log_softmax(logits, dim=1)
    n = X.shape[0]
    nll = -logp[torch.arange(n, device=X.device), y].sum()
    loss = nll + 0.5 * l2 * (coef * coef).sum()
    p = torch.exp(logp)
    p[torch.arange(n, device=X.device), y] -= 1.0
    grad_coef = p.T @ X + l2 * coef
    grad_intercept = p.sum(dim=0)
    return loss, grad_coef, grad_intercept


# ----------------------------------------------------------------------
# Gaussian NB (N5)
# ----------------------------------------------------------------------


def gnb_joint_loglik(
    X: torch.Tensor, theta: torch.Tensor, var: torch.Tensor, class_prior: torch.Tensor
) -> torch.Tensor:
    """Per-class joint log-likelihood (sklearn GaussianNB._joint_log_likelihood)."""
    theta = theta.to(X.dtype)
    var = var.to(X.dtype)
    class_prior = class_prior.to(X.dtype)
    # const[c] = log prior[c] - 0.5 * sum_j log(2*pi*var[c,j])
    const = torch.log(class_prior) - 0.5 * torch.log(2.0 * torch.pi * var).sum(dim=1)
    # quad[n,c] = -0.5 * sum_j (x[n,j]-theta[c,j])^2 / var[c,j]
    diff = X.unsqueeze(1) - theta.unsqueeze(0)  # (n, C, F)
    quad = -0.5 * (diff * diff / var.unsqueeze(0)).sum(dim=2)
    return quad + const


def gnb_argmax(
    X: torch.Tensor, theta: torch.Tensor, var: torch.Tensor, class_prior: torch.Tensor
) -> torch.Tensor:
    return torch.argmax(gnb_joint_loglik(X, theta, var, class_prior), dim=1).to(torch.int32)


def gnb_fit_stats(
    X: torch.Tensor, y: torch.Tensor, n_classes: int
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Per-class sufficient statistics (count, sum, sum-of-squares)."""
    F = X.shape[1]
    count = torch.zeros(n_classes, dtype=X.dtype, device=X.device)
    s = torch.zeros(n_classes, F, dtype=X.dtype, device=X.device)
    sq = torch.zeros(n_classes, F, dtype=X.dtype, device=X.device)
    count.index_add_(0, y, torch.ones_like(y, dtype=X.dtype))
    s.index_add_(0, y, X)
    sq.index_add_(0, y, X * X)
    return count, s, sq


# ----------------------------------------------------------------------
# KMeans (N6)
# ----------------------------------------------------------------------


def pairwise_sqdist(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    """||a-b||^2 via the expanded form (the GEMM-shaped formulation the MFMA
    kernel uses); clamped at 0 for numerical safety."""
    an = (A * A).sum(dim=1, keepdim=True)
    bn = (B * B).sum(dim=1, keepdim=True).T
    d = an + bn - 2.0 * (A @ B.T)
    return torch.clamp(d, min=0.0)


def kmeans_assign(
    X: torch.Tensor, centers: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """Lloyd assignment step. Returns (labels, counts[K], sums[K,F], inertia)."""
    centers = centers.to(X.dtype)
    d = pairwise_sqdist(X, centers)
    dmin, labels = torch.min(d, dim=1)
    K, F = centers.shape
    counts = torch.zeros(K, dtype=X.dtype, device=X.device)
    sums = torch.zeros(K, F, dtype=X.dtype, device=X.device)
    counts.index_add_(0, labels, torch.ones_like(dmin))
    sums.index_add_(0, labels, X)
    return labels.to(torch.int32), counts, sums, dmin.sum()


# ----------------------------------------------------------------------
# KNN (N3)
# ----------------------------------------------------------------------


def knn_topk(Q: torch.Tensor, R: torch.Tensor, k: int, approx: bool = False) -> Tuple[torch.Tensor, torch.Tensor]:
    # approx is a GPU-path option (bf16 coarse pass); the CPU oracle is exact
    """Brute-force k smallest squared distances. Returns (dist[nq,k], idx[nq,k]),
    sorted ascending (ties by lower index, matching sklearn's ordering)."""
    d = pairwise_sqdist(Q, R)
    dist, idx = torch.topk(d, k, dim=1, largest=False, sorted=True)
    return dist, idx


def knn_classify(
    Q: torch.Tensor, R: torch.Tensor, y: torch.Tensor, k: int, n_classes: int, idx_base: int = 0
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Fused top-k + uniform vote (CPU oracle for the fused GPU kernel)."""
    dist, idx = knn_topk(Q, R, k)
    lab = knn_vote(idx, y.long(), n_classes)
    return dist, idx + idx_base, lab


def knn_vote(idx: torch.Tensor, y: torch.Tensor, n_classes: int) -> torch.Tensor:
    """Uniform-weight majority vote; ties -> lowest class index (sklearn mode)."""
    labels = y[idx]  # (nq, k)
    counts = torch.zeros(idx.shape[0], n_classes, dtype=torch.int32, device=idx.device)
    counts.scatter_add_(1, labels.long(), torch.ones_like(labels, dtype=torch.int32))
    return torch.argmax(counts, dim=1).to(torch.int32)


def rf_hist(bins: torch.Tensor, y: torch.Tensor, nid: torch.Tensor, n_nodes: int, n_classes: int) -> torch.Tensor:
    """Per-node per-feature class histograms for the level-synchronous tree
    builder: hist[node, f, bin, class] = #rows of that class with that bin.
    Rows with nid < 0 (finalised) are skipped.  CPU oracle of the HIP
    scatter kernel (csrc rf_hist_kernel)."""
    F = bins.shape[1]
    mask = nid >= 0
    b = bins[mask].long()
    yv = y[mask].long()
    nd = nid[mask].long()
    feats = torch.arange(F, dtype=torch.long, device=bins.device)
    codes = ((nd.unsqueeze(1) * F + feats) * 256 + b) * n_classes + yv.unsqueeze(1)
    hist = torch.bincount(codes.reshape(-1), minlength=n_nodes * F * 256 * n_classes)
    return hist.reshape(n_nodes, F, 256, n_classes).to(torch.int32)


# ----------------------------------------------------------------------
# SVC RBF OVO (N2)
# ----------------------------------------------------------------------


def rbf_kernel(X: torch.Tensor, SV: torch.Tensor, gamma: float) -> torch.Tensor:
    return torch.exp(-gamma * pairwise_sqdist(X, SV))


def svc_ovo_decision(
    K: torch.Tensor,
    dual_coef: torch.Tensor,
    intercept: torch.Tensor,
    n_support: torch.Tensor,
) -> torch.Tensor:
    """libsvm one-vs-one decision values from a kernel matrix.

    ``dual_coef`` is (C-1, n_SV) in libsvm layout; SVs are grouped by class
    (sizes ``n_support``).  Pair p=(i,j), i<j:
      dec[p] = sum_{sv in class i} dual_coef[j-1, sv] * K[:, sv]
             + sum_{sv in class j} dual_coef[i,   sv] * K[:, sv] + intercept[p]
    """
    C = int(n_support.numel())
    starts = torch.zeros(C + 1, dtype=torch.long)
    starts[1:] = torch.cumsum(n_support.cpu(), dim=0)
    # acc[n, c, o] = sum over SVs of class c of dual_coef[o', sv] * K[n, sv]
    # with o' = o for o < c else o-1  (the 30-accumulator form the HIP kernel
    # uses)
    n = K.shape[0]
    decs = []
    p = 0
    for i in range(C):
        for j in range(i + 1, C):
            si, ei = int(starts[i]), int(starts[i + 1])
            sj, ej = int(starts[j]), int(starts[j + 1])
            d = (
                K[:, si:ei] @ dual_coef[j - 1, si:ei]
                + K[:, sj:ej] @ dual_coef[i, sj:ej]
                + intercept[p]
            )
            decs.append(d)
            p += 1
    return torch.stack(decs, dim=1)  # (n, C*(C-1)/2)


def svc_vote(dec: torch.Tensor, n_classes: int) -> torch.Tensor:
    """libsvm majority vote over OVO decisions; ties -> first max."""
    n = dec.shape[0]
    votes = torch.zeros(n, n_classes, dtype=torch.int32, device=dec.device)
    p = 0
    for i in range(n_classes):
        for j in range(i + 1, n_classes):
            win_i = dec[:, p] > 0
            votes[:, i] += win_i.to(torch.int32)
            votes[:, j] += (~win_i).to(torch.int32)
            p += 1
    return torch.argmax(votes, dim=1).to(torch.int32)


def svc_predict(
    X: torch.Tensor,
    SV: torch.Tensor,
    dual_coef: torch.Tensor,
    intercept: torch.Tensor,
    n_support: torch.Tensor,
    gamma: float,
    svclass: torch.Tensor = None,
) -> torch.Tensor:
    K = rbf_kernel(X, SV.to(X.dtype), gamma)
    dec = svc_ovo_decision(K, dual_coef.to(X.dtype), intercept.to(X.dtype), n_support)
    return svc_vote(dec, int(n_support.numel()))


# ----------------------------------------------------------------------
# Random forest predict (N4)
# ----------------------------------------------------------------------


def rf_flatten(trees, n_classes: int) -> Dict[str, torch.Tensor]:
    """Pack per-tree arrays into the SoA layout shared by CPU and HIP
    traversal kernels.

    Layout (per node, 8 bytes in the packed form):
      - non-leaf: threshold f32, right-child u16 (tree-local), feature u8
      - leaf:     payload u32 = row index into ``leaf_proba`` (f32[,(C)])
    sklearn's depth-first builder guarantees left_child == node_index + 1,
    which the packed form relies on (checked here).
    """
    all_thr = []
    all_right = []
    all_feat = []
    all_leafidx = []
    leaf_probs = []
    offsets = [0]
    n_leaves = 0
    for t in trees:
        left = np.asarray(t["left"], dtype=np.int64)
        right = np.asarray(t["right"], dtype=np.int64)
        feat = np.asarray(t["feature"], dtype=np.int64)
        thr = np.asarray(t["threshold"], dtype=np.float64)
        values = np.asarray(t["values"], dtype=np.float64)  # (nodes, C)
        n = left.shape[0]
        is_leaf = left == -1
        # depth-first invariant: left child immediately follows its parent
        inner = ~is_leaf
        if not np.all(left[inner] == np.nonzero(inner)[0] + 1):
            raise ValueError("tree violates left_child == index+1 invariant")
        # normalized per-leaf class distribution (sklearn predict_proba)
        probs = values / np.clip(values.sum(axis=1, keepdims=True), 1e-30, None)
        leaf_idx = np.full(n, -1, dtype=np.int64)
        leaf_idx[is_leaf] = n_leaves + np.arange(int(is_leaf.sum()))
        n_leaves += int(is_leaf.sum())
        leaf_probs.append(probs[is_leaf])
        all_thr.append(np.where(is_leaf, 0.0, thr).astype(np.float32))
        all_right.append(np.where(is_leaf, 0, right).astype(np.int32))
        all_feat.append(np.where(is_leaf, -1, feat).astype(np.int32))
        all_leafidx.append(leaf_idx.astype(np.int32))
        offsets.append(offsets[-1] + n)
    return {
        "threshold": torch.from_numpy(np.concatenate(all_thr)),
        "right": torch.from_numpy(np.concatenate(all_right)),
        "feature": torch.from_numpy(np.concatenate(all_feat)),
        "leaf_index": torch.from_numpy(np.concatenate(all_leafidx)),
        "leaf_proba": torch.from_numpy(np.concatenate(leaf_probs).astype(np.float32)),
        "tree_offset": torch.tensor(offsets, dtype=torch.int32),
        "n_classes": torch.tensor(n_classes, dtype=torch.int32),
    }


def rf_predict_proba(X: torch.Tensor, forest: Dict[str, torch.Tensor]) -> torch.Tensor:
    """Vectorised level-synchronous traversal: all rows advance one level per
    iteration (CPU oracle for the HIP per-lane traversal kernel)."""
    n = X.shape[0]
    device = X.device
    thr = forest["threshold"].to(device)
    right = forest["right"].to(device)
    feat = forest["feature"].to(device)
    leaf_index = forest["leaf_index"].to(device)
    leaf_proba = forest["leaf_proba"].to(device)
    offsets = forest["tree_offset"].to(device)
    C = int(forest["n_classes"])
    n_trees = offsets.numel() - 1
    acc = torch.zeros(n, C, dtype=torch.float32, device=device)
    rows = torch.arange(n, device=device)
    for t in range(n_trees):
        base = int(offsets[t])
        idx = torch.zeros(n, dtype=torch.long, device=device)
        active = torch.ones(n, dtype=torch.bool, device=device)
        while bool(active.any()):
            g = base + idx
            f = feat[g]
            leaf = f < 0
            done = active & leaf
            if bool(done.any()):
                acc[rows[done]] += leaf_proba[leaf_index[g[done]].long()]
                active = active & ~leaf
            still = active
            if bool(still.any()):
                gs = g[still]
                go_left = X[still, f[still].long()] <= thr[gs]
                nxt = torch.where(go_left, idx[still] + 1, right[gs].long())
                idx = idx.clone()
                idx[still] = nxt
    return acc / n_trees


def rf_argmax(X: torch.Tensor, forest: Dict[str, torch.Tensor]) -> torch.Tensor:
    return torch.argmax(rf_predict_proba(X, forest), dim=1).to(torch.int32)


# ----------------------------------------------------------------------
# Serve-path feature extraction (CPU oracle of the GPU kernel; mirrors the
# reference Flow update math, traffic_classifier.py:63-96)
# ----------------------------------------------------------------------


def flow_features(cur: torch.Tensor, prev: torch.Tensor, times: torch.Tensor) -> torch.Tensor:
    """counters -> 12-feature rows (mirrors Flow update math,
    traffic_classifier.py:63-96).

    cur/prev: (n,4) cumulative [fwd_pkts, fwd_bytes, rev_pkts, rev_bytes] at
    the latest / previous update of each direction;
    times: (n,6) [tf_cur, tf_prev, tr_cur, tr_prev, t_start, pad].
    Division guards: a zero time delta leaves the rate at 0 (the reference
    skips the update; with prev==cur at creation the result matches).
    """
    d = cur - prev
    df = times[:, 0] - times[:, 1]
    dr = times[:, 2] - times[:, 3]
    lf = times[:, 0] - times[:, 4]
    lr = times[:, 2] - times[:, 4]

    def safe(num, den):
        out = torch.zeros_like(num)
        nz = den != 0
        out[nz] = num[nz] / den[nz]
        return out

    cols = [
        d[:, 0],
        d[:, 1],
        safe(d[:, 0], df),
        safe(cur[:, 0], lf),
        safe(d[:, 1], df),
        safe(cur[:, 1], lf),
        d[:, 2],
        d[:, 3],
        safe(d[:, 2], dr),
        safe(cur[:, 2], lr),
        safe(d[:, 3], dr),
        safe(cur[:, 3], lr),
    ]
    return torch.stack(cols, dim=1).to(torch.float32)


def kmeans_labels(X: torch.Tensor, centers: torch.Tensor) -> torch.Tensor:
    """Assignment only (predict path; no partial-sum update)."""
    return kmeans_assign(X, centers)[0]
