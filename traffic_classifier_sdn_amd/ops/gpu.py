"""GPU op bindings: hand-written CDNA4 HIP kernels (csrc/*.hip).

Every op the models call on CUDA tensors is bound here explicitly.  Hot ops
run the HIP kernels; a handful of cold diagnostic ops (decision functions,
seeding distances) deliberately alias the torch implementations — they are
not part of any benched path.  Import fails loudly when the extension is
missing (see ops.__init__)."""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

from . import cpu as _cpu
from . import _tcsdn_hip as _ext  # type: ignore  # built by setup.py

# cold / diagnostic aliases (torch ops, not benched)
linear_logits = _cpu.linear_logits
gnb_joint_loglik = _cpu.gnb_joint_loglik
pairwise_sqdist = _cpu.pairwise_sqdist
rbf_kernel = _cpu.rbf_kernel
svc_ovo_decision = _cpu.svc_ovo_decision
svc_vote = _cpu.svc_vote
knn_vote = _cpu.knn_vote
rf_flatten = _cpu.rf_flatten


def _f32(t: torch.Tensor) -> torch.Tensor:
    return t.to(torch.float32).contiguous()


# Model parameters are fixed between fits, but a `.to(float32)` per predict
# call launches a conversion kernel every time — inside the hipGraph-captured
# serve path that was ~9 stray elementwise kernels per replay.  Cache the
# converted copies keyed by source buffer address, and keep a reference to
# the SOURCE tensor in the entry: torch's caching allocator reuses freed
# addresses, so a data_ptr key alone could serve a stale cast after a model
# is freed and a same-shape tensor lands at the recycled address (ADVICE
# r01).  Pinning the source keeps its address out of the free pool for the
# lifetime of the entry, which makes the (ptr, numel, dtype) key unambiguous
# (the models never mutate parameters in place).
_cast_cache: Dict[Tuple[int, int, torch.dtype], Tuple[torch.Tensor, torch.Tensor]] = {}


def _f32_cached(t: torch.Tensor) -> torch.Tensor:
    if t.dtype == torch.float32 and t.is_contiguous():
        return t
    key = (t.data_ptr(), t.numel(), t.dtype)
    ent = _cast_cache.get(key)
    if ent is None:
        if len(_cast_cache) > 64:
            _cast_cache.clear()
        ent = (t, t.to(torch.float32).contiguous())
        _cast_cache[key] = ent
    return ent[1]


def _f64(t: torch.Tensor) -> torch.Tensor:
    return t.to(torch.float64).contiguous()


# ----------------------------------------------------------------------
# predict ops
# ----------------------------------------------------------------------


def linear_argmax(X: torch.Tensor, coef: torch.Tensor, intercept: torch.Tensor) -> torch.Tensor:
    if X.shape[1] != 12:  # HIP kernels are specialised for the 12-feature
        return _cpu.linear_argmax(X, coef, intercept)  # schema; torch ops run
    return _ext.linear_argmax(_f32(X), _f32_cached(coef), _f32_cached(intercept))


# entry = (var_ref, prior_ref, theta32, inv_var, const32): the first two pin
# the source buffers so their addresses cannot be recycled while cached
# (same aliasing hazard as _cast_cache)
_gnb_cache: Dict[Tuple[int, int], Tuple[torch.Tensor, ...]] = {}


def gnb_argmax(
    X: torch.Tensor, theta: torch.Tensor, var: torch.Tensor, class_prior: torch.Tensor
) -> torch.Tensor:
    # per-class constants derived in f64 for stability, cached per model so
    # the serve path replays no derivation kernels
    if X.shape[1] != 12:
        return _cpu.gnb_argmax(X, theta, var, class_prior)
    key = (var.data_ptr(), class_prior.data_ptr())
    ent = _gnb_cache.get(key)
    if ent is None:
        if len(_gnb_cache) > 32:
            _gnb_cache.clear()
        var64 = var.double()
        const = (torch.log(class_prior.double()) - 0.5 * torch.log(2.0 * torch.pi * var64).sum(dim=1))
        ent = (var, class_prior, _f32(theta), (1.0 / var64).float().contiguous(), const.float().contiguous())
        _gnb_cache[key] = ent
    _, _, theta32, inv_var, const32 = ent
    return _ext.gnb_predict(_f32(X), theta32, inv_var, const32)


def kmeans_assign(
    X: torch.Tensor, centers: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    if X.shape[1] != 12:
        return _cpu.kmeans_assign(X, centers)
    labels, counts, sums, inertia = _ext.kmeans_assign(_f32(X), _f32(centers), True)
    dt = X.dtype if X.dtype in (torch.float32, torch.float64) else torch.float32
    return labels, counts.to(dt), sums.to(dt), inertia[0].to(dt)


# MFMA path cutover: below this reference-set size the scalar kernel's
# launch simplicity wins; above it the v_mfma_f32_32x32x2_f32 distance
# GEMM dominates (csrc/knn_mfma.hip)
_KNN_MFMA_MIN_ROWS = 100_000
# entry = (R_ref, cmean): R_ref pins the reference buffer against address
# recycling
_knn_cmean_cache: Dict[Tuple[int, int], Tuple[torch.Tensor, torch.Tensor]] = {}


def _knn_cmean(R: torch.Tensor) -> torch.Tensor:
    """Reference column means (centering makes the expanded-form distance
    cancellation-safe); cached per reference buffer."""
    key = (R.data_ptr(), R.shape[0])
    ent = _knn_cmean_cache.get(key)
    if ent is None:
        if len(_knn_cmean_cache) > 16:
            _knn_cmean_cache.clear()
        ent = (R, R.mean(dim=0).float().contiguous())
        _knn_cmean_cache[key] = ent
    return ent[1]


def _knn_shards(nr: int) -> int:
    # enough (query-block x shard) workgroups to fill 256 CUs at 2 WG/CU,
    # with shards of at least ~256K rows to amortize seeding
    return max(1, min(64, nr // 262_144))


def knn_topk(Q: torch.Tensor, R: torch.Tensor, k: int, approx: bool = False) -> Tuple[torch.Tensor, torch.Tensor]:
    if Q.shape[1] != 12:
        return _cpu.knn_topk(Q, R, k)
    Qf, Rf = _f32(Q), _f32(R)
    if (approx or R.shape[0] >= _KNN_MFMA_MIN_ROWS) and k <= 8:
        dist, idx = _ext.knn_topk_mfma(
            Qf, Rf, _knn_cmean(Rf), None, k, 0, 0, _knn_shards(R.shape[0]),
            1 if approx else 0,
        )
        return dist, idx
    dist, idx = _ext.knn_topk(Qf, Rf, None, k, 0, 0)
    return dist, idx


def knn_classify(
    Q: torch.Tensor, R: torch.Tensor, y: torch.Tensor, k: int, n_classes: int,
    idx_base: int = 0, approx: bool = False
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Fused top-k + uniform vote (labels also returned for sharded merge).

    ``approx=True`` selects the bf16 coarse-pass kernel: 16x the f32 MFMA
    rate, exact-f32 refine over a 16-candidate pool per query (measured
    recall, not proven — see csrc/knn_mfma.hip)."""
    if Q.shape[1] != 12:
        return _cpu.knn_classify(Q, R, y, k, n_classes, idx_base)
    y8 = y.to(torch.uint8).contiguous()
    Qf, Rf = _f32(Q), _f32(R)
    if (approx or R.shape[0] >= _KNN_MFMA_MIN_ROWS) and k <= 8 and n_classes <= 16:
        dist, idx, lab = _ext.knn_topk_mfma(
            Qf, Rf, _knn_cmean(Rf), y8, k, n_classes, idx_base,
            _knn_shards(R.shape[0]), 1 if approx else 0
        )
        return dist, idx, lab
    dist, idx, lab = _ext.knn_topk(Qf, Rf, y8, k, n_classes, idx_base)
    return dist, idx, lab


def svc_predict(
    X: torch.Tensor,
    SV: torch.Tensor,
    dual_coef: torch.Tensor,
    intercept: torch.Tensor,
    n_support: torch.Tensor,
    gamma: float,
    svclass: torch.Tensor = None,
) -> torch.Tensor:
    if X.shape[1] != 12:
        return _cpu.svc_predict(X, SV, dual_coef, intercept, n_support, gamma, svclass)
    if svclass is None:
        # static per model; callers on the hipGraph path precompute it
        # (repeat_interleave is not stream-capture safe)
        svclass = torch.repeat_interleave(
            torch.arange(n_support.numel(), device=X.device), n_support.to(X.device)
        ).to(torch.uint8)
    return _ext.svc_predict(
        _f32(X), _f32_cached(SV), _f32_cached(dual_coef), svclass.contiguous(),
        _f32_cached(intercept), float(gamma)
    )


def rf_hist(bins: torch.Tensor, y: torch.Tensor, nid: torch.Tensor, n_nodes: int,
            n_classes: int, fsel: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Per-node per-feature class histograms.  With ``fsel`` (u8
    [n_nodes,12] mtry mask) only the selected feature planes are scattered
    — 3 atomics per row instead of 12 at sklearn's max_features=3; the
    unselected planes stay zero and must not be read."""
    if bins.shape[1] != 12:  # HIP kernel is specialised for the 12-feature schema
        return _cpu.rf_hist(bins, y, nid, n_nodes, n_classes)
    hist = torch.zeros(n_nodes, 12, 256, n_classes, dtype=torch.int32, device=bins.device)
    _ext.rf_hist(bins.contiguous(), y.contiguous(), nid.contiguous(), hist,
                 None if fsel is None else fsel.contiguous())
    return hist


def rf_pack(forest: Dict[str, torch.Tensor], device) -> Dict[str, torch.Tensor]:
    """Pack the SoA forest (ops.cpu.rf_flatten layout) into the traversal
    kernel's uint2 node format:
      word0 = f32 threshold bits (inner) | leaf-probability row (mixed leaf)
      word1 = (right_child_GLOBAL << 8) | feature byte
    Feature byte: 0..0xef = inner-node feature id; 0xf0|class = PURE leaf
    (one-hot distribution, counted in-register by the kernel); 0xff = mixed
    leaf (distribution row in word0).  Forests up to 1023 trees and 16
    classes; features up to 0xef.
    """
    thr = forest["threshold"].cpu()
    right = forest["right"].cpu().to(torch.int64)
    feat = forest["feature"].cpu().to(torch.int64)
    leaf_index = forest["leaf_index"].cpu().to(torch.int64)
    offsets = forest["tree_offset"].cpu().to(torch.int64)
    n_nodes = thr.numel()
    n_trees = offsets.numel() - 1
    base = torch.repeat_interleave(offsets[:-1], offsets[1:] - offsets[:-1])
    is_leaf = feat < 0
    probs = forest["leaf_proba"].cpu()
    # pure leaf: exactly one class carries the whole mass
    pmax, pcls = probs.max(dim=1)
    leaf_pure = torch.zeros_like(is_leaf)
    leaf_cls = torch.zeros_like(feat)
    valid = leaf_index >= 0
    leaf_pure[valid] = pmax[leaf_index[valid]] >= 1.0
    leaf_cls[valid] = pcls[leaf_index[valid]].to(torch.int64)
    w0 = torch.where(
        is_leaf, leaf_index.to(torch.int32), thr.view(torch.int32)
    ).to(torch.int32)
    right_global = (right + base) << 8
    leaf_byte = torch.where(leaf_pure, 0xF0 | leaf_cls, torch.tensor(0xFF, dtype=torch.int64))
    w1 = torch.where(is_leaf, leaf_byte, right_global | feat).to(torch.int32)
    nodes = torch.stack([w0, w1], dim=1).contiguous()
    return {
        "nodes": nodes.to(device),
        "roots": offsets[:-1].to(torch.int32).to(device),
        "leaf_proba": forest["leaf_proba"].to(torch.float32).to(device).contiguous(),
        "n_leaves": int(forest["leaf_proba"].shape[0]),
        "n_classes": int(forest["n_classes"]),
    }


def rf_argmax(X: torch.Tensor, forest: Dict[str, torch.Tensor]) -> torch.Tensor:
    packed = forest.get("_packed")
    if packed is None or packed["nodes"].device != X.device:
        packed = rf_pack(forest, X.device)
        forest["_packed"] = packed
    return _ext.rf_predict(
        _f32(X),
        packed["nodes"],
        packed["roots"],
        packed["leaf_proba"],
        packed["n_leaves"],
        packed["n_classes"],
    )


# ----------------------------------------------------------------------
# fit ops
# ----------------------------------------------------------------------


def gnb_fit_stats(
    X: torch.Tensor, y: torch.Tensor, n_classes: int
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    if X.shape[1] != 12:
        return _cpu.gnb_fit_stats(X, y, n_classes)
    count, s, sq = _ext.gnb_fit_stats(_f64(X), y.to(torch.int64).contiguous(), n_classes)
    return count.to(X.dtype), s.to(X.dtype), sq.to(X.dtype)


def logistic_loss_grad(
    X: torch.Tensor,
    y: torch.Tensor,
    coef: torch.Tensor,
    intercept: torch.Tensor,
    l2: float = 1.0,
    sample_range=None,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    if X.shape[1] != 12:  # e.g. the PCA-projection fit in analysis.py
        return _cpu.logistic_loss_grad(X, y, coef, intercept, l2=l2)
    grad, loss = _ext.logistic_grad(
        _f64(X), y.to(torch.int64).contiguous(), _f64(coef), _f64(intercept)
    )
    loss = loss[0] + 0.5 * l2 * (coef.double() * coef.double()).sum()
    g_coef = grad[:, :12] + l2 * coef.double()
    g_b = grad[:, 12]
    return loss.to(X.dtype), g_coef.to(X.dtype), g_b.to(X.dtype)


def flow_features(cur: torch.Tensor, prev: torch.Tensor, times: torch.Tensor) -> torch.Tensor:
    return _ext.flow_features(_f64(cur), _f64(prev), _f64(times))


def kmeans_labels(X: torch.Tensor, centers: torch.Tensor) -> torch.Tensor:
    if X.shape[1] != 12:
        return _cpu.kmeans_labels(X, centers)
    labels, _, _, _ = _ext.kmeans_assign(_f32(X), _f32_cached(centers), False)
    return labels


# diagnostic-only path (torch eager; not benched)
rf_predict_proba = _cpu.rf_predict_proba
