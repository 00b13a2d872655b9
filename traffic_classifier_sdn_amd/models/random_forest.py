"""Random forest classifier (reference estimator N4) — the flagship model.

predict is the framework's headline benchmark path (BASELINE.json config #2:
100-tree 6-class predict on 10M synthetic flow rows per MI355X): the packed
forest (8-byte nodes + leaf-probability table, ``ops.cpu.rf_flatten``) lives
in LDS on the GPU and every wavefront lane walks all trees for its rows.

fit is a depth-first exact-gini builder matching sklearn 1.0.1 defaults
(bootstrap, max_features=sqrt(F)->3, min_samples_split=2, min_samples_leaf=1,
no depth cap), producing trees with the same left_child==index+1 invariant
the packed layout relies on.  Distributed fit trains disjoint tree subsets
per rank on the broadcast dataset and all-gathers the trees (forests are
embarrassingly parallel across trees; per-node histogram all-reduce is the
row-sharded alternative for >HBM datasets).
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

import numpy as np
import torch

from .. import ops
from ..parallel import dist
from .base import ArrayLike, Estimator, as_tensor, encode_labels


def _build_tree(
    X: np.ndarray,
    y: np.ndarray,
    n_classes: int,
    rng: np.random.Generator,
    max_features: int,
    min_samples_split: int = 2,
    min_samples_leaf: int = 1,
    max_depth: int = 10**9,
) -> Dict[str, np.ndarray]:
    """Depth-first exact best-split (gini) CART builder.

    Emits nodes in sklearn's depth-first order (left child == parent+1) so
    the packed GPU layout applies directly.
    """
    n, F = X.shape
    left: List[int] = []
    right: List[int] = []
    feature: List[int] = []
    threshold: List[float] = []
    values: List[np.ndarray] = []

    def gini_counts(counts: np.ndarray) -> float:
        tot = counts.sum()
        if tot == 0:
            return 0.0
        p = counts / tot
        return 1.0 - float((p * p).sum())

    def new_node() -> int:
        left.append(-1)
        right.append(-1)
        feature.append(-1)
        threshold.append(0.0)
        values.append(None)  # filled on finalize
        return len(left) - 1

    def build(rows: np.ndarray, depth: int) -> int:
        node = new_node()
        counts = np.bincount(y[rows], minlength=n_classes).astype(np.float64)
        values[node] = counts
        n_rows = rows.shape[0]
        if (
            n_rows < min_samples_split
            or depth >= max_depth
            or (counts > 0).sum() <= 1
        ):
            return node
        best = None  # (impurity_sum, feat, thr, sorted_rows, split_pos)
        feats = rng.permutation(F)
        tried = 0
        for f in feats:
            # sklearn semantics: draw max_features candidates, but keep
            # searching past constant features until a valid split exists
            if tried >= max_features and best is not None:
                break
            tried += 1
            xv = X[rows, f]
            order = np.argsort(xv, kind="stable")
            xs = xv[order]
            if xs[0] == xs[-1]:
                continue
            ys = y[rows[order]]
            # cumulative class counts at each boundary
            onehot = np.zeros((n_rows, n_classes), dtype=np.float64)
            onehot[np.arange(n_rows), ys] = 1.0
            cum = np.cumsum(onehot, axis=0)
            # valid split positions: between distinct adjacent values, and
            # respecting min_samples_leaf
            pos = np.nonzero(xs[1:] != xs[:-1])[0]  # split after index p
            pos = pos[(pos + 1 >= min_samples_leaf) & (n_rows - pos - 1 >= min_samples_leaf)]
            if pos.size == 0:
                continue
            lc = cum[pos]  # (P, C) left counts
            tot = cum[-1]
            rc = tot - lc
            nl = lc.sum(axis=1)
            nr = rc.sum(axis=1)
            gl = 1.0 - ((lc / nl[:, None]) ** 2).sum(axis=1)
            gr = 1.0 - ((rc / nr[:, None]) ** 2).sum(axis=1)
            imp = (nl * gl + nr * gr) / n_rows
            bi = int(np.argmin(imp))
            if best is None or imp[bi] < best[0] - 1e-15:
                thr = (xs[pos[bi]] + xs[pos[bi] + 1]) / 2.0
                best = (float(imp[bi]), int(f), thr, rows[order], int(pos[bi]) + 1)
        if best is None:
            return node
        _, f, thr, sorted_rows, split = best
        feature[node] = f
        threshold[node] = thr
        l = build(sorted_rows[:split], depth + 1)
        left[node] = l
        r = build(sorted_rows[split:], depth + 1)
        right[node] = r
        return node

    root = build(np.arange(n), 0)
    assert root == 0
    return {
        "left": np.asarray(left, dtype=np.int32),
        "right": np.asarray(right, dtype=np.int32),
        "feature": np.asarray(feature, dtype=np.int32),
        "threshold": np.asarray(threshold, dtype=np.float64),
        "values": np.stack(values).astype(np.float64),
    }


class RandomForestClassifier(Estimator):
    kind = "random_forest"

    def __init__(
        self,
        n_estimators: int = 100,
        max_features: str | int = "sqrt",
        bootstrap: bool = True,
        seed: Optional[int] = 0,
        device: Optional[str] = None,
        builder: str = "auto",
    ):
        super().__init__(device)
        self.n_estimators = n_estimators
        self.max_features = max_features
        self.bootstrap = bootstrap
        self.seed = seed
        # "exact": depth-first exact-gini CPU builder (sklearn semantics);
        # "hist": level-synchronous 256-bin histogram builder (HIP scatter
        # kernel on GPU, models/rf_hist_fit.py); "auto" = hist on CUDA
        self.builder = builder
        self.trees_: Optional[List[Dict[str, np.ndarray]]] = None
        self._forest = None  # packed SoA (lazy)

    def fit(self, X: ArrayLike, y: ArrayLike):
        Xn = np.asarray(as_tensor(X, torch.device("cpu"), torch.float64).numpy())
        self.classes_, y_idx = encode_labels(y)
        yn = y_idx.numpy().astype(np.int64)
        C = len(self.classes_)
        F = Xn.shape[1]
        mf = (
            max(1, int(np.sqrt(F)))
            if self.max_features in ("sqrt", "auto")
            else int(self.max_features)
        )
        builder = self.builder
        if builder == "auto":
            builder = "hist" if self.device.type == "cuda" else "exact"
        # tree-parallel across ranks: rank r builds trees r, r+W, r+2W, ...
        W, R = dist.world_size(), dist.rank()
        my_trees = []
        if builder == "hist":
            from .rf_hist_fit import build_forest_hist

            Xt = as_tensor(X, self.device, torch.float32)
            yt = y_idx.to(self.device)
            ids = list(range(R, self.n_estimators, W))
            built = build_forest_hist(
                Xt, yt, C, self.n_estimators, max_features=mf,
                seed=self.seed, bootstrap=self.bootstrap, tree_range=ids,
            )
            my_trees = list(zip(ids, built))
        else:
            for t in range(R, self.n_estimators, W):
                rng = np.random.default_rng(None if self.seed is None else self.seed + t)
                if self.bootstrap:
                    rows = rng.integers(0, Xn.shape[0], size=Xn.shape[0])
                    Xb, yb = Xn[rows], yn[rows]
                else:
                    Xb, yb = Xn, yn
                my_trees.append((t, _build_tree(Xb, yb, C, rng, mf)))
        if dist.is_initialized():
            gathered: List[List] = [None] * W
            torch.distributed.all_gather_object(gathered, my_trees)
            all_trees = [t for part in gathered for t in part]
        else:
            all_trees = my_trees
        all_trees.sort(key=lambda p: p[0])
        self.trees_ = [t for _, t in all_trees]
        self._forest = None
        return self

    @property
    def forest(self):
        if self._forest is None:
            C = len(self.classes_)
            self._forest = ops.cpu.rf_flatten(self.trees_, C)
            self._forest = {
                k: (v.to(self.device) if isinstance(v, torch.Tensor) else v)
                for k, v in self._forest.items()
            }
        return self._forest

    def predict_index(self, X: ArrayLike) -> torch.Tensor:
        Xt = as_tensor(X, self.device, torch.float32)
        return ops.rf_argmax(Xt, self.forest)

    def predict_proba(self, X: ArrayLike) -> torch.Tensor:
        Xt = as_tensor(X, self.device, torch.float32)
        return ops.rf_predict_proba(Xt, self.forest)

    # -- checkpointing -------------------------------------------------
    def to_params(self) -> Dict[str, Any]:
        return {
            "kind": self.kind,
            "classes": np.asarray(self.classes_, dtype=object),
            "n_classes": len(self.classes_),
            "trees": self.trees_,
        }

    @classmethod
    def from_params(cls, params: Dict[str, Any], device: Optional[str] = None):
        m = cls(n_estimators=len(params["trees"]), device=device)
        m.classes_ = np.asarray([str(c) for c in params["classes"]], dtype=object)
        m.trees_ = [
            {k: np.asarray(v) for k, v in t.items()} for t in params["trees"]
        ]
        return m
