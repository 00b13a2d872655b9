"""KMeans clustering (reference estimator N6).

Lloyd iterations with kmeans++ seeding and n_init restarts (sklearn 1.0.1
``KMeans`` behaviour; the shipped checkpoint converged in 13 elkan iters —
Lloyd and elkan produce the same fixed points, SURVEY.md §7 hard parts).
The assignment step is the fused distance+argmin+partial-sum op (MFMA distance
GEMM on GPU); per-iteration (cluster_size, cluster_sum) are all-reduced for
data-parallel fit.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch

from .. import ops
from ..parallel import dist
from .base import ArrayLike, Estimator, as_tensor


class KMeans(Estimator):
    kind = "kmeans"

    def __init__(
        self,
        n_clusters: int = 6,
        n_init: int = 10,
        max_iter: int = 300,
        tol: float = 1e-4,
        seed: Optional[int] = 0,
        device: Optional[str] = None,
    ):
        super().__init__(device)
        self.n_clusters = n_clusters
        self.n_init = n_init
        self.max_iter = max_iter
        self.tol = tol
        self.seed = seed
        self.cluster_centers_: Optional[torch.Tensor] = None
        self.inertia_: float = float("inf")
        self.n_iter_: int = 0

    # -- seeding -------------------------------------------------------
    def _kmeanspp(self, X: torch.Tensor, rng: np.random.Generator) -> torch.Tensor:
        """kmeans++ with n_local_trials (sklearn's greedy variant)."""
        n = X.shape[0]
        k = self.n_clusters
        n_trials = 2 + int(np.log(k))
        first = int(rng.integers(n))
        centers = [X[first]]
        closest = ops.pairwise_sqdist(X, X[first].unsqueeze(0)).squeeze(1)
        for _ in range(1, k):
            total = float(closest.sum())
            if total == 0.0:
                cand = torch.as_tensor(rng.integers(n, size=n_trials))
            else:
                r = rng.random(n_trials) * total
                cum = torch.cumsum(closest, dim=0).cpu().numpy()
                cand = torch.as_tensor(np.searchsorted(cum, r).clip(0, n - 1))
            dc = ops.pairwise_sqdist(X, X[cand.to(X.device)])
            pot = torch.minimum(closest.unsqueeze(1), dc).sum(dim=0)
            best = int(torch.argmin(pot))
            centers.append(X[int(cand[best])])
            closest = torch.minimum(closest, dc[:, best])
        return torch.stack(centers)

    def _lloyd(self, X: torch.Tensor, centers: torch.Tensor):
        prev_shift = None
        n_iter = 0
        inertia = float("inf")
        for it in range(self.max_iter):
            labels, counts, sums, inert = ops.kmeans_assign(X, centers)
            dist.allreduce_flat([counts, sums, inert.reshape(1)])
            inertia = float(inert)
            new_centers = torch.where(
                counts.unsqueeze(1) > 0, sums / counts.clamp(min=1.0).unsqueeze(1), centers
            )
            shift = float(((new_centers - centers) ** 2).sum())
            centers = new_centers
            n_iter = it + 1
            if shift <= self.tol:
                break
        labels, _, _, inert = ops.kmeans_assign(X, centers)
        dist.allreduce_flat([inert.reshape(1)])
        return centers, labels, float(inert), n_iter

    def fit(self, X: ArrayLike, y: ArrayLike = None, sharded: bool = False):
        Xt = as_tensor(X, self.device, torch.float64)
        rng = np.random.default_rng(self.seed)
        # sklearn tol is relative to the mean feature variance of X
        var = Xt.var(dim=0, unbiased=False).mean()
        tol_abs = float(self.tol * var)
        best = None
        for init in range(self.n_init):
            if dist.is_initialized():
                # seed on rank 0's shard, broadcast (all ranks iterate the
                # same centers over their own row shards)
                centers = self._kmeanspp(Xt, rng)
                dist.broadcast_(centers)
            else:
                centers = self._kmeanspp(Xt, rng)
            saved_tol = self.tol
            self.tol = tol_abs
            centers, labels, inertia, n_iter = self._lloyd(Xt, centers)
            self.tol = saved_tol
            if best is None or inertia < best[2]:
                best = (centers, labels, inertia, n_iter)
        self.cluster_centers_ = best[0]
        self.labels_ = best[1]
        self.inertia_ = best[2]
        self.n_iter_ = best[3]
        self.classes_ = None  # unsupervised: predictions are cluster ids
        return self

    def fit_predict(self, X: ArrayLike, y: ArrayLike = None):
        self.fit(X, y)
        return self.labels_.cpu().numpy()

    def predict_index(self, X: ArrayLike) -> torch.Tensor:
        Xt = as_tensor(X, self.device, torch.float32)
        return ops.kmeans_labels(Xt, self.cluster_centers_)

    # -- checkpointing -------------------------------------------------
    def to_params(self) -> Dict[str, Any]:
        p = {
            "kind": self.kind,
            "centers": self.cluster_centers_.double().cpu().numpy(),
            "n_clusters": int(self.n_clusters),
            "inertia": float(self.inertia_),
            "n_iter": int(self.n_iter_),
        }
        if getattr(self, "cluster_label_names_", None) is not None:
            # mode-based cluster -> class-name map (fit.py, nb1 cell 125)
            p["cluster_label_names"] = np.asarray(self.cluster_label_names_, dtype=object)
        return p

    @classmethod
    def from_params(cls, params: Dict[str, Any], device: Optional[str] = None):
        centers = np.asarray(params["centers"], dtype=np.float64)
        m = cls(n_clusters=int(params.get("n_clusters", centers.shape[0])), device=device)
        m.cluster_centers_ = torch.as_tensor(centers).to(m.device)
        m.inertia_ = float(params.get("inertia", 0.0))
        m.n_iter_ = int(params.get("n_iter", 0))
        if params.get("cluster_label_names") is not None:
            m.cluster_label_names_ = np.asarray(params["cluster_label_names"], dtype=object)
        return m
