"""Gaussian naive Bayes (reference estimator N5).

fit = per-class (count, sum, sum-of-squares) sufficient statistics — a
segmented reduction op, all-reduced across ranks for data-parallel fit —
followed by the sklearn variance smoothing (epsilon = 1e-9 * max feature
variance, sklearn 1.0.1 GaussianNB semantics; checkpoint shapes per
SURVEY.md §2.3).  predict = fused joint-log-likelihood + argmax op.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch

from .. import ops
from ..parallel import dist
from .base import ArrayLike, Estimator, as_tensor, encode_labels


class GaussianNB(Estimator):
    kind = "gaussian_nb"

    def __init__(self, var_smoothing: float = 1e-9, device: Optional[str] = None):
        super().__init__(device)
        self.var_smoothing = var_smoothing
        self.theta_: Optional[torch.Tensor] = None
        self.var_: Optional[torch.Tensor] = None
        self.class_prior_: Optional[torch.Tensor] = None
        self.class_count_: Optional[torch.Tensor] = None
        self.epsilon_: float = 0.0

    def fit(self, X: ArrayLike, y: ArrayLike, sharded: bool = False):
        Xt = as_tensor(X, self.device, torch.float64)
        if sharded and dist.is_initialized():
            classes_local = np.unique(np.asarray(y).ravel())
            # label space must be global: gather class sets via object list
            all_classes = [None] * dist.world_size()
            torch.distributed.all_gather_object(all_classes, list(classes_local))
            self.classes_ = np.unique(np.concatenate([np.asarray(c) for c in all_classes])).astype(object)
            lut = {c: i for i, c in enumerate(self.classes_)}
            y_idx = torch.tensor([lut[v] for v in np.asarray(y).ravel()], dtype=torch.int64)
        else:
            self.classes_, y_idx = encode_labels(y)
        y_t = y_idx.to(self.device)
        C = len(self.classes_)
        count, s, sq = ops.gnb_fit_stats(Xt, y_t, C)
        # global max-variance for epsilon needs the global moments, so reduce
        # the sufficient stats first (one fused all-reduce)
        dist.allreduce_flat([count, s, sq])
        n_total = count.sum()
        theta = s / count.unsqueeze(1)
        var = sq / count.unsqueeze(1) - theta * theta
        # sklearn: epsilon_ = var_smoothing * max over features of Var(X)
        gx_sum = s.sum(dim=0)
        gx_sq = sq.sum(dim=0)
        global_var = gx_sq / n_total - (gx_sum / n_total) ** 2
        self.epsilon_ = float(self.var_smoothing * global_var.max())
        var = var + self.epsilon_
        self.theta_ = theta
        self.var_ = var
        self.class_count_ = count
        self.class_prior_ = count / n_total
        return self

    def predict_index(self, X: ArrayLike) -> torch.Tensor:
        Xt = as_tensor(X, self.device, torch.float32)
        return ops.gnb_argmax(
            Xt,
            self.theta_,
            self.var_,
            self.class_prior_,
        )

    def joint_log_likelihood(self, X: ArrayLike) -> torch.Tensor:
        Xt = as_tensor(X, self.device, torch.float64)
        return ops.gnb_joint_loglik(Xt, self.theta_, self.var_, self.class_prior_)

    def predict_proba(self, X: ArrayLike) -> np.ndarray:
        """exp-normalised joint log-likelihoods (sklearn GaussianNB API)."""
        Xt = as_tensor(X, self.device, torch.float64)
        ll = ops.gnb_joint_loglik(Xt, self.theta_, self.var_, self.class_prior_)
        return torch.softmax(ll, dim=1).cpu().numpy()

    # -- checkpointing -------------------------------------------------
    def to_params(self) -> Dict[str, Any]:
        return {
            "kind": self.kind,
            "classes": np.asarray(self.classes_, dtype=object),
            "theta": self.theta_.double().cpu().numpy(),
            "var": self.var_.double().cpu().numpy(),
            "class_prior": self.class_prior_.double().cpu().numpy(),
            "class_count": self.class_count_.double().cpu().numpy(),
            "epsilon": float(self.epsilon_),
        }

    @classmethod
    def from_params(cls, params: Dict[str, Any], device: Optional[str] = None):
        m = cls(device=device)
        m.classes_ = np.asarray([str(c) for c in params["classes"]], dtype=object)
        to = lambda k: torch.as_tensor(np.asarray(params[k], dtype=np.float64)).to(m.device)
        m.theta_ = to("theta")
        m.var_ = to("var")
        m.class_prior_ = to("class_prior")
        m.class_count_ = to("class_count")
        m.epsilon_ = float(params["epsilon"])
        return m
