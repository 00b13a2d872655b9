"""Multinomial logistic regression (reference estimator N1).

Matches sklearn 1.0.1 ``LogisticRegression`` defaults (l2, C=1.0, lbfgs,
max_iter=100, multinomial for >=3 classes) closely enough to reproduce the
reference's 96.47% 6-class accuracy.  The loss/gradient is the
framework op :func:`ops.logistic_loss_grad` (fused HIP kernel on GPU); the
L-BFGS driver is scipy on the host, with per-step gradient all-reduce for
data-parallel fit.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch

from .. import ops
from ..parallel import dist
from .base import ArrayLike, Estimator, as_tensor, encode_labels


class LogisticRegression(Estimator):
    kind = "logistic"

    def __init__(self, C: float = 1.0, max_iter: int = 100, tol: float = 1e-4, device: Optional[str] = None):
        super().__init__(device)
        self.C = C
        self.max_iter = max_iter
        self.tol = tol
        self.coef_: Optional[torch.Tensor] = None
        self.intercept_: Optional[torch.Tensor] = None

    def fit(self, X: ArrayLike, y: ArrayLike, sharded: bool = False):
        """Fit with scipy L-BFGS on the device-computed loss/grad.

        ``sharded=True`` means X/y are this rank's row shard; gradients and
        loss are summed across ranks each L-BFGS step (one fused all-reduce).
        """
        from scipy.optimize import minimize

        Xt = as_tensor(X, self.device, torch.float64)
        self.classes_, y_idx = encode_labels(y)
        y_t = y_idx.to(self.device)
        n_classes = len(self.classes_)
        F = Xt.shape[1]
        l2 = 1.0 / self.C

        def fun(w: np.ndarray):
            wt = torch.from_numpy(w.reshape(n_classes, F + 1)).to(self.device)
            coef = wt[:, :F].contiguous()
            b = wt[:, F].contiguous()
            loss, g_coef, g_b = ops.logistic_loss_grad(Xt, y_t, coef, b, l2=l2)
            if dist.is_initialized():
                # the l2 term is replicated on every rank: after the summed
                # all-reduce keep exactly one copy of it
                w_ = dist.world_size()
                reg_l = 0.5 * l2 * float((coef * coef).sum())
                buf = [loss.reshape(1).clone(), g_coef, g_b]
                dist.allreduce_flat(buf)
                loss = buf[0][0] - (w_ - 1) * reg_l
                g_coef = buf[1] - (w_ - 1) * l2 * coef
                g_b = buf[2]
            g = torch.cat([g_coef, g_b.unsqueeze(1)], dim=1)
            return float(loss), g.cpu().numpy().ravel()

        w0 = np.zeros(n_classes * (F + 1))
        res = minimize(
            fun,
            w0,
            jac=True,
            method="L-BFGS-B",
            options={"maxiter": self.max_iter, "gtol": self.tol, "maxls": 50},
        )
        wt = torch.from_numpy(res.x.reshape(n_classes, F + 1))
        self.coef_ = wt[:, :F].to(self.device, torch.float32).contiguous()
        self.intercept_ = wt[:, F].to(self.device, torch.float32).contiguous()
        self._coef64 = wt[:, :F].contiguous()
        self._intercept64 = wt[:, F].contiguous()
        self.n_iter_ = int(res.nit)
        return self

    def predict_index(self, X: ArrayLike) -> torch.Tensor:
        Xt = as_tensor(X, self.device, torch.float32)
        return ops.linear_argmax(Xt, self.coef_, self.intercept_)

    def decision_function(self, X: ArrayLike) -> torch.Tensor:
        Xt = as_tensor(X, self.device, torch.float32)
        return ops.linear_logits(Xt, self.coef_, self.intercept_)

    def predict_proba(self, X: ArrayLike) -> np.ndarray:
        """Softmax class probabilities (sklearn LogisticRegression API)."""
        logits = self.decision_function(X)
        return torch.softmax(logits.double(), dim=1).cpu().numpy()

    # -- checkpointing -------------------------------------------------
    def to_params(self) -> Dict[str, Any]:
        coef = getattr(self, "_coef64", self.coef_.double()).cpu().numpy()
        intercept = getattr(self, "_intercept64", self.intercept_.double()).cpu().numpy()
        return {
            "kind": self.kind,
            "classes": np.asarray(self.classes_, dtype=object),
            "coef": np.asarray(coef, dtype=np.float64),
            "intercept": np.asarray(intercept, dtype=np.float64),
        }

    @classmethod
    def from_params(cls, params: Dict[str, Any], device: Optional[str] = None):
        m = cls(device=device)
        m.classes_ = np.asarray([str(c) for c in params["classes"]], dtype=object)
        m.coef_ = torch.as_tensor(np.asarray(params["coef"], dtype=np.float32)).to(m.device)
        m.intercept_ = torch.as_tensor(np.asarray(params["intercept"], dtype=np.float32)).to(m.device)
        return m
