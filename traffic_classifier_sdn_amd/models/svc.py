"""RBF-kernel support-vector classifier (reference estimator N2).

predict: fused RBF-Gram + one-vs-one vote op on the libsvm checkpoint layout
(support vectors grouped by class, ``dual_coef`` (C-1, nSV), 15 OVO
intercepts — SURVEY.md §2.3).

fit: one-vs-one SMO (libsvm WSS-1 working-set selection, analytic pair
update) over precomputed kernel blocks — the per-pair subproblems are a few
thousand rows for the reference dataset.  Kernel-matrix tiles are computed by
the distance-GEMM op (MFMA on GPU).  Distributed fit shards rows and
all-reduces kernel-row partial sums (planned; single-process fit is complete).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch

from .. import ops
from .base import ArrayLike, Estimator, as_tensor, encode_labels


def _smo_binary(K: torch.Tensor, y: torch.Tensor, C: float, tol: float, max_iter: int):
    """libsvm-style SMO for one binary subproblem.

    K: (n, n) kernel matrix (f64), y: (n,) in {-1, +1} (f64).
    Returns (alpha, b).  Working-set selection is the maximal-violating-pair
    rule; the pair update is the analytic two-variable solution.
    """
    n = K.shape[0]
    alpha = torch.zeros(n, dtype=K.dtype, device=K.device)
    grad = -torch.ones(n, dtype=K.dtype, device=K.device)  # G = Q@alpha - 1
    Qdiag = torch.diagonal(K).clone()  # y_i^2 * K_ii
    minus_yG = None
    for _ in range(max_iter):
        minus_yG = -y * grad
        up = ((y > 0) & (alpha < C)) | ((y < 0) & (alpha > 0))
        low = ((y > 0) & (alpha > 0)) | ((y < 0) & (alpha < C))
        m_up = torch.where(up, minus_yG, torch.tensor(-np.inf, dtype=K.dtype, device=K.device))
        m_low = torch.where(low, minus_yG, torch.tensor(np.inf, dtype=K.dtype, device=K.device))
        i = int(torch.argmax(m_up))
        j = int(torch.argmin(m_low))
        if float(m_up[i]) - float(m_low[j]) < tol:
            break
        yi, yj = float(y[i]), float(y[j])
        Qi = y[i] * y * K[i]  # row i of Q
        Qj = y[j] * y * K[j]
        a = float(Qdiag[i] + Qdiag[j] - 2.0 * yi * yj * K[i, j])
        if a <= 0:
            a = 1e-12
        b_ = float(m_up[i] - m_low[j])
        # step along the pair direction, then clip to the box
        d = b_ / a
        ai_old, aj_old = float(alpha[i]), float(alpha[j])
        ai = ai_old + yi * d
        aj = aj_old - yj * d
        # clip: keep yi*ai + yj*aj constant
        s = yi * ai_old + yj * aj_old
        ai = min(max(ai, 0.0), C)
        aj = yj * (s - yi * ai)
        aj = min(max(aj, 0.0), C)
        ai = yi * (s - yj * aj)
        ai = min(max(ai, 0.0), C)
        dai, daj = ai - ai_old, aj - aj_old
        if abs(dai) < 1e-16 and abs(daj) < 1e-16:
            break
        alpha[i] = ai
        alpha[j] = aj
        grad += Qi * dai + Qj * daj
    # intercept: average -y*G over free vectors, else midpoint of bounds
    minus_yG = -y * grad
    free = (alpha > 1e-12) & (alpha < C - 1e-12)
    if bool(free.any()):
        rho = -float(minus_yG[free].mean())
    else:
        up = ((y > 0) & (alpha < C)) | ((y < 0) & (alpha > 0))
        low = ((y > 0) & (alpha > 0)) | ((y < 0) & (alpha < C))
        hi = float(torch.where(up, minus_yG, torch.tensor(-np.inf, dtype=K.dtype)).max())
        lo = float(torch.where(low, minus_yG, torch.tensor(np.inf, dtype=K.dtype)).min())
        rho = -(hi + lo) / 2.0
    return alpha, -rho  # decision uses +b with b = -rho


class SVC(Estimator):
    kind = "svc"

    def __init__(
        self,
        C: float = 1.0,
        gamma: str | float = "scale",
        tol: float = 1e-3,
        max_iter: int = 200_000,
        device: Optional[str] = None,
    ):
        super().__init__(device)
        self.C = C
        self.gamma = gamma
        self.tol = tol
        self.max_iter = max_iter
        self.support_vectors_: Optional[torch.Tensor] = None
        self.dual_coef_: Optional[torch.Tensor] = None
        self.intercept_: Optional[torch.Tensor] = None
        self.n_support_: Optional[torch.Tensor] = None
        self.gamma_: float = 0.0

    def fit(self, X: ArrayLike, y: ArrayLike):
        Xt = as_tensor(X, self.device, torch.float64)
        self.classes_, y_idx = encode_labels(y)
        y_t = y_idx.to(self.device)
        C_cls = len(self.classes_)
        n, F = Xt.shape
        if self.gamma == "scale":
            xv = float(Xt.var(unbiased=False))
            self.gamma_ = 1.0 / (F * xv) if xv > 0 else 1.0
        elif self.gamma == "auto":
            self.gamma_ = 1.0 / F
        else:
            self.gamma_ = float(self.gamma)

        # per-pair SMO; remember alphas per (pair, global row)
        alphas = {}
        intercepts = []
        for i in range(C_cls):
            for j in range(i + 1, C_cls):
                sel = (y_t == i) | (y_t == j)
                idx = torch.nonzero(sel, as_tuple=False).squeeze(1)
                Xp = Xt[idx]
                yp = torch.where(y_t[idx] == i, 1.0, -1.0).to(Xt.dtype)
                Kp = ops.rbf_kernel(Xp, Xp, self.gamma_)
                a, b = _smo_binary(Kp, yp, self.C, self.tol, self.max_iter)
                alphas[(i, j)] = (idx, a * yp)  # signed coefficients
                intercepts.append(b)

        # assemble libsvm layout: SVs = rows with any nonzero coef, grouped
        # by class in class order; dual_coef (C-1, nSV)
        is_sv = torch.zeros(n, dtype=torch.bool)
        for (i, j), (idx, sc) in alphas.items():
            nz = sc.abs() > 1e-12
            is_sv[idx[nz].cpu()] = True
        order = []
        n_support = []
        for c in range(C_cls):
            rows = torch.nonzero((y_t.cpu() == c) & is_sv, as_tuple=False).squeeze(1)
            order.append(rows)
            n_support.append(int(rows.numel()))
        order_t = torch.cat(order)
        nSV = int(order_t.numel())
        pos_of = {int(r): p for p, r in enumerate(order_t.tolist())}
        dual = torch.zeros(C_cls - 1, nSV, dtype=torch.float64)
        for (i, j), (idx, sc) in alphas.items():
            for local, g in enumerate(idx.cpu().tolist()):
                v = float(sc[local])
                if abs(v) <= 1e-12 or g not in pos_of:
                    continue
                c = int(y_t[g])  # class of this SV: i or j
                other = j if c == i else i
                row = other if other < c else other - 1
                dual[row, pos_of[g]] = v
        self.support_ = order_t.to(torch.int64)
        self.support_vectors_ = Xt[order_t.to(Xt.device)].contiguous()
        self.dual_coef_ = dual.to(self.device)
        self.intercept_ = torch.tensor(intercepts, dtype=torch.float64, device=self.device)
        self.n_support_ = torch.tensor(n_support, dtype=torch.int64, device=self.device)
        return self

    def predict_index(self, X: ArrayLike) -> torch.Tensor:
        Xt = as_tensor(X, self.device, torch.float32)
        return ops.svc_predict(
            Xt,
            self.support_vectors_.to(Xt.dtype),
            self.dual_coef_.to(Xt.dtype),
            self.intercept_.to(Xt.dtype),
            self.n_support_,
            self.gamma_,
        )

    def decision_function_ovo(self, X: ArrayLike) -> torch.Tensor:
        Xt = as_tensor(X, self.device, torch.float64)
        K = ops.rbf_kernel(Xt, self.support_vectors_.to(Xt.dtype), self.gamma_)
        return ops.svc_ovo_decision(
            K, self.dual_coef_.double(), self.intercept_.double(), self.n_support_
        )

    # -- checkpointing -------------------------------------------------
    def to_params(self) -> Dict[str, Any]:
        return {
            "kind": self.kind,
            "classes": np.asarray(self.classes_, dtype=object),
            "support_vectors": self.support_vectors_.double().cpu().numpy(),
            "dual_coef": self.dual_coef_.double().cpu().numpy(),
            "intercept": self.intercept_.double().cpu().numpy(),
            "n_support": self.n_support_.cpu().numpy(),
            "gamma": float(self.gamma_),
            "support": getattr(self, "support_", torch.zeros(0, dtype=torch.int64)).cpu().numpy(),
        }

    @classmethod
    def from_params(cls, params: Dict[str, Any], device: Optional[str] = None):
        m = cls(device=device)
        m.classes_ = np.asarray([str(c) for c in params["classes"]], dtype=object)
        m.support_vectors_ = torch.as_tensor(np.asarray(params["support_vectors"], dtype=np.float64)).to(m.device)
        m.dual_coef_ = torch.as_tensor(np.asarray(params["dual_coef"], dtype=np.float64)).to(m.device)
        m.intercept_ = torch.as_tensor(np.asarray(params["intercept"], dtype=np.float64)).to(m.device)
        m.n_support_ = torch.as_tensor(np.asarray(params["n_support"], dtype=np.int64)).to(m.device)
        m.gamma_ = float(params["gamma"])
        return m
