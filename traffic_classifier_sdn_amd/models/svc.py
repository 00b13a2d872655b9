"""RBF-kernel support-vector classifier (reference estimator N2).

predict: fused RBF-Gram + one-vs-one vote op on the libsvm checkpoint layout
(support vectors grouped by class, ``dual_coef`` (C-1, nSV), 15 OVO
intercepts — SURVEY.md §2.3).

fit: one-vs-one SMO (libsvm WSS-1 working-set selection, analytic pair
update) with on-the-fly fused kernel rows — no Gram matrix is materialised,
so the fit scales to the BASELINE 1M-row config; rows shard across ranks
with one candidate all-gather per iteration (models/svc_fit.py).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch

from .. import ops
from ..parallel import dist
from .base import ArrayLike, Estimator, as_tensor, encode_labels
from .svc_fit import smo_fit_pair


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

    def fit(self, X: ArrayLike, y: ArrayLike, sharded: bool = False):
        """One-vs-one SMO fit (models/svc_fit.py: fused on-the-fly kernel
        rows, HIP kernels on GPU, row-sharded across ranks when ``sharded``
        with one candidate all-gather per iteration)."""
        Xt = as_tensor(X, self.device, torch.float32)
        n, F = Xt.shape
        if sharded and dist.is_initialized():
            classes_local = np.unique(np.asarray(y).ravel())
            all_classes = [None] * dist.world_size()
            torch.distributed.all_gather_object(all_classes, list(classes_local))
            self.classes_ = np.unique(
                np.concatenate([np.asarray(c) for c in all_classes])
            ).astype(object)
            lut = {c: i for i, c in enumerate(self.classes_)}
            y_idx = torch.tensor([lut[v] for v in np.asarray(y).ravel()], dtype=torch.int64)
        else:
            self.classes_, y_idx = encode_labels(y)
        y_t = y_idx.to(self.device)
        C_cls = len(self.classes_)

        if self.gamma == "scale":
            # sklearn: 1 / (F * Var(all elements of X)); global when sharded
            # device tensor: NCCL collectives reject CPU tensors
            st = torch.tensor(
                [float(Xt.numel()), float(Xt.double().sum()), float((Xt.double() ** 2).sum())],
                dtype=torch.float64, device=self.device,
            )
            dist.allreduce_(st)
            xv = float(st[2] / st[0] - (st[1] / st[0]) ** 2)
            self.gamma_ = 1.0 / (F * xv) if xv > 0 else 1.0
        elif self.gamma == "auto":
            self.gamma_ = 1.0 / F
        else:
            self.gamma_ = float(self.gamma)

        n_pairs = C_cls * (C_cls - 1) // 2
        pair_alpha = torch.zeros(n, n_pairs, dtype=torch.float64)  # signed, local rows
        intercepts = []
        p = 0
        self.n_iter_ = []
        for i in range(C_cls):
            for j in range(i + 1, C_cls):
                sel = (y_t == i) | (y_t == j)
                idx = torch.nonzero(sel, as_tuple=False).squeeze(1)
                Xp = Xt[idx].contiguous()
                yp = torch.where(y_t[idx] == i, 1.0, -1.0).to(torch.float32)
                alpha, b, iters = smo_fit_pair(
                    Xp, yp, C=self.C, gamma=self.gamma_, tol=self.tol, max_iter=self.max_iter
                )
                pair_alpha[idx.cpu(), p] = (alpha.cpu() * yp.double().cpu())
                intercepts.append(b)
                self.n_iter_.append(iters)
                p += 1

        # assemble the libsvm layout (SVs grouped by class); gather shards
        sv_mask = pair_alpha.abs().sum(dim=1) > 1e-12
        Xsv_l = Xt[sv_mask.to(self.device)].double().cpu().numpy()
        ysv_l = y_t[sv_mask.to(self.device)].cpu().numpy()
        asv_l = pair_alpha[sv_mask].numpy()
        if sharded and dist.is_initialized():
            parts = [None] * dist.world_size()
            torch.distributed.all_gather_object(parts, (Xsv_l, ysv_l, asv_l))
            Xsv = np.concatenate([q[0] for q in parts])
            ysv = np.concatenate([q[1] for q in parts])
            asv = np.concatenate([q[2] for q in parts])
        else:
            Xsv, ysv, asv = Xsv_l, ysv_l, asv_l
        order = np.argsort(ysv, kind="stable")
        Xsv, ysv, asv = Xsv[order], ysv[order], asv[order]
        nSV = Xsv.shape[0]
        n_support = np.bincount(ysv, minlength=C_cls)
        dual = np.zeros((C_cls - 1, nSV), dtype=np.float64)
        p = 0
        for i in range(C_cls):
            for j in range(i + 1, C_cls):
                col = asv[:, p]
                for c, other in ((i, j), (j, i)):
                    row = other if other < c else other - 1
                    mask = ysv == c
                    dual[row, mask] = col[mask]
                p += 1
        self.support_vectors_ = torch.as_tensor(Xsv).to(self.device)
        self.dual_coef_ = torch.as_tensor(dual).to(self.device)
        self.intercept_ = torch.tensor(intercepts, dtype=torch.float64, device=self.device)
        self.n_support_ = torch.as_tensor(n_support, dtype=torch.int64).to(self.device)
        self.support_ = torch.nonzero(sv_mask, as_tuple=False).squeeze(1)
        return self

    def predict_index(self, X: ArrayLike) -> torch.Tensor:
        Xt = as_tensor(X, self.device, torch.float32)
        sc = getattr(self, "_svclass", None)
        if sc is None or sc.device != Xt.device:
            # SV -> class byte map; precomputed so the hipGraph-captured
            # serve path never runs repeat_interleave inside capture
            sc = torch.repeat_interleave(
                torch.arange(self.n_support_.numel(), device=Xt.device),
                self.n_support_.to(Xt.device),
            ).to(torch.uint8).contiguous()
            self._svclass = sc
        # raw (f64) params: the op layer caches the f32 copies so the
        # hipGraph serve path replays no conversion kernels
        return ops.svc_predict(
            Xt,
            self.support_vectors_,
            self.dual_coef_,
            self.intercept_,
            self.n_support_,
            self.gamma_,
            svclass=sc,
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
