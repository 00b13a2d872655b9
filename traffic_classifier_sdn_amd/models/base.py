"""Estimator base: device-aware fit/predict with sklearn-compatible surface.

Every estimator exposes ``fit(X, y)`` / ``predict(X)`` accepting numpy arrays,
lists or torch tensors (the reference serve loop calls ``model.predict`` on a
1-row nested list, traffic_classifier.py:106 — that works here too), plus
``to(device)``, ``to_params()`` / ``from_params()`` for checkpointing, and
string-label handling identical to sklearn (classes sorted, predictions
returned as the original label dtype).
"""

from __future__ import annotations

from typing import Any, Dict, Optional, Sequence, Union

import numpy as np
import torch

ArrayLike = Union[np.ndarray, torch.Tensor, Sequence]


def as_tensor(X: ArrayLike, device: torch.device, dtype=torch.float32) -> torch.Tensor:
    if isinstance(X, torch.Tensor):
        return X.to(device=device, dtype=dtype)
    arr = np.asarray(X, dtype=np.float64)
    if arr.ndim == 1:
        arr = arr.reshape(1, -1)
    return torch.from_numpy(arr).to(device=device, dtype=dtype)


def encode_labels(y: ArrayLike):
    """sorted-unique label encoding (sklearn semantics).
    Returns (classes ndarray[object], y_idx int64 tensor)."""
    y_arr = np.asarray(y).ravel()
    classes, y_idx = np.unique(y_arr, return_inverse=True)
    return classes.astype(object), torch.from_numpy(y_idx.astype(np.int64))


class Estimator:
    kind: str = ""

    def __init__(self, device: Optional[str] = None) -> None:
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.classes_: Optional[np.ndarray] = None

    # -- public API ----------------------------------------------------
    def fit(self, X: ArrayLike, y: Optional[ArrayLike] = None):
        raise NotImplementedError

    def predict(self, X: ArrayLike):
        idx = self.predict_index(X)
        if self.classes_ is None:
            return idx.cpu().numpy()
        return self.classes_[idx.cpu().numpy()]

    def predict_index(self, X: ArrayLike) -> torch.Tensor:
        """Integer class-index predictions (device tensor)."""
        raise NotImplementedError

    def to(self, device: str):
        self.device = torch.device(device)
        for name, val in list(self.__dict__.items()):
            if isinstance(val, torch.Tensor):
                setattr(self, name, val.to(self.device))
        return self

    # -- checkpointing -------------------------------------------------
    def to_params(self) -> Dict[str, Any]:
        raise NotImplementedError

    @classmethod
    def from_params(cls, params: Dict[str, Any], device: Optional[str] = None):
        raise NotImplementedError

    def save(self, path: str) -> None:
        from ..utils import checkpoint as ckpt

        if str(path).endswith(".npz"):
            ckpt.save_params_npz(self.to_params(), path)
        else:
            ckpt.save_sklearn_pickle(self.to_params(), path)

    # -- misc ----------------------------------------------------------
    def score(self, X: ArrayLike, y: ArrayLike) -> float:
        pred = self.predict(X)
        y_arr = np.asarray(y).ravel()
        return float((pred == y_arr).mean())
