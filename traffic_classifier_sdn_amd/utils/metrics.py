"""Evaluation metrics (the notebooks' accuracy_score / confusion_matrix)."""

from __future__ import annotations

import numpy as np


def accuracy(y_true, y_pred) -> float:
    a = np.asarray(y_true).ravel()
    b = np.asarray(y_pred).ravel()
    return float((a == b).mean())


def confusion_matrix(y_true, y_pred, labels=None) -> np.ndarray:
    a = np.asarray(y_true).ravel()
    b = np.asarray(y_pred).ravel()
    if labels is None:
        labels = np.unique(np.concatenate([a, b]))
    lut = {l: i for i, l in enumerate(labels)}
    m = np.zeros((len(labels), len(labels)), dtype=np.int64)
    for t, p in zip(a, b):
        m[lut[t], lut[p]] += 1
    return m
