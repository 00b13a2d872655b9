"""The six estimators of the reference, MI355X-native."""

from typing import Any, Dict, Optional

from .base import Estimator
from .gaussian_nb import GaussianNB
from .kmeans import KMeans
from .kneighbors import KNeighborsClassifier
from .logistic import LogisticRegression
from .random_forest import RandomForestClassifier
from .svc import SVC

_KIND_TO_CLASS = {
    "logistic": LogisticRegression,
    "gaussian_nb": GaussianNB,
    "kmeans": KMeans,
    "kneighbors": KNeighborsClassifier,
    "svc": SVC,
    "random_forest": RandomForestClassifier,
}

# reference CLI subcommand -> (checkpoint file name, kind)
# (reference: traffic_classifier.py:229-240; the reference's 'knearest'
# CLI alias is honored here — its loader bug is fixed)
ALGO_TO_CHECKPOINT = {
    "logistic": ("LogisticRegression", "logistic"),
    "kmeans": ("KMeans_Clustering", "kmeans"),
    "svm": ("SVC", "svc"),
    "knearest": ("KNeighbors", "kneighbors"),
    "kneighbors": ("KNeighbors", "kneighbors"),
    "Randomforest": ("RandomForestClassifier", "random_forest"),
    "gaussiannb": ("GaussianNB", "gaussian_nb"),
}


def from_params(params: Dict[str, Any], device: Optional[str] = None) -> Estimator:
    kind = params["kind"]
    cls = _KIND_TO_CLASS.get(kind)
    if cls is None:
        raise ValueError(f"unknown estimator kind {kind!r}")
    return cls.from_params(params, device=device)


def load_model(path: str, device: Optional[str] = None) -> Estimator:
    """Load any checkpoint: reference sklearn pickle or framework .npz."""
    from ..utils import checkpoint as ckpt

    if str(path).endswith(".npz"):
        params = ckpt.load_params_npz(path)
    else:
        params = ckpt.load_reference_checkpoint(path)
    return from_params(params, device=device)


__all__ = [
    "Estimator",
    "GaussianNB",
    "KMeans",
    "KNeighborsClassifier",
    "LogisticRegression",
    "RandomForestClassifier",
    "SVC",
    "from_params",
    "load_model",
    "ALGO_TO_CHECKPOINT",
]
