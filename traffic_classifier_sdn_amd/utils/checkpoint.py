"""Checkpoint I/O: sklearn-pickle-compatible model persistence.

The reference persists every trained estimator as a raw ``pickle.dump`` of an
sklearn-1.0.1 object (reference: traffic_classifier.py:243 loads them;
notebooks/1_log_Kmeans.ipynb cell 55 etc. dump them).  Those pickles embed
Cython types whose binary layouts have since changed — the sklearn shipped in
this image (1.7.x) refuses to load two of the six (``Tree`` node dtype,
``KDTree``/dist-metrics module moves) — so this framework reads the pickles
*itself* with a shadow unpickler that reconstructs the numeric state without
importing any sklearn class, and converts it to plain-tensor parameter dicts
(the framework's native checkpoint form).

Write path: parameter dicts are exported back to genuine sklearn estimator
objects (using the installed sklearn) so new fits remain loadable by sklearn,
and to the framework's own ``.npz`` format for GPU-side loading.
"""

from __future__ import annotations

import io
import pickle
from typing import Any, Dict

import numpy as np

# ----------------------------------------------------------------------
# Shadow unpickling
# ----------------------------------------------------------------------


class ShadowObject:
    """Stand-in for any non-numpy class found in a checkpoint pickle.

    Captures constructor args (``_reduce_args``) and ``__setstate__`` payloads
    (dict payloads merge into ``__dict__``; tuple/other payloads are stored in
    ``_state``) so every numeric attribute of the original object is
    reachable without the original class.
    """

    _shadow_module: str = ""
    _shadow_name: str = ""

    def __new__(cls, *args: Any) -> "ShadowObject":
        obj = object.__new__(cls)
        obj.__dict__["_reduce_args"] = args
        return obj

    def __init__(self, *args: Any) -> None:  # REDUCE-style construction
        self.__dict__.setdefault("_reduce_args", args)

    def __setstate__(self, state: Any) -> None:
        if isinstance(state, dict):
            self.__dict__.update(state)
        elif (
            isinstance(state, tuple)
            and len(state) == 2
            and (state[0] is None or isinstance(state[0], dict))
            and (state[1] is None or isinstance(state[1], dict))
        ):
            if state[0]:
                self.__dict__.update(state[0])
            if state[1]:
                self.__dict__.update(state[1])
        else:
            self.__dict__["_state"] = state

    @property
    def shadow_class(self) -> str:
        return f"{self._shadow_module}.{self._shadow_name}"

    def __repr__(self) -> str:  # pragma: no cover
        return f"<shadow {self.shadow_class}>"


_ALLOWED_MODULE_PREFIXES = ("numpy", "builtins", "collections", "copyreg", "_codecs")


class ShadowUnpickler(pickle.Unpickler):
    """Unpickler that resolves numpy/builtin globals normally and shadows
    everything else (sklearn/scipy classes) with :class:`ShadowObject`
    subclasses, so version-incompatible Cython layouts never run."""

    def find_class(self, module: str, name: str) -> Any:
        if module.split(".")[0] in [p.split(".")[0] for p in _ALLOWED_MODULE_PREFIXES]:
            return super().find_class(module, name)
        key = (module, name)
        cls = _shadow_class_cache.get(key)
        if cls is None:
            cls = type(
                f"Shadow_{name}",
                (ShadowObject,),
                {"_shadow_module": module, "_shadow_name": name},
            )
            _shadow_class_cache[key] = cls
        return cls


_shadow_class_cache: Dict[Any, type] = {}


def shadow_load(path: str) -> ShadowObject:
    with open(path, "rb") as f:
        return ShadowUnpickler(f).load()


def shadow_loads(data: bytes) -> ShadowObject:
    return ShadowUnpickler(io.BytesIO(data)).load()


# ----------------------------------------------------------------------
# Shadow estimator -> parameter dict
# ----------------------------------------------------------------------

# sklearn 1.0.1 Tree node structured-array field order
# (left_child, right_child, feature, threshold, impurity,
#  n_node_samples, weighted_n_node_samples) — SURVEY.md §2.3.
_TREE_FIELDS = ("left_child", "right_child", "feature", "threshold")


def _labels_to_str(arr: np.ndarray) -> np.ndarray:
    return np.asarray([str(x) for x in np.asarray(arr).ravel()], dtype=object)


def params_from_shadow(obj: ShadowObject) -> Dict[str, Any]:
    """Convert a shadow-loaded sklearn estimator into the framework's plain
    parameter dict.  Dispatch is on the pickled class name."""
    name = obj._shadow_name
    if name == "LogisticRegression":
        return {
            "kind": "logistic",
            "classes": _labels_to_str(obj.classes_),
            "coef": np.asarray(obj.coef_, dtype=np.float64),
            "intercept": np.asarray(obj.intercept_, dtype=np.float64),
        }
    if name == "GaussianNB":
        return {
            "kind": "gaussian_nb",
            "classes": _labels_to_str(obj.classes_),
            "theta": np.asarray(obj.theta_, dtype=np.float64),
            "var": np.asarray(getattr(obj, "var_", getattr(obj, "sigma_", None)), dtype=np.float64),
            "class_prior": np.asarray(obj.class_prior_, dtype=np.float64),
            "class_count": np.asarray(obj.class_count_, dtype=np.float64),
            "epsilon": float(obj.epsilon_),
        }
    if name == "KMeans":
        return {
            "kind": "kmeans",
            "centers": np.asarray(obj.cluster_centers_, dtype=np.float64),
            "n_clusters": int(obj.n_clusters),
            "inertia": float(obj.inertia_),
            "n_iter": int(obj.n_iter_),
        }
    if name == "SVC":
        return {
            "kind": "svc",
            "classes": _labels_to_str(obj.classes_),
            "support_vectors": np.asarray(obj.support_vectors_, dtype=np.float64),
            "dual_coef": np.asarray(obj.dual_coef_, dtype=np.float64),
            "intercept": np.asarray(obj._intercept_, dtype=np.float64),
            "n_support": np.asarray(obj._n_support, dtype=np.int64),
            "gamma": float(obj._gamma),
            "support": np.asarray(obj.support_, dtype=np.int64),
        }
    if name == "KNeighborsClassifier":
        return {
            "kind": "kneighbors",
            "classes": _labels_to_str(obj.classes_),
            "fit_X": np.asarray(obj._fit_X, dtype=np.float64),
            "y": np.asarray(obj._y, dtype=np.int64),
            "n_neighbors": int(obj.n_neighbors),
        }
    if name == "RandomForestClassifier":
        return _params_from_forest(obj)
    raise ValueError(f"unsupported checkpoint class: {obj.shadow_class}")


def _params_from_forest(obj: ShadowObject) -> Dict[str, Any]:
    """Flatten a shadow RandomForestClassifier into SoA tree arrays.

    Per-tree state comes from each DecisionTreeClassifier's embedded
    ``sklearn.tree._tree.Tree`` shadow: ``_reduce_args`` =
    (n_features, n_classes array, n_outputs) and ``__setstate__`` dict with
    'nodes' (structured) and 'values' (node_count, 1, n_classes).
    """
    classes = _labels_to_str(obj.classes_)
    n_classes = len(classes)
    trees = []
    for est in obj.estimators_:
        t = est.tree_
        nodes = np.asarray(t.nodes)
        values = np.asarray(t.values)  # (node_count, 1, n_classes)
        trees.append(
            {
                "left": nodes["left_child"].astype(np.int32),
                "right": nodes["right_child"].astype(np.int32),
                "feature": nodes["feature"].astype(np.int32),
                "threshold": nodes["threshold"].astype(np.float64),
                "values": values.reshape(values.shape[0], -1).astype(np.float64),
            }
        )
    return {
        "kind": "random_forest",
        "classes": classes,
        "n_classes": n_classes,
        "trees": trees,
    }


def load_reference_checkpoint(path: str) -> Dict[str, Any]:
    """Read any of the six reference pickle checkpoints into params."""
    return params_from_shadow(shadow_load(path))


# ----------------------------------------------------------------------
# Framework-native .npz checkpoint format
# ----------------------------------------------------------------------


def save_params_npz(params: Dict[str, Any], path: str) -> None:
    """Serialize a parameter dict to a flat .npz (framework-native format,
    loadable without pickle and identical on CPU/GPU hosts)."""
    flat: Dict[str, Any] = {"kind": np.asarray(params["kind"])}
    for key, val in params.items():
        if key == "kind":
            continue
        if key == "trees":
            flat["n_trees"] = np.asarray(len(val))
            for i, tree in enumerate(val):
                for tkey, tval in tree.items():
                    flat[f"tree{i}/{tkey}"] = tval
        elif isinstance(val, np.ndarray) and val.dtype == object:
            flat[key] = val.astype(str)
        else:
            flat[key] = np.asarray(val)
    np.savez_compressed(path, **flat)


def load_params_npz(path: str) -> Dict[str, Any]:
    with np.load(path, allow_pickle=False) as z:
        kind = str(z["kind"])
        params: Dict[str, Any] = {"kind": kind}
        if "n_trees" in z:
            n_trees = int(z["n_trees"])
            trees = []
            for i in range(n_trees):
                prefix = f"tree{i}/"
                trees.append(
                    {k[len(prefix):]: z[k] for k in z.files if k.startswith(prefix)}
                )
            params["trees"] = trees
        for k in z.files:
            if k in ("kind", "n_trees") or "/" in k:
                continue
            arr = z[k]
            if arr.ndim == 0:
                val = arr.item()
                params[k] = val
            else:
                params[k] = arr
        if "classes" in params:
            params["classes"] = np.asarray([str(c) for c in params["classes"]], dtype=object)
        return params


# ----------------------------------------------------------------------
# Export to genuine sklearn objects (write path)
# ----------------------------------------------------------------------


def params_to_sklearn(params: Dict[str, Any]):
    """Build a fitted sklearn estimator (installed version) from params, so
    framework fits can be pickled for sklearn consumers.  RandomForest export
    rebuilds each Tree via sklearn's own buffer protocol."""
    kind = params["kind"]
    if kind == "logistic":
        from sklearn.linear_model import LogisticRegression

        est = LogisticRegression()
        est.classes_ = np.asarray(params["classes"])
        est.coef_ = np.asarray(params["coef"], dtype=np.float64)
        est.intercept_ = np.asarray(params["intercept"], dtype=np.float64)
        est.n_features_in_ = est.coef_.shape[1]
        est.n_iter_ = np.asarray([100])
        return est
    if kind == "gaussian_nb":
        from sklearn.naive_bayes import GaussianNB

        est = GaussianNB()
        est.classes_ = np.asarray(params["classes"])
        est.theta_ = np.asarray(params["theta"], dtype=np.float64)
        est.var_ = np.asarray(params["var"], dtype=np.float64)
        est.class_prior_ = np.asarray(params["class_prior"], dtype=np.float64)
        est.class_count_ = np.asarray(params["class_count"], dtype=np.float64)
        est.epsilon_ = float(params["epsilon"])
        est.n_features_in_ = est.theta_.shape[1]
        return est
    if kind == "kmeans":
        from sklearn.cluster import KMeans

        centers = np.asarray(params["centers"], dtype=np.float64)
        est = KMeans(n_clusters=int(params["n_clusters"]))
        est.cluster_centers_ = centers
        est.inertia_ = float(params.get("inertia", 0.0))
        est.n_iter_ = int(params.get("n_iter", 0))
        est.n_features_in_ = centers.shape[1]
        est._n_threads = 1
        return est
    if kind == "svc":
        from sklearn.svm import SVC as SkSVC

        SV = np.ascontiguousarray(params["support_vectors"], dtype=np.float64)
        dual = np.ascontiguousarray(params["dual_coef"], dtype=np.float64)
        intercept = np.ascontiguousarray(params["intercept"], dtype=np.float64)
        n_support = np.ascontiguousarray(params["n_support"], dtype=np.int32)
        classes = np.asarray(params["classes"])
        gamma = float(params["gamma"])
        est = SkSVC(kernel="rbf", gamma=gamma)
        est.classes_ = classes
        est.support_vectors_ = SV
        support = np.asarray(params.get("support", np.arange(SV.shape[0])))
        est.support_ = np.ascontiguousarray(support, dtype=np.int32)
        est._n_support = n_support
        # sklearn negates the public dual/intercept for binary problems
        # (sklearn/svm/_base.py fit); params hold the public layout
        binary = len(classes) == 2
        est.dual_coef_ = dual
        est._dual_coef_ = -dual if binary else dual
        est.intercept_ = intercept
        est._intercept_ = -intercept if binary else intercept
        est._probA = np.empty(0, dtype=np.float64)
        est._probB = np.empty(0, dtype=np.float64)
        est._gamma = gamma
        est._sparse = False
        est.shape_fit_ = (int(params.get("n_fit_rows", SV.shape[0])), SV.shape[1])
        est.fit_status_ = 0
        est.n_features_in_ = SV.shape[1]
        est._num_iter = np.ones(max(1, len(intercept)), dtype=np.int32)
        return est
    if kind == "kneighbors":
        from sklearn.neighbors import KNeighborsClassifier

        est = KNeighborsClassifier(n_neighbors=int(params["n_neighbors"]), algorithm="brute")
        est.fit(np.asarray(params["fit_X"], dtype=np.float64), np.asarray(params["classes"])[np.asarray(params["y"])])
        return est
    if kind == "random_forest":
        from sklearn.ensemble import RandomForestClassifier as SkRF
        from sklearn.tree import DecisionTreeClassifier
        from sklearn.tree._tree import Tree

        classes = np.asarray(params["classes"])
        C = len(classes)
        trees = params["trees"]
        F = int(params.get("n_features", max(int(t["feature"].max()) for t in trees) + 1))
        # version-robust templates: sklearn's node struct dtype and the
        # count-vs-fraction convention of tree values (changed in 1.4)
        tmpl = DecisionTreeClassifier(max_depth=1).fit([[0.0], [1.0]], [0, 1])
        tstate = tmpl.tree_.__getstate__()
        node_dtype = tstate["nodes"].dtype
        values_normalized = abs(float(tstate["values"][0].sum()) - 1.0) < 1e-9
        ests = []
        for t in trees:
            feat = np.asarray(t["feature"], dtype=np.int64)
            n = len(feat)
            left = np.asarray(t["left"], dtype=np.int64)
            right = np.asarray(t["right"], dtype=np.int64)
            thr = np.asarray(t["threshold"], dtype=np.float64)
            counts = np.asarray(t["values"], dtype=np.float64)
            nns = counts.sum(axis=1)
            p = counts / np.maximum(nns[:, None], 1e-300)
            is_leaf = feat < 0
            nodes = np.zeros(n, dtype=node_dtype)
            nodes["left_child"] = np.where(is_leaf, -1, left)
            nodes["right_child"] = np.where(is_leaf, -1, right)
            nodes["feature"] = np.where(is_leaf, -2, feat)
            nodes["threshold"] = np.where(is_leaf, -2.0, thr)
            nodes["impurity"] = 1.0 - (p ** 2).sum(axis=1)
            nodes["n_node_samples"] = nns.astype(np.int64)
            nodes["weighted_n_node_samples"] = nns
            # depth per node via parent scan (children follow parents)
            depth = np.zeros(n, dtype=np.int64)
            for i in range(n):
                if feat[i] >= 0:
                    depth[left[i]] = depth[i] + 1
                    depth[right[i]] = depth[i] + 1
            tree = Tree(F, np.asarray([C], dtype=np.intp), 1)
            tree.__setstate__(
                {
                    "max_depth": int(depth.max()),
                    "node_count": n,
                    "nodes": nodes,
                    "values": (p if values_normalized else counts).reshape(n, 1, C),
                }
            )
            dt = DecisionTreeClassifier()
            dt.tree_ = tree
            dt.classes_ = classes
            dt.n_classes_ = C
            dt.n_features_in_ = F
            dt.n_outputs_ = 1
            dt.max_features_ = max(1, int(np.sqrt(F)))
            ests.append(dt)
        est = SkRF(n_estimators=len(trees))
        est.estimators_ = ests
        est.classes_ = classes
        est.n_classes_ = C
        est.n_features_in_ = F
        est.n_outputs_ = 1
        return est
    raise ValueError(f"unknown params kind {kind}")


def save_sklearn_pickle(params: Dict[str, Any], path: str) -> None:
    est = params_to_sklearn(params)
    with open(path, "wb") as f:
        pickle.dump(est, f, protocol=4)
