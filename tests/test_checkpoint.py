"""Checkpoint layer: shadow-unpickle the six reference sklearn-1.0.1
pickles (SURVEY.md §2.3 shapes), convert to params, round-trip the
framework-native .npz format, and export back to sklearn objects."""

import os

import numpy as np
import pytest

from traffic_classifier_sdn_amd.utils import checkpoint as ckpt

REF_MODELS = "/root/reference/models"
needs_ref = pytest.mark.skipif(not os.path.isdir(REF_MODELS), reason="reference mount absent")


@needs_ref
def test_logistic_shapes():
    p = ckpt.load_reference_checkpoint(os.path.join(REF_MODELS, "LogisticRegression"))
    assert p["kind"] == "logistic"
    assert list(p["classes"]) == ["dns", "ping", "telnet", "voice"]
    assert p["coef"].shape == (4, 12)
    assert p["intercept"].shape == (4,)


@needs_ref
def test_gnb_shapes():
    p = ckpt.load_reference_checkpoint(os.path.join(REF_MODELS, "GaussianNB"))
    assert p["theta"].shape == (6, 12)
    assert p["var"].shape == (6, 12)
    assert abs(p["class_prior"].sum() - 1.0) < 1e-12
    assert abs(p["epsilon"] - 0.0987020) < 1e-4


@needs_ref
def test_kmeans_shapes():
    p = ckpt.load_reference_checkpoint(os.path.join(REF_MODELS, "KMeans_Clustering"))
    assert p["centers"].shape == (4, 12)
    assert p["n_clusters"] == 4
    assert p["n_iter"] == 13


@needs_ref
def test_svc_shapes():
    p = ckpt.load_reference_checkpoint(os.path.join(REF_MODELS, "SVC"))
    assert p["support_vectors"].shape == (2281, 12)
    assert p["dual_coef"].shape == (5, 2281)
    assert p["intercept"].shape == (15,)
    assert list(p["n_support"]) == [579, 516, 759, 115, 199, 113]
    assert abs(p["gamma"] - 5.516893602700757e-09) < 1e-20


@needs_ref
def test_kneighbors_shapes():
    p = ckpt.load_reference_checkpoint(os.path.join(REF_MODELS, "KNeighbors"))
    assert p["fit_X"].shape == (4448, 12)
    assert p["y"].shape == (4448,)
    assert p["n_neighbors"] == 5
    assert len(p["classes"]) == 6


@needs_ref
def test_random_forest_structure():
    p = ckpt.load_reference_checkpoint(os.path.join(REF_MODELS, "RandomForestClassifier"))
    assert len(p["trees"]) == 100
    for t in p["trees"][:5]:
        n = t["left"].shape[0]
        assert 25 <= n <= 101
        assert t["values"].shape == (n, 6)
        inner = t["feature"] >= 0
        # depth-first invariant the packed GPU layout needs
        assert np.all(t["left"][inner] == np.nonzero(inner)[0] + 1)


@needs_ref
@pytest.mark.parametrize("name", ["LogisticRegression", "GaussianNB", "KMeans_Clustering", "SVC", "KNeighbors", "RandomForestClassifier"])
def test_npz_round_trip(tmp_path, name):
    p = ckpt.load_reference_checkpoint(os.path.join(REF_MODELS, name))
    path = str(tmp_path / f"{name}.npz")
    ckpt.save_params_npz(p, path)
    q = ckpt.load_params_npz(path)
    assert q["kind"] == p["kind"]
    for k, v in p.items():
        if k in ("kind", "trees"):
            continue
        if isinstance(v, np.ndarray):
            if v.dtype == object:
                assert [str(a) for a in v] == [str(b) for b in q[k]]
            else:
                np.testing.assert_array_equal(v, q[k])
        else:
            assert q[k] == pytest.approx(v)
    if "trees" in p:
        assert len(q["trees"]) == len(p["trees"])
        for a, b in zip(p["trees"], q["trees"]):
            for k in a:
                np.testing.assert_array_equal(a[k], b[k])


@needs_ref
def test_export_to_sklearn_pickle(tmp_path):
    import pickle
    import warnings

    p = ckpt.load_reference_checkpoint(os.path.join(REF_MODELS, "GaussianNB"))
    out = str(tmp_path / "GaussianNB")
    ckpt.save_sklearn_pickle(p, out)
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        est = pickle.load(open(out, "rb"))
    X = np.random.default_rng(0).normal(size=(16, 12)) * 100
    pred = est.predict(X)
    assert pred.shape == (16,)
