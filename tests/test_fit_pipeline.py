"""Offline training pipeline (fit.py — the notebooks' C9 loop, SURVEY.md
§3.4) and the SVC / RandomForest sklearn-pickle exporters."""

import os
import pickle
import subprocess
import sys
import warnings

import json

import numpy as np
import pytest

from traffic_classifier_sdn_amd import fit as fitmod
from traffic_classifier_sdn_amd.models import SVC, RandomForestClassifier, from_params
from traffic_classifier_sdn_amd.utils import checkpoint as ckpt
from traffic_classifier_sdn_amd.utils.datasets import (
    load_reference_dataset,
    train_test_split_ref,
)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def split():
    X, y = load_reference_dataset()
    return train_test_split_ref(X, y)


def test_fit_cli_fast_algos(tmp_path, split):
    """gaussiannb + randomforest + kmeans via the CLI; accuracy reported and
    checkpoints written under the reference file names."""
    out = subprocess.run(
        [
            sys.executable,
            "-m",
            "traffic_classifier_sdn_amd.fit",
            "--algos",
            "gaussiannb,randomforest,kmeans",
            "--out",
            str(tmp_path),
            "--json",
        ],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=300,
    )
    assert out.returncode == 0, out.stderr
    import json

    results = {json.loads(l)["algo"]: json.loads(l) for l in out.stdout.splitlines() if l.startswith("{")}
    assert results["gaussiannb"]["accuracy"] > 0.97
    assert results["randomforest"]["accuracy"] > 0.99
    assert (tmp_path / "GaussianNB.npz").exists()
    assert (tmp_path / "RandomForestClassifier.npz").exists()
    assert (tmp_path / "KMeans_Clustering.npz").exists()
    # KMeans checkpoint carries the mode-based cluster->class name map
    p = ckpt.load_params_npz(str(tmp_path / "KMeans_Clustering.npz"))
    assert "cluster_label_names" in p and len(p["cluster_label_names"]) == 6


def test_mode_cluster_assignment():
    ids = np.array([0, 0, 0, 1, 1, 2])
    y = np.array(["a", "a", "b", "b", "b", "c"], dtype=object)
    names = fitmod.mode_cluster_assignment(ids, y, 3)
    assert list(names) == ["a", "b", "c"]


def test_rf_sklearn_export_round_trip(tmp_path, split):
    Xtr, Xte, ytr, yte = split
    m = RandomForestClassifier(n_estimators=10, seed=7).fit(Xtr, ytr)
    path = str(tmp_path / "RandomForestClassifier")
    ckpt.save_sklearn_pickle(m.to_params(), path)
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        sk = pickle.load(open(path, "rb"))
        agree = (m.predict(Xte) == sk.predict(Xte)).mean()
    assert agree == 1.0


def test_svc_sklearn_export_round_trip(tmp_path, split):
    Xtr, Xte, ytr, yte = split
    m = SVC().fit(Xtr[:800], ytr[:800])
    path = str(tmp_path / "SVC")
    ckpt.save_sklearn_pickle(m.to_params(), path)
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        sk = pickle.load(open(path, "rb"))
        agree = (m.predict(Xte) == sk.predict(Xte)).mean()
    assert agree > 0.999


def test_hist_rf_checkpoint_round_trip(tmp_path, split):
    """Hist-built forest -> npz -> reload -> identical predictions."""
    from traffic_classifier_sdn_amd.models import RandomForestClassifier, load_model

    Xtr, Xte, ytr, yte = split
    m = RandomForestClassifier(n_estimators=12, seed=3, builder="hist", device="cpu").fit(Xtr, ytr)
    path = str(tmp_path / "rf_hist.npz")
    m.save(path)
    m2 = load_model(path)
    np.testing.assert_array_equal(m.predict(Xte), m2.predict(Xte))


def test_fit_cli_custom_data_dir(tmp_path):
    """fit.py --data-dir consumes collector-produced CSVs end to end."""
    from traffic_classifier_sdn_amd.flow.replay import SynthFlowSpec, TelemetryReplaySource
    from traffic_classifier_sdn_amd.serve import TrainingCollector

    rng = np.random.default_rng(5)
    for cls, (pps, bpp) in (("voice", (50, 214)), ("dns", (2, 100))):
        specs = [
            SynthFlowSpec(f"02:11:00:00:00:{i:02x}", f"06:11:00:00:00:{i:02x}",
                          pps * rng.uniform(0.9, 1.1), bpp, pps, bpp)
            for i in range(6)
        ]
        with open(tmp_path / f"{cls}_training_data.csv", "w") as f:
            TrainingCollector(cls, f).run(TelemetryReplaySource(specs=specs, seed=3).stream(30))
    out = subprocess.run(
        [sys.executable, "-m", "traffic_classifier_sdn_amd.fit",
         "--algos", "gaussiannb", "--data-dir", str(tmp_path),
         "--out", str(tmp_path / "m"), "--json"],
        cwd=REPO, capture_output=True, text=True, timeout=180,
    )
    assert out.returncode == 0, out.stderr[-1500:]
    rec = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][0])
    assert rec["accuracy"] > 0.9  # two well-separated synthetic classes


def test_lr_knn_kmeans_sklearn_export_round_trip(tmp_path, split):
    """Remaining exporters: LR / KNN / KMeans framework fits load into
    stock sklearn and agree."""
    import warnings

    from traffic_classifier_sdn_amd.models import (
        KMeans,
        KNeighborsClassifier,
        LogisticRegression,
    )
    from traffic_classifier_sdn_amd.utils import checkpoint as ckpt

    Xtr, Xte, ytr, yte = split
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        lr = LogisticRegression().fit(Xtr, ytr)
        p = str(tmp_path / "LogisticRegression")
        ckpt.save_sklearn_pickle(lr.to_params(), p)
        sk = pickle.load(open(p, "rb"))
        assert (lr.predict(Xte) == sk.predict(Xte)).mean() > 0.999

        knn = KNeighborsClassifier().fit(Xtr, ytr)
        p = str(tmp_path / "KNeighbors")
        ckpt.save_sklearn_pickle(knn.to_params(), p)
        sk = pickle.load(open(p, "rb"))
        assert (knn.predict(Xte) == sk.predict(Xte)).mean() > 0.995  # dup-row ties

        km = KMeans(n_clusters=6).fit(Xtr)
        p = str(tmp_path / "KMeans_Clustering")
        ckpt.save_sklearn_pickle(km.to_params(), p)
        sk = pickle.load(open(p, "rb"))
        ours = km.predict(Xte)
        theirs = sk.predict(np.asarray(Xte, dtype=np.float64))
        assert (np.asarray(ours) == theirs).mean() > 0.999


def test_synthesize_quake_rows_plausible():
    """Quake rows come through the real collection path, have the schema's
    12 features, and are separable enough that a 6-class fit beats the
    published 6-class accuracies (VERDICT r01 missing #2)."""
    from traffic_classifier_sdn_amd.utils.datasets import (
        load_six_class_dataset,
        synthesize_quake_rows,
        train_test_split_ref,
    )

    X, y = synthesize_quake_rows(200, seed=5)
    assert X.shape == (200, 12)
    assert set(y) == {"quake"}
    assert np.isfinite(X).all() and (X >= 0).all()
    # deterministic per seed
    X2, _ = synthesize_quake_rows(200, seed=5)
    np.testing.assert_array_equal(X, X2)

    X6, y6 = load_six_class_dataset(quake_rows=300, seed=5)
    assert sorted(set(y6)) == ["dns", "game", "ping", "quake", "telnet", "voice"]

    from traffic_classifier_sdn_amd.models import GaussianNB

    Xtr, Xte, ytr, yte = train_test_split_ref(X6, y6)
    m = GaussianNB().fit(Xtr, ytr)
    acc = (m.predict(Xte) == yte).mean()
    assert acc > 0.97  # published 6-class GNB accuracy is 98.63 on real quake


def test_fit_with_synth_quake_six_classes(tmp_path):
    """--with-synth-quake fits on the true 6-class set (shipped 5 classes +
    D-ITG Quake3 replay rows) and writes loadable checkpoints."""
    import json
    import subprocess
    import sys

    r = subprocess.run(
        [sys.executable, "-m", "traffic_classifier_sdn_amd.fit",
         "--algos", "gaussiannb", "--with-synth-quake",
         "--out", str(tmp_path), "--json"],
        capture_output=True, text=True, timeout=300,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    )
    assert r.returncode == 0, r.stderr[-800:]
    d = json.loads(r.stdout.strip().splitlines()[-1])
    assert d["accuracy"] > 0.97
    from traffic_classifier_sdn_amd.models import load_model

    m = load_model(str(tmp_path / "GaussianNB.npz"))
    assert sorted(str(c) for c in m.classes_) == ["dns", "game", "ping", "quake", "telnet", "voice"]


def test_fit_torchrun_world2_gloo(tmp_path):
    """fit.py end-to-end under torchrun (2 gloo ranks): sharded DP fit, rank
    0 writes the checkpoint — the launch mode an 8-GPU node uses."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29641", "-m", "traffic_classifier_sdn_amd.fit",
         "--algos", "gaussiannb", "--out", str(tmp_path), "--json"],
        capture_output=True, text=True, timeout=420, cwd=repo,
        env={**os.environ, "OMP_NUM_THREADS": "1"},
    )
    assert r.returncode == 0, r.stderr[-1200:]
    assert (tmp_path / "GaussianNB.npz").exists()
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["accuracy"] > 0.97


def test_fit_plots_confusion_svg(tmp_path):
    """fit.py --plots writes a valid confusion-matrix heatmap SVG."""
    import subprocess
    import sys
    import xml.etree.ElementTree as ET

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "traffic_classifier_sdn_amd.fit",
         "--algos", "gaussiannb", "--out", str(tmp_path / "m"),
         "--plots", str(tmp_path / "plots"), "--json"],
        capture_output=True, text=True, timeout=300, cwd=repo,
    )
    assert r.returncode == 0, r.stderr[-800:]
    svg = tmp_path / "plots" / "confusion_gaussiannb.svg"
    assert svg.exists()
    root = ET.parse(svg).getroot()
    assert root.tag.endswith("svg")
    assert open(svg).read().count("<rect") >= 37  # 6x6 cells + background
