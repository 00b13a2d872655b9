"""GPU serve engine: counter-snapshot feature parity (CPU) and the
hipGraph-captured classify path (GPU)."""

import numpy as np
import pytest
import torch

from traffic_classifier_sdn_amd.flow.replay import TelemetryReplaySource
from traffic_classifier_sdn_amd.flow.parser import replay
from traffic_classifier_sdn_amd.ops import cpu as oc
from traffic_classifier_sdn_amd.serve_gpu import GpuServeEngine


def _table(polls=6, seed=7):
    return replay(TelemetryReplaySource(seed=seed).stream(polls))


def test_counters_snapshot_matches_feature_matrix():
    table = _table()
    cur, prev, times = table.counters_snapshot()
    X_inc = table.feature_matrix(dtype=np.float32)
    X_gpu_path = oc.flow_features(
        torch.from_numpy(cur), torch.from_numpy(prev), torch.from_numpy(times)
    ).numpy()
    np.testing.assert_allclose(X_gpu_path, X_inc, rtol=1e-6, atol=1e-6)


def test_engine_cpu_matches_direct_predict(dataset):
    from traffic_classifier_sdn_amd.models import GaussianNB, LogisticRegression

    X, y = dataset
    models = {
        "gnb": GaussianNB(device="cpu").fit(X, y),
        "lr": LogisticRegression(device="cpu").fit(X, y),
    }
    table = _table()
    eng = GpuServeEngine(models, capacity=64, use_graph=False, device="cpu")
    out = eng.classify(table)
    Xf = table.feature_matrix(dtype=np.float32)
    for name, m in models.items():
        expect = m.predict_index(Xf).numpy()
        np.testing.assert_array_equal(out[name], expect)


@pytest.mark.gpu
def test_engine_gpu_graph(dataset):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import os

    from traffic_classifier_sdn_amd.models import load_model

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    names = ["RandomForestClassifier", "GaussianNB", "LogisticRegression", "SVC", "KMeans_Clustering"]
    models = {
        n: load_model(os.path.join(REPO, "data", "ref_models", n + ".npz"), device="cuda")
        for n in names
    }
    table = _table(polls=8)
    eng = GpuServeEngine(models, capacity=4096, use_graph=True)
    out1 = eng.classify(table)
    # graph is captured now; replay again and compare with non-graph engine
    out2 = eng.classify(table)
    eng_plain = GpuServeEngine(models, capacity=4096, use_graph=False)
    out3 = eng_plain.classify(table)
    n = len(table)
    for name in names:
        assert out1[name].shape == (n,)
        np.testing.assert_array_equal(out1[name], out2[name])
        np.testing.assert_array_equal(out2[name], out3[name])
    # the 1 ms poll-cadence budget (BASELINE config #5): graph replay end to
    # end (snapshot + H2D + 6 model predicts + D2H) under 5 ms, replay-only
    # well under 1 ms measured separately below
    import time

    eng.classify(table)
    t0 = time.perf_counter()
    reps = 50
    for _ in range(reps):
        eng._graph.replay()
    torch.cuda.synchronize()
    per_replay = (time.perf_counter() - t0) / reps
    assert per_replay < 1e-3, f"graph replay {per_replay*1e3:.3f} ms"


@pytest.mark.gpu
def test_engine_gpu_matches_cpu_predictions(dataset):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import os

    from traffic_classifier_sdn_amd.models import load_model

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    table = _table(polls=8)
    name = "RandomForestClassifier"
    gm = load_model(os.path.join(REPO, "data", "ref_models", name + ".npz"), device="cuda")
    cm = load_model(os.path.join(REPO, "data", "ref_models", name + ".npz"), device="cpu")
    eng = GpuServeEngine({"rf": gm}, capacity=512, use_graph=True)
    out = eng.classify(table)
    expect = cm.predict_index(table.feature_matrix(dtype=np.float32)).numpy()
    np.testing.assert_array_equal(out["rf"], expect)


def test_engine_capacity_fallback(dataset):
    """Flows beyond the graph capacity take the unbounded (non-graph) path
    and must produce identical labels."""
    from traffic_classifier_sdn_amd.models import GaussianNB

    X, y = dataset
    models = {"gnb": GaussianNB(device="cpu").fit(X, y)}
    table = _table(polls=6)
    n = len(table)
    assert n > 2
    small = GpuServeEngine(models, capacity=2, use_graph=False, device="cpu")
    big = GpuServeEngine(models, capacity=64, use_graph=False, device="cpu")
    out_small = small.classify(table)  # n > capacity -> unbounded path
    out_big = big.classify(table)
    np.testing.assert_array_equal(out_small["gnb"], out_big["gnb"])


@pytest.mark.gpu
def test_cli_serve_gpu_end_to_end(tmp_path):
    """The user-facing CLI path on GPU: replay telemetry -> checkpoint load
    -> device predict -> rendered flow table."""
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "traffic_classifier_sdn_amd", "Randomforest",
         "--source", "replay", "--replay-polls", "25", "--device", "cuda",
         "--models-dir", os.path.join(repo, "data", "ref_models"), "--stats"],
        cwd=repo, capture_output=True, text=True, timeout=180,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "Traffic Type" in out.stdout  # rendered table header
    assert any(c in out.stdout for c in ("dns", "game", "ping", "telnet", "voice", "quake"))
    assert '"predict_ms"' in out.stderr  # --stats JSON metrics emitted
