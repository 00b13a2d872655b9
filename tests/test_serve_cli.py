"""Serve engine + CLI tests over the fake telemetry source (SURVEY.md §4:
the monitor -> feature -> predict path must test without Ryu/Mininet/root)."""

import io
import os
import subprocess
import sys

import numpy as np
import pytest

from traffic_classifier_sdn_amd.flow.replay import TelemetryReplaySource
from traffic_classifier_sdn_amd.models import GaussianNB, KMeans
from traffic_classifier_sdn_amd.serve import (
    RealtimeClassifier,
    TrainingCollector,
    map_cluster_labels,
    render_flow_table,
)
from traffic_classifier_sdn_amd.utils.schema import CSV_HEADER

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def fitted_gnb(request):
    from traffic_classifier_sdn_amd.utils.datasets import load_reference_dataset

    X, y = load_reference_dataset()
    return GaussianNB(device="cpu").fit(X, y)


def test_realtime_classifier_prints_table(fitted_gnb):
    out = io.StringIO()
    rc = RealtimeClassifier(fitted_gnb, predict_every=8, out=out)
    src = TelemetryReplaySource(seed=3)
    rc.run(src.stream(4))
    text = out.getvalue()
    assert "Flow ID" in text and "Traffic Type" in text
    assert "00:00:00:00:00:01" in text
    # statuses rendered
    assert "ACTIVE" in text


def test_cluster_label_mapping():
    out = map_cluster_labels(np.asarray([0, 5, 2, 99]))
    assert list(out) == ["dns", "voice", "ping", "unknown"]


def test_unsupervised_serve_path():
    # KMeans predictions are cluster ids mapped through the reference's
    # int->name table (traffic_classifier.py:109-114)
    from traffic_classifier_sdn_amd.utils.datasets import load_reference_dataset

    X, y = load_reference_dataset()
    m = KMeans(n_clusters=4, n_init=2, seed=0, device="cpu").fit(X[:2000])
    out = io.StringIO()
    rc = RealtimeClassifier(m, predict_every=8, out=out)
    rc.run(TelemetryReplaySource(seed=3).stream(3))
    text = out.getvalue()
    assert any(c in text for c in ("dns", "game", "ping", "quake", "telnet", "voice"))


def test_training_collector_csv_format(tmp_path):
    f = io.StringIO()
    tc = TrainingCollector("voice", f)
    src = TelemetryReplaySource(seed=4)
    tc.run(src.stream(3))
    lines = f.getvalue().splitlines()
    assert lines[0] + "\n" == CSV_HEADER
    # one row per tracked flow per accepted record
    assert len(lines) > 5
    row = lines[-1].split("\t")
    assert len(row) == 17
    assert row[-1] == "voice"


def test_cli_replay_end_to_end(tmp_path):
    # full process-level run: gaussiannb over replay source using the
    # converted reference checkpoint
    models_dir = os.path.join(REPO, "data", "ref_models")
    if not os.path.exists(os.path.join(models_dir, "GaussianNB.npz")):
        pytest.skip("converted checkpoints absent")
    env = dict(os.environ)
    r = subprocess.run(
        [
            sys.executable,
            "-m",
            "traffic_classifier_sdn_amd",
            "gaussiannb",
            "--source",
            "replay",
            "--replay-polls",
            "12",
            "--models-dir",
            models_dir,
            "--device",
            "cpu",
        ],
        capture_output=True,
        text=True,
        timeout=300,
        cwd=REPO,
        env=env,
    )
    assert r.returncode == 0, r.stderr
    assert "Flow ID" in r.stdout


def test_cli_train_replay(tmp_path):
    r = subprocess.run(
        [
            sys.executable,
            "-m",
            "traffic_classifier_sdn_amd",
            "train",
            "testcls",
            "--source",
            "replay",
            "--replay-polls",
            "5",
        ],
        capture_output=True,
        text=True,
        timeout=300,
        cwd=str(tmp_path),
        env={**os.environ, "PYTHONPATH": REPO},
    )
    assert r.returncode == 0, r.stderr
    out_csv = tmp_path / "testcls_training_data.csv"
    assert out_csv.exists()
    content = out_csv.read_text().splitlines()
    assert content[0] + "\n" == CSV_HEADER
    assert content[-1].endswith("testcls")


def test_cli_rejects_unknown_subcommand():
    r = subprocess.run(
        [sys.executable, "-m", "traffic_classifier_sdn_amd", "nonsense"],
        capture_output=True,
        text=True,
        timeout=120,
        cwd=REPO,
    )
    assert r.returncode != 0


def test_stats_flag_emits_json_metrics():
    import io
    import json

    from traffic_classifier_sdn_amd.flow.replay import TelemetryReplaySource
    from traffic_classifier_sdn_amd.models import GaussianNB
    from traffic_classifier_sdn_amd.serve import RealtimeClassifier
    from traffic_classifier_sdn_amd.utils.datasets import load_reference_dataset

    X, y = load_reference_dataset()
    model = GaussianNB(device="cpu").fit(X[:500].astype("float64"), y[:500])
    out, err = io.StringIO(), io.StringIO()
    rc = RealtimeClassifier(model, out=out, stats=True, stats_out=err)
    rc.run(TelemetryReplaySource(seed=1).stream(20))
    lines = [json.loads(l) for l in err.getvalue().splitlines()]
    assert lines, "no stats lines emitted"
    assert lines[-1]["flows"] > 0 and lines[-1]["predict_ms"] > 0


def test_kmeans_serve_uses_learned_cluster_names():
    import io

    import numpy as np

    from traffic_classifier_sdn_amd.flow.replay import TelemetryReplaySource
    from traffic_classifier_sdn_amd.models import KMeans
    from traffic_classifier_sdn_amd.serve import RealtimeClassifier
    from traffic_classifier_sdn_amd.utils.datasets import load_reference_dataset

    X, _ = load_reference_dataset()
    km = KMeans(n_clusters=6, device="cpu").fit(X[:2000])
    km.cluster_label_names_ = np.asarray(
        ["c0", "c1", "c2", "c3", "c4", "c5"], dtype=object
    )
    out = io.StringIO()
    rc = RealtimeClassifier(km, out=out)
    rc.run(TelemetryReplaySource(seed=1).stream(20))
    body = out.getvalue()
    assert any(f"c{i}" in body for i in range(6))


def test_cli_falls_back_to_packaged_checkpoints(tmp_path):
    """A fresh clone with no models/ dir still serves (data/ref_models)."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "traffic_classifier_sdn_amd", "gaussiannb",
         "--source", "replay", "--replay-polls", "12",
         "--models-dir", str(tmp_path / "nonexistent")],
        cwd=repo, capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stderr[-1500:]
    assert "Traffic Type" in out.stdout


def test_cli_subprocess_source_with_custom_monitor(tmp_path):
    """The reference's process model (Popen + stdout scrape,
    traffic_classifier.py:228): spawn a fake monitor command, classify its
    stream, and shut the process group down cleanly."""
    emitter = tmp_path / "emit.py"
    emitter.write_text(
        "import sys, time\n"
        "t = 1600000000\n"
        "for poll in range(25):\n"
        "    t += 1\n"
        "    for i in range(3):\n"
        "        print(f'data\\t{t}\\t1\\t1\\t02:00:00:00:00:{i:02x}\\t"
        "06:00:00:00:00:{i:02x}\\t2\\t{poll*50+i}\\t{poll*5000+i}')\n"
        "    sys.stdout.flush()\n"
    )
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "traffic_classifier_sdn_amd", "gaussiannb",
         "--source", "subprocess", "--monitor-cmd", f"{sys.executable} {emitter}",
         "--models-dir", os.path.join(repo, "data", "ref_models")],
        cwd=repo, capture_output=True, text=True, timeout=120,
    )
    assert r.returncode == 0, r.stderr[-1500:]
    assert "Traffic Type" in r.stdout
    assert r.stdout.count("Flow ID") >= 2  # multiple prediction passes


def test_prometheus_metrics_endpoint():
    """--prometheus exposes flow/latency metrics on a scrape port."""
    import urllib.request

    pytest.importorskip("prometheus_client")
    from traffic_classifier_sdn_amd.flow.replay import TelemetryReplaySource
    from traffic_classifier_sdn_amd.models import GaussianNB
    from traffic_classifier_sdn_amd.serve import RealtimeClassifier
    from traffic_classifier_sdn_amd.utils.datasets import load_reference_dataset

    X, y = load_reference_dataset()
    m = GaussianNB().fit(X[:500], y[:500])
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    rc = RealtimeClassifier(m, out=io.StringIO(), prometheus_port=port)
    for line in TelemetryReplaySource(seed=1).stream(6):
        rc.feed(line)
    body = urllib.request.urlopen(f"http://127.0.0.1:{port}/metrics", timeout=10).read().decode()
    assert "tcsdn_flows_tracked" in body
    assert "tcsdn_predict_passes_total" in body
    assert "tcsdn_predict_seconds_bucket" in body
    assert "tcsdn_class_flows" in body


def test_serve_models_tolerate_absurd_magnitudes():
    """A hostile/buggy switch can report absurd counters; every serve-path
    model must classify (any label) without crashing, NaN-ing or hanging."""
    from traffic_classifier_sdn_amd.models import load_model

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    X = np.zeros((8, 12), dtype=np.float64)
    X[0] = 1e38          # near-f32-max rates
    X[1] = -1e38         # nonsense negative counters
    X[2] = 1e-40         # subnormal territory
    X[3, 5] = np.inf     # a single inf feature
    X[4] = 0.0
    for name in ("RandomForestClassifier", "GaussianNB", "LogisticRegression",
                 "SVC", "KMeans_Clustering"):
        m = load_model(os.path.join(repo, "data", "ref_models", name + ".npz"))
        pred = m.predict_index(X)
        assert pred.shape[0] == 8
        assert int(pred.min()) >= 0, name  # a real class, never a sentinel
