"""Full product loop: telemetry collection -> offline fit -> real-time
serve, all through the user-facing components (C4 -> C9 -> C5)."""

import io
import os

import numpy as np

from traffic_classifier_sdn_amd.flow.replay import SynthFlowSpec, TelemetryReplaySource
from traffic_classifier_sdn_amd.models import GaussianNB, load_model
from traffic_classifier_sdn_amd.serve import RealtimeClassifier, TrainingCollector
from traffic_classifier_sdn_amd.utils.datasets import load_reference_dataset
from traffic_classifier_sdn_amd.utils.metrics import accuracy


def _specs(kind, n, seed):
    rng = np.random.default_rng(seed)
    out = []
    for i in range(n):
        if kind == "voice":  # steady mid-rate small packets
            pps, bpp = rng.uniform(45, 55), rng.uniform(200, 230)
        else:  # "dns": sparse tiny flows
            pps, bpp = rng.uniform(1, 3), rng.uniform(70, 130)
        out.append(
            SynthFlowSpec(
                f"02:00:00:00:{seed:02x}:{i:02x}", f"06:00:00:00:{seed:02x}:{i:02x}",
                pps, bpp, pps * rng.uniform(0.8, 1.2), bpp * rng.uniform(0.8, 1.2),
            )
        )
    return out


def test_collect_fit_serve_loop(tmp_path):
    # 1) COLLECT: two traffic classes through the training collector
    for cls, seed in (("voice", 1), ("dns", 2)):
        src = TelemetryReplaySource(specs=_specs(cls, 8, seed), seed=seed)
        with open(tmp_path / f"{cls}_training_data.csv", "w") as f:
            TrainingCollector(cls, f).run(src.stream(40))

    # 2) FIT: pipeline loader reads the collected CSVs by their file names
    X, y = load_reference_dataset(data_dir=str(tmp_path))
    assert set(np.unique(y)) == {"voice", "dns"}
    assert len(X) > 200
    model = GaussianNB().fit(X, y)
    ckpt = tmp_path / "GaussianNB.npz"
    model.save(str(ckpt))

    # 3) SERVE: fresh checkpoint classifies a new replay stream
    m2 = load_model(str(ckpt))
    out = io.StringIO()
    rc = RealtimeClassifier(m2, out=out)
    rc.run(TelemetryReplaySource(specs=_specs("voice", 6, 9), seed=9).stream(30))
    labels = rc.classify_now()
    assert len(labels) == 6
    assert (labels == "voice").mean() > 0.8  # voice-shaped flows -> voice
