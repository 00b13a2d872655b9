"""30-second end-to-end installation check.

    python tools/verify_install.py

Covers: package import, native extensions, a CPU fit on the shipped rows,
the replay serve loop, checkpoint round-trip — and, when a GPU is visible,
the HIP kernel smoke.  Exits non-zero on any failure.
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> int:
    import numpy as np
    import torch

    from traffic_classifier_sdn_amd import ops  # noqa: F401  (dispatch facade)
    from traffic_classifier_sdn_amd.flow.replay import TelemetryReplaySource
    from traffic_classifier_sdn_amd.models import GaussianNB, load_model
    from traffic_classifier_sdn_amd.serve import RealtimeClassifier
    from traffic_classifier_sdn_amd.utils.datasets import (
        load_reference_dataset,
        train_test_split_ref,
    )

    print("[1/5] dataset ...", end=" ")
    X, y = load_reference_dataset()
    Xtr, Xte, ytr, yte = train_test_split_ref(X, y)
    print(f"ok ({X.shape[0]} rows)")

    print("[2/5] native flow table ...", end=" ")
    try:
        from traffic_classifier_sdn_amd.flow.native import NativePollParser  # noqa: F401

        print("ok (C++ parser)")
    except Exception as e:
        print(f"FALLBACK (python parser): {e!r}")

    print("[3/5] CPU fit ...", end=" ")
    m = GaussianNB().fit(Xtr, ytr)
    acc = (m.predict(Xte).astype(str) == yte.astype(str)).mean()
    assert acc > 0.97, acc
    print(f"ok (GNB acc {acc:.4f})")

    print("[4/5] checkpoint round-trip + replay serve ...", end=" ")
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "GaussianNB.npz")
        m.save(path)
        m2 = load_model(path)
        import io

        rc = RealtimeClassifier(m2, out=io.StringIO())
        for line in TelemetryReplaySource(seed=0).stream(12):
            rc.feed(line)
        assert rc.parser.records > 0
    print("ok")

    print("[5/5] GPU ...", end=" ")
    if torch.cuda.is_available():
        from traffic_classifier_sdn_amd.ops import gpu as og  # noqa: F401

        mg = GaussianNB(device="cuda").fit(Xtr, ytr)
        accg = (mg.predict(Xte).astype(str) == yte.astype(str)).mean()
        assert accg > 0.97, accg
        print(f"ok (HIP kernels on {torch.cuda.get_device_name(0)}, acc {accg:.4f})")
    else:
        print("skipped (no GPU visible; CPU paths verified)")
    print("ALL CHECKS PASSED")
    return 0


if __name__ == "__main__":
    sys.exit(main())
