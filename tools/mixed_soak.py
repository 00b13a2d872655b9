"""Production-mix soak: cycles of serve + KNN (exact & bf16) + fits of all
families on one GPU, tracking latency and device-memory stability.

    python tools/mixed_soak.py [seconds]
"""
import os
import sys
import time

os.environ.setdefault("OMP_NUM_THREADS", "4")
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def main() -> int:
    budget = float(sys.argv[1]) if len(sys.argv) > 1 else 480.0
    assert torch.cuda.is_available()
    from traffic_classifier_sdn_amd.flow.native import NativePollParser
    from traffic_classifier_sdn_amd.flow.replay import SynthFlowSpec, TelemetryReplaySource
    from traffic_classifier_sdn_amd.models import (
        GaussianNB,
        KNeighborsClassifier,
        LogisticRegression,
        RandomForestClassifier,
        SVC,
        load_model,
    )
    from traffic_classifier_sdn_amd.serve_gpu import GpuServeEngine
    from traffic_classifier_sdn_amd.utils.datasets import synthetic_flow_rows, load_reference_dataset

    rng = np.random.default_rng(0)
    specs = [
        SynthFlowSpec(
            "02:%02x:%02x:%02x:%02x:%02x" % tuple(int(v) for v in rng.integers(0, 256, 5)),
            "06:%02x:%02x:%02x:%02x:%02x" % tuple(int(v) for v in rng.integers(0, 256, 5)),
            float(rng.uniform(1, 60)), float(rng.uniform(60, 1200)),
            float(rng.uniform(1, 60)), float(rng.uniform(60, 1200)),
        )
        for _ in range(8192)
    ]
    src = TelemetryReplaySource(specs=specs, seed=0)
    parser = NativePollParser()
    parser.feed_buffer("\n".join(src.stream(2)) + "\n")
    names = ["RandomForestClassifier", "GaussianNB", "LogisticRegression", "SVC", "KMeans_Clustering"]
    models = {n: load_model(f"data/ref_models/{n}.npz", device="cuda") for n in names}
    eng = GpuServeEngine(models, capacity=8192, use_graph=True, device="cuda")

    X_real, y_real = load_reference_dataset()
    Xfit = synthetic_flow_rows(100_000, seed=1, reference_X=X_real)
    yfit = rng.integers(0, 6, size=100_000)
    Rknn = torch.from_numpy(synthetic_flow_rows(2_000_000, seed=2, reference_X=X_real)).float().cuda()
    yknn = rng.integers(0, 6, size=2_000_000)
    Q = torch.from_numpy(synthetic_flow_rows(4096, seed=3, reference_X=X_real)).float().cuda()

    t_end = time.time() + budget
    cycle = 0
    mem0 = None
    serve_lat = []
    while time.time() < t_end:
        cycle += 1
        for _ in range(500):
            eng.classify(parser.table)
            serve_lat.append(eng.last_latency_s)
        km = KNeighborsClassifier(n_neighbors=5, device="cuda").fit(
            Rknn.cpu().numpy() if cycle == -1 else Rknn, yknn)
        km.predict_index(Q)
        km.approx = True
        km.predict_index(Q)
        GaussianNB(device="cuda").fit(Xfit, yfit)
        LogisticRegression(device="cuda", max_iter=12).fit(Xfit, yfit)
        SVC(device="cuda", max_iter=1500).fit(Xfit[:20_000], yfit[:20_000])
        RandomForestClassifier(n_estimators=5, builder="hist", device="cuda").fit(Xfit, yfit)
        torch.cuda.synchronize()
        mem = torch.cuda.memory_allocated()
        if mem0 is None:
            mem0 = mem
        print(f"cycle {cycle}: mem {mem/2**20:.0f} MiB (d {(mem-mem0)/2**20:+.1f})", flush=True)
    lat = np.asarray(serve_lat[100:])
    print(
        f"SOAK OK: {cycle} cycles, serve p50 {np.percentile(lat,50)*1e3:.3f} ms "
        f"p99.9 {np.percentile(lat,99.9)*1e3:.3f} ms, "
        f"mem growth {(mem-mem0)/2**20:.1f} MiB"
    )
    assert (mem - mem0) / 2**20 < 512, "device memory grew suspiciously"
    return 0


if __name__ == "__main__":
    sys.exit(main())
