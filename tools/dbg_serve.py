"""Phase-level latency breakdown of GpuServeEngine.classify (tail hunt)."""
import sys, time, os
sys.path.insert(0, '.')
import numpy as np, torch, gc
from traffic_classifier_sdn_amd.flow.parser import replay
from traffic_classifier_sdn_amd.flow.replay import SynthFlowSpec, TelemetryReplaySource
from traffic_classifier_sdn_amd.serve_gpu import GpuServeEngine
from traffic_classifier_sdn_amd.models import load_model

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
rng = np.random.default_rng(0)
specs = [SynthFlowSpec("02:%02x:%02x:%02x:%02x:%02x" % tuple(int(v) for v in rng.integers(0,256,5)),
                       "06:%02x:%02x:%02x:%02x:%02x" % tuple(int(v) for v in rng.integers(0,256,5)),
                       float(rng.uniform(1,60)), float(rng.uniform(60,1200)),
                       float(rng.uniform(1,60)), float(rng.uniform(60,1200))) for _ in range(8192)]
table = replay(TelemetryReplaySource(specs=specs, seed=0).stream(2))
names = ["RandomForestClassifier","GaussianNB","LogisticRegression","SVC","KMeans_Clustering"]
models = {n: load_model(os.path.join(REPO,"data","ref_models",n+".npz"), device="cuda") for n in names}
eng = GpuServeEngine(models, capacity=8192, use_graph=True)
eng.classify(table)  # capture
phases = {k: [] for k in ("snapshot","h2d","replay","d2h","total")}
mode = sys.argv[1] if len(sys.argv) > 1 else "gc-on"
if mode == "blocking":
    import ctypes
    h = ctypes.CDLL("libamdhip64.so")
    # hipDeviceScheduleBlockingSync = 0x4 (must precede context use... set anyway)
    print("hipSetDeviceFlags rc:", h.hipSetDeviceFlags(ctypes.c_uint(4)))
if mode == "gc-off":
    gc.collect(); gc.freeze(); gc.disable()
for it in range(200):
    t0 = time.perf_counter()
    cur, prev, times = table.counters_snapshot()
    n = len(table)
    eng.h_cur[:n] = torch.from_numpy(cur); eng.h_prev[:n] = torch.from_numpy(prev); eng.h_times[:n] = torch.from_numpy(times)
    t1 = time.perf_counter()
    eng.d_cur.copy_(eng.h_cur, non_blocking=True); eng.d_prev.copy_(eng.h_prev, non_blocking=True); eng.d_times.copy_(eng.h_times, non_blocking=True)
    torch.cuda.synchronize(); t2 = time.perf_counter()
    eng._graph.replay(); torch.cuda.synchronize(); t3 = time.perf_counter()
    for name in models: eng.h_labels[name].copy_(eng.d_labels[name], non_blocking=True)
    torch.cuda.synchronize(); t4 = time.perf_counter()
    phases["snapshot"].append(t1-t0); phases["h2d"].append(t2-t1)
    phases["replay"].append(t3-t2); phases["d2h"].append(t4-t3); phases["total"].append(t4-t0)
for k,v in phases.items():
    a = np.array(v)*1e3
    print(f"{k:9s} p50={np.percentile(a,50):8.3f} p90={np.percentile(a,90):8.3f} p99={np.percentile(a,99):8.3f} max={a.max():8.3f} ms")
print("gc counts:", gc.get_count(), "gc thresholds:", gc.get_threshold())
# which iterations were slow?
tot = np.array(phases["total"])*1e3
slow = np.nonzero(tot > 5)[0]
print(mode, "slow iters:", slow[:20].tolist(), "values:", tot[slow[:10]].round(1).tolist())
