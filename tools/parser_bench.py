"""Telemetry ingestion throughput: Python parser vs native C++ bulk parser
(the C3 'line-rate' claim).  CPU-only; run anywhere."""
import sys, time
sys.path.insert(0, ".")
import numpy as np
from traffic_classifier_sdn_amd.flow.native import HAVE_NATIVE, NativePollParser
from traffic_classifier_sdn_amd.flow.parser import PollStreamParser
from traffic_classifier_sdn_amd.flow.replay import SynthFlowSpec, TelemetryReplaySource

n_flows, polls = 2000, 50
rng = np.random.default_rng(0)
specs = [SynthFlowSpec(f"02:{i>>8:02x}:{i&255:02x}:00:00:01", f"06:{i>>8:02x}:{i&255:02x}:00:00:02",
                       float(rng.uniform(1, 60)), float(rng.uniform(60, 1200)),
                       float(rng.uniform(1, 60)), float(rng.uniform(60, 1200)))
         for i in range(n_flows)]
lines = list(TelemetryReplaySource(specs=specs, seed=0).stream(polls))
buf = "\n".join(lines) + "\n"
print(f"{len(lines)} telemetry lines, {len(buf)/1e6:.1f} MB")

p = PollStreamParser()
t0 = time.perf_counter(); p.feed_many(lines); dt_py = time.perf_counter() - t0
print(f"python per-line parser : {len(lines)/dt_py/1e6:.2f} M lines/s ({dt_py*1e3:.0f} ms)")

if HAVE_NATIVE:
    q = NativePollParser()
    t0 = time.perf_counter(); q.feed_buffer(buf); dt_nat = time.perf_counter() - t0
    print(f"native bulk parser     : {len(lines)/dt_nat/1e6:.2f} M lines/s ({dt_nat*1e3:.0f} ms)  [{dt_py/dt_nat:.1f}x]")
    assert len(q.table) == len(p.table)
