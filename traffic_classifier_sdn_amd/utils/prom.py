"""Optional Prometheus metrics endpoint for the serve path.

The reference's only observability is its TSV log and PrettyTable dump
(SURVEY.md §5 metrics/logging); `--stats` already adds structured JSON
lines.  For production serving this module exposes the same counters as a
Prometheus scrape target (`prometheus_client` is in the image):

    python -m traffic_classifier_sdn_amd Randomforest --prometheus 9101

Metrics:
    tcsdn_flows_tracked            live flows in the table (gauge)
    tcsdn_records_total            accepted telemetry records (counter)
    tcsdn_predict_passes_total     prediction passes run (counter)
    tcsdn_predict_seconds          per-pass predict latency (histogram)
    tcsdn_class_flows{label=...}   flows per predicted class (gauge)

Entirely optional: importing this module without prometheus_client
installed raises ImportError only when the flag is actually used.
"""

from __future__ import annotations

from typing import Optional, Sequence


class PromServeMetrics:
    def __init__(self, port: int, addr: str = "0.0.0.0"):
        from prometheus_client import (
            Counter,
            Gauge,
            Histogram,
            start_http_server,
        )

        self.flows = Gauge("tcsdn_flows_tracked", "live flows in the flow table")
        self.records = Counter("tcsdn_records_total", "accepted telemetry records")
        self.passes = Counter("tcsdn_predict_passes_total", "prediction passes run")
        self.latency = Histogram(
            "tcsdn_predict_seconds",
            "per-pass predict latency",
            buckets=(1e-4, 2.5e-4, 5e-4, 1e-3, 2.5e-3, 5e-3, 1e-2, 5e-2, 0.25, 1.0),
        )
        self.class_flows = Gauge(
            "tcsdn_class_flows", "flows per predicted class", ["label"]
        )
        start_http_server(port, addr=addr)

    def observe_pass(self, n_flows: int, n_records: int, predict_s: float,
                     labels: Optional[Sequence] = None) -> None:
        self.flows.set(n_flows)
        # counters are monotone: feed the delta since the last observation
        prev = getattr(self, "_last_records", 0)
        if n_records > prev:
            self.records.inc(n_records - prev)
        self._last_records = n_records
        self.passes.inc()
        self.latency.observe(predict_s)
        if labels is not None and len(labels):
            import numpy as np

            vals, counts = np.unique(np.asarray(labels).astype(str), return_counts=True)
            for v, c in zip(vals, counts):
                self.class_flows.labels(label=str(v)).set(int(c))
