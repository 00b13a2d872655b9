"""Real-time classification engine (the reference's serve path, rebuilt).

Pipeline per poll cycle (reference: traffic_classifier.py:144-171):
telemetry line -> flow-table update; every N accepted records (N=10, the
reference's PrettyTable cadence) -> batched feature matrix -> ONE device
predict over all live flows -> rendered table.

Differences from the reference, by design: prediction is
one batched kernel launch over every live flow instead of a Python loop of
batch-1 ``model.predict`` calls; on GPU the launch sequence is hipGraph-
captured (ops.gpu) to amortise launch overhead at high poll rates.
"""

from __future__ import annotations

import json
import sys
import time
from typing import Iterable, TextIO

import numpy as np

from .flow.parser import PollStreamParser
from .flow.state import FlowTable
from .models.base import Estimator
from .utils.schema import CLASS_NAMES, CSV_HEADER
from .utils.table import Table


def render_flow_table(table: FlowTable, labels) -> str:
    """The reference's console table (traffic_classifier.py:100-118)."""
    t = Table(["Flow ID", "Src MAC", "Dest MAC", "Traffic Type", "Forward Status", "Reverse Status"])
    metas = table.metas()
    statuses = table.statuses()
    for i, (meta, (fs, rs)) in enumerate(zip(metas, statuses)):
        t.add_row([i, meta.ethsrc, meta.ethdst, labels[i], fs, rs])
    return str(t)


def map_cluster_labels(idx: np.ndarray) -> np.ndarray:
    """Integer cluster ids -> class names (reference label map,
    traffic_classifier.py:109-114)."""
    names = np.asarray(CLASS_NAMES, dtype=object)
    idx = np.asarray(idx).ravel().astype(np.int64)
    out = np.empty(idx.shape, dtype=object)
    in_range = (idx >= 0) & (idx < len(names))
    out[in_range] = names[idx[in_range]]
    out[~in_range] = "unknown"
    return out


class RealtimeClassifier:
    """Streaming classify loop over any line source."""

    def __init__(
        self,
        model: Estimator,
        predict_every: int = 10,
        out: TextIO = sys.stdout,
        stats: bool = False,
        stats_out: TextIO = sys.stderr,
        prometheus_port: int = 0,
    ) -> None:
        self.model = model
        self.parser = PollStreamParser()
        self.predict_every = predict_every
        self.out = out
        self.stats = stats
        self.stats_out = stats_out
        self._last_batch = 0
        self.prom = None
        if prometheus_port:
            from .utils.prom import PromServeMetrics

            self.prom = PromServeMetrics(prometheus_port)

    def classify_now(self) -> np.ndarray:
        table = self.parser.table
        if len(table) == 0:
            return np.asarray([], dtype=object)
        X = table.feature_matrix(dtype=np.float32)
        pred = self.model.predict(X)
        if self.model.classes_ is None:  # unsupervised: cluster ids -> names
            names = getattr(self.model, "cluster_label_names_", None)
            if names is not None:
                # mode-based map learned at fit time (fit.py); the reference's
                # fixed index->name map silently mislabels
                idx = np.asarray(pred).ravel().astype(np.int64)
                pred = np.asarray(names, dtype=object)[idx]
            else:
                pred = map_cluster_labels(pred)
        return pred

    def feed(self, line) -> bool:
        """Feed one telemetry line; returns True when a prediction pass ran.

        The cadence counts accepted data records, mirroring the reference's
        per-line counter (traffic_classifier.py:167-171: its `time % 10`
        ticks once per flow row, not per second).
        """
        before = self.parser.records
        self.parser.feed(line)
        if self.parser.records == before:
            return False
        if (self.parser.records - self._last_batch) >= self.predict_every:
            self._last_batch = self.parser.records
            t0 = time.perf_counter()
            labels = self.classify_now()
            predict_s = time.perf_counter() - t0
            self.out.write(render_flow_table(self.parser.table, labels) + "\n")
            self.out.flush()
            if self.prom is not None:
                self.prom.observe_pass(
                    len(self.parser.table), self.parser.records, predict_s, labels
                )
            if self.stats:
                n = len(self.parser.table)
                self.stats_out.write(
                    json.dumps(
                        {
                            "flows": n,
                            "records": self.parser.records,
                            "predict_ms": predict_s * 1e3,
                            "flows_per_sec": n / predict_s if predict_s > 0 else None,
                            "device": str(getattr(self.model, "device", "cpu")),
                        }
                    )
                    + "\n"
                )
                self.stats_out.flush()
            return True
        return False

    def run(self, lines: Iterable) -> None:
        for line in lines:
            self.feed(line)


class TrainingCollector:
    """Training-data collection loop (reference: traffic_classifier.py:209-223):
    writes the 17-column header then one TSV row per tracked flow per
    accepted telemetry record."""

    def __init__(self, traffic_type: str, f: TextIO) -> None:
        self.traffic_type = traffic_type
        self.f = f
        self.parser = PollStreamParser()
        self.f.write(CSV_HEADER)

    def feed(self, line) -> None:
        before = self.parser.records
        self.parser.feed(line)
        if self.parser.records != before:
            for row in self.parser.table.training_rows(self.traffic_type):
                self.f.write(row + "\n")

    def run(self, lines: Iterable) -> None:
        for line in lines:
            self.feed(line)
