"""GPU real-time serve engine (BASELINE.json config #5).

Per poll cycle: the host flow table tracks raw counters only; the GPU runs
feature extraction (flow_features kernel: the reference's rate math,
traffic_classifier.py:63-96) fused back-to-back with the ensemble's predict
kernels, all captured once in a hipGraph (torch.cuda.CUDAGraph == hipGraph
on ROCm) and replayed per poll — one graph launch instead of K kernel
launches, sized for a 1 ms poll cadence.

Buffers are fixed-capacity (graphs need static shapes); flows beyond
capacity fall back to a plain (non-graph) launch path.
"""

from __future__ import annotations

import time
from typing import Dict, Optional

import numpy as np
import torch

from .flow.state import FlowTable
from .models.base import Estimator


class GpuServeEngine:
    def __init__(
        self,
        models: Dict[str, Estimator],
        capacity: int = 8192,
        use_graph: bool = True,
        device: str = "cuda",
    ) -> None:
        self.models = models
        self.capacity = capacity
        self.device = torch.device(device)
        self.use_graph = use_graph and self.device.type == "cuda"
        # pinned staging + device buffers (static for graph capture)
        pin = self.device.type == "cuda"
        self.h_cur = torch.zeros(capacity, 4, dtype=torch.float64, pin_memory=pin)
        self.h_prev = torch.zeros(capacity, 4, dtype=torch.float64, pin_memory=pin)
        self.h_times = torch.zeros(capacity, 6, dtype=torch.float64, pin_memory=pin)
        self.d_cur = torch.zeros(capacity, 4, dtype=torch.float64, device=self.device)
        self.d_prev = torch.zeros(capacity, 4, dtype=torch.float64, device=self.device)
        self.d_times = torch.zeros(capacity, 6, dtype=torch.float64, device=self.device)
        self.d_labels: Dict[str, torch.Tensor] = {}
        self.h_labels: Dict[str, torch.Tensor] = {}
        self._graph: Optional[torch.cuda.CUDAGraph] = None
        self.last_latency_s = 0.0

    # -- compute pipeline (captured) -----------------------------------
    def _compute(self) -> None:
        from . import ops

        X = ops.flow_features(self.d_cur, self.d_prev, self.d_times)
        for name, model in self.models.items():
            labels = model.predict_index(X)
            if name in self.d_labels:
                self.d_labels[name].copy_(labels)
            else:
                self.d_labels[name] = labels
                self.h_labels[name] = torch.empty_like(
                    labels, device="cpu", pin_memory=self.device.type == "cuda"
                )

    def _capture(self) -> None:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):  # warmup allocations outside the graph
                self._compute()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._compute()
        self._graph = g

    # -- per-poll entry -------------------------------------------------
    def classify(self, table: FlowTable) -> Dict[str, np.ndarray]:
        """Run one classification pass over every live flow; returns
        {model_name: labels[n] int32} (class indices)."""
        n = len(table)
        t0 = time.perf_counter()
        if n > self.capacity:
            return self._classify_unbounded(table)
        cur, prev, times = table.counters_snapshot()
        self.h_cur[:n] = torch.from_numpy(cur)
        self.h_prev[:n] = torch.from_numpy(prev)
        self.h_times[:n] = torch.from_numpy(times)
        if self.device.type == "cuda":
            self.d_cur.copy_(self.h_cur, non_blocking=True)
            self.d_prev.copy_(self.h_prev, non_blocking=True)
            self.d_times.copy_(self.h_times, non_blocking=True)
            if self.use_graph:
                if self._graph is None:
                    self._capture()
                self._graph.replay()
            else:
                self._compute()
            out = {}
            for name in self.models:
                self.h_labels[name].copy_(self.d_labels[name], non_blocking=True)
            torch.cuda.synchronize()
            for name in self.models:
                out[name] = self.h_labels[name][:n].numpy().copy()
        else:
            self.d_cur.copy_(self.h_cur)
            self.d_prev.copy_(self.h_prev)
            self.d_times.copy_(self.h_times)
            self._compute()
            out = {name: self.d_labels[name][:n].numpy().copy() for name in self.models}
        self.last_latency_s = time.perf_counter() - t0
        return out

    def _classify_unbounded(self, table: FlowTable) -> Dict[str, np.ndarray]:
        from . import ops

        t0 = time.perf_counter()
        cur, prev, times = table.counters_snapshot()
        to = lambda a: torch.from_numpy(a).to(self.device)
        X = ops.flow_features(to(cur), to(prev), to(times))
        out = {}
        for name, model in self.models.items():
            out[name] = model.predict_index(X).cpu().numpy()
        self.last_latency_s = time.perf_counter() - t0
        return out
