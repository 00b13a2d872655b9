"""Driver-contract guard: every bench workload emits one JSON line with the
required schema fields on rank 0 (the round driver parses this exactly)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = [
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
]

CASES = {
    "rf": ["--steps", "2", "--warmup", "1"],
    "knn": ["--workload", "knn", "--steps", "2", "--warmup", "1", "--knn-queries", "2048"],
    "svc-fit": ["--workload", "svc-fit", "--steps", "2", "--warmup", "1", "--svc-iters-per-step", "5"],
    "serve": ["--workload", "serve", "--steps", "3", "--warmup", "1"],
    "rf-fit": ["--workload", "rf-fit", "--steps", "2", "--warmup", "1"],
}


@pytest.mark.parametrize("name", sorted(CASES))
def test_bench_emits_driver_contract(name):
    r = subprocess.run(
        [sys.executable, "bench.py"] + CASES[name],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stderr[-1500:]
    line = r.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for k in REQUIRED:
        assert k in d, k
    assert d["n_gpus"] == 1
    assert d["value"] > 0
    assert isinstance(d["config"], dict) and d["config"]
    assert d["scaling"] in ("weak", "strong")
    assert d["data"] == "synthetic"
