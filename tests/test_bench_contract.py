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
    "knn-approx": ["--workload", "knn", "--knn-approx", "--steps", "2", "--warmup", "1",
                   "--knn-queries", "2048"],
    "svc-fit": ["--workload", "svc-fit", "--steps", "2", "--warmup", "1", "--svc-iters-per-step", "5"],
    "svc-fit-full": ["--workload", "svc-fit-full", "--steps", "1", "--warmup", "0",
                     "--svc-rows", "1500", "--svc-full-max-iter", "60"],
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


def _run_multirank(extra, n, timeout=420):
    """Bare `python bench.py --gpus N` must self-spawn N ranks (VERDICT r01
    weak #1: the flag used to be parsed and ignored, so the driver's SCALE
    run would have measured 1 rank at every N)."""
    env = dict(os.environ)
    env.pop("WORLD_SIZE", None)
    env.pop("RANK", None)
    env["OMP_NUM_THREADS"] = "1"
    r = subprocess.run(
        [sys.executable, "bench.py", "--gpus", str(n)] + extra,
        cwd=REPO, capture_output=True, text=True, timeout=timeout, env=env,
    )
    assert r.returncode == 0, (r.stdout[-800:], r.stderr[-1500:])
    lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout[-800:]  # exactly one JSON line (rank 0)
    d = json.loads(lines[0])
    for k in REQUIRED:
        assert k in d, k
    assert d["n_gpus"] == n
    assert d["value"] > 0
    return d


def test_gpus_flag_spawns_ranks():
    d = _run_multirank(["--steps", "2", "--warmup", "1", "--rows-per-gpu", "2000"], 4)
    assert d["config"]["parallelism"] == "dp4"


@pytest.mark.parametrize(
    "extra",
    [
        ["--steps", "2", "--warmup", "1", "--rows-per-gpu", "1000"],
        ["--workload", "knn", "--steps", "2", "--warmup", "1",
         "--knn-queries", "256", "--knn-ref-rows-per-gpu", "1000"],
        ["--workload", "svc-fit", "--steps", "2", "--warmup", "1",
         "--svc-iters-per-step", "3", "--svc-rows", "4000"],
        ["--workload", "svc-fit-full", "--steps", "1", "--warmup", "0",
         "--svc-rows", "2400", "--svc-full-max-iter", "30"],
        ["--workload", "serve", "--steps", "2", "--warmup", "1", "--serve-flows", "64"],
        ["--workload", "rf-fit", "--steps", "1", "--warmup", "0",
         "--rf-fit-rows", "1000", "--rf-fit-trees", "2"],
    ],
    ids=["rf", "knn", "svc-fit", "svc-fit-full", "serve", "rf-fit"],
)
def test_world8_dress_rehearsal(extra):
    """Every workload's control flow at world size 8 (gloo, tiny shapes) so
    the first 8x MI355X driver run has no untested branch."""
    _run_multirank(extra, 8)
