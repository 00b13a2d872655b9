"""Flagship benchmark: RandomForest (100 trees, 6-class) flow classification
throughput — flows/sec on synthetic flow-stat rows (BASELINE.json config #2),
weak-scaled over N GPUs (one rank per GPU, RCCL).

    python bench.py --gpus N --steps K --warmup W [--rows-per-gpu R]

Per step, every rank classifies its resident R-row shard with the packed-
forest HIP traversal kernel (ops.gpu.rf_argmax).  The timed region is
bracketed by a barrier + torch.cuda.synchronize on both sides; elapsed time
is the MAX over ranks; rank 0 prints one JSON line.

The forest is the reference's own 100-tree 6-class checkpoint (converted to
data/ref_models/RandomForestClassifier.npz), so tree shapes/depths match the
named config; rows are drawn from the real flow-stat rows with jitter.  The
model's 6-class accuracy on the real held-out split is reported alongside
(99.87%).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from traffic_classifier_sdn_amd.models import load_model
from traffic_classifier_sdn_amd.parallel import dist
from traffic_classifier_sdn_amd.utils.datasets import (
    load_reference_dataset,
    synthetic_flow_rows,
    train_test_split_ref,
)
from traffic_classifier_sdn_amd.utils.metrics import accuracy

REPO = os.path.dirname(os.path.abspath(__file__))
RF_CKPT = os.path.join(REPO, "data", "ref_models", "RandomForestClassifier.npz")


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--rows-per-gpu", type=int, default=10_000_000)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    use_gpu = torch.cuda.is_available()
    rank, world = dist.init_from_env("nccl" if use_gpu else "gloo")
    if use_gpu:
        torch.cuda.set_device(dist.env_local_rank())
        device = f"cuda:{dist.env_local_rank()}"
    else:
        device = "cpu"
        args.rows_per_gpu = min(args.rows_per_gpu, 20_000)

    model = load_model(RF_CKPT, device=device)

    # measured (non-timed) accuracy on the real data, reference split
    X_real, y_real = load_reference_dataset()
    _, Xte, _, yte = train_test_split_ref(X_real, y_real)
    acc = accuracy(yte, model.predict(Xte))

    # resident synthetic shard, seeded per rank
    Xn = synthetic_flow_rows(
        args.rows_per_gpu, seed=args.seed + 1000 * rank, reference_X=X_real
    )
    X = torch.from_numpy(Xn).to(device)
    if use_gpu:
        torch.cuda.synchronize()

    def step():
        out = model.predict_index(X)
        return out

    for _ in range(args.warmup):
        step()
    if use_gpu:
        torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if use_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    # MAX over ranks
    e = torch.tensor([elapsed], dtype=torch.float64, device=device if use_gpu else "cpu")
    if dist.is_initialized():
        torch.distributed.all_reduce(e, op=torch.distributed.ReduceOp.MAX)
    dist.barrier()
    elapsed = float(e[0])

    total_rows = args.rows_per_gpu * world * args.steps
    flows_per_sec = total_rows / elapsed
    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "flows/sec, RandomForest 100-tree 6-class predict on synthetic flow-stat rows",
                    "value": flows_per_sec,
                    "unit": "flows/s",
                    "n_gpus": world,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": elapsed / args.steps * 1000.0,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "fp32",
                    "data": "synthetic",
                    "config": {
                        "model": "RandomForestClassifier-100trees-6class",
                        "global_batch": args.rows_per_gpu * world,
                        "seq_len": 12,
                        "parallelism": f"dp{world}",
                        "accuracy_6class": acc,
                        "accuracy_published_ref": 0.9987,
                    },
                }
            )
        )
    if dist.is_initialized():
        torch.distributed.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
