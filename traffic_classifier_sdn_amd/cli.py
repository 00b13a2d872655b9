"""CLI with the reference's exact subcommand surface
(reference: traffic_classifier.py:174-246), plus flags for the new engine.

    python -m traffic_classifier_sdn_amd train <TypeOfData> [--source ...]
    python -m traffic_classifier_sdn_amd <algo>              [--source ...]

Subcommands: train, logistic, kmeans, knearest, svm, Randomforest,
gaussiannb (SUBCOMMANDS at traffic_classifier.py:189; the reference's
unreachable-`knearest` loader bug, SURVEY.md §2.1, is fixed: `knearest`
loads the KNeighbors checkpoint).

Telemetry sources:
  --source subprocess   spawn a monitor command (default: the in-package
                        OpenFlow-1.3 monitor) and scrape its stdout — the
                        reference's process model (traffic_classifier.py:228)
  --source stdin        read `data\t` TSV lines from stdin
  --source replay       built-in synthetic telemetry (no network needed)
"""

from __future__ import annotations

import argparse
import os
import signal
import subprocess
import sys
from typing import Iterable, Optional

SUBCOMMANDS = ("train", "logistic", "kmeans", "knearest", "svm", "Randomforest", "gaussiannb")

DEFAULT_TIMEOUT = 15 * 60  # training-collection window (traffic_classifier.py:27)


def _monitor_cmd() -> str:
    # the in-package native OpenFlow 1.3 monitor (no Ryu dependency)
    return f"{sys.executable} -m traffic_classifier_sdn_amd.flow.monitor"


def _line_source(args) -> Iterable:
    if args.source == "stdin":
        return sys.stdin
    if args.source == "replay":
        from .flow.replay import TelemetryReplaySource

        return TelemetryReplaySource(seed=args.seed).stream(args.replay_polls)
    # subprocess: spawn monitor, scrape stdout (reference process model)
    cmd = args.monitor_cmd or _monitor_cmd()
    p = subprocess.Popen(
        cmd, shell=True, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        preexec_fn=os.setsid,
    )

    def gen():
        try:
            for line in p.stdout:
                yield line
        finally:
            try:
                os.killpg(os.getpgid(p.pid), signal.SIGTERM)
            except (ProcessLookupError, PermissionError):
                pass

    return gen()


def _checkpoint_path(algo: str, models_dir: str) -> str:
    """Resolve the checkpoint like the reference loader (models/<Name>),
    trying the engine .npz first, then the sklearn pickle, then the
    in-repo converted reference checkpoints (data/ref_models) so a fresh
    clone serves out of the box."""
    from .models import ALGO_TO_CHECKPOINT

    fname, _ = ALGO_TO_CHECKPOINT[algo]
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for d in (models_dir, os.path.join(repo, "data", "ref_models")):
        for cand in (os.path.join(d, fname + ".npz"), os.path.join(d, fname)):
            if os.path.exists(cand):
                return cand
    return os.path.join(models_dir, fname)


def main(argv: Optional[list] = None) -> int:
    parser = argparse.ArgumentParser(
        prog="traffic_classifier_sdn_amd",
        description=__doc__,
        formatter_class=argparse.RawDescriptionHelpFormatter,
    )
    parser.add_argument("subcommand", choices=SUBCOMMANDS)
    parser.add_argument("traffic_type", nargs="?", help="traffic class (train mode)")
    parser.add_argument("--source", choices=("subprocess", "stdin", "replay"), default="subprocess")
    parser.add_argument("--monitor-cmd", default=None, help="override monitor command")
    parser.add_argument("--models-dir", default="models", help="checkpoint directory")
    parser.add_argument("--device", default=None, help="cpu / cuda (default: auto)")
    parser.add_argument("--timeout", type=int, default=DEFAULT_TIMEOUT, help="train collection seconds")
    parser.add_argument("--replay-polls", type=int, default=30)
    parser.add_argument("--seed", type=int, default=0)
    parser.add_argument("--prometheus", type=int, default=0, metavar="PORT",
                        help="expose serve metrics on a Prometheus scrape port")
    parser.add_argument("--stats", action="store_true",
                        help="emit one JSON perf line (flows, predict_ms, flows/sec) per prediction pass on stderr")
    args = parser.parse_args(argv)

    if args.subcommand == "train":
        if not args.traffic_type:
            print("ERROR: specify traffic type.\n", file=sys.stderr)
            return 2
        from .serve import TrainingCollector

        out_path = f"{args.traffic_type}_training_data.csv"
        lines = _line_source(args)
        with open(out_path, "w") as f:
            collector = TrainingCollector(args.traffic_type, f)

            if args.source == "subprocess":
                # the reference's 15-min SIGALRM window
                # (traffic_classifier.py:214-215)
                def _alarm(signum, frame):
                    raise KeyboardInterrupt

                signal.signal(signal.SIGALRM, _alarm)
                signal.alarm(args.timeout)
            try:
                collector.run(lines)
            except KeyboardInterrupt:
                print("Finished collecting data.")
        return 0

    # serve mode: load checkpoint, stream, classify
    from .models import load_model
    from .serve import RealtimeClassifier

    path = _checkpoint_path(args.subcommand, args.models_dir)
    if not os.path.exists(path):
        print(f"ERROR: checkpoint {path} not found", file=sys.stderr)
        return 2
    model = load_model(path, device=args.device)
    rc = RealtimeClassifier(model, stats=args.stats, prometheus_port=args.prometheus)
    try:
        rc.run(_line_source(args))
    except KeyboardInterrupt:
        pass
    return 0


if __name__ == "__main__":
    sys.exit(main())
