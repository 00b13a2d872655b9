"""D-ITG traffic-script generation (reference: D-IGT_scripts/*, SURVEY.md
§2.1 C13) — emits ITGSend multi-flow script files for each traffic class
used to collect training data on the Mininet testbed.

    python -m traffic_classifier_sdn_amd.ditg [--out D-IGT_scripts]
        [--dst 10.0.0.1] [--classes voice,quake,telnet,game,dns,all]

Each line is one ITGSend flow: ``-a <dst> -rp <port> <application> [opts]``.
Applications mirror the reference's choices (all_script_file:1-5): VoIP with
G.711 ×2 over RTP with voice-activity detection (voice), Quake3 (quake),
Telnet (telnet), CSa = Counter-Strike active player (game), DNS (dns).
The ping class needs no script — it is plain ICMP echo from the hosts
(reference README.md workflow).
"""

from __future__ import annotations

import argparse
import os
from typing import Dict, List, Optional, Sequence

# class -> (receive port, ITGSend application spec)
FLOW_SPECS: Dict[str, tuple] = {
    "voice": (10001, "VoIP -x G.711.2 -h RTP -VAD"),
    "quake": (10002, "Quake3"),
    "telnet": (10002, "Telnet"),
    "game": (10002, "CSa"),
    "dns": (10003, "DNS"),
}

# the reference's multi-flow mix (all_script_file line order)
ALL_ORDER = ("voice", "quake", "telnet", "game", "dns")


def script_lines(cls: str, dst: str = "10.0.0.1") -> List[str]:
    """ITGSend script lines for one traffic class ('all' = the 5-flow mix)."""
    if cls == "all":
        return [script_lines(c, dst)[0] for c in ALL_ORDER]
    if cls not in FLOW_SPECS:
        raise KeyError(f"no D-ITG spec for class {cls!r} (choices: {', '.join(FLOW_SPECS)}, all)")
    port, app = FLOW_SPECS[cls]
    return [f"-a {dst} -rp {port} {app}"]


def write_scripts(out_dir: str, dst: str = "10.0.0.1", classes: Optional[Sequence[str]] = None) -> List[str]:
    """Write <class>_script_file for each class (+ all_script_file); returns
    the written paths.  File naming matches the reference directory."""
    classes = list(classes) if classes else list(ALL_ORDER) + ["all"]
    os.makedirs(out_dir, exist_ok=True)
    written = []
    for cls in classes:
        path = os.path.join(out_dir, f"{cls}_script_file")
        with open(path, "w") as f:
            f.write("\n".join(script_lines(cls, dst)) + "\n")
        written.append(path)
    return written


def main(argv: Optional[Sequence[str]] = None) -> int:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--out", default="D-IGT_scripts")
    ap.add_argument("--dst", default="10.0.0.1")
    ap.add_argument("--classes", default=None, help="comma list (default: all five + the mix)")
    args = ap.parse_args(argv)
    classes = [c.strip() for c in args.classes.split(",")] if args.classes else None
    for p in write_scripts(args.out, dst=args.dst, classes=classes):
        print(p)
    return 0


if __name__ == "__main__":
    import sys

    sys.exit(main())
