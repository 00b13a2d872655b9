"""Fake telemetry sources for tests and offline benchmarking.

The reference can only be exercised against a live Mininet/OVS testbed
(reference: README.md:26-34).  This module synthesises the same ``data\\t``
TSV stream (wire format of simple_monitor_13.py:66) from simple statistical
traffic models, so the whole monitor -> feature -> predict path runs in CI
without root, Ryu, or a network.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Iterator, List, Optional, Sequence

import numpy as np

from .parser import format_record


@dataclass
class SynthFlowSpec:
    """One synthetic bidirectional flow: mean per-second rates per direction."""

    eth_src: str
    eth_dst: str
    fwd_pps: float
    fwd_Bpp: float  # bytes per packet
    rev_pps: float
    rev_Bpp: float
    in_port: int = 1
    out_port: int = 2
    datapath: int = 1


def default_specs() -> List[SynthFlowSpec]:
    """A small mixed-traffic scenario (rates loosely shaped like the
    reference's D-ITG classes: VoIP, DNS, telnet, game)."""
    return [
        SynthFlowSpec("00:00:00:00:00:01", "00:00:00:00:00:02", 50.0, 214.0, 50.0, 214.0),   # voice-like
        SynthFlowSpec("00:00:00:00:00:03", "00:00:00:00:00:04", 2.0, 80.0, 2.0, 120.0),      # dns-like
        SynthFlowSpec("00:00:00:00:00:05", "00:00:00:00:00:06", 5.0, 60.0, 5.0, 1000.0),     # telnet-like
        SynthFlowSpec("00:00:00:00:00:07", "00:00:00:00:00:08", 30.0, 90.0, 28.0, 90.0),     # game-like
    ]


class TelemetryReplaySource:
    """Generates poll cycles of ``data\\t`` TSV lines for a set of flows.

    Each poll advances wall-clock by ``interval`` seconds and each flow's
    cumulative counters by a Poisson-ish draw around its mean rates.  Both
    directions of a flow appear as separate lines (as OVS reports two
    unidirectional entries), so the parser's forward/reverse key resolution
    (traffic_classifier.py:157-165) is exercised.
    """

    def __init__(
        self,
        specs: Optional[Sequence[SynthFlowSpec]] = None,
        interval: int = 1,
        t0: int = 1_600_000_000,
        seed: int = 0,
        jitter: bool = True,
    ) -> None:
        self.specs = list(specs) if specs is not None else default_specs()
        self.interval = interval
        self.t = t0
        self.rng = np.random.default_rng(seed)
        self.jitter = jitter
        # cumulative (pkts, bytes) per direction per flow
        self._state: List[List[int]] = [[0, 0, 0, 0] for _ in self.specs]

    def poll(self) -> List[str]:
        """Advance one poll cycle; return the telemetry lines."""
        self.t += self.interval
        lines: List[str] = []
        for spec, st in zip(self.specs, self._state):
            for direction in (0, 1):
                pps = spec.fwd_pps if direction == 0 else spec.rev_pps
                bpp = spec.fwd_Bpp if direction == 0 else spec.rev_Bpp
                mean_pkts = pps * self.interval
                dp = int(self.rng.poisson(mean_pkts)) if self.jitter else int(round(mean_pkts))
                db = int(dp * bpp)
                st[direction * 2 + 0] += dp
                st[direction * 2 + 1] += db
                src = spec.eth_src if direction == 0 else spec.eth_dst
                dst = spec.eth_dst if direction == 0 else spec.eth_src
                lines.append(
                    format_record(
                        self.t,
                        spec.datapath,
                        spec.in_port if direction == 0 else spec.out_port,
                        src,
                        dst,
                        spec.out_port if direction == 0 else spec.in_port,
                        st[direction * 2 + 0],
                        st[direction * 2 + 1],
                    )
                )
        return lines

    def stream(self, polls: int) -> Iterator[str]:
        for _ in range(polls):
            yield from self.poll()
