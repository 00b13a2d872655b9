"""Poll-stream parser: telemetry ``data\\t...`` lines -> flow-table updates.

The telemetry wire contract (one TSV line per live flow per poll) is the
load-bearing API of the reference monitor (reference: simple_monitor_13.py:66):

    data\\t<epoch>\\t<dpid>\\t<in_port>\\t<eth_src>\\t<eth_dst>\\t<out_port>\\t<packet_count>\\t<byte_count>

This module parses that stream into :class:`FlowTable` updates, mirroring the
field handling of the reference driver loop (reference:
traffic_classifier.py:147-165) — including ignoring non-``data`` lines.
"""

from __future__ import annotations

from typing import Iterable, Optional, Union

from .state import FlowTable

Line = Union[str, bytes]


class PollStreamParser:
    """Incremental parser feeding a :class:`FlowTable`.

    ``feed(line)`` returns the updated slot index for ``data`` lines and
    ``None`` for everything else.  ``records`` counts accepted data lines
    (the reference's per-line ``time`` counter used for the every-10-lines
    prediction cadence, traffic_classifier.py:167-171).
    """

    def __init__(self, table: Optional[FlowTable] = None) -> None:
        self.table = table if table is not None else FlowTable()
        self.records = 0
        self.bad_lines = 0

    def feed(self, line: Line) -> Optional[int]:
        if isinstance(line, bytes):
            if not line.startswith(b"data"):
                return None
            try:
                text = line.decode("utf-8", errors="strict")
            except UnicodeDecodeError:
                self.bad_lines += 1
                return None
        else:
            if not line.startswith("data"):
                return None
            text = line
        fields = text.rstrip("\r\n").split("\t")[1:]
        if len(fields) < 8:
            self.bad_lines += 1
            return None
        try:
            time = int(fields[0])
            packets = int(fields[6])
            bytes_ = int(fields[7])
        except ValueError:
            self.bad_lines += 1
            return None
        slot = self.table.update(
            time,
            fields[1],  # datapath id
            fields[2],  # in_port
            fields[3],  # eth_src
            fields[4],  # eth_dst
            fields[5],  # out_port
            packets,
            bytes_,
        )
        self.records += 1
        return slot

    def feed_many(self, lines: Iterable[Line]) -> int:
        n = 0
        for line in lines:
            if self.feed(line) is not None:
                n += 1
        return n


def format_record(
    time: int,
    datapath: int,
    in_port: int,
    eth_src: str,
    eth_dst: str,
    out_port: int,
    packets: int,
    bytes_: int,
) -> str:
    """Render one telemetry line exactly as the reference monitor logs it
    (simple_monitor_13.py:66: dpid/ports in hex, counters in decimal)."""
    return "data\t%s\t%x\t%x\t%s\t%s\t%x\t%d\t%d" % (
        time,
        datapath,
        in_port,
        eth_src,
        eth_dst,
        out_port,
        packets,
        bytes_,
    )


def replay(lines: Iterable[Line], table: Optional[FlowTable] = None) -> FlowTable:
    """Parse a full canned telemetry stream (tests / offline replay)."""
    p = PollStreamParser(table)
    p.feed_many(lines)
    return p.table
