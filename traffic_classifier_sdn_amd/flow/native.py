"""Python facade over the native C++ flow table (csrc/flowtable.cpp).

``best_flow_table()`` returns the native implementation when the extension
is built (it is built in-tree by setup.py alongside the HIP extension) and
the pure-Python :class:`FlowTable` otherwise.  Both expose the same batch
read-out surface (feature_matrix / counters_snapshot / statuses / metas).
"""

from __future__ import annotations

from typing import Optional

try:
    from ._tcsdn_native import NativeFlowTable  # type: ignore

    HAVE_NATIVE = True
except ImportError:  # pragma: no cover - depends on build
    NativeFlowTable = None  # type: ignore
    HAVE_NATIVE = False

from .state import FlowTable


def best_flow_table(prefer_native: bool = True):
    if prefer_native and HAVE_NATIVE:
        return NativeFlowTable()
    return FlowTable()


class NativePollParser:
    """Drop-in for PollStreamParser backed by the C++ bulk parser."""

    def __init__(self) -> None:
        if not HAVE_NATIVE:
            raise RuntimeError("_tcsdn_native extension not built")
        self.table = NativeFlowTable()

    @property
    def records(self) -> int:
        return self.table.records

    @property
    def bad_lines(self) -> int:
        return self.table.bad_lines

    def feed(self, line) -> Optional[int]:
        if isinstance(line, bytes):
            line = line.decode("utf-8", errors="replace")
        slot = self.table.feed_line(line)
        return None if slot < 0 else slot

    def feed_many(self, lines) -> int:
        n = 0
        for line in lines:
            if self.feed(line) is not None:
                n += 1
        return n

    def feed_buffer(self, buf) -> int:
        if isinstance(buf, bytes):
            buf = buf.decode("utf-8", errors="replace")
        return self.table.feed_buffer(buf)
