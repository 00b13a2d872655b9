"""Bidirectional per-flow state tracking.

Reimplements, with identical arithmetic, the flow-statistics object and flow
table of the reference driver (reference: traffic_classifier.py:29-96 for the
``Flow`` math, :144-171 for the table/key semantics):

* cumulative packet/byte counters per direction,
* per-poll deltas,
* instantaneous rates   = delta   / (t - t_last)    (guard: skip when equal),
* average rates         = counter / (t - t_start)   (guard: skip when equal),
* ACTIVE/INACTIVE status per direction (INACTIVE when either delta is zero),
* forward/reverse direction resolution: the first observed (src, dst)
  direction of a flow is "forward"; a line whose (dst, src) key matches an
  existing flow updates that flow's reverse direction.

A vectorised, array-backed ``FlowTable`` is the framework-native container:
flow state lives in parallel numpy arrays so that the feature matrix for a
whole poll cycle is a zero-copy slice handed to the GPU predict path, instead
of a per-flow Python-object walk (reference hot loop #2,
traffic_classifier.py:103-118).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Iterator, List, Optional, Tuple

import numpy as np



ACTIVE = "ACTIVE"
INACTIVE = "INACTIVE"

# Column indices of the per-flow state matrix (float64).
_TIME_START = 0
# forward block
_F_PKTS = 1
_F_BYTES = 2
_F_DELTA_PKTS = 3
_F_DELTA_BYTES = 4
_F_INST_PPS = 5
_F_AVG_PPS = 6
_F_INST_BPS = 7
_F_AVG_BPS = 8
_F_LAST_TIME = 9
# reverse block
_R_PKTS = 10
_R_BYTES = 11
_R_DELTA_PKTS = 12
_R_DELTA_BYTES = 13
_R_INST_PPS = 14
_R_AVG_PPS = 15
_R_INST_BPS = 16
_R_AVG_BPS = 17
_R_LAST_TIME = 18
_F_ACTIVE = 19
_R_ACTIVE = 20
# raw counter snapshots for the GPU feature-extraction path: previous
# cumulative counters and the update times bracketing the last change of
# each direction (serve_gpu hands these to the flow_features kernel)
_F_PREV_PKTS = 21
_F_PREV_BYTES = 22
_F_PREV_TIME = 23
_R_PREV_PKTS = 24
_R_PREV_BYTES = 25
_R_PREV_TIME = 26
_STATE_COLS = 27

# Feature-matrix column order == utils.schema.FEATURE_NAMES order.
_FEATURE_COLS = np.array(
    [
        _F_DELTA_PKTS,
        _F_DELTA_BYTES,
        _F_INST_PPS,
        _F_AVG_PPS,
        _F_INST_BPS,
        _F_AVG_BPS,
        _R_DELTA_PKTS,
        _R_DELTA_BYTES,
        _R_INST_PPS,
        _R_AVG_PPS,
        _R_INST_BPS,
        _R_AVG_BPS,
    ],
    dtype=np.int64,
)

# 16-column training-row order (reference: traffic_classifier.py:124-141).
_TRAIN_COLS = np.array(
    [
        _F_PKTS,
        _F_BYTES,
        _F_DELTA_PKTS,
        _F_DELTA_BYTES,
        _F_INST_PPS,
        _F_AVG_PPS,
        _F_INST_BPS,
        _F_AVG_BPS,
        _R_PKTS,
        _R_BYTES,
        _R_DELTA_PKTS,
        _R_DELTA_BYTES,
        _R_INST_PPS,
        _R_AVG_PPS,
        _R_INST_BPS,
        _R_AVG_BPS,
    ],
    dtype=np.int64,
)


@dataclass
class FlowMeta:
    """Identity of one tracked flow (first-observed = forward direction)."""

    datapath: str
    inport: str
    ethsrc: str
    ethdst: str
    outport: str


class FlowTable:
    """Array-backed flow table with reference-identical update semantics.

    Keys are ``(datapath, ethsrc, ethdst)`` tuples (the reference hashes the
    concatenated strings, traffic_classifier.py:157; a tuple key is the
    collision-free equivalent and stable across processes).
    """

    def __init__(self, capacity: int = 1024) -> None:
        self._index: Dict[Tuple[str, str, str], int] = {}
        self._meta: List[FlowMeta] = []
        self._state = np.zeros((capacity, _STATE_COLS), dtype=np.float64)
        self._n = 0

    def __len__(self) -> int:
        return self._n

    def _grow(self) -> None:
        cap = self._state.shape[0]
        new = np.zeros((cap * 2, _STATE_COLS), dtype=np.float64)
        new[: self._n] = self._state[: self._n]
        self._state = new

    # ------------------------------------------------------------------
    # update path (one telemetry record)
    # ------------------------------------------------------------------
    def update(
        self,
        time: int,
        datapath: str,
        inport: str,
        ethsrc: str,
        ethdst: str,
        outport: str,
        packets: int,
        bytes_: int,
    ) -> int:
        """Apply one poll record; returns the flow's slot index.

        Mirrors the create/update-forward/update-reverse resolution of
        reference traffic_classifier.py:157-165.
        """
        key = (datapath, ethsrc, ethdst)
        slot = self._index.get(key)
        if slot is not None:
            self._update_forward(slot, packets, bytes_, time)
            return slot
        rev_key = (datapath, ethdst, ethsrc)
        slot = self._index.get(rev_key)
        if slot is not None:
            self._update_reverse(slot, packets, bytes_, time)
            return slot
        return self._create(key, time, datapath, inport, ethsrc, ethdst, outport, packets, bytes_)

    def _create(
        self,
        key: Tuple[str, str, str],
        time: int,
        datapath: str,
        inport: str,
        ethsrc: str,
        ethdst: str,
        outport: str,
        packets: int,
        bytes_: int,
    ) -> int:
        if self._n == self._state.shape[0]:
            self._grow()
        slot = self._n
        self._n += 1
        self._index[key] = slot
        self._meta.append(FlowMeta(datapath, inport, ethsrc, ethdst, outport))
        s = self._state[slot]
        s[:] = 0.0
        s[_TIME_START] = time
        s[_F_PKTS] = packets
        s[_F_BYTES] = bytes_
        s[_F_LAST_TIME] = time
        s[_R_LAST_TIME] = time
        # forward starts ACTIVE, reverse INACTIVE (traffic_classifier.py:47,59)
        s[_F_ACTIVE] = 1.0
        s[_R_ACTIVE] = 0.0
        s[_F_PREV_PKTS] = packets
        s[_F_PREV_BYTES] = bytes_
        s[_F_PREV_TIME] = time
        s[_R_PREV_PKTS] = 0.0
        s[_R_PREV_BYTES] = 0.0
        s[_R_PREV_TIME] = time
        return slot

    def _update_forward(self, slot: int, packets: int, bytes_: int, time: int) -> None:
        # reference: traffic_classifier.py:63-78
        s = self._state[slot]
        s[_F_PREV_PKTS] = s[_F_PKTS]
        s[_F_PREV_BYTES] = s[_F_BYTES]
        s[_F_PREV_TIME] = s[_F_LAST_TIME]
        s[_F_DELTA_PKTS] = packets - s[_F_PKTS]
        s[_F_PKTS] = packets
        if time != s[_TIME_START]:
            s[_F_AVG_PPS] = packets / float(time - s[_TIME_START])
        if time != s[_F_LAST_TIME]:
            s[_F_INST_PPS] = s[_F_DELTA_PKTS] / float(time - s[_F_LAST_TIME])
        s[_F_DELTA_BYTES] = bytes_ - s[_F_BYTES]
        s[_F_BYTES] = bytes_
        if time != s[_TIME_START]:
            s[_F_AVG_BPS] = bytes_ / float(time - s[_TIME_START])
        if time != s[_F_LAST_TIME]:
            s[_F_INST_BPS] = s[_F_DELTA_BYTES] / float(time - s[_F_LAST_TIME])
        s[_F_LAST_TIME] = time
        s[_F_ACTIVE] = 0.0 if (s[_F_DELTA_BYTES] == 0 or s[_F_DELTA_PKTS] == 0) else 1.0

    def _update_reverse(self, slot: int, packets: int, bytes_: int, time: int) -> None:
        # reference: traffic_classifier.py:81-96
        s = self._state[slot]
        s[_R_PREV_PKTS] = s[_R_PKTS]
        s[_R_PREV_BYTES] = s[_R_BYTES]
        s[_R_PREV_TIME] = s[_R_LAST_TIME]
        s[_R_DELTA_PKTS] = packets - s[_R_PKTS]
        s[_R_PKTS] = packets
        if time != s[_TIME_START]:
            s[_R_AVG_PPS] = packets / float(time - s[_TIME_START])
        if time != s[_R_LAST_TIME]:
            s[_R_INST_PPS] = s[_R_DELTA_PKTS] / float(time - s[_R_LAST_TIME])
        s[_R_DELTA_BYTES] = bytes_ - s[_R_BYTES]
        s[_R_BYTES] = bytes_
        if time != s[_TIME_START]:
            s[_R_AVG_BPS] = bytes_ / float(time - s[_TIME_START])
        if time != s[_R_LAST_TIME]:
            s[_R_INST_BPS] = s[_R_DELTA_BYTES] / float(time - s[_R_LAST_TIME])
        s[_R_LAST_TIME] = time
        s[_R_ACTIVE] = 0.0 if (s[_R_DELTA_BYTES] == 0 or s[_R_DELTA_PKTS] == 0) else 1.0

    # ------------------------------------------------------------------
    # batch read-out
    # ------------------------------------------------------------------
    def feature_matrix(self, dtype=np.float32) -> np.ndarray:
        """(n_flows, 12) feature matrix in canonical schema order."""
        return self._state[: self._n][:, _FEATURE_COLS].astype(dtype)

    def training_matrix(self) -> np.ndarray:
        """(n_flows, 16) training-row matrix (cumulative + 12 features)."""
        return self._state[: self._n][:, _TRAIN_COLS]

    def statuses(self) -> List[Tuple[str, str]]:
        """(forward_status, reverse_status) per flow."""
        out = []
        for i in range(self._n):
            s = self._state[i]
            out.append(
                (
                    ACTIVE if s[_F_ACTIVE] else INACTIVE,
                    ACTIVE if s[_R_ACTIVE] else INACTIVE,
                )
            )
        return out

    def counters_snapshot(self):
        """Raw per-flow counter state for the GPU feature-extraction kernel:
        (cur[n,4], prev[n,4], times[n,6]) float64 arrays with layouts
        cur/prev = [fwd_pkts, fwd_bytes, rev_pkts, rev_bytes],
        times = [tf_cur, tf_prev, tr_cur, tr_prev, t_start, 0].
        ops.flow_features(cur, prev, times) reproduces feature_matrix()
        exactly (same math, same division guards)."""
        st = self._state[: self._n]
        cur = st[:, [_F_PKTS, _F_BYTES, _R_PKTS, _R_BYTES]].copy()
        prev = st[:, [_F_PREV_PKTS, _F_PREV_BYTES, _R_PREV_PKTS, _R_PREV_BYTES]].copy()
        times = np.zeros((self._n, 6), dtype=np.float64)
        times[:, 0] = st[:, _F_LAST_TIME]
        times[:, 1] = st[:, _F_PREV_TIME]
        times[:, 2] = st[:, _R_LAST_TIME]
        times[:, 3] = st[:, _R_PREV_TIME]
        times[:, 4] = st[:, _TIME_START]
        return cur, prev, times

    def metas(self) -> List[FlowMeta]:
        return list(self._meta)

    def keys(self) -> Iterator[Tuple[str, str, str]]:
        return iter(self._index.keys())

    def slot_of(self, key: Tuple[str, str, str]) -> Optional[int]:
        return self._index.get(key)

    def training_rows(self, traffic_type: str) -> List[str]:
        """TSV training rows, one per tracked flow, formatted exactly as the
        reference collector writes them (traffic_classifier.py:121-142:
        ``str()`` of ints for counters/deltas, ``str()`` of floats for rates).
        """
        rows = []
        mat = self._state[: self._n]
        for i in range(self._n):
            s = mat[i]
            vals = [
                str(int(s[_F_PKTS])),
                str(int(s[_F_BYTES])),
                str(int(s[_F_DELTA_PKTS])),
                str(int(s[_F_DELTA_BYTES])),
                str(s[_F_INST_PPS]),
                str(s[_F_AVG_PPS]),
                str(s[_F_INST_BPS]),
                str(s[_F_AVG_BPS]),
                str(int(s[_R_PKTS])),
                str(int(s[_R_BYTES])),
                str(int(s[_R_DELTA_PKTS])),
                str(int(s[_R_DELTA_BYTES])),
                str(s[_R_INST_PPS]),
                str(s[_R_AVG_PPS]),
                str(s[_R_INST_BPS]),
                str(s[_R_AVG_BPS]),
                str(traffic_type),
            ]
            rows.append("\t".join(vals))
        return rows


class Flow:
    """Single-flow object API mirroring the reference ``Flow`` class
    (reference: traffic_classifier.py:29-96), backed by a one-slot
    :class:`FlowTable`.  Provided for API parity; the batch path should use
    :class:`FlowTable` directly.
    """

    def __init__(self, time_start, datapath, inport, ethsrc, ethdst, outport, packets, bytes):
        self._t = FlowTable(capacity=1)
        self._t.update(int(time_start), str(datapath), str(inport), str(ethsrc), str(ethdst), str(outport), int(packets), int(bytes))
        self.datapath = datapath
        self.inport = inport
        self.ethsrc = ethsrc
        self.ethdst = ethdst
        self.outport = outport

    def updateforward(self, packets, bytes, curr_time):
        self._t._update_forward(0, int(packets), int(bytes), int(curr_time))

    def updatereverse(self, packets, bytes, curr_time):
        self._t._update_reverse(0, int(packets), int(bytes), int(curr_time))

    def _get(self, col):
        return self._t._state[0][col]

    # attribute views matching the reference's names
    time_start = property(lambda self: self._get(_TIME_START))
    forward_packets = property(lambda self: int(self._get(_F_PKTS)))
    forward_bytes = property(lambda self: int(self._get(_F_BYTES)))
    forward_delta_packets = property(lambda self: int(self._get(_F_DELTA_PKTS)))
    forward_delta_bytes = property(lambda self: int(self._get(_F_DELTA_BYTES)))
    forward_inst_pps = property(lambda self: self._get(_F_INST_PPS))
    forward_avg_pps = property(lambda self: self._get(_F_AVG_PPS))
    forward_inst_bps = property(lambda self: self._get(_F_INST_BPS))
    forward_avg_bps = property(lambda self: self._get(_F_AVG_BPS))
    forward_status = property(lambda self: ACTIVE if self._get(_F_ACTIVE) else INACTIVE)
    reverse_packets = property(lambda self: int(self._get(_R_PKTS)))
    reverse_bytes = property(lambda self: int(self._get(_R_BYTES)))
    reverse_delta_packets = property(lambda self: int(self._get(_R_DELTA_PKTS)))
    reverse_delta_bytes = property(lambda self: int(self._get(_R_DELTA_BYTES)))
    reverse_inst_pps = property(lambda self: self._get(_R_INST_PPS))
    reverse_avg_pps = property(lambda self: self._get(_R_AVG_PPS))
    reverse_inst_bps = property(lambda self: self._get(_R_INST_BPS))
    reverse_avg_bps = property(lambda self: self._get(_R_AVG_BPS))
    reverse_status = property(lambda self: ACTIVE if self._get(_R_ACTIVE) else INACTIVE)

    def features(self) -> np.ndarray:
        """The 12-feature serve vector (reference: traffic_classifier.py:104)."""
        return self._t.feature_matrix(dtype=np.float64)[0]
