"""Flow-state math parity with the reference Flow class
(traffic_classifier.py:29-96) and table key semantics (:144-171)."""

import numpy as np

from traffic_classifier_sdn_amd.flow.state import ACTIVE, INACTIVE, Flow, FlowTable


def test_create_initial_state():
    f = Flow(100, "1", "1", "aa", "bb", "2", 10, 1000)
    assert f.forward_packets == 10
    assert f.forward_bytes == 1000
    assert f.forward_delta_packets == 0
    assert f.forward_status == ACTIVE
    assert f.reverse_status == INACTIVE
    assert f.forward_inst_pps == 0.0


def test_update_forward_math():
    f = Flow(100, "1", "1", "aa", "bb", "2", 10, 1000)
    f.updateforward(30, 3000, 102)
    # deltas
    assert f.forward_delta_packets == 20
    assert f.forward_delta_bytes == 2000
    # avg uses cumulative / (t - t_start)
    assert f.forward_avg_pps == 30 / 2.0
    assert f.forward_avg_bps == 3000 / 2.0
    # inst uses delta / (t - t_last)
    assert f.forward_inst_pps == 20 / 2.0
    assert f.forward_inst_bps == 2000 / 2.0
    assert f.forward_status == ACTIVE


def test_division_guards():
    # same timestamp: rates untouched (guards at traffic_classifier.py:66-67)
    f = Flow(100, "1", "1", "aa", "bb", "2", 10, 1000)
    f.updateforward(30, 3000, 100)
    assert f.forward_inst_pps == 0.0
    assert f.forward_avg_pps == 0.0
    assert f.forward_delta_packets == 20


def test_inactive_on_zero_delta():
    f = Flow(100, "1", "1", "aa", "bb", "2", 10, 1000)
    f.updateforward(10, 1000, 105)
    assert f.forward_status == INACTIVE
    f.updateforward(11, 1100, 106)
    assert f.forward_status == ACTIVE


def test_reverse_direction_resolution():
    t = FlowTable()
    t.update(100, "1", "1", "aa", "bb", "2", 10, 1000)
    # reversed (src,dst) updates the same flow's reverse side
    slot = t.update(101, "1", "2", "bb", "aa", "1", 7, 700)
    assert len(t) == 1
    assert slot == 0
    st = t.statuses()[0]
    assert st == (ACTIVE, ACTIVE)
    feats = t.feature_matrix(dtype=np.float64)[0]
    assert feats[6] == 7  # delta reverse packets (first reverse obs)
    assert feats[7] == 700


def test_distinct_flows_get_slots():
    t = FlowTable()
    t.update(100, "1", "1", "aa", "bb", "2", 1, 10)
    t.update(100, "1", "1", "cc", "dd", "2", 2, 20)
    t.update(100, "2", "1", "aa", "bb", "2", 3, 30)  # different datapath
    assert len(t) == 3


def test_feature_matrix_order():
    t = FlowTable()
    t.update(100, "1", "1", "aa", "bb", "2", 10, 1000)
    t.update(102, "1", "1", "aa", "bb", "2", 30, 3000)
    t.update(103, "1", "2", "bb", "aa", "1", 5, 500)
    f = t.feature_matrix(dtype=np.float64)[0]
    # schema order: dFp, dFb, Fipps, Fapps, FiBps, FaBps, dRp, dRb, ...
    assert f[0] == 20 and f[1] == 2000
    assert f[2] == 10.0 and f[3] == 15.0
    assert f[4] == 1000.0 and f[5] == 1500.0
    assert f[6] == 5 and f[7] == 500
    np.testing.assert_allclose(f[8], 5 / 3.0)


def test_training_rows_format():
    t = FlowTable()
    t.update(100, "1", "1", "aa", "bb", "2", 10, 1000)
    rows = t.training_rows("dns")
    parts = rows[0].split("\t")
    assert len(parts) == 17
    assert parts[0] == "10" and parts[1] == "1000"
    assert parts[-1] == "dns"
    assert parts[4] == "0.0"  # float formatting like the reference str()


def test_grow_beyond_capacity():
    t = FlowTable(capacity=2)
    for i in range(10):
        t.update(100, "1", "1", f"src{i}", f"dst{i}", "2", i, i * 10)
    assert len(t) == 10
    assert t.feature_matrix().shape == (10, 12)
