"""C++ flow table parity with the Python FlowTable on identical streams."""

import os

import numpy as np
import pytest

from traffic_classifier_sdn_amd.flow import native
from traffic_classifier_sdn_amd.flow.parser import PollStreamParser
from traffic_classifier_sdn_amd.flow.replay import TelemetryReplaySource

pytestmark = pytest.mark.skipif(not native.HAVE_NATIVE, reason="native ext not built")


def _both(polls=10, seed=5):
    lines = list(TelemetryReplaySource(seed=seed).stream(polls))
    p_py = PollStreamParser()
    p_py.feed_many(lines)
    p_nat = native.NativePollParser()
    p_nat.feed_buffer("\n".join(lines) + "\n")
    return p_py, p_nat


def test_feature_matrix_parity():
    p_py, p_nat = _both()
    a = p_py.table.feature_matrix(dtype=np.float32)
    b = p_nat.table.feature_matrix()
    assert a.shape == b.shape
    np.testing.assert_allclose(b, a, rtol=1e-7, atol=1e-7)
    assert p_nat.records == p_py.records
    assert len(p_nat.table) == len(p_py.table)


def test_counters_snapshot_parity():
    p_py, p_nat = _both(seed=6)
    for x, y in zip(p_py.table.counters_snapshot(), p_nat.table.counters_snapshot()):
        np.testing.assert_allclose(y, x)


def test_statuses_and_metas():
    p_py, p_nat = _both(seed=7)
    assert p_nat.table.statuses() == p_py.table.statuses()
    mp = [(m.datapath, m.inport, m.ethsrc, m.ethdst, m.outport) for m in p_py.table.metas()]
    assert p_nat.table.metas() == mp


def test_malformed_lines_counted():
    p = native.NativePollParser()
    assert p.feed("data\t100\t1") is None
    assert p.feed("garbage") is None
    assert p.feed("data\tx\t1\t1\ta\tb\t2\t1\t2") is None
    assert p.bad_lines == 2
    assert p.records == 0


def test_bulk_vs_per_line():
    lines = list(TelemetryReplaySource(seed=8).stream(5))
    a = native.NativePollParser()
    a.feed_many(lines)
    b = native.NativePollParser()
    b.feed_buffer("\n".join(lines))
    np.testing.assert_allclose(b.table.feature_matrix(), a.table.feature_matrix())


def test_sanitizer_harness(tmp_path):
    """Build the pure-C++ core under ASan+UBSan and run the adversarial /
    randomized / semantic property harness (SURVEY.md §5 sanitizer parity)."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    src = os.path.join(repo, "tools", "flowtable_san.cpp")
    exe = str(tmp_path / "flowtable_san")
    build = subprocess.run(
        ["g++", "-std=c++17", "-O1", "-g", "-fsanitize=address,undefined",
         "-fno-sanitize-recover=all", src, "-o", exe],
        capture_output=True, text=True, timeout=180,
    )
    assert build.returncode == 0, build.stderr
    run = subprocess.run([exe], capture_output=True, text=True, timeout=120)
    assert run.returncode == 0, run.stdout + run.stderr
    assert "SANITIZER HARNESS OK" in run.stdout


def test_differential_python_vs_native_hypothesis():
    """Property test: the Python FlowTable and the C++ core produce
    identical feature matrices on arbitrary telemetry streams, including
    same-timestamp polls (division guards), counter resets (negative
    deltas), and interleaved forward/reverse observations."""
    hyp = pytest.importorskip("hypothesis")
    from hypothesis import given, settings, strategies as st

    from traffic_classifier_sdn_amd.flow.native import HAVE_NATIVE, NativePollParser

    if not HAVE_NATIVE:
        pytest.skip("native extension not built")

    rec = st.tuples(
        st.integers(min_value=0, max_value=3),      # time step (0 = repeat)
        st.integers(min_value=0, max_value=2),      # dpid
        st.integers(min_value=0, max_value=3),      # src host
        st.integers(min_value=0, max_value=3),      # dst host
        st.integers(min_value=0, max_value=10**6),  # packets
        st.integers(min_value=0, max_value=10**9),  # bytes
    )

    @settings(max_examples=60, deadline=None)
    @given(st.lists(rec, min_size=1, max_size=60))
    def run(records):
        py = PollStreamParser()
        nat = NativePollParser()
        t = 1000
        for dt, dp, s, d, pk, by in records:
            if s == d:
                continue
            t += dt
            line = f"data\t{t}\t{dp}\t1\th{s:02d}\th{d:02d}\t2\t{pk}\t{by}"
            py.feed(line)
            nat.feed(line)
        assert len(py.table) == len(nat.table)
        if len(py.table):
            a = py.table.feature_matrix(dtype=np.float32)
            b = nat.table.feature_matrix()
            np.testing.assert_allclose(a, b, rtol=1e-6, atol=1e-6)

    run()
