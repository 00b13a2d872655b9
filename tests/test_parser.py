"""Telemetry TSV parser + replay source tests (wire contract of
simple_monitor_13.py:66, driver parse loop traffic_classifier.py:147-165)."""

import numpy as np

from traffic_classifier_sdn_amd.flow.parser import PollStreamParser, format_record, replay
from traffic_classifier_sdn_amd.flow.replay import TelemetryReplaySource


def test_format_record_wire_format():
    line = format_record(1600000000, 1, 1, "00:aa", "00:bb", 2, 15, 1500)
    assert line == "data\t1600000000\t1\t1\t00:aa\t00:bb\t2\t15\t1500"


def test_parser_accepts_bytes_and_str():
    p = PollStreamParser()
    assert p.feed("data\t100\t1\t1\taa\tbb\t2\t10\t1000") == 0
    assert p.feed(b"data\t101\t1\t1\taa\tbb\t2\t20\t2000") == 0
    assert p.records == 2
    assert len(p.table) == 1


def test_parser_ignores_non_data_lines():
    p = PollStreamParser()
    assert p.feed("time\tdatapath\tin-port") is None
    assert p.feed(b"loading app simple_monitor_13.py") is None
    assert p.feed("") is None
    assert p.records == 0


def test_parser_rejects_malformed():
    p = PollStreamParser()
    assert p.feed("data\t100\t1") is None
    assert p.feed("data\tnotanumber\t1\t1\taa\tbb\t2\t10\t1000") is None
    assert p.bad_lines == 2


def test_replay_round_trip():
    src = TelemetryReplaySource(seed=1)
    table = replay(src.stream(5))
    # each spec produces a forward and reverse entry resolved to one flow
    assert len(table) == len(src.specs)
    X = table.feature_matrix()
    assert X.shape == (len(src.specs), 12)
    assert np.isfinite(X).all()
    # after several polls, rates are populated
    assert (X[:, 3] > 0).all()  # forward avg pps


def test_replay_reverse_resolution():
    src = TelemetryReplaySource(seed=2)
    p = PollStreamParser()
    p.feed_many(src.poll())
    st = p.table.statuses()
    # reverse direction was observed for every flow
    assert all(s[1] in ("ACTIVE", "INACTIVE") for s in st)
    assert p.records == 2 * len(src.specs)
