"""Canonical feature schema for SDN flow classification.

This is the framework's core dtype: every estimator consumes rows of the
12-feature vector below, in exactly this order.  The order matches the
serve-time vector built by the reference driver
(reference: traffic_classifier.py:104) and the ``feature_names_in_``
attribute of every shipped sklearn checkpoint.

The 17-column training CSV layout (16 features + label) matches the header
written by the reference collector (reference: traffic_classifier.py:217).
"""

from __future__ import annotations

# The 6 traffic classes, in sklearn's sorted-label order
# (reference: traffic_classifier.py:109-114).
CLASS_NAMES = ("dns", "game", "ping", "quake", "telnet", "voice")
NUM_CLASSES = len(CLASS_NAMES)
CLASS_TO_INDEX = {name: i for i, name in enumerate(CLASS_NAMES)}

# Canonical 12 model features, in serve order (reference:
# traffic_classifier.py:104).  Names are byte-for-byte the CSV header
# fields so that checkpoints' feature_names_in_ round-trips, including the
# reference's "DeltaReverse" typo in feature 8.
FEATURE_NAMES = (
    "Delta Forward Packets",
    "Delta Forward Bytes",
    "Forward Instantaneous Packets per Second",
    "Forward Average Packets per second",
    "Forward Instantaneous Bytes per Second",
    "Forward Average Bytes per second",
    "Delta Reverse Packets",
    "Delta Reverse Bytes",
    "DeltaReverse Instantaneous Packets per Second",
    "Reverse Average Packets per second",
    "Reverse Instantaneous Bytes per Second",
    "Reverse Average Bytes per second",
)
NUM_FEATURES = len(FEATURE_NAMES)

# Cumulative counter columns present in training CSVs but dropped before
# fitting (reference: notebooks/1_log_Kmeans.ipynb cell 18).
CUMULATIVE_NAMES = (
    "Forward Packets",
    "Forward Bytes",
    "Reverse Packets",
    "Reverse Bytes",
)

# Full 17-column training CSV header, byte-for-byte the reference header
# string (reference: traffic_classifier.py:217).
CSV_HEADER_COLUMNS = (
    "Forward Packets",
    "Forward Bytes",
    "Delta Forward Packets",
    "Delta Forward Bytes",
    "Forward Instantaneous Packets per Second",
    "Forward Average Packets per second",
    "Forward Instantaneous Bytes per Second",
    "Forward Average Bytes per second",
    "Reverse Packets",
    "Reverse Bytes",
    "Delta Reverse Packets",
    "Delta Reverse Bytes",
    "DeltaReverse Instantaneous Packets per Second",
    "Reverse Average Packets per second",
    "Reverse Instantaneous Bytes per Second",
    "Reverse Average Bytes per second",
    "Traffic Type",
)
CSV_HEADER = "\t".join(CSV_HEADER_COLUMNS) + "\n"

LABEL_COLUMN = "Traffic Type"
