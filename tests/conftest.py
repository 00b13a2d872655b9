import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

REFERENCE_DIR = "/root/reference"


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an MI355X GPU")


def has_reference():
    return os.path.isdir(REFERENCE_DIR)


@pytest.fixture(scope="session")
def dataset():
    """(X[n,12] f64, y[n] object) — reference CSVs or packed in-repo copy."""
    from traffic_classifier_sdn_amd.utils import datasets

    try:
        return datasets.load_reference_dataset()
    except FileNotFoundError:
        pytest.skip("no dataset available")


@pytest.fixture(scope="session")
def split(dataset):
    from traffic_classifier_sdn_amd.utils.datasets import train_test_split_ref

    X, y = dataset
    return train_test_split_ref(X, y)


@pytest.fixture(scope="session")
def rng():
    return np.random.default_rng(42)
