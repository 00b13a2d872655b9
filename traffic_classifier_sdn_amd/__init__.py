"""MI355X-native SDN traffic-flow classification framework.

A from-scratch rebuild of the capabilities of ashwinn-v/Traffic-classifier-SDN
(see SURVEY.md): OpenFlow-1.3 flow-stats telemetry -> bidirectional flow
feature extraction -> six flow classifiers (LogisticRegression, RBF-SVC,
GaussianNB, KNeighbors, RandomForest, KMeans), each with hand-written CDNA4
HIP kernels for fit and predict, RCCL-over-xGMI data parallelism, and
sklearn-pickle-compatible checkpoints.
"""

__version__ = "0.1.0"

from .utils import schema  # noqa: F401
from .utils.schema import CLASS_NAMES, FEATURE_NAMES, NUM_CLASSES, NUM_FEATURES  # noqa: F401


def __getattr__(name):
    # lazy imports keep `import traffic_classifier_sdn_amd` light (torch is
    # only pulled in when models/ops are touched)
    if name in ("models", "ops", "flow", "parallel", "utils", "serve"):
        import importlib

        return importlib.import_module(f".{name}", __name__)
    if name == "load_model":
        from .models import load_model

        return load_model
    raise AttributeError(name)
