"""Convert the read-only reference assets into framework-native artifacts.

Produces (committed to the repo so GPU boxes / CI work without the
/root/reference mount):
  data/flows.npz            - the shipped training CSVs, packed
  data/ref_models/<name>.npz - the six sklearn checkpoints, converted via the
                               shadow reader to the framework .npz format

Run:  python tools/convert_reference.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from traffic_classifier_sdn_amd.utils import checkpoint as ckpt
from traffic_classifier_sdn_amd.utils import datasets

REF_MODELS = "/root/reference/models"
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

NAMES = [
    "LogisticRegression",
    "GaussianNB",
    "KMeans_Clustering",
    "SVC",
    "KNeighbors",
    "RandomForestClassifier",
]


def main() -> None:
    os.makedirs(os.path.join(REPO, "data", "ref_models"), exist_ok=True)
    datasets.save_packed_dataset()
    print("wrote", datasets.PACKED_DATASET)
    for name in NAMES:
        src = os.path.join(REF_MODELS, name)
        dst = os.path.join(REPO, "data", "ref_models", name + ".npz")
        params = ckpt.load_reference_checkpoint(src)
        ckpt.save_params_npz(params, dst)
        print("wrote", dst)


if __name__ == "__main__":
    main()
