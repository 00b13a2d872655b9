"""Dataset analysis — the reference's PCA / cluster study as a runnable
module (reference: notebooks/1_log_Kmeans.ipynb cells 63-131, SURVEY.md
§2.1 C10).

    python -m traffic_classifier_sdn_amd.analysis [--device cuda|cpu] [--json]

Reproduces, on the engine's own tensor stack (device-resident when a GPU is
present; the 12-dim covariance eigenproblem is tiny so the hot work is the
standardize + projection passes over the rows):

- StandardScaler + PCA: explained-variance ratio of the leading components
  (notebook: PC1 47.68% + PC2 33.43% = 81.11% on the 6-class dataset)
- LogisticRegression trained on the 2 leading PCs: held-out accuracy
  (notebook cell 91: 83.03%)
- KMeans on the standardized rows with the mode-based cluster→class
  assignment and its supervised accuracy (notebook cells 104-125)

``--plots DIR`` additionally writes the notebook's three figures as SVG
(PCA scatter / LR-on-2PC decision boundary / KMeans cluster scatter —
notebook cells 85, 98, 126) via the dependency-free utils.svgplot writer;
without the flag only the numbers the notebook reads off its figures are
returned.
"""

from __future__ import annotations

import argparse
import json
from typing import Dict, Optional, Sequence

import numpy as np
import torch

from .models import KMeans, LogisticRegression
from .utils.datasets import load_reference_dataset, train_test_split_ref
from .utils.metrics import accuracy
from .utils.schema import CLASS_NAMES


def standardize(X: torch.Tensor):
    """StandardScaler fit_transform (ddof=0, sklearn semantics)."""
    mean = X.mean(dim=0)
    std = X.std(dim=0, unbiased=False)
    std = torch.where(std == 0, torch.ones_like(std), std)
    return (X - mean) / std, mean, std


def pca(X: torch.Tensor, n_components: int = 2):
    """PCA via eigendecomposition of the feature covariance (12×12 — the
    row pass is the only O(n) work).  Returns (components [k,F],
    explained_variance_ratio [k], projected rows [n,k])."""
    Xc = X - X.mean(dim=0)
    n = Xc.shape[0]
    cov = (Xc.T @ Xc) / (n - 1)
    evals, evecs = torch.linalg.eigh(cov)  # ascending
    order = torch.argsort(evals, descending=True)
    evals = evals[order]
    evecs = evecs[:, order]
    ratio = evals / evals.sum()
    comps = evecs[:, :n_components].T.contiguous()
    # sklearn sign convention: largest-|.| entry of each component positive
    signs = torch.sign(comps[torch.arange(comps.shape[0]), comps.abs().argmax(dim=1)])
    comps = comps * signs[:, None]
    proj = Xc @ comps.T
    return comps, ratio[:n_components], proj


def run_analysis(device: Optional[str] = None, seed: int = 101, plots: Optional[str] = None) -> Dict:
    dev = device or ("cuda" if torch.cuda.is_available() else "cpu")
    X, y = load_reference_dataset()
    Xt = torch.as_tensor(X, dtype=torch.float64, device=dev)
    Xs, _, _ = standardize(Xt)

    comps, ratio, proj = pca(Xs, n_components=2)
    out: Dict = {
        "n_rows": int(X.shape[0]),
        "n_features": int(X.shape[1]),
        "device": str(Xt.device),
        "pca_explained_variance_ratio": [float(r) for r in ratio],
        "pca_explained_variance_total": float(ratio.sum()),
    }

    # LR on the 2 leading PCs, the notebook's 50/50 split
    P = proj.cpu().numpy()
    Ptr, Pte, ytr, yte = train_test_split_ref(P, y, random_state=seed)
    lr2 = LogisticRegression(device=dev).fit(Ptr, ytr)
    out["lr_accuracy_on_2pc"] = accuracy(yte, lr2.predict(Pte))

    # KMeans on standardized rows + mode-based cluster->class assignment
    from .fit import mode_cluster_assignment

    km = KMeans(n_clusters=len(CLASS_NAMES), device=dev)
    km.fit(Xs.cpu().numpy() if dev == "cpu" else Xs)
    cluster_ids = km.predict_index(Xs.to(torch.float32)).cpu().numpy()
    names = mode_cluster_assignment(cluster_ids, y, km.n_clusters)
    out["kmeans_inertia"] = float(km.inertia_)
    out["kmeans_n_iter"] = int(km.n_iter_)
    out["kmeans_mode_assignment"] = [str(n) for n in names]
    out["kmeans_supervised_accuracy"] = accuracy(y, names[cluster_ids])

    if plots:
        import os

        from .utils.svgplot import scatter_svg

        os.makedirs(plots, exist_ok=True)
        classes = sorted(set(str(v) for v in y))
        lut = {c: i for i, c in enumerate(classes)}
        y_idx = np.asarray([lut[str(v)] for v in y])
        written = [
            scatter_svg(
                os.path.join(plots, "pca_scatter.svg"), P, y_idx, classes,
                "PCA of flow features (true class)",
            ),
            scatter_svg(
                os.path.join(plots, "pca_decision_boundary.svg"), Pte,
                np.asarray([lut[str(v)] for v in yte]), classes,
                f"LR on 2 PCs — decision regions (acc {out['lr_accuracy_on_2pc']:.3f})",
                decision_fn=lambda G: np.asarray(
                    [np.searchsorted(lr2.classes_.astype(str), str(v)) for v in lr2.predict(G)]
                ),
            ),
            scatter_svg(
                os.path.join(plots, "kmeans_clusters.svg"), P, cluster_ids,
                [f"cluster {i} -> {names[i]}" for i in range(km.n_clusters)],
                "KMeans clusters in PC space (mode-assigned class)",
            ),
        ]
        out["plots"] = written
    return out


def main(argv: Optional[Sequence[str]] = None) -> int:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--device", default=None)
    ap.add_argument("--json", action="store_true")
    ap.add_argument("--plots", default=None, metavar="DIR",
                    help="write the notebook's three figures as SVG into DIR")
    args = ap.parse_args(argv)
    res = run_analysis(device=args.device, plots=args.plots)
    if args.json:
        print(json.dumps(res))
    else:
        print(f"rows={res['n_rows']} features={res['n_features']} device={res['device']}")
        r = res["pca_explained_variance_ratio"]
        print(
            f"PCA explained variance: PC1 {r[0]*100:.2f}% + PC2 {r[1]*100:.2f}% "
            f"= {res['pca_explained_variance_total']*100:.2f}%  (notebook: 47.68+33.43=81.11%)"
        )
        print(f"LogisticRegression on 2 PCs: acc={res['lr_accuracy_on_2pc']:.4f} (notebook: 0.8303)")
        print(
            f"KMeans: inertia={res['kmeans_inertia']:.4g} iters={res['kmeans_n_iter']} "
            f"mode-assignment acc={res['kmeans_supervised_accuracy']:.4f}"
        )
        print(f"cluster -> class: {res['kmeans_mode_assignment']}")
    return 0


if __name__ == "__main__":
    import sys

    sys.exit(main())
