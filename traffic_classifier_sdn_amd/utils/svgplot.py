"""Minimal dependency-free SVG plotting for the analysis module.

The reference notebooks render three matplotlib figures (SURVEY.md §2.1
C10: notebooks/1_log_Kmeans.ipynb cells 85 / 98 / 126 — PCA scatter,
LR-on-2PC decision boundary, KMeans cluster scatter).  matplotlib is not in
the image, so this writes the equivalent figures as plain SVG: a scatter
plot and a decision-region raster, both tiny hand-rolled documents.
"""

from __future__ import annotations

from typing import Callable, Optional, Sequence

import numpy as np

# 6-class palette (colorblind-safe-ish hues)
PALETTE = ["#4477aa", "#ee6677", "#228833", "#ccbb44", "#66ccee", "#aa3377",
           "#bbbbbb", "#000000"]

W, H, PAD = 640, 480, 48


def _scale(v: np.ndarray, lo: float, hi: float, out_lo: float, out_hi: float) -> np.ndarray:
    span = (hi - lo) or 1.0
    return out_lo + (v - lo) / span * (out_hi - out_lo)


def _axes(title: str, xlabel: str, ylabel: str) -> list:
    return [
        f'<svg xmlns="http://www.w3.org/2000/svg" width="{W}" height="{H}" '
        f'viewBox="0 0 {W} {H}">',
        f'<rect width="{W}" height="{H}" fill="white"/>',
        f'<text x="{W/2}" y="20" text-anchor="middle" font-size="15" '
        f'font-family="sans-serif">{title}</text>',
        f'<text x="{W/2}" y="{H-8}" text-anchor="middle" font-size="12" '
        f'font-family="sans-serif">{xlabel}</text>',
        f'<text x="14" y="{H/2}" text-anchor="middle" font-size="12" '
        f'font-family="sans-serif" transform="rotate(-90 14 {H/2})">{ylabel}</text>',
        f'<rect x="{PAD}" y="{PAD/2+8}" width="{W-2*PAD}" height="{H-2*PAD}" '
        f'fill="none" stroke="#999"/>',
    ]


def _legend(parts: list, labels: Sequence[str]) -> None:
    for i, lab in enumerate(labels):
        y = PAD / 2 + 24 + i * 16
        parts.append(
            f'<circle cx="{W-PAD+14}" cy="{y}" r="4" fill="{PALETTE[i % len(PALETTE)]}"/>'
            f'<text x="{W-PAD+22}" y="{y+4}" font-size="11" '
            f'font-family="sans-serif">{lab}</text>'
        )


def scatter_svg(
    path: str,
    xy: np.ndarray,
    labels: np.ndarray,
    label_names: Sequence[str],
    title: str,
    xlabel: str = "PC1",
    ylabel: str = "PC2",
    max_points: int = 2500,
    seed: int = 0,
    decision_fn: Optional[Callable[[np.ndarray], np.ndarray]] = None,
    grid: int = 64,
) -> str:
    """Scatter of xy[n,2] colored by integer ``labels``; when
    ``decision_fn`` is given (rows[m,2] -> class index[m]) the background is
    rasterized into decision regions first (the notebook's contourf)."""
    xy = np.asarray(xy, dtype=np.float64)
    labels = np.asarray(labels)
    if xy.shape[0] > max_points:
        idx = np.random.default_rng(seed).choice(xy.shape[0], max_points, replace=False)
        xy, labels = xy[idx], labels[idx]
    x_lo, x_hi = np.percentile(xy[:, 0], [0.5, 99.5])
    y_lo, y_hi = np.percentile(xy[:, 1], [0.5, 99.5])
    parts = _axes(title, xlabel, ylabel)

    if decision_fn is not None:
        gx = np.linspace(x_lo, x_hi, grid)
        gy = np.linspace(y_lo, y_hi, grid)
        GX, GY = np.meshgrid(gx, gy)
        cls = np.asarray(decision_fn(np.stack([GX.ravel(), GY.ravel()], axis=1)))
        cls = cls.reshape(grid, grid)
        cw = (W - 2 * PAD) / grid
        ch = (H - 2 * PAD) / grid
        for iy in range(grid):
            for ix in range(grid):
                c = PALETTE[int(cls[iy, ix]) % len(PALETTE)]
                px = PAD + ix * cw
                py = PAD / 2 + 8 + (H - 2 * PAD) - (iy + 1) * ch
                parts.append(
                    f'<rect x="{px:.1f}" y="{py:.1f}" width="{cw + 0.5:.1f}" '
                    f'height="{ch + 0.5:.1f}" fill="{c}" fill-opacity="0.25"/>'
                )

    px = _scale(np.clip(xy[:, 0], x_lo, x_hi), x_lo, x_hi, PAD, W - PAD)
    py = _scale(np.clip(xy[:, 1], y_lo, y_hi), y_lo, y_hi, H - PAD, PAD / 2 + 8)
    for i in range(xy.shape[0]):
        c = PALETTE[int(labels[i]) % len(PALETTE)]
        parts.append(f'<circle cx="{px[i]:.1f}" cy="{py[i]:.1f}" r="2.2" fill="{c}" fill-opacity="0.75"/>')
    _legend(parts, label_names)
    parts.append("</svg>")
    doc = "\n".join(parts)
    with open(path, "w") as f:
        f.write(doc)
    return path


def confusion_svg(path: str, cm, labels: Sequence[str], title: str) -> str:
    """Confusion-matrix heatmap (the notebooks' seaborn heatmap, e.g.
    notebooks/1_log_Kmeans.ipynb cell 59) as a standalone SVG."""
    cm = np.asarray(cm, dtype=np.float64)
    n = cm.shape[0]
    cell = min(64, (W - 2 * PAD) // n)
    x0 = PAD + 40
    y0 = PAD / 2 + 24
    parts = [
        f'<svg xmlns="http://www.w3.org/2000/svg" width="{W}" height="{H}" '
        f'viewBox="0 0 {W} {H}">',
        f'<rect width="{W}" height="{H}" fill="white"/>',
        f'<text x="{W/2}" y="18" text-anchor="middle" font-size="15" '
        f'font-family="sans-serif">{title}</text>',
    ]
    vmax = cm.max() or 1.0
    for i in range(n):
        for j in range(n):
            v = cm[i, j] / vmax
            # white -> blue ramp
            r = int(255 * (1 - 0.75 * v))
            g = int(255 * (1 - 0.55 * v))
            x, y = x0 + j * cell, y0 + i * cell
            parts.append(
                f'<rect x="{x}" y="{y}" width="{cell}" height="{cell}" '
                f'fill="rgb({r},{g},255)" stroke="#ddd"/>'
            )
            tcol = "#000" if v < 0.6 else "#fff"
            parts.append(
                f'<text x="{x + cell/2}" y="{y + cell/2 + 4}" text-anchor="middle" '
                f'font-size="11" font-family="sans-serif" fill="{tcol}">{int(cm[i, j])}</text>'
            )
    for k, lab in enumerate(labels):
        parts.append(
            f'<text x="{x0 + k*cell + cell/2}" y="{y0 + n*cell + 14}" '
            f'text-anchor="middle" font-size="10" font-family="sans-serif">{lab[:6]}</text>'
        )
        parts.append(
            f'<text x="{x0 - 6}" y="{y0 + k*cell + cell/2 + 3}" text-anchor="end" '
            f'font-size="10" font-family="sans-serif">{lab[:6]}</text>'
        )
    parts.append(
        f'<text x="{x0 + n*cell/2}" y="{y0 + n*cell + 30}" text-anchor="middle" '
        f'font-size="11" font-family="sans-serif">predicted</text>'
    )
    parts.append("</svg>")
    with open(path, "w") as f:
        f.write("\n".join(parts))
    return path
