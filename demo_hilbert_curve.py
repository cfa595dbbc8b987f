#!/usr/bin/env python3
"""Visualize Hilbert / zigzag patch scan orders (reference
demo_hilbert_curve.py:1-40+ and hilbert.py:373-473 visualizations).

    python demo_hilbert_curve.py [--grid 8] [--out hilbert_demo.png]
"""
import argparse

import numpy as np

from flaxdiff_amd.models.hilbert import hilbert_indices, zigzag_indices


def plot_scan(ax, idx, h, w, title):
    rows, cols = idx.numpy() // w, idx.numpy() % w
    ax.plot(cols, rows, "-o", markersize=3, linewidth=1)
    ax.set_title(title)
    ax.invert_yaxis()
    ax.set_aspect("equal")
    for i in (0, len(idx) - 1):
        ax.annotate(str(i), (cols[i], rows[i]), fontsize=8, color="red")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--grid", type=int, default=8)
    ap.add_argument("--out", type=str, default="hilbert_demo.png")
    args = ap.parse_args()

    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    h = w = args.grid
    fig, axes = plt.subplots(1, 3, figsize=(13, 4.2))
    plot_scan(axes[0], hilbert_indices(h, w), h, w, f"Hilbert {h}x{w}")
    plot_scan(axes[1], zigzag_indices(h, w), h, w, f"Zigzag (ZigMa) {h}x{w}")
    raster = hilbert_indices(h, w).sort().values  # 0..n-1 row-major
    plot_scan(axes[2], raster, h, w, f"Raster {h}x{w}")

    # locality stat: mean 2-D distance between sequence neighbors
    for name, idx in (("hilbert", hilbert_indices(h, w)),
                      ("zigzag", zigzag_indices(h, w)), ("raster", raster)):
        r, c = idx.numpy() // w, idx.numpy() % w
        d = np.abs(np.diff(r)) + np.abs(np.diff(c))
        print(f"{name:8s} mean |Δ2d| between neighbors: {d.mean():.3f}  "
              f"max: {d.max()}")

    fig.tight_layout()
    fig.savefig(args.out, dpi=120)
    print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
