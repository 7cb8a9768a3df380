"""Offline plotting of training logs and per-generation fitness dumps.

Mirrors the reference's viz utilities (``src/utils/viz.py:28-79``):
``graph_log`` parses the ``k:v`` lines of a ``saved/<run>/es.log`` (or any
reporter log) and plots selected series; ``graph_fits`` plots the per-gen
fitness ``.npy`` dumps written by DefaultReporterSet.
"""
from __future__ import annotations

import os
import re
from typing import Dict, List, Optional

import numpy as np


def parse_log(path: str) -> Dict[str, List[float]]:
    series: Dict[str, List[float]] = {}
    pat = re.compile(r"(?:INFO:root:)?([\w\- ]+):(-?[\d.]+(?:e-?\d+)?)$")
    with open(path) as f:
        for line in f:
            m = pat.match(line.strip())
            if m:
                k, v = m.group(1), m.group(2)
                try:
                    series.setdefault(k, []).append(float(v))
                except ValueError:
                    pass
    return series


def graph_log(path: str, keys=("rew", "dist", "time"), out: Optional[str] = None):
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    series = parse_log(path)
    fig, axes = plt.subplots(len(keys), 1, figsize=(8, 3 * len(keys)), squeeze=False)
    for ax, k in zip(axes[:, 0], keys):
        if k in series:
            ax.plot(series[k])
            ax.set_ylabel(k)
        ax.set_xlabel("generation")
    fig.tight_layout()
    out = out or path + ".png"
    fig.savefig(out)
    return out


def graph_fits(fit_folder: str, out: Optional[str] = None):
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    files = sorted((f for f in os.listdir(fit_folder) if f.endswith(".npy")),
                   key=lambda f: int(f.split(".")[0]))
    means, maxs = [], []
    for f in files:
        fits = np.load(os.path.join(fit_folder, f))
        col = fits[:, 0] if fits.ndim > 1 else fits
        means.append(col.mean())
        maxs.append(col.max())
    fig, ax = plt.subplots(figsize=(8, 4))
    ax.plot(means, label="mean fitness")
    ax.plot(maxs, label="max fitness")
    ax.set_xlabel("generation")
    ax.legend()
    fig.tight_layout()
    out = out or os.path.join(fit_folder, "fits.png")
    fig.savefig(out)
    return out
