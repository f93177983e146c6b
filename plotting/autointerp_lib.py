"""Shared machinery for the autointerp comparison plots.

The reference's five ``plot_autointerp_*.py`` scripts (C28) are near-copies
of two shapes: (a) a per-transform violin+means figure over one results
folder (plot_autointerp_violins.py:60-127), and (b) a per-layer errorbar
series for a chosen transform list (plot_autointerp_vs_baselines.py /
_vs_topk_baselines.py / _across_size.py / _across_chunks.py).  This module
holds both shapes once; the named scripts set their transform lists.

Score folders follow the protocol layout written by
``interpret/protocol.py`` (feature_N/explanation.txt), read through
``interpret/drivers.read_scores``.
"""

from __future__ import annotations

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import os
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np

from sparse_coding_amd.interpret.drivers import read_scores  # noqa: F401 (re-export)

COLORS = ["red", "blue", "green", "orange", "purple", "pink", "black",
          "brown", "cyan", "magenta", "grey", "yellow", "lime"]
MARKERS = ["o", "v", "^", "*", "x", "<", ">", "s", "p", "P", "h", "H", "+", "X", "D", "d"]


def _plt():
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    return plt


def violin_with_means(results_folder: str, score_mode: str = "top",
                      save_path: Optional[str] = None, title: str = ""):
    """Reference plot_autointerp_violins.read_results (:60-127): violin per
    transform on the fixed -0.2..0.6 axis, 95% CI errorbars on the means."""
    plt = _plt()
    scores = read_scores(results_folder, score_mode)
    if not scores:
        print(f"No scores found for {results_folder}")
        return None
    transforms = list(scores.keys())
    fig, ax = plt.subplots()
    ax.set_ylim(-0.2, 0.6)
    ax.set_yticks(np.arange(-0.2, 0.6, 0.1))
    ax.grid(axis="y", color="grey", linestyle="-", linewidth=0.5, alpha=0.3)
    scores_list = [scores[t][1] for t in transforms if len(scores[t][1]) > 0]
    parts = ax.violinplot(scores_list, showmeans=False, showextrema=False)
    for i, pc in enumerate(parts["bodies"]):
        pc.set_facecolor(COLORS[i % len(COLORS)])
        pc.set_edgecolor(COLORS[i % len(COLORS)])
        pc.set_alpha(0.3)
    ax.set_xticks(np.arange(1, len(transforms) + 1))
    ax.set_xticklabels(transforms, rotation=90)
    for i, t in enumerate(transforms):
        vals = scores[t][1]
        ci = 1.96 * np.std(vals, ddof=1) / np.sqrt(len(vals)) if len(vals) > 1 else 0.0
        ax.errorbar(i + 1, np.mean(vals), yerr=ci, fmt="o",
                    color=COLORS[i % len(COLORS)], elinewidth=2, capsize=20)
    ax.set_title(title or f"{os.path.basename(results_folder)} {score_mode}")
    ax.set_xlabel("Transform")
    ax.set_ylabel("autointerp score")
    ax.axhline(y=0, linestyle="-", color="black", linewidth=1)
    fig.tight_layout()
    if save_path:
        fig.savefig(save_path)
    return fig


def collect_layer_scores(base_path: str, activation_names: Sequence[str],
                         score_mode: str) -> List[Dict[str, Tuple[List[int], List[float]]]]:
    """One read_scores() dict per layer folder (reference :20-26)."""
    return [read_scores(os.path.join(base_path, name), score_mode)
            for name in activation_names]


def layer_errorbar(all_scores: List[Dict], transforms: Sequence[str],
                   save_path: Optional[str] = None, top: float = 0.35,
                   xlabel: str = "Layer", title: str = "",
                   xtick_labels: Optional[Sequence[str]] = None):
    """Reference grouped-errorbar shape (vs_baselines :55-160): per x-position
    (layer/size/chunk), one 95%-CI errorbar per transform."""
    plt = _plt()
    n_x = len(all_scores)
    fig, ax = plt.subplots()
    ax.set_ylim(bottom=0, top=top)
    ax.set_yticks(np.arange(0, top, 0.1))
    ax.grid(axis="y", color="grey", linestyle="-", linewidth=0.5, alpha=0.3)
    ax.set_xticks(np.arange(1, n_x + 1))
    ax.set_xticklabels(xtick_labels if xtick_labels is not None else [str(i) for i in range(n_x)])
    plotted = set()
    for i in range(n_x):
        for j, t in enumerate(transforms):
            if t not in all_scores[i] or not all_scores[i][t][1]:
                continue
            vals = all_scores[i][t][1]
            ci = 1.96 * np.std(vals, ddof=1) / np.sqrt(len(vals)) if len(vals) > 1 else 0.0
            ax.errorbar(i + 1 + j * 0.07, np.mean(vals), yerr=ci, fmt="o",
                        color=COLORS[j % len(COLORS)], elinewidth=1, capsize=0,
                        markersize=8, marker=MARKERS[j % len(MARKERS)],
                        label=t if t not in plotted else None)
            plotted.add(t)
    ax.set_xlabel(xlabel)
    ax.set_ylabel("autointerp score")
    if title:
        ax.set_title(title)
    ax.legend(fontsize=7)
    fig.tight_layout()
    if save_path:
        fig.savefig(save_path)
    return fig
