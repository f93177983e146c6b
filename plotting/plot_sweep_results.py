"""Sweep-grid plots: loss / FVU / L0 across the (l1_alpha, dict_size) grid.

Covers reference ``plotting/plot_sweep_results.py``: heatmap grids of final
metrics per hyperparameter setting from a learned_dicts.pt checkpoint.
"""

from __future__ import annotations

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import argparse
from collections import defaultdict

import numpy as np
import torch

from sparse_coding_amd.metrics import standard_metrics as sm


def grid_metrics(learned_dicts_path: str, sample: torch.Tensor, device: str = "cpu"):
    dicts = torch.load(learned_dicts_path, map_location="cpu", weights_only=False)
    l1s = sorted({hp["l1_alpha"] for _, hp in dicts})
    sizes = sorted({hp.get("dict_size", 0) for _, hp in dicts})
    fvu = np.full((len(l1s), len(sizes)), np.nan)
    l0 = np.full((len(l1s), len(sizes)), np.nan)
    dead = np.full((len(l1s), len(sizes)), np.nan)
    for ld, hp in dicts:
        ld.to_device(device)
        i = l1s.index(hp["l1_alpha"])
        j = sizes.index(hp.get("dict_size", 0))
        s = sample.to(device)
        fvu[i, j] = sm.fraction_variance_unexplained(ld, s).item()
        l0[i, j] = sm.mean_l0(ld, s).item()
        dead[i, j] = sm.dead_feature_fraction(ld, s)
    return l1s, sizes, {"fvu": fvu, "l0": l0, "dead_frac": dead}


def plot_grids(l1s, sizes, grids, save_prefix: str = "sweep"):
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    for name, grid in grids.items():
        fig, ax = plt.subplots()
        im = ax.imshow(grid, aspect="auto", cmap="viridis")
        ax.set_xticks(range(len(sizes)))
        ax.set_xticklabels(sizes)
        ax.set_yticks(range(len(l1s)))
        ax.set_yticklabels([f"{v:.1e}" for v in l1s])
        ax.set_xlabel("dict_size")
        ax.set_ylabel("l1_alpha")
        ax.set_title(name)
        fig.colorbar(im)
        fig.tight_layout()
        fig.savefig(f"{save_prefix}_{name}.png", dpi=120)
    return True


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--learned-dicts", required=True)
    p.add_argument("--chunk", required=True)
    p.add_argument("--n-samples", type=int, default=10000)
    p.add_argument("--device", default="cpu")
    p.add_argument("--out-prefix", default="sweep")
    args = p.parse_args()
    chunk = torch.load(args.chunk, map_location="cpu").float()
    idx = np.random.choice(len(chunk), size=min(args.n_samples, len(chunk)), replace=False)
    l1s, sizes, grids = grid_metrics(args.learned_dicts, chunk[idx], args.device)
    plot_grids(l1s, sizes, grids, args.out_prefix)


if __name__ == "__main__":
    main()
