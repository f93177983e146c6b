"""Scatter of active-neuron count vs sparsity, colored by log10(l1), one
colormap per dict ratio (reference plotting/num_dead_plot.py)."""

from __future__ import annotations

import argparse
import math
import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import numpy as np
import torch

from sparse_coding_amd.metrics import standard_metrics as sm


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--dict-files", required=True,
                   help="comma-separated ratio:path pairs, e.g. 0.5:out_r05/_9/learned_dicts.pt,1:out_r1/_9/learned_dicts.pt")
    p.add_argument("--chunk", required=True)
    p.add_argument("--n-samples", type=int, default=100000)
    p.add_argument("--device", default="cpu")
    p.add_argument("--out", default="num_dead_plot.png")
    args = p.parse_args(argv)

    chunk = torch.load(args.chunk, map_location="cpu", weights_only=False).float()
    idx = np.random.choice(len(chunk), size=min(args.n_samples, len(chunk)), replace=False)
    sample = chunk[idx].to(args.device)

    datapoints = []
    labels = []
    for pair in args.dict_files.split(","):
        label, path = pair.split(":", 1)
        series = []
        for ld, hp in torch.load(path, map_location="cpu", weights_only=False):
            ld.to_device(args.device)
            mean_nz = sm.mean_nonzero_activations(ld, sample)
            sparsity = mean_nz.sum().item()
            num_dead = mean_nz.count_nonzero().item()
            series.append((num_dead, sparsity, hp["l1_alpha"]))
        datapoints.append(series)
        labels.append(label)

    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    colors = ["Purples", "Blues", "Greens", "Oranges", "Reds"]
    fig = plt.figure()
    ax = fig.add_subplot(111)
    for i, series in enumerate(datapoints):
        if not series:
            continue
        num_dead, sparsity, l1s = zip(*series)
        ax.scatter(num_dead, sparsity,
                   c=[math.log10(max(x, 1e-6)) for x in l1s],
                   label=labels[i], cmap=colors[i % len(colors)], vmin=-5, vmax=-1)
    ax.set_xlabel("Sparsity")
    ax.set_ylabel("Num Active Neurons")
    ax.set_xscale("log")
    ax.set_yscale("log")
    ax.legend()
    plt.savefig(args.out)
    print(f"saved {args.out}")


if __name__ == "__main__":
    main()
