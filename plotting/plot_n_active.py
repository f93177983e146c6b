"""Active-feature counts across dict size / training time / layer.

Covers reference ``plotting/plot_n_active*.py`` and ``num_dead_plot.py``:
number of ever-active features (and dead fraction) per saved checkpoint.
"""

from __future__ import annotations

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import argparse
import os
import re

import numpy as np
import torch

from sparse_coding_amd.metrics import standard_metrics as sm


def n_active_over_checkpoints(output_folder: str, sample: torch.Tensor, device: str = "cpu"):
    """Walk the _{i}/learned_dicts.pt checkpoints of one sweep output folder
    and return {chunk_idx: [(hyperparams, n_active, dead_frac), ...]}."""
    results = {}
    for entry in sorted(os.listdir(output_folder)):
        m = re.fullmatch(r"_(\d+)", entry)
        path = os.path.join(output_folder, entry, "learned_dicts.pt")
        if not m or not os.path.exists(path):
            continue
        dicts = torch.load(path, map_location="cpu", weights_only=False)
        rows = []
        for ld, hp in dicts:
            ld.to_device(device)
            s = sample.to(device)
            n_act = sm.batched_calc_feature_n_ever_active(ld, s, threshold=1)
            dead = sm.dead_feature_fraction(ld, s)
            rows.append((hp, n_act, dead))
        results[int(m.group(1))] = rows
    return results


def plot_n_active(results, save_path: str = "n_active.png"):
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    fig, (ax1, ax2) = plt.subplots(1, 2, figsize=(10, 4))
    chunks = sorted(results.keys())
    n_dicts = len(results[chunks[0]]) if chunks else 0
    for di in range(n_dicts):
        hp = results[chunks[0]][di][0]
        label = f"l1={hp.get('l1_alpha', 0):.1e}"
        ax1.plot(chunks, [results[c][di][1] for c in chunks], "o-", label=label)
        ax2.plot(chunks, [results[c][di][2] for c in chunks], "o-", label=label)
    ax1.set_xlabel("chunks trained")
    ax1.set_ylabel("n ever-active features")
    ax2.set_xlabel("chunks trained")
    ax2.set_ylabel("dead fraction")
    ax1.legend(fontsize=6)
    fig.tight_layout()
    fig.savefig(save_path, dpi=120)
    return fig


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--output-folder", required=True)
    p.add_argument("--chunk", required=True)
    p.add_argument("--n-samples", type=int, default=10000)
    p.add_argument("--device", default="cpu")
    p.add_argument("--out", default="n_active.png")
    args = p.parse_args()
    chunk = torch.load(args.chunk, map_location="cpu").float()
    idx = np.random.choice(len(chunk), size=min(args.n_samples, len(chunk)), replace=False)
    results = n_active_over_checkpoints(args.output_folder, chunk[idx], args.device)
    plot_n_active(results, args.out)


if __name__ == "__main__":
    main()
