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


# ---------------------------------------------------------------------------
# shared helpers for the named n-active family scripts (reference
# plot_n_active*.py all compute: per saved checkpoint, per learned dict,
# the fraction of features active >threshold times on one chunk sample)
# ---------------------------------------------------------------------------

def frac_alive_series(dicts_path: str, sample: torch.Tensor, device: str = "cpu",
                      threshold: int = 10, batch_size: int = 16384):
    """[(l1_alpha, frac_alive, n_alive, n_feats), ...] for one
    learned_dicts.pt (reference plot_n_active_over_time.py:52-77)."""
    out = []
    dicts = torch.load(dicts_path, map_location="cpu", weights_only=False)
    for ld, hp in dicts:
        ld.to_device(device)
        s = sample.to(device)
        n_active_count = torch.zeros(ld.n_feats, device=device)
        for i in range(0, len(s), batch_size):
            code = ld.encode(s[i : i + batch_size])
            n_active_count += (code > 0).sum(dim=0).float()
        n_alive = int((n_active_count > threshold).sum().item())
        l1a = hp.get("l1_alpha", 0) or 8e-5  # reference maps l1=0 -> 8e-5
        out.append((l1a, n_alive / ld.n_feats, n_alive, ld.n_feats))
    return out


def series_over_checkpoints(load_dir: str, folder: str, epochs, sample,
                            device: str = "cpu", threshold: int = 10):
    """[(epoch, [(l1, frac), ...]), ...] over _{epoch}/learned_dicts.pt."""
    data = []
    for epoch in epochs:
        path = os.path.join(load_dir, folder, f"_{epoch}", "learned_dicts.pt")
        if not os.path.exists(path):
            continue
        rows = frac_alive_series(path, sample, device, threshold)
        data.append((epoch, [(l1, frac) for l1, frac, *_ in rows]))
    return data


def two_panel_alive_plot(series, save_path: str, title: str, abs_scale: float = 1.0):
    """Reference's two-panel layout (:100-119): fraction + absolute counts
    vs l1, one line per series key."""
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    fig, (ax, ax2) = plt.subplots(1, 2, figsize=plt.figaspect(1 / 3))
    for key, rows in series:
        if not rows:
            continue
        ax.plot(*zip(*rows), label=str(key))
        ax2.plot(*zip(*[(l1, abs_scale * f) for l1, f in rows]), label=str(key))
    for a, ylab in ((ax, "Fraction of features alive"), (ax2, "Number of features alive")):
        a.set_xscale("log")
        a.set_xlabel("L1 Alpha")
        a.set_ylabel(ylab)
        a.legend(fontsize=7)
    ax.set_title(title)
    fig.tight_layout()
    fig.savefig(save_path)
    return fig


def load_sample(chunk_path: str, n_samples: int = 50000) -> torch.Tensor:
    chunk = torch.load(chunk_path, map_location="cpu", weights_only=False).float()
    idx = np.random.choice(len(chunk), size=min(n_samples, len(chunk)), replace=False)
    return chunk[idx]
