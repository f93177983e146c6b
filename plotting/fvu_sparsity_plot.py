"""FVU-vs-L0 pareto curves with baseline overlays.

Parity with reference ``plotting/fvu_sparsity_plot.py`` (score registry :20-37,
AUC-of-curve :40): for each learned_dicts.pt, score every dict on a held-out
chunk and plot FVU against mean L0, one curve per dict family, with PCA /
ICA / identity baselines as reference points.
"""

from __future__ import annotations

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import argparse
import os
from collections import defaultdict
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from sparse_coding_amd.metrics import standard_metrics as sm


def score_dict(learned_dict, sample: torch.Tensor) -> Tuple[float, float]:
    """(mean L0, FVU) of one dict on a sample (reference score_dict :20)."""
    l0 = sm.mean_l0(learned_dict, sample).item()
    fvu = sm.fraction_variance_unexplained(learned_dict, sample).item()
    return l0, fvu


def pareto_auc(points: List[Tuple[float, float]]) -> float:
    """Area under the (sorted-by-L0) FVU curve — lower is better
    (reference :40)."""
    pts = sorted(points)
    if len(pts) < 2:
        return float("nan")
    auc = 0.0
    for (x0, y0), (x1, y1) in zip(pts, pts[1:]):
        auc += (x1 - x0) * (y0 + y1) / 2
    span = pts[-1][0] - pts[0][0]
    return auc / span if span > 0 else float("nan")


def score_learned_dicts(
    learned_dicts_path: str,
    sample: torch.Tensor,
    device: str = "cpu",
    group_by: str = "dict_size",
) -> Dict[str, List[Tuple[float, float, dict]]]:
    dicts = torch.load(learned_dicts_path, map_location="cpu", weights_only=False)
    curves = defaultdict(list)
    for ld, hp in dicts:
        ld.to_device(device)
        l0, fvu = score_dict(ld, sample.to(device))
        key = f"{group_by}_{hp.get(group_by, '?')}"
        curves[key].append((l0, fvu, hp))
    return dict(curves)


def plot_fvu_sparsity(
    curves: Dict[str, List[Tuple[float, float, dict]]],
    baselines: Optional[Dict[str, Tuple[float, float]]] = None,
    save_path: str = "fvu_sparsity.png",
    title: str = "FVU vs mean L0",
):
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    fig, ax = plt.subplots(figsize=(7, 5))
    for name, pts in sorted(curves.items()):
        pts_sorted = sorted((l0, fvu) for l0, fvu, _ in pts)
        xs = [p[0] for p in pts_sorted]
        ys = [p[1] for p in pts_sorted]
        ax.plot(xs, ys, "o-", label=f"{name} (auc={pareto_auc(pts_sorted):.3f})")
    if baselines:
        for name, (l0, fvu) in baselines.items():
            ax.scatter([l0], [fvu], marker="*", s=150, label=name)
    ax.set_xlabel("mean L0 (active features/example)")
    ax.set_ylabel("fraction variance unexplained")
    ax.set_yscale("log")
    ax.set_title(title)
    ax.legend(fontsize=7)
    fig.tight_layout()
    fig.savefig(save_path, dpi=120)
    return fig


def main(argv=None, default_out="fvu_sparsity.png", center_default=False):
    p = argparse.ArgumentParser()
    p.add_argument("--learned-dicts", required=True)
    p.add_argument("--chunk", required=True, help="held-out activation chunk .pt")
    p.add_argument("--device", default="cuda:0" if torch.cuda.is_available() else "cpu")
    p.add_argument("--n-samples", type=int, default=20000)
    p.add_argument("--out", default=default_out)
    p.add_argument("--baseline-dir", default="", help="folder of saved baseline dicts")
    p.add_argument("--center", action="store_true", default=center_default,
                   help="subtract the chunk mean before scoring (the mean-centered "
                        "MLP variant, reference fvu_sparsity_plot_mlp_center.py)")
    args = p.parse_args(argv)

    chunk = torch.load(args.chunk, map_location="cpu").float()
    idx = np.random.choice(len(chunk), size=min(args.n_samples, len(chunk)), replace=False)
    sample = chunk[idx]
    if args.center:
        sample = sample - chunk.mean(dim=0)

    curves = score_learned_dicts(args.learned_dicts, sample, device=args.device)

    baselines = {}
    if args.baseline_dir and os.path.isdir(args.baseline_dir):
        for fname in os.listdir(args.baseline_dir):
            if fname.endswith(".pt"):
                try:
                    bd = torch.load(os.path.join(args.baseline_dir, fname), map_location="cpu", weights_only=False)
                    bd.to_device(args.device)
                    baselines[fname[:-3]] = score_dict(bd, sample.to(args.device))
                except Exception as e:  # noqa: BLE001
                    print(f"skipping baseline {fname}: {e}")

    plot_fvu_sparsity(curves, baselines, save_path=args.out)
    print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
