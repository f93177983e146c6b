"""Erasure bottleneck curves (reference plotting/bottleneck_plot.py +
erasure_plot.py): probe AUROC vs number of erased features/rank, for
dict-feature ablation vs LEACE."""

from __future__ import annotations

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import argparse

import torch

from sparse_coding_amd.sweep.erasure import erasure_curves, plot_erasure


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--learned-dict", required=True, help="single saved LearnedDict .pt")
    p.add_argument("--activations", required=True, help=".pt with {'acts': [N,d], 'labels': [N]}")
    p.add_argument("--out", default="bottleneck.png")
    p.add_argument("--ks", default="0,1,2,4,8,16,32")
    args = p.parse_args()

    ld = torch.load(args.learned_dict, map_location="cpu", weights_only=False)
    data = torch.load(args.activations, map_location="cpu", weights_only=False)
    acts, labels = data["acts"].float(), data["labels"].long()
    ks = [int(k) for k in args.ks.split(",")]
    curves = erasure_curves(ld, acts, labels, ks=ks)
    plot_erasure(curves, save_path=args.out)
    for k, a, b in zip(curves["k"], curves["dict_ablation"], curves["leace"]):
        print(f"k={k:3d}  dict_ablation={a:.3f}  leace={b:.3f}")


if __name__ == "__main__":
    main()
