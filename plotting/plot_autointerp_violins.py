"""Violin+means of autointerp scores per transform, every activation folder x
score mode (reference plotting/plot_autointerp_violins.py:131-140)."""

from __future__ import annotations

import argparse
import os

from autointerp_lib import violin_with_means


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--base-path", required=True,
                   help="root of autointerp results: {base}/{activation_name}/{transform}/feature_N/")
    p.add_argument("--plots-folder", default=None, help="default: save next to the scores")
    p.add_argument("--score-modes", default="top,random,top_random")
    args = p.parse_args(argv)

    names = [x for x in os.listdir(args.base_path)
             if os.path.isdir(os.path.join(args.base_path, x))]
    for name in names:
        folder = os.path.join(args.base_path, name)
        for mode in args.score_modes.split(","):
            out_dir = args.plots_folder or folder
            os.makedirs(out_dir, exist_ok=True)
            save = os.path.join(out_dir, f"{name}_{mode}_means_and_violin.png" if args.plots_folder
                                else f"{mode}_means_and_violin.png")
            fig = violin_with_means(folder, mode, save_path=save)
            if fig is not None:
                print(f"saved {save}")


if __name__ == "__main__":
    main()
