"""Autointerp scores across dictionary-size ratios at fixed L1 (reference
plotting/plot_autointerp_across_size.py: tied_r{0.5..32}_l1a0.00086)."""

from __future__ import annotations

import argparse

from autointerp_lib import collect_layer_scores, layer_errorbar


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--base-path", required=True)
    p.add_argument("--layers", default="0,1,2,3,4,5")
    p.add_argument("--layer-loc", default="residual")
    p.add_argument("--score-mode", default="top")
    p.add_argument("--ratios", default="0.5,1.0,2.0,4.0,8.0,16.0,32.0")
    p.add_argument("--l1-tag", default="0.00086")
    p.add_argument("--out", default="autointerp_across_size.png")
    args = p.parse_args(argv)

    layers = [int(x) for x in args.layers.split(",")]
    names = [f"l{i}_{args.layer_loc}" for i in layers]
    all_scores = collect_layer_scores(args.base_path, names, args.score_mode)
    transforms = [f"tied_r{r}_l1a{args.l1_tag}" for r in args.ratios.split(",")]
    layer_errorbar(all_scores, transforms, save_path=args.out, top=0.34,
                   xtick_labels=[str(i) for i in layers],
                   title=f"autointerp across dict size ({args.layer_loc}, {args.score_mode})")
    print(f"saved {args.out}")


if __name__ == "__main__":
    main()
