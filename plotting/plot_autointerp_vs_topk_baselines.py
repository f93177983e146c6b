"""Autointerp scores per layer: trained SAE vs sparsity-matched TopK-PCA /
TopK-ICA baselines (reference plotting/plot_autointerp_vs_topk_baselines.py)."""

from __future__ import annotations

import argparse

from autointerp_lib import collect_layer_scores, layer_errorbar


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--base-path", required=True)
    p.add_argument("--layers", default="0,1,2,3,4,5")
    p.add_argument("--layer-loc", default="residual")
    p.add_argument("--score-mode", default="top")
    p.add_argument("--sae-transform", default="tied_r2.0_l1a0.00086")
    p.add_argument("--baselines", default="pca_topk,ica_topk,random_topk",
                   help="sparsity-matched top-k baseline transform names")
    p.add_argument("--out", default="autointerp_vs_topk_baselines.png")
    args = p.parse_args(argv)

    layers = [int(x) for x in args.layers.split(",")]
    names = [f"l{i}_{args.layer_loc}" for i in layers]
    all_scores = collect_layer_scores(args.base_path, names, args.score_mode)
    transforms = [args.sae_transform] + args.baselines.split(",")
    top = 0.2 if args.score_mode == "random" else 0.35
    layer_errorbar(all_scores, transforms, save_path=args.out, top=top,
                   xtick_labels=[str(i) for i in layers],
                   title=f"autointerp vs top-k baselines ({args.layer_loc}, {args.score_mode})")
    print(f"saved {args.out}")


if __name__ == "__main__":
    main()
