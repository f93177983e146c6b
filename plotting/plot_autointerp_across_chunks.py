"""Autointerp scores across training-chunk counts (reference
plotting/plot_autointerp_across_chunks.py: tied_r{R}_nc{1,4,16,32}_l1a{V},
the layout written by interpret/drivers.interpret_across_chunks)."""

from __future__ import annotations

import argparse

from autointerp_lib import collect_layer_scores, layer_errorbar


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--base-path", required=True)
    p.add_argument("--layers", default="2")
    p.add_argument("--layer-loc", default="residual")
    p.add_argument("--score-mode", default="top")
    p.add_argument("--ratio", default="2.0")
    p.add_argument("--chunks", default="1,4,16,32")
    p.add_argument("--l1-tag", default="0.00072")
    p.add_argument("--out", default="autointerp_across_chunks.png")
    args = p.parse_args(argv)

    layers = [int(x) for x in args.layers.split(",")]
    names = [f"l{i}_{args.layer_loc}" for i in layers]
    all_scores = collect_layer_scores(args.base_path, names, args.score_mode)
    transforms = [f"tied_r{args.ratio}_nc{c}_l1a{args.l1_tag}" for c in args.chunks.split(",")]
    layer_errorbar(all_scores, transforms, save_path=args.out, top=0.34,
                   xtick_labels=[str(i) for i in layers],
                   title=f"autointerp across chunks ({args.layer_loc}, {args.score_mode})")
    print(f"saved {args.out}")


if __name__ == "__main__":
    main()
