"""Fraction of alive features vs L1 across training epochs, per layer
(reference plotting/plot_n_active_over_time.py)."""

from __future__ import annotations

import argparse
import os

from plot_n_active import load_sample, series_over_checkpoints, two_panel_alive_plot


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--load-dir", required=True, help="sweep outputs root")
    p.add_argument("--chunk", required=True, help="activation chunk .pt for the alive counts")
    p.add_argument("--layer", type=int, default=2)
    p.add_argument("--layer-loc", default="residual")
    p.add_argument("--tied", default="tied")
    p.add_argument("--ratio", default="1.0")
    p.add_argument("--epochs", default="0,10,20,30,40,50,59")
    p.add_argument("--folder-template", default="{tied}_{loc}_l{layer}_r{ratio}",
                   help="sweep folder name template")
    p.add_argument("--device", default="cpu")
    p.add_argument("--out", default=None)
    args = p.parse_args(argv)

    folder = args.folder_template.format(tied=args.tied, loc=args.layer_loc,
                                         layer=args.layer, ratio=args.ratio)
    sample = load_sample(args.chunk)
    epochs = [int(e) for e in args.epochs.split(",")]
    series = series_over_checkpoints(args.load_dir, folder, epochs, sample, args.device)
    out = args.out or f"active_plot_{args.tied}_l{args.layer}_{args.layer_loc}_overtime{args.ratio}.png"
    two_panel_alive_plot(series, out,
                         f"% active features over time ({args.layer_loc} layer {args.layer})",
                         abs_scale=sample.shape[1])
    print(f"saved {out}")


if __name__ == "__main__":
    main()
