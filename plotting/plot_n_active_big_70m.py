"""Fraction of alive features vs L1 across dict-size ratios at one
checkpoint, Pythia-70m layout (reference plotting/plot_n_active_big_70m.py)."""

from __future__ import annotations

import argparse

from plot_n_active import frac_alive_series, load_sample, two_panel_alive_plot
import os


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--load-dir", required=True)
    p.add_argument("--chunk", required=True)
    p.add_argument("--layer", type=int, default=2)
    p.add_argument("--layer-loc", default="residual")
    p.add_argument("--tied", default="tied")
    p.add_argument("--ratios", default="0.5,1.0,2.0,4.0,8.0")
    p.add_argument("--epoch", type=int, default=9, help="checkpoint index (_N folder)")
    p.add_argument("--device", default="cpu")
    p.add_argument("--out", default=None)
    args = p.parse_args(argv)

    sample = load_sample(args.chunk)
    series = []
    for ratio in args.ratios.split(","):
        path = os.path.join(args.load_dir, f"{args.tied}_{args.layer_loc}_l{args.layer}_r{ratio}",
                            f"_{args.epoch}", "learned_dicts.pt")
        if not os.path.exists(path):
            continue
        rows = frac_alive_series(path, sample, args.device)
        series.append((f"r{ratio}", [(l1, frac) for l1, frac, *_ in rows]))
    out = args.out or f"active_plot_{args.tied}_l{args.layer}_{args.layer_loc}_by_ratio.png"
    two_panel_alive_plot(series, out,
                         f"% active features by dict ratio ({args.layer_loc} layer {args.layer})",
                         abs_scale=sample.shape[1])
    print(f"saved {out}")


if __name__ == "__main__":
    main()
