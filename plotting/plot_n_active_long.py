"""Long-run variant of the n-active-over-time plot (reference
plotting/plot_n_active_long.py: the *_long sweep folders, epoch list to 59)."""

from __future__ import annotations

import argparse

from plot_n_active import load_sample, series_over_checkpoints, two_panel_alive_plot


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--load-dir", required=True)
    p.add_argument("--chunk", required=True)
    p.add_argument("--layer", type=int, default=2)
    p.add_argument("--layer-loc", default="residual")
    p.add_argument("--tied", default="tied")
    p.add_argument("--ratio", default="1.0")
    p.add_argument("--epochs", default="0,10,20,30,40,50,59")
    p.add_argument("--device", default="cpu")
    p.add_argument("--out", default=None)
    args = p.parse_args(argv)

    folder = f"{args.tied}_{args.layer_loc}_l{args.layer}_r{args.ratio}_long"
    sample = load_sample(args.chunk)
    epochs = [int(e) for e in args.epochs.split(",")]
    series = series_over_checkpoints(args.load_dir, folder, epochs, sample, args.device)
    out = args.out or f"active_plot_{args.tied}_l{args.layer}_{args.layer_loc}_long.png"
    two_panel_alive_plot(series, out,
                         f"% active features, long run ({args.layer_loc} layer {args.layer})",
                         abs_scale=sample.shape[1])
    print(f"saved {out}")


if __name__ == "__main__":
    main()
