"""n-active plot for the GPT-2-small MLP sweeps (reference
plotting/plot_n_active_gpt2sm.py: same computation, gpt2sm folder naming)."""

from __future__ import annotations

import argparse
import os

from plot_n_active import frac_alive_series, load_sample, two_panel_alive_plot


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--load-dir", required=True)
    p.add_argument("--chunk", required=True)
    p.add_argument("--layer", type=int, default=5)
    p.add_argument("--layer-loc", default="mlpout")
    p.add_argument("--ratios", default="1.0,2.0,4.0,8.0")
    p.add_argument("--epoch", type=int, default=9)
    p.add_argument("--folder-template", default="gpt2sm_{loc}_l{layer}_r{ratio}")
    p.add_argument("--device", default="cpu")
    p.add_argument("--out", default=None)
    args = p.parse_args(argv)

    sample = load_sample(args.chunk)
    series = []
    for ratio in args.ratios.split(","):
        folder = args.folder_template.format(loc=args.layer_loc, layer=args.layer, ratio=ratio)
        path = os.path.join(args.load_dir, folder, f"_{args.epoch}", "learned_dicts.pt")
        if not os.path.exists(path):
            continue
        rows = frac_alive_series(path, sample, args.device)
        series.append((f"r{ratio}", [(l1, frac) for l1, frac, *_ in rows]))
    out = args.out or f"active_plot_gpt2sm_l{args.layer}_{args.layer_loc}.png"
    two_panel_alive_plot(series, out,
                         f"% active features, GPT-2-small ({args.layer_loc} layer {args.layer})",
                         abs_scale=sample.shape[1])
    print(f"saved {out}")


if __name__ == "__main__":
    main()
