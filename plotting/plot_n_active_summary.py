"""Summary: fraction of alive features vs L1, one line per LAYER at a fixed
ratio/checkpoint (reference plotting/plot_n_active_summary.py)."""

from __future__ import annotations

import argparse
import os

from plot_n_active import frac_alive_series, load_sample, two_panel_alive_plot


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--load-dir", required=True)
    p.add_argument("--chunk-template", required=True,
                   help="per-layer chunk path template with {layer}, e.g. chunks/l{layer}_residual/0.pt")
    p.add_argument("--layers", default="0,1,2,3,4,5")
    p.add_argument("--layer-loc", default="residual")
    p.add_argument("--tied", default="tied")
    p.add_argument("--ratio", default="2.0")
    p.add_argument("--epoch", type=int, default=9)
    p.add_argument("--device", default="cpu")
    p.add_argument("--out", default="n_active_summary.png")
    args = p.parse_args(argv)

    series = []
    d_act = 0
    for layer in (int(x) for x in args.layers.split(",")):
        path = os.path.join(args.load_dir,
                            f"{args.tied}_{args.layer_loc}_l{layer}_r{args.ratio}",
                            f"_{args.epoch}", "learned_dicts.pt")
        chunk_path = args.chunk_template.format(layer=layer)
        if not (os.path.exists(path) and os.path.exists(chunk_path)):
            continue
        sample = load_sample(chunk_path)
        d_act = sample.shape[1]
        rows = frac_alive_series(path, sample, args.device)
        series.append((f"layer {layer}", [(l1, frac) for l1, frac, *_ in rows]))
    two_panel_alive_plot(series, args.out,
                         f"% active features by layer ({args.layer_loc}, r{args.ratio})",
                         abs_scale=d_act)
    print(f"saved {args.out}")


if __name__ == "__main__":
    main()
