"""FVU-vs-L0 pareto on mean-centered MLP activations (reference
plotting/fvu_sparsity_plot_mlp_center.py — scoring against the centered
chunk)."""

from fvu_sparsity_plot import main as _main


def main(argv=None):
    _main(argv, default_out="fvu_sparsity_mlp_center.png", center_default=True)


if __name__ == "__main__":
    main()
