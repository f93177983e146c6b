"""FVU-vs-L0 pareto for the GPT-2-small sweeps (reference
plotting/fvu_sparsity_plot_gpt2sm.py — same scoring, gpt2sm artifacts)."""

from fvu_sparsity_plot import main as _main


def main(argv=None):
    _main(argv, default_out="fvu_sparsity_gpt2sm.png")


if __name__ == "__main__":
    main()
