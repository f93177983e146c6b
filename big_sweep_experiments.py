"""Shim + CLI: reference big_sweep_experiments.py -> sparse_coding_amd.sweep.experiments.

The reference's __main__ toggles a hard-coded experiment by editing the file
(:1270-1279); here any catalogue entry is a subcommand:

    python big_sweep_experiments.py run_dense_l1_range --use_synthetic_dataset ...
    python big_sweep_experiments.py run_topk --n_chunks 4 ...
    python big_sweep_experiments.py list          # show available experiments

Flags after the experiment name populate EnsembleArgs/SyntheticEnsembleArgs
(the reference's auto-argparse contract, opt-in from_cli).
"""

from sparse_coding_amd.sweep.experiments import *  # noqa: F401,F403


def main(argv=None):
    import sys

    import sparse_coding_amd.sweep.experiments as _exps
    from sparse_coding_amd.config import EnsembleArgs, SyntheticEnsembleArgs

    args = list(sys.argv[1:] if argv is None else argv)
    names = sorted(n for n in dir(_exps) if n.startswith("run_"))
    if not args or args[0] in ("list", "-h", "--help"):
        print("experiments:\n  " + "\n  ".join(names))
        return
    name = args[0]
    if name not in names:
        raise SystemExit(f"unknown experiment {name!r}; `list` shows the catalogue")
    rest = args[1:]
    synthetic = "--use_synthetic_dataset" in rest or "--use-synthetic-dataset" in rest
    cls = SyntheticEnsembleArgs if synthetic else EnsembleArgs
    cfg = cls.from_cli([a.replace("--use-synthetic-dataset", "--use_synthetic_dataset")
                        for a in rest])
    return getattr(_exps, name)(cfg)


if __name__ == "__main__":
    main()
