from sparse_coding_amd.sweep.basic_l1_sweep import basic_l1_sweep, SweepArgs  # noqa: F401

if __name__ == "__main__":
    basic_l1_sweep(SweepArgs.from_cli())
