from sparse_coding_amd.sweep.baselines import (  # noqa: F401
    run_layer_baselines, run_ica, resave_change_sparsity, run_all,
)
