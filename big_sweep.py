"""Top-level shim: reference entry `from big_sweep import sweep`."""
from sparse_coding_amd.sweep.big_sweep import (  # noqa: F401
    sweep, ensemble_train_loop, unstacked_to_learned_dicts, get_model,
    init_model_dataset, init_synthetic_dataset, log_standard_metrics,
    make_hyperparam_name, filter_learned_dicts, format_hyperparam_val,
)
