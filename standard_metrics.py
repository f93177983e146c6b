"""Top-level shim: reference code does `import standard_metrics`."""
from sparse_coding_amd.metrics.standard_metrics import *  # noqa: F401,F403
from sparse_coding_amd.metrics.standard_metrics import (  # noqa: F401
    mcs_duplicates, mmcs, mcs_to_fixed, mmcs_to_fixed, mmcs_from_list,
    mean_nonzero_activations, fraction_variance_unexplained, r_squared,
    calc_moments_streaming, batched_calc_feature_n_ever_active,
)
