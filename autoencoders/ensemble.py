from sparse_coding_amd.engine.ensemble import (  # noqa: F401
    FunctionalEnsemble, stack_dict, unstack_dict, optim_str_to_func,
)
from sparse_coding_amd.models.sae_signatures import DictSignature  # noqa: F401
