from sparse_coding_amd.engine.ensemble import FunctionalEnsemble, stack_dict, unstack_dict, optim_str_to_func
