from sparse_coding_amd.functional.optim import adam, sgd, apply_updates, optim_str_to_func
