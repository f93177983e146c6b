from sparse_coding_amd.models.ica import ICAEncoder, NNegICAEncoder  # noqa: F401
