from sparse_coding_amd.models.semilinear import FFLayer, SemiLinearSAE  # noqa: F401
