from sparse_coding_amd.models.pca import (  # noqa: F401
    BatchedMean, BatchedPCA, PCAEncoder, calc_mean, calc_pca,
)
