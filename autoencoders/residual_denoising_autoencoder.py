from sparse_coding_amd.models.lista import (  # noqa: F401
    LISTALayer, FunctionalLISTADenoisingSAE, LISTADenoisingSAE,
    ResidualDenoisingLayer, FunctionalResidualDenoisingSAE, ResidualDenoisingSAE,
    shrinkage,
)
