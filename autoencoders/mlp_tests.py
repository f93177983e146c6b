from sparse_coding_amd.models.positive import (  # noqa: F401
    TiedPositiveSAE, UntiedPositiveSAE, FunctionalPositiveTiedSAE,
)
