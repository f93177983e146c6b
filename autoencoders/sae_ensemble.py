from sparse_coding_amd.models.sae_signatures import (  # noqa: F401
    FunctionalSAE, FunctionalTiedSAE, FunctionalTiedCenteredSAE,
    FunctionalThresholdingSAE, ThresholdingSAE, FunctionalMaskedTiedSAE,
    FunctionalMaskedSAE, FunctionalReverseSAE,
)
