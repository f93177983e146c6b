from sparse_coding_amd.models.learned_dict import (
    LearnedDict, Identity, IdentityPositive, IdentityReLU, RandomDict,
    UntiedSAE, TiedSAE, ReverseSAE, AddedNoise, Rotation, normalize_rows,
)
from sparse_coding_amd.models.sae_signatures import (
    DictSignature, FunctionalSAE, FunctionalTiedSAE, FunctionalTiedCenteredSAE,
    FunctionalThresholdingSAE, ThresholdingSAE, FunctionalMaskedTiedSAE,
    FunctionalMaskedSAE, FunctionalReverseSAE,
)
from sparse_coding_amd.models.topk import TopKEncoder, TopKLearnedDict
