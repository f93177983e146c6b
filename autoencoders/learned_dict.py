from sparse_coding_amd.models.learned_dict import (  # noqa: F401
    LearnedDict, Identity, IdentityPositive, IdentityReLU, RandomDict,
    UntiedSAE, TiedSAE, ReverseSAE, AddedNoise, Rotation, normalize_rows,
)
