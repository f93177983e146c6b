from sparse_coding_amd.data.random_dataset import (  # noqa: F401
    RandomDatasetGenerator, SparseMixDataset, generate_rand_dataset,
    generate_correlated_dataset, generate_noise_dataset, generate_rand_feats,
    generate_corr_matrix,
)
