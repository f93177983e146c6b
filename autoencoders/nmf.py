from sparse_coding_amd.models.nmf import NMFEncoder  # noqa: F401
