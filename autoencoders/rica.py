from sparse_coding_amd.models.rica import RICA  # noqa: F401
