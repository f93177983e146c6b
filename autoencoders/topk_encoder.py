from sparse_coding_amd.models.topk import TopKEncoder, TopKLearnedDict  # noqa: F401
