from sparse_coding_amd.sweep.cluster_runs import (  # noqa: F401
    dispatch_job_on_chunk, dispatch_lite, collect_lite, job_wrapper,
)
