from sparse_coding_amd.sweep.experiments import *  # noqa: F401,F403
