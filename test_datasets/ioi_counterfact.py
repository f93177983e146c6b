"""Shim: reference test_datasets/ioi_counterfact.py (the Redwood IOIDataset
port).  The rebuild's prompt-pair generator covers the same eval role; see
sparse_coding_amd/data/eval_prompts.py."""

from sparse_coding_amd.data.eval_prompts import *  # noqa: F401,F403
