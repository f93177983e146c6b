"""Shim: reference test_datasets/ioi.py (:11-67)."""

from sparse_coding_amd.data.eval_prompts import (  # noqa: F401
    filter_single_token,
    generate_ioi_dataset,
)
