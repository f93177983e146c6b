"""Shim: reference test_datasets/preprocess_gender_dataset.py."""

from sparse_coding_amd.data.eval_prompts import generate_gender_dataset  # noqa: F401
