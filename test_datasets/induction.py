"""Shim: reference test_datasets/induction.py (empty in the reference; the
rebuild fills it in)."""

from sparse_coding_amd.data.eval_prompts import generate_induction_dataset  # noqa: F401
