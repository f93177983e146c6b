"""Shim package: reference `test_datasets/` (C26: IOI / counterfact / gender
/ induction prompt sets).  Implementations live in
sparse_coding_amd/data/eval_prompts.py."""
