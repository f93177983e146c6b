"""Shim: reference test_datasets/ioi_counterfact.py (the Redwood IOIDataset
port) -> sparse_coding_amd/data/ioi_counterfact.py (full semantics:
templates, per-prompt metadata, word_idx maps, flipped sets)."""

from sparse_coding_amd.data.ioi_counterfact import *  # noqa: F401,F403
from sparse_coding_amd.data.ioi_counterfact import (  # noqa: F401
    ABBA_TEMPLATES,
    ABC_TEMPLATES,
    BABA_TEMPLATES,
    BAC_TEMPLATES,
    IOIDataset,
    NAMES,
    NOUNS_DICT,
    OBJECTS,
    PLACES,
    gen_ioi_dataset,
    gen_prompt_counterfact,
    multiple_replace,
)
