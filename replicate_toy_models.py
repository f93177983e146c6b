"""Shim: reference entry point `replicate_toy_models.py` (C23).

The MI355X implementation lives in sparse_coding_amd/sweep/toy_models.py.
"""

from sparse_coding_amd.sweep.toy_models import *  # noqa: F401,F403
from sparse_coding_amd.sweep.toy_models import main, run_single_go  # noqa: F401

if __name__ == "__main__":
    main()
