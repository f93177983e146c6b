"""HIP extension loader for the gfx950 kernels.

The extension is built IN-TREE (``sparse_coding_amd/ops/hip`` →
``sparse_coding_amd/ops/_sae_hip.so``) by ``__graft_entry__.build()`` /
``python -m sparse_coding_amd.ops.build`` so the .so ships with the repo
snapshot to GPU boxes.  Policy: on a CUDA/ROCm device the HIP path is THE
path — a missing extension raises instead of silently falling back to eager
PyTorch (the eager path is reserved for CPU and for explicit
``backend="torch"`` requests).
"""

from __future__ import annotations

import importlib
import os
from typing import Optional

_ext = None
_ext_err: Optional[Exception] = None


def _try_load():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return
    try:
        _ext = importlib.import_module("sparse_coding_amd.ops._sae_hip")
    except Exception as e:  # noqa: BLE001
        _ext_err = e


def extension_available() -> bool:
    _try_load()
    return _ext is not None


def get_extension(required: bool = True):
    """Return the compiled HIP extension module.

    required=True raises a loud error when it cannot be imported — this is
    what keeps a GPU run from silently training on the eager fallback.
    """
    _try_load()
    if _ext is None and required:
        raise RuntimeError(
            "sparse_coding_amd HIP extension (_sae_hip) is not built/importable. "
            "Build it with `python -m sparse_coding_amd.ops.build` (hipcc, "
            "--offload-arch=gfx950). Refusing to silently fall back to eager "
            f"PyTorch on a GPU device. Original error: {_ext_err!r}"
        )
    return _ext
