"""Build the gfx950 HIP extension in-tree.

`python -m sparse_coding_amd.ops.build` produces
``sparse_coding_amd/ops/_sae_hip.so`` (kept next to the sources so the repo
snapshot carries it to GPU boxes; .so files are git-ignored).

Uses torch.utils.cpp_extension's compiler driver (hipcc underneath on ROCm)
with an explicit gfx950 offload arch; cross-compiles fine on CPU-only
machines.
"""

from __future__ import annotations

import os
import shutil
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(HERE, "hip", "sae_ops.hip")
OUT = os.path.join(HERE, "_sae_hip.so")
BUILD_DIR = os.path.join(HERE, "hip", "build")


def build(verbose: bool = True) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)

    from torch.utils.cpp_extension import load

    mod = load(
        name="_sae_hip",
        sources=[SRC],
        build_directory=BUILD_DIR,
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950", "-std=c++17"],
        verbose=verbose,
        is_python_module=False,
        keep_intermediates=True,
    )
    built = os.path.join(BUILD_DIR, "_sae_hip.so")
    shutil.copy2(built, OUT)
    return OUT


if __name__ == "__main__":
    path = build()
    print(f"built {path}")
