"""sparse_coding_amd: MI355X-native sparse-autoencoder training framework.

A from-scratch rebuild of the capabilities of HoagyC/sparse_coding
(reference layout in SURVEY.md) designed for AMD Instinct MI355X (gfx950):
PyTorch-ROCm host/glue, hand-written CDNA4 HIP kernels for the SAE ensemble
training hot loop, RCCL over xGMI for multi-GPU data parallelism.
"""

__version__ = "0.1.0"

from sparse_coding_amd.engine.ensemble import FunctionalEnsemble
from sparse_coding_amd.functional.optim import adam, sgd, optim_str_to_func
