"""Runtime kernel-variant configuration for the fused SAE step.

The GEMM kernels are compiled in two tile depths (see sae_kernels.hip):

  bk=32  128x128 tile, 64-66 KB LDS, 2 blocks/CU  (default)
  bk=16  128x128 tile, 32-33 KB LDS, 3 blocks/CU  (more co-resident blocks
         to hide barrier skew; <=80 VGPRs via __launch_bounds__(512, 6))

plus a `prio` flag (s_setprio(1) on the second-dispatched wave half) and a
`staging` choice for the encoder/code-grad GEMMs:

  staging="pre"  pre-transpose x/r/What once per step (k_transpose_scale),
                 then all GEMM operands stage direct (float4 LDS writes)
  staging="t"    transpose inside the GEMM staging (stride-BM+1 LDS, b32
                 writes), no separate transpose kernels

Defaults come from env (SC_AMD_BK / SC_AMD_PRIO / SC_AMD_STAGING) and can be
overridden per-process via set_kernel_config() — scripts/ktune.py sweeps the
combinations on a GPU box and the measured winner is baked into DEFAULTS.
"""

from __future__ import annotations

import os


# Measured winners (MI355X, flagship bench shape; see profiles/README.md).
DEFAULTS = {
    # ktune sweep r01 (profiles/README.md): t/bk16 639k acts/s vs t/bk32
    # 628k, pre/bk16 632k, pre/bk32 629k; prio neutral.  bk16 = 4 blocks/CU.
    "bk": 16,
    "prio": False,
    "staging": "t",
    # output-tile width of the GEMM kernels: 128 (default) or 256
    # (128x256 tiles: 4 accumulators/wave, 2 blocks/CU — measured slower
    # than bk16's 4-block co-residency on the flagship shape, kept as a
    # variant; see profiles/README.md)
    "bn": 128,
    # per-kernel overrides, None -> use "bk".  dec_fwd reduces over the
    # dictionary (K = n, long): the deeper TBK=32 tile wins there even
    # though TBK=16's occupancy wins the short-K kernels (643.6k vs 639.0k
    # acts/s on the flagship bench).
    "bk_grad_w": None,
    "bk_dec": 32,
    # per-kernel output-tile width for the SHORT-K kernels (enc_fwd/gc,
    # K = d): None -> "bn".  PMC r02: enc/gc sit at 0.52-0.53 MFMA-busy vs
    # dec's 0.83; the 128x256 tile halves their column panels (fewer
    # barriers per FLOP) at 2 blocks/CU — sweep via SC_AMD_BN_ENC.
    "bn_enc": None,
}

_cfg = dict(DEFAULTS)

if os.environ.get("SC_AMD_BK"):
    _cfg["bk"] = int(os.environ["SC_AMD_BK"])
if os.environ.get("SC_AMD_BN"):
    _cfg["bn"] = int(os.environ["SC_AMD_BN"])
if os.environ.get("SC_AMD_PRIO"):
    _cfg["prio"] = os.environ["SC_AMD_PRIO"] == "1"
if os.environ.get("SC_AMD_STAGING"):
    _cfg["staging"] = os.environ["SC_AMD_STAGING"]
if os.environ.get("SC_AMD_BK_GRAD_W"):
    _cfg["bk_grad_w"] = int(os.environ["SC_AMD_BK_GRAD_W"])
if os.environ.get("SC_AMD_BK_DEC"):
    _cfg["bk_dec"] = int(os.environ["SC_AMD_BK_DEC"])
if os.environ.get("SC_AMD_BN_ENC"):
    _cfg["bn_enc"] = int(os.environ["SC_AMD_BN_ENC"])


def kernel_config() -> dict:
    return dict(_cfg)


def set_kernel_config(**kwargs) -> None:
    for k, v in kwargs.items():
        if k not in _cfg:
            raise KeyError(f"unknown kernel-config key {k!r}; valid: {sorted(_cfg)}")
        _cfg[k] = v
    assert _cfg["bk"] in (16, 32) and _cfg["staging"] in ("pre", "t")
    assert _cfg["bn"] in (128, 256)
    assert _cfg["bn_enc"] in (None, 128, 256)
