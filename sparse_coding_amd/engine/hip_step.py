"""Fused CDNA4 training step for SAE ensembles.

Implements the reference's vmapped grad+Adam `step_batch`
(autoencoders/ensemble.py:119-123,175-193 + sae_ensemble.py losses) as a
pipeline of hand-written gfx950 HIP kernels (sources in
``sparse_coding_amd/ops/hip/``):

  k_row_norms    : per-row 1/max(||w||, 1e-8) of the dictionary      [M,n]
  k_fwd_gc       : encoder-GEMM → +bias → ReLU → decoder-GEMM →
                   residual, MSE/L1 partials, fired counts, and the
                   code-gradient GEMM → relu-mask → g_pre   (f32 MFMA)
  k_grad_w       : grad GEMMs over the batch dim → g_Ŵ (+ g_Wenc) + g_bias
  k_project_adam : analytic gradient of w/max(||w||,eps) (the in-forward
                   renormalization), fused Adam update of all params

Numerics: fp32 end to end (the reference trains fp32 — BASELINE.md), using
the exact-f32 MFMA path (`v_mfma_f32_*_f32`).  Validated against the
torch/vmap oracle in tests/test_hip_numerics.py.
"""

from __future__ import annotations

import math
from typing import Optional

import torch

from sparse_coding_amd import ops as _ops
from sparse_coding_amd.models import sae_signatures as sigs


def _identity_centering(buffers, n_models: int) -> bool:
    rot = buffers.get("center_rot")
    scale = buffers.get("center_scale")
    trans = buffers.get("center_trans")
    if rot is None and scale is None and trans is None:
        return True
    d = rot.shape[-1]
    eye = torch.eye(d, device=rot.device, dtype=rot.dtype).expand_as(rot)
    return (
        torch.equal(rot, eye)
        and bool((scale == 1).all())
        and bool((trans == 0).all())
    )


def maybe_make_step(ensemble, required: bool = False) -> Optional["HipSAEStep"]:
    """Return a HipSAEStep if (sig, device, buffers) are supported.

    On a CUDA device with a supported signature, a missing extension is a
    hard error (required or not): GPU runs must not silently fall back to
    eager (round-end check records which .so files were loaded).
    """
    import os

    if os.environ.get("SPARSE_CODING_AMD_FORCE_TORCH") == "1":
        return None

    dev = torch.device(ensemble.device) if not isinstance(ensemble.device, torch.device) else ensemble.device
    if dev.type != "cuda":
        if required:
            raise RuntimeError(f"backend='hip' requested but ensemble device is {dev}")
        return None

    sig = ensemble.sig
    tied = sig is sigs.FunctionalTiedSAE
    untied = sig is sigs.FunctionalSAE
    if not (tied or untied):
        if required:
            raise RuntimeError(f"backend='hip' requested but signature {sig.__name__} has no fused step yet")
        return None
    if tied and not _identity_centering(ensemble.buffers, ensemble.n_models):
        if required:
            raise RuntimeError("fused tied step requires identity centering buffers")
        return None

    ext = _ops.get_extension(required=True)  # loud on GPU
    return HipSAEStep(ensemble, ext, tied=tied)


class HipSAEStep:
    """Holds workspaces + launches the fused step for one ensemble."""

    def __init__(self, ensemble, ext, tied: bool):
        self.ens = ensemble
        self.ext = ext
        self.tied = tied
        self._ws = {}

        p = ensemble.params
        self.n_models, self.n_dict, self.d_act = p["encoder"].shape
        if not tied:
            assert p["decoder"].shape == p["encoder"].shape

        # per-model hyperparams as [M] fp32 device tensors
        b = ensemble.buffers
        self.l1_alpha = b["l1_alpha"].to(torch.float32).reshape(self.n_models).contiguous()
        bd = b.get("bias_decay")
        if bd is None:
            bd = torch.zeros(self.n_models, device=p["encoder"].device)
        self.bias_decay = bd.to(torch.float32).reshape(self.n_models).contiguous()

        opt = ensemble.optimizer_kwargs
        self.lr = float(opt.get("lr", 1e-3))
        betas = opt.get("betas", (0.9, 0.999))
        self.beta1, self.beta2 = float(betas[0]), float(betas[1])
        self.eps = float(opt.get("eps", 1e-8))
        if ensemble.optimizer_func is not None:
            name = getattr(ensemble.optimizer_func, "__name__", "adam")
            if name not in ("adam",):
                raise RuntimeError(f"fused step supports adam only, got {name}")

    def _workspace(self, name, shape, dtype=torch.float32):
        ws = self._ws.get(name)
        if ws is None or ws.shape != torch.Size(shape):
            ws = torch.empty(shape, device=self.ens.params["encoder"].device, dtype=dtype)
            self._ws[name] = ws
        return ws

    def step(self, minibatches: torch.Tensor, expand_dims: bool = True):
        ens = self.ens
        M, n, d = self.n_models, self.n_dict, self.d_act
        if expand_dims:
            x = minibatches  # [B, d] shared across models
            shared_x = True
        else:
            x = minibatches  # [M, B, d]
            shared_x = False
            raise NotImplementedError("per-model batches not yet supported by the HIP step")
        B = x.shape[0]
        x = x.contiguous()

        p = ens.params
        st = ens.optim_states
        enc = p["encoder"]
        bias = p["encoder_bias"]
        dec = enc if self.tied else p["decoder"]

        c = self._workspace("c", (M, B, n))
        gpre = self._workspace("gpre", (M, B, n))
        r = self._workspace("r", (M, B, d))
        inv_norms = self._workspace("inv_norms", (M, n))
        loss_parts = self._workspace("loss_parts", (M, 2))  # mse_sum, l1_sum
        fired = self._workspace("fired", (M, n))
        gw_dec = self._workspace("gw_dec", (M, n, d))
        gw_enc = gw_dec if self.tied else self._workspace("gw_enc", (M, n, d))
        g_bias = self._workspace("g_bias", (M, n))

        step_no = st["step"]  # [M] float
        mu_enc = st["mu"]["encoder"]
        nu_enc = st["nu"]["encoder"]
        mu_bias = st["mu"]["encoder_bias"]
        nu_bias = st["nu"]["encoder_bias"]
        if self.tied:
            mu_dec, nu_dec = mu_enc, nu_enc
        else:
            mu_dec, nu_dec = st["mu"]["decoder"], st["nu"]["decoder"]

        self.ext.sae_step_f32(
            x, enc, dec, bias,
            self.l1_alpha, self.bias_decay,
            c, gpre, r, inv_norms, loss_parts, fired,
            gw_dec, gw_enc, g_bias,
            mu_enc, nu_enc, mu_dec, nu_dec, mu_bias, nu_bias,
            step_no,
            self.lr, self.beta1, self.beta2, self.eps,
            self.tied,
        )

        mse = loss_parts[:, 0] / (B * d)
        l1 = loss_parts[:, 1] / B * self.l1_alpha
        bias_norm = torch.norm(bias, 2, dim=-1)
        l_bd = self.bias_decay * bias_norm
        total = mse + l1 + l_bd
        loss_data = {
            "loss": total,
            "l_reconstruction": mse,
            "l_l1": l1,
            "l_bias_decay": l_bd,
        }
        aux_data = {"c": c}
        return loss_data, aux_data
