"""Trainable SAE signatures (init / loss / to_learned_dict static trios).

Semantics parity with reference ``autoencoders/sae_ensemble.py`` (each class
cited below).  A signature's ``loss`` is pure/functional so that
``torch.vmap(torch.func.grad(loss))`` batches it over the ensemble dimension —
that is the eager oracle; the fused HIP step in
``sparse_coding_amd.engine.hip_step`` implements the same math (including the
gradient of the in-forward decoder row renormalization) and is validated
against it.

Shared math lives in module-level helpers instead of being repeated per class.
"""

from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn
import torch.nn.functional as F

from sparse_coding_amd.models.learned_dict import (
    LearnedDict,
    ReverseSAE,
    TiedSAE,
    UntiedSAE,
    normalize_rows,
)


class DictSignature:
    """init/loss/to_learned_dict static trio (reference ensemble.py:15-22)."""

    @staticmethod
    def to_learned_dict(params, buffers):  # pragma: no cover - interface
        raise NotImplementedError

    @staticmethod
    def loss(params, buffers, batch):  # pragma: no cover - interface
        raise NotImplementedError


# ---------------------------------------------------------------------------
# shared building blocks
# ---------------------------------------------------------------------------

def _xavier(shape, device=None, dtype=None) -> torch.Tensor:
    w = torch.empty(shape, device=device, dtype=dtype)
    nn.init.xavier_uniform_(w)
    return w


def _scalar_buf(val, device=None, dtype=None) -> torch.Tensor:
    return torch.tensor(val, device=device, dtype=dtype)


def _encode_relu(weight, bias, batch):
    return torch.clamp(batch @ weight.T + bias, min=0.0)


def _decode(dictionary, c):
    return c @ dictionary


def _sae_losses(x_hat, target, c, l1_alpha, bias=None, bias_decay=None):
    """MSE over all elements + l1_alpha * mean_b ||c_b||_1 (+ bias decay).

    Reference: sae_ensemble.py:63-65 (untied), :148-150 (tied).
    """
    l_reconstruction = (x_hat - target).pow(2).mean()
    l_l1 = l1_alpha * torch.norm(c, 1, dim=-1).mean()
    total = l_reconstruction + l_l1
    loss_data = {
        "loss": total,
        "l_reconstruction": l_reconstruction,
        "l_l1": l_l1,
    }
    if bias_decay is not None:
        l_bias_decay = bias_decay * torch.norm(bias, 2)
        total = total + l_bias_decay
        loss_data["loss"] = total
        loss_data["l_bias_decay"] = l_bias_decay
    return total, (loss_data, {"c": c})


# ---------------------------------------------------------------------------
# signatures
# ---------------------------------------------------------------------------

class FunctionalSAE(DictSignature):
    """Untied SAE: raw encoder; decoder row-renormalized inside the forward.
    Reference: sae_ensemble.py:13-78."""

    @staticmethod
    def init(activation_size, n_dict_components, l1_alpha, bias_decay=0.0, device=None, dtype=None):
        params = {
            "encoder": _xavier((n_dict_components, activation_size), device, dtype),
            "encoder_bias": torch.zeros(n_dict_components, device=device, dtype=dtype),
            "decoder": _xavier((n_dict_components, activation_size), device, dtype),
        }
        buffers = {
            "l1_alpha": _scalar_buf(l1_alpha, device, dtype),
            "bias_decay": _scalar_buf(bias_decay, device, dtype),
        }
        return params, buffers

    @staticmethod
    def to_learned_dict(params, buffers):
        return UntiedSAE(params["encoder"], params["decoder"], params["encoder_bias"])

    @staticmethod
    def encode(params, buffers, batch):
        return _encode_relu(params["encoder"], params["encoder_bias"], batch)

    @staticmethod
    def loss(params, buffers, batch):
        c = _encode_relu(params["encoder"], params["encoder_bias"], batch)
        learned_dict = normalize_rows(params["decoder"])
        x_hat = _decode(learned_dict, c)
        return _sae_losses(
            x_hat, batch, c, buffers["l1_alpha"],
            bias=params["encoder_bias"], bias_decay=buffers["bias_decay"],
        )


class FunctionalTiedSAE(DictSignature):
    """Tied SAE: one weight matrix, row-normalized, used for encode AND
    decode; optional affine whitening-centering buffers.
    Reference: sae_ensemble.py:81-162."""

    @staticmethod
    def init(
        activation_size,
        n_dict_components,
        l1_alpha,
        device=None,
        dtype=None,
        bias_decay=0.0,
        translation=None,
        rotation=None,
        scaling=None,
    ):
        params = {
            "encoder": _xavier((n_dict_components, activation_size), device, dtype),
            "encoder_bias": torch.zeros(n_dict_components, device=device, dtype=dtype),
        }
        buffers = {
            "l1_alpha": _scalar_buf(l1_alpha, device, dtype),
            # NOTE: reference FunctionalTiedSAE.init drops its bias_decay arg
            # while .loss reads buffers["bias_decay"] (sae_ensemble.py:90,150)
            # — a latent KeyError there; we store it (default 0.0 keeps the
            # loss value identical).
            "bias_decay": _scalar_buf(bias_decay, device, dtype),
            "center_rot": rotation if rotation is not None else torch.eye(activation_size, device=device, dtype=dtype),
            "center_trans": translation if translation is not None else torch.zeros(activation_size, device=device, dtype=dtype),
            "center_scale": scaling if scaling is not None else torch.ones(activation_size, device=device, dtype=dtype),
        }
        return params, buffers

    @staticmethod
    def to_learned_dict(params, buffers):
        return TiedSAE(
            params["encoder"],
            params["encoder_bias"],
            centering=(buffers["center_trans"], buffers["center_rot"], buffers["center_scale"]),
            norm_encoder=True,
        )

    @staticmethod
    def center(buffers, batch):
        return (batch - buffers["center_trans"]) @ buffers["center_rot"].T * buffers["center_scale"]

    @staticmethod
    def uncenter(buffers, batch):
        return (batch / buffers["center_scale"]) @ buffers["center_rot"] + buffers["center_trans"]

    @staticmethod
    def loss(params, buffers, batch):
        learned_dict = normalize_rows(params["encoder"])
        batch_c = FunctionalTiedSAE.center(buffers, batch)
        c = _encode_relu(learned_dict, params["encoder_bias"], batch_c)
        x_hat_c = _decode(learned_dict, c)
        # loss is computed in centered space (reference sae_ensemble.py:148);
        # includes the bias-decay term of sae_ensemble.py:150
        return _sae_losses(
            x_hat_c, batch_c, c, buffers["l1_alpha"],
            bias=params["encoder_bias"], bias_decay=buffers["bias_decay"],
        )


class FunctionalTiedCenteredSAE(DictSignature):
    """Tied SAE with a *learnable* centering translation.
    Reference: sae_ensemble.py:164-230."""

    @staticmethod
    def init(activation_size, n_dict_components, l1_alpha, center=None, device=None, dtype=None):
        params = {
            "center": center if center is not None else torch.zeros(activation_size, device=device, dtype=dtype),
            "encoder": _xavier((n_dict_components, activation_size), device, dtype),
            "encoder_bias": torch.zeros(n_dict_components, device=device, dtype=dtype),
        }
        buffers = {"l1_alpha": _scalar_buf(l1_alpha, device, dtype)}
        return params, buffers

    @staticmethod
    def to_learned_dict(params, buffers):
        return TiedSAE(
            params["encoder"], params["encoder_bias"],
            centering=(params["center"], None, None), norm_encoder=True,
        )

    @staticmethod
    def loss(params, buffers, batch):
        learned_dict = normalize_rows(params["encoder"])
        batch_c = batch - params["center"]
        c = _encode_relu(learned_dict, params["encoder_bias"], batch_c)
        x_hat_c = _decode(learned_dict, c)
        return _sae_losses(x_hat_c, batch_c, c, buffers["l1_alpha"])


class FunctionalThresholdingSAE(DictSignature):
    """Soft-thresholding SAE with a relu6 gate (reference sae_ensemble.py:232-289)."""

    @staticmethod
    def init(activation_size, n_dict_components, l1_alpha, device=None, dtype=None):
        params = {
            "encoder": _xavier((n_dict_components, activation_size), device, dtype),
            "activation_scale": torch.ones(n_dict_components, device=device, dtype=dtype),
            "activation_gain": torch.zeros(n_dict_components, device=device, dtype=dtype),
        }
        buffers = {"l1_alpha": _scalar_buf(l1_alpha, device, dtype)}
        return params, buffers

    @staticmethod
    def gate(c, scale, gain):
        """The thresholding nonlinearity (sae_ensemble.py:256-259)."""
        a_sq = scale.pow(2)
        c = (c + gain) / torch.clamp(a_sq, 1e-8)
        c = F.relu6(60.0 * (c - 0.9)) / 6.0 + F.relu(c - 1.0)
        return c * a_sq

    @staticmethod
    def encode(params, batch, learned_dict):
        c = batch @ learned_dict.T
        return FunctionalThresholdingSAE.gate(c, params["activation_scale"], params["activation_gain"])

    @staticmethod
    def loss(params, buffers, batch):
        learned_dict = normalize_rows(params["encoder"])
        c = FunctionalThresholdingSAE.encode(params, batch, learned_dict)
        x_hat = _decode(learned_dict, c)
        return _sae_losses(x_hat, batch, c, buffers["l1_alpha"])

    @staticmethod
    def to_learned_dict(params, buffers):
        return ThresholdingSAE(params)


class ThresholdingSAE(LearnedDict):
    """Eval wrapper for FunctionalThresholdingSAE (sae_ensemble.py:292-305)."""

    def __init__(self, params):
        self.params = params
        self.n_feats, self.activation_size = params["encoder"].shape

    def get_learned_dict(self):
        return normalize_rows(self.params["encoder"])

    def encode(self, batch):
        return FunctionalThresholdingSAE.encode(self.params, batch, self.get_learned_dict())

    def to_device(self, device):
        self.params = {k: v.to(device) for k, v in self.params.items()}


class FunctionalMaskedTiedSAE(DictSignature):
    """Tied SAE padded to a common stack width with a coefficient mask, so
    ensembles of different dict sizes stack (reference sae_ensemble.py:309-373)."""

    @staticmethod
    def init(activation_size, n_dict_components, n_components_stack, l1_alpha, bias_decay=0.0, device=None, dtype=None):
        params = {
            "encoder": _xavier((n_components_stack, activation_size), device, dtype),
            "encoder_bias": torch.zeros(n_components_stack, device=device, dtype=dtype),
        }
        mask = torch.ones(n_components_stack, device=device, dtype=torch.bool)
        mask[:n_dict_components] = False
        buffers = {
            "l1_alpha": _scalar_buf(l1_alpha, device, dtype),
            "bias_decay": _scalar_buf(bias_decay, device, dtype),
            "dict_size": torch.tensor(n_dict_components, device=device, dtype=torch.long),
            "coef_mask": mask,
        }
        return params, buffers

    @staticmethod
    def to_learned_dict(params, buffers):
        k = buffers["dict_size"].item()
        return TiedSAE(params["encoder"][:k], params["encoder_bias"][:k], norm_encoder=True)

    @staticmethod
    def loss(params, buffers, batch):
        learned_dict = normalize_rows(params["encoder"])
        c = _encode_relu(learned_dict, params["encoder_bias"], batch)
        c = c.masked_fill(buffers["coef_mask"], 0.0)
        x_hat = _decode(learned_dict, c)
        return _sae_losses(x_hat, batch, c, buffers["l1_alpha"])


class FunctionalMaskedSAE(DictSignature):
    """Untied masked variant (reference sae_ensemble.py:377-444)."""

    @staticmethod
    def init(activation_size, n_dict_components, n_components_stack, l1_alpha, bias_decay=0.0, device=None, dtype=None):
        params = {
            "encoder": _xavier((n_components_stack, activation_size), device, dtype),
            "encoder_bias": torch.zeros(n_components_stack, device=device, dtype=dtype),
            "decoder": _xavier((n_components_stack, activation_size), device, dtype),
        }
        mask = torch.ones(n_components_stack, device=device, dtype=torch.bool)
        mask[:n_dict_components] = False
        buffers = {
            "l1_alpha": _scalar_buf(l1_alpha, device, dtype),
            "bias_decay": _scalar_buf(bias_decay, device, dtype),
            "dict_size": torch.tensor(n_dict_components, device=device, dtype=torch.long),
            "coef_mask": mask,
        }
        return params, buffers

    @staticmethod
    def to_learned_dict(params, buffers):
        k = buffers["dict_size"].item()
        return UntiedSAE(params["encoder"][:k], params["decoder"][:k], params["encoder_bias"][:k])

    @staticmethod
    def loss(params, buffers, batch):
        learned_dict = normalize_rows(params["decoder"])
        c = _encode_relu(params["encoder"], params["encoder_bias"], batch)
        c = c.masked_fill(buffers["coef_mask"], 0.0)
        x_hat = _decode(learned_dict, c)
        return _sae_losses(x_hat, batch, c, buffers["l1_alpha"])


class FunctionalReverseSAE(DictSignature):
    """Tied SAE that removes the bias from active features before decoding
    (reference sae_ensemble.py:447-503)."""

    @staticmethod
    def init(activation_size, n_dict_components, l1_alpha, bias_decay=0.0, device=None, dtype=None):
        params = {
            "encoder": _xavier((n_dict_components, activation_size), device, dtype),
            "encoder_bias": torch.zeros(n_dict_components, device=device, dtype=dtype),
        }
        buffers = {
            "l1_alpha": _scalar_buf(l1_alpha, device, dtype),
            "bias_decay": _scalar_buf(bias_decay, device, dtype),
        }
        return params, buffers

    @staticmethod
    def to_learned_dict(params, buffers):
        return ReverseSAE(params["encoder"], params["encoder_bias"], norm_encoder=True)

    @staticmethod
    def loss(params, buffers, batch):
        learned_dict = normalize_rows(params["encoder"])
        c = _encode_relu(learned_dict, params["encoder_bias"], batch)
        on = c > 0.0
        c = torch.where(on, c - params["encoder_bias"], c)
        x_hat = _decode(learned_dict, c)
        return _sae_losses(
            x_hat, batch, c, buffers["l1_alpha"],
            bias=params["encoder_bias"], bias_decay=buffers["bias_decay"],
        )


# Pickle compatibility with reference checkpoints: classes
# resolve as autoencoders.sae_ensemble.<Name>.
for _cls in (
    FunctionalSAE,
    FunctionalTiedSAE,
    FunctionalTiedCenteredSAE,
    FunctionalThresholdingSAE,
    ThresholdingSAE,
    FunctionalMaskedTiedSAE,
    FunctionalMaskedSAE,
    FunctionalReverseSAE,
):
    _cls.__module__ = "autoencoders.sae_ensemble"
# DictSignature itself lives in autoencoders.ensemble in the reference
DictSignature.__module__ = "autoencoders.ensemble"
